"""RCCL fan-out module tests (MI355X). World=1 exercises the full
init/group/all-to-all/sync path (self send/recv); multi-GPU scheduling is
covered by the driver's 8-GPU bench run (bench.py --transport rccl)."""
import pytest

pytestmark = pytest.mark.gpu

pytest.importorskip("torch")
import torch  # noqa: E402


def test_rccl_unique_id_len():
    from starway_amd import rccl as swr

    uid = swr.unique_id()
    assert isinstance(uid, bytes) and len(uid) == 128


def test_rccl_self_alltoall_exact():
    from starway_amd import rccl as swr

    uid = swr.unique_id()
    mesh = swr.RcclMesh(uid, rank=0, world=1, device=0)
    n = 1 << 20
    send = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    recv = torch.zeros_like(send)
    torch.cuda.synchronize()
    mesh.all_to_all(send, recv)
    mesh.synchronize()
    assert torch.equal(send, recv)


def test_rccl_self_sendrecv():
    from starway_amd import rccl as swr

    uid = swr.unique_id()
    mesh = swr.RcclMesh(uid, rank=0, world=1, device=0)
    send = torch.arange(4096, dtype=torch.uint8, device="cuda") % 251
    recv = torch.zeros_like(send)
    torch.cuda.synchronize()
    mesh.sendrecv(send, recv, peer=0)
    mesh.synchronize()
    assert torch.equal(send, recv)


def test_rccl_rejects_cpu_tensor():
    from starway_amd import rccl as swr

    uid = swr.unique_id()
    mesh = swr.RcclMesh(uid, rank=0, world=1, device=0)
    with pytest.raises(ValueError):
        mesh.send(torch.zeros(16, dtype=torch.uint8), 0)
