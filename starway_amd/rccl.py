"""RCCL fan-out: ncclSend/ncclRecv groups over the xGMI mesh.

The tagged engine delivers each message with a single hipIpc pull — one
xGMI link per pair. For all-pairs / collective-shaped traffic RCCL
schedules across all 7 links per MI355X, so this module is the right tool
when every endpoint talks to every other endpoint at once (the
multi-endpoint fan-out the reference delegated to UCX's multi-transport
layer).

Usage (one process per GPU):

    import starway_amd.rccl as swr
    uid = swr.unique_id() if rank == 0 else None
    uid = <broadcast uid via starway tagged messages / gloo / any oob>
    mesh = swr.RcclMesh(uid, rank=rank, world=world, device=local_rank)
    mesh.all_to_all(send_tensor, recv_tensor)   # peer-major equal chunks
    mesh.synchronize()

Rendezvous of the 128-byte unique id is out of band by design — e.g. over
a starway Server/Client pair, which is what `bootstrap_from_messaging`
does.
"""
from __future__ import annotations

import ctypes
from importlib.util import find_spec
from pathlib import Path
from typing import Any

# Same loader policy as the HIP runtime in __init__: prefer torch's bundled
# librccl (soname librccl.so.1) so torch + our module share one copy.
_spec = find_spec("torch")
if _spec and _spec.origin:
    _p = Path(_spec.origin).parent / "lib" / "librccl.so"
    if _p.exists():
        try:
            ctypes.CDLL(str(_p), mode=ctypes.RTLD_GLOBAL)
        except OSError:
            pass

from ._rccl import RcclGroup, unique_id  # noqa: E402


def _devptr(t: Any) -> tuple[int, int]:
    """(ptr, nbytes) of a contiguous HIP tensor."""
    if not (hasattr(t, "is_cuda") and t.is_cuda):
        raise ValueError("RCCL fan-out needs HIP device tensors")
    if not t.is_contiguous():
        raise ValueError("tensor must be contiguous")
    return t.data_ptr(), t.numel() * t.element_size()


class RcclMesh:
    """Convenience wrapper over RcclGroup for torch tensors."""

    def __init__(self, uid: bytes, rank: int, world: int, device: int):
        self._g = RcclGroup(uid, rank, world, device)
        self.rank = rank
        self.world = world

    def all_to_all(self, send: Any, recv: Any) -> None:
        """Equal-chunk all-to-all: send/recv are peer-major tensors whose
        byte size is divisible by world; chunk i goes to/comes from peer i
        (including self)."""
        sptr, snb = _devptr(send)
        rptr, rnb = _devptr(recv)
        if snb != rnb or snb % self.world:
            raise ValueError("send/recv sizes must match and divide by world")
        self._g.all_to_all(sptr, rptr, snb // self.world)

    def sendrecv(self, send: Any, recv: Any, peer: int) -> None:
        """Bidirectional exchange with one peer (grouped, deadlock-free)."""
        sptr, snb = _devptr(send)
        rptr, rnb = _devptr(recv)
        self._g.group_start()
        self._g.send(sptr, snb, peer)
        self._g.recv(rptr, rnb, peer)
        self._g.group_end()

    def send(self, t: Any, peer: int) -> None:
        ptr, nb = _devptr(t)
        self._g.send(ptr, nb, peer)

    def recv(self, t: Any, peer: int) -> None:
        ptr, nb = _devptr(t)
        self._g.recv(ptr, nb, peer)

    def group_start(self) -> None:
        self._g.group_start()

    def group_end(self) -> None:
        self._g.group_end()

    def synchronize(self) -> None:
        self._g.synchronize()


async def bootstrap_from_messaging(role: str, endpoint, *, rank: int,
                                   world: int, device: int,
                                   tag: int = 0x7CC1) -> RcclMesh:
    """Exchange the ncclUniqueId over an established starway connection.

    role="root": `endpoint` is a (Server, [eps]) tuple — generates the id
    and tag-sends it to every ep. role="peer": `endpoint` is a connected
    Client — receives it.
    """
    import numpy as np

    if role == "root":
        server, eps = endpoint
        uid = unique_id()
        buf = np.frombuffer(uid, dtype=np.uint8).copy()
        for ep in eps:
            await server.asend(ep, buf, tag)
        await server.aflush()
    elif role == "peer":
        client = endpoint
        buf = np.zeros(128, dtype=np.uint8)
        _, length = await client.arecv(buf, tag, (1 << 64) - 1)
        uid = bytes(buf[:length])
    else:
        raise ValueError("role must be 'root' or 'peer'")
    return RcclMesh(uid, rank=rank, world=world, device=device)


__all__ = ["RcclGroup", "RcclMesh", "unique_id", "bootstrap_from_messaging"]
