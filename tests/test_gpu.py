"""GPU-path tests (MI355X): gfx950 copy-kernel numerics, device-tensor
tagged messaging (same-process raw-ptr path and cross-process hipIpc path),
mixed host/device transfers, flush-with-GPU-sends, truncation RECV_FAIL.

Numerics policy: message delivery is a byte-exact copy, so every test
compares against the untouched source tensor (the PyTorch reference of a
copy op is the tensor itself).
"""
import asyncio
import contextlib
import multiprocessing as mp
import os
import random

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

pytest.importorskip("torch")
import torch  # noqa: E402

import starway_amd as sw  # noqa: E402
from starway_amd import _core  # noqa: E402



@contextlib.asynccontextmanager
async def loopback():
    server = sw.Server()
    client = sw.Client()
    addr = server.listen_address()
    await client.aconnect_address(addr)
    try:
        yield server, client
    finally:
        await client.aclose()
        await server.aclose()


# =============================================================================
# Copy-kernel numerics (direct)
# =============================================================================


@pytest.mark.parametrize("nbytes", [1, 16, 4096, 1 << 20, (1 << 20) + 13])
def test_copy_kernel_exact(nbytes):
    src = torch.randint(0, 256, (nbytes,), dtype=torch.uint8, device="cuda")
    dst = torch.zeros_like(src)
    torch.cuda.synchronize()
    _core._copy_device_sync(dst.data_ptr(), src.data_ptr(), nbytes, 0)
    assert torch.equal(src, dst)


def test_copy_kernel_unaligned_slices():
    base = torch.randint(0, 256, (1 << 16,), dtype=torch.uint8, device="cuda")
    out = torch.zeros_like(base)
    torch.cuda.synchronize()
    # co-aligned but offset from the 16B boundary
    for off, ln in [(1, 4097), (3, 1 << 14), (7, 255), (15, 16)]:
        _core._copy_device_sync(
            out.data_ptr() + off, base.data_ptr() + off, ln, 0
        )
        assert torch.equal(base[off : off + ln], out[off : off + ln])


def test_copy_kernel_large_nt_path():
    # >64 MiB goes through the non-temporal kernel variant.
    n = 96 << 20
    src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
    dst = torch.zeros_like(src)
    torch.cuda.synchronize()
    _core._copy_device_sync(dst.data_ptr(), src.data_ptr(), n, 0)
    assert torch.equal(src, dst)


# =============================================================================
# Same-process device messaging (raw-ptr RTS path)
# =============================================================================


@pytest.mark.parametrize("nbytes", [64, 4096, 1 << 20, 64 << 20])
async def test_device_send_recv_same_process(nbytes):
    async with loopback() as (server, client):
        src = torch.randint(0, 256, (nbytes,), dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        fut = server.arecv(dst, 0, 0)
        await client.asend(src, 5)
        tag, length = await fut
        torch.cuda.synchronize()
        assert tag == 5 and length == nbytes
        assert torch.equal(src, dst)


async def test_device_send_before_recv_posted():
    # RTS lands in the unexpected queue; matched when the recv is posted.
    async with loopback() as (server, client):
        src = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        send_fut = client.asend(src, 77)
        await asyncio.sleep(0.05)
        tag, length = await server.arecv(dst, 77, (1 << 64) - 1)
        await send_fut
        torch.cuda.synchronize()
        assert tag == 77 and torch.equal(src, dst)


async def test_device_server_to_client():
    async with loopback() as (server, client):
        ep = next(iter(server.list_clients()))
        src = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        fut = client.arecv(dst, 0, 0)
        await asyncio.sleep(0.01)
        await server.asend(ep, src, 9)
        tag, _ = await fut
        torch.cuda.synchronize()
        assert tag == 9 and torch.equal(src, dst)


async def test_device_duplex_concurrent():
    async with loopback() as (server, client):
        ep = next(iter(server.list_clients()))
        n = 8 << 20
        a = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        b = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        ra = torch.zeros_like(a)
        rb = torch.zeros_like(b)
        torch.cuda.synchronize()
        futs = await asyncio.gather(
            server.arecv(ra, 1, (1 << 64) - 1),
            client.arecv(rb, 2, (1 << 64) - 1),
            client.asend(a, 1),
            server.asend(ep, b, 2),
        )
        torch.cuda.synchronize()
        assert futs[0][0] == 1 and futs[1][0] == 2
        assert torch.equal(a, ra) and torch.equal(b, rb)


# =============================================================================
# Mixed host/device
# =============================================================================


async def test_gpu_to_cpu_recv():
    async with loopback() as (server, client):
        src = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda")
        dst = np.zeros(1 << 20, dtype=np.uint8)
        torch.cuda.synchronize()
        fut = server.arecv(dst, 0, 0)
        await client.asend(src, 3)
        tag, length = await fut
        assert tag == 3 and length == 1 << 20
        np.testing.assert_array_equal(src.cpu().numpy(), dst)


async def test_cpu_to_gpu_recv():
    async with loopback() as (server, client):
        src = np.random.randint(0, 256, 1 << 20, dtype=np.uint8)
        dst = torch.zeros(1 << 20, dtype=torch.uint8, device="cuda")
        fut = server.arecv(dst, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(src, 4)
        tag, length = await fut
        torch.cuda.synchronize()
        assert tag == 4 and length == 1 << 20
        np.testing.assert_array_equal(src, dst.cpu().numpy())


# =============================================================================
# Flush / cancel / truncation semantics with GPU sends
# =============================================================================


async def test_gpu_send_completion_means_delivery():
    async with loopback() as (server, client):
        src = torch.full((1 << 16,), 7, dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        fut = server.arecv(dst, 0, 0)
        await client.asend(src, 1)  # completes only after RECV_DONE
        await fut
        torch.cuda.synchronize()
        assert torch.equal(src, dst)
        await client.aflush()  # nothing pending; must resolve


async def test_gpu_truncation_fails_both_sides():
    # Rendezvous-size message (past the inbox cutoff): truncation is
    # reported back to the sender via RECV_FAIL.
    async with loopback() as (server, client):
        src = torch.zeros(64 << 10, dtype=torch.uint8, device="cuda")
        dst = torch.zeros(128, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        recv_fut = server.arecv(dst, 0, 0)
        send_fut = client.asend(src, 1)
        with pytest.raises(Exception, match="truncated"):
            await recv_fut
        with pytest.raises(Exception, match="truncated"):
            await send_fut


async def test_gpu_small_truncation_fails_receiver_only():
    # Inbox-size message: eager semantics (like the CPU eager path) — the
    # receiver fails with truncation, the sender's push completes.
    async with loopback() as (server, client):
        src = torch.zeros(1024, dtype=torch.uint8, device="cuda")
        dst = torch.zeros(128, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        recv_fut = server.arecv(dst, 0, 0)
        send_fut = client.asend(src, 1)
        with pytest.raises(Exception, match="truncated"):
            await recv_fut
        await send_fut  # completes: payload captured by the push kernel


async def test_gpu_pending_send_canceled_on_close():
    server = sw.Server()
    client = sw.Client()
    addr = server.listen_address()
    await client.aconnect_address(addr)
    src = torch.zeros(1 << 20, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    send_fut = client.asend(src, 123)  # no recv posted: RTS never acked
    await asyncio.sleep(0.05)
    close_fut = client.aclose()
    with pytest.raises(Exception, match="cancel|reset"):
        await send_fut
    await close_fut
    await server.aclose()


# =============================================================================
# Cross-process hipIpc path (two processes, one GPU)
# =============================================================================


def _ipc_child_server(port: int, nbytes: int, ready):
    import torch

    import starway_amd as sw

    async def inner():
        server = sw.Server()
        server.listen("127.0.0.1", port)
        connected = asyncio.Event()
        loop = asyncio.get_running_loop()
        server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(connected.set))
        ready.set()  # listening: parent may connect now
        await connected.wait()
        ep = next(iter(server.list_clients()))
        src = torch.arange(nbytes, dtype=torch.uint8, device="cuda") % 251
        src = src.contiguous()
        torch.cuda.synchronize()
        await server.asend(ep, src, 21)  # completes on RECV_DONE (delivery)
        await server.aflush_ep(ep)
        await server.aclose()

    asyncio.run(inner())


async def test_cross_process_ipc_device_transfer(port):
    nbytes = 32 << 20
    ctx = mp.get_context("spawn")
    ready = ctx.Event()
    p = ctx.Process(target=_ipc_child_server, args=(port, nbytes, ready))
    p.start()
    try:
        assert ready.wait(120), "child server did not come up"
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        dst = torch.zeros(nbytes, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        tag, length = await client.arecv(dst, 0, 0)
        torch.cuda.synchronize()
        assert tag == 21 and length == nbytes
        expect = torch.arange(nbytes, dtype=torch.uint8, device="cuda") % 251
        assert torch.equal(dst, expect)
        await client.aclose()
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()


def _ipc_child_client(port: int, nbytes: int):
    import torch

    import starway_amd as sw

    async def inner():
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        src = (torch.arange(nbytes, dtype=torch.uint8, device="cuda") * 3) % 241
        src = src.contiguous()
        torch.cuda.synchronize()
        await client.asend(src, 33)
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_cross_process_ipc_client_to_server(port):
    nbytes = 8 << 20
    server = sw.Server()
    server.listen("127.0.0.1", port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_ipc_child_client, args=(port, nbytes))
    p.start()
    try:
        dst = torch.zeros(nbytes, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        tag, length = await server.arecv(dst, 0, 0)
        torch.cuda.synchronize()
        assert tag == 33 and length == nbytes
        expect = (torch.arange(nbytes, dtype=torch.uint8, device="cuda") * 3) % 241
        assert torch.equal(dst, expect)
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
        await server.aclose()


# =============================================================================
# Benchmark-path sanity: bench.py loopback step on GPU
# =============================================================================


async def test_bench_loopback_step():
    async with loopback() as (server, client):
        n = 16 << 20
        src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        for step in range(3):
            tag = (1 << 60) | step
            fut = server.arecv(dst, tag, (1 << 64) - 1)
            await client.asend(src, tag)
            await fut
        torch.cuda.synchronize()
        assert torch.equal(src, dst)


# =============================================================================
# Strided (non-contiguous) device tensors: pack/unpack in the pull kernel
# =============================================================================


async def test_strided_send_to_contiguous_recv():
    async with loopback() as (server, client):
        base = torch.randint(0, 256, (64, 256), dtype=torch.uint8,
                             device="cuda")
        sl = base[:, 16:144]  # 2D row-strided slice, 64 x 128
        dst = torch.zeros(64, 128, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        fut = server.arecv(dst, 0, 0)
        await client.asend(sl, 8)
        tag, ln = await fut
        torch.cuda.synchronize()
        assert tag == 8 and ln == 64 * 128
        assert torch.equal(sl.contiguous(), dst)


async def test_contiguous_send_to_strided_recv():
    async with loopback() as (server, client):
        src = torch.randint(0, 256, (32, 64), dtype=torch.uint8,
                            device="cuda")
        frame = torch.zeros(32, 256, dtype=torch.uint8, device="cuda")
        window = frame[:, 100:164]  # strided destination view
        torch.cuda.synchronize()
        fut = server.arecv(window, 0, 0)
        await client.asend(src, 9)
        tag, ln = await fut
        torch.cuda.synchronize()
        assert tag == 9 and ln == 32 * 64
        assert torch.equal(src, window.contiguous())
        assert frame[:, :100].eq(0).all() and frame[:, 164:].eq(0).all()


async def test_strided_to_strided_roundtrip():
    async with loopback() as (server, client):
        a = torch.randint(0, 256, (16, 512), dtype=torch.uint8, device="cuda")
        b = torch.zeros(16, 512, dtype=torch.uint8, device="cuda")
        ssl = a[:, 7:263]
        dsl = b[:, 33:289]
        torch.cuda.synchronize()
        fut = server.arecv(dsl, 0, 0)
        await client.asend(ssl, 10)
        await fut
        torch.cuda.synchronize()
        assert torch.equal(ssl.contiguous(), dsl.contiguous())


async def test_strided_geometry_mismatch_fails():
    async with loopback() as (server, client):
        src = torch.zeros(8, 64, dtype=torch.uint8, device="cuda")
        frame = torch.zeros(8, 256, dtype=torch.uint8, device="cuda")
        wrong = frame[:, :32]  # strided dst, wrong total size
        torch.cuda.synchronize()
        fut = server.arecv(wrong, 0, 0)
        sfut = client.asend(src[:, ::1], 11)
        with pytest.raises(Exception, match="geometry|truncated"):
            await fut
        # Small (inbox-eager) send: the sender is not notified of the
        # receiver-side geometry failure — eager semantics.
        await sfut


async def test_cross_host_gpu_bounce_path():
    """Cross-host GPU sends stage D2H and ship as eager (hipIpc cannot
    cross hosts). Forced via STARWAY_FORCE_XHOST; receiver H2D-bounces into
    the posted device buffer."""
    os.environ["STARWAY_FORCE_XHOST"] = "1"
    try:
        async with loopback() as (server, client):
            src = torch.randint(0, 256, (4 << 20,), dtype=torch.uint8,
                                device="cuda")
            dst = torch.zeros_like(src)
            torch.cuda.synchronize()
            fut = server.arecv(dst, 0, 0)
            await client.asend(src, 12)
            await client.aflush()  # must cover the staged wire bytes
            tag, ln = await fut
            torch.cuda.synchronize()
            assert tag == 12 and ln == 4 << 20
            assert torch.equal(src, dst)
    finally:
        del os.environ["STARWAY_FORCE_XHOST"]

def _strided_ipc_child(port, ready):
    import torch

    import starway_amd as sw

    async def inner():
        server = sw.Server()
        server.listen("127.0.0.1", port)
        connected = asyncio.Event()
        loop = asyncio.get_running_loop()
        server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(connected.set))
        ready.set()
        await connected.wait()
        ep = next(iter(server.list_clients()))
        base = (torch.arange(64 * 512, dtype=torch.int32, device="cuda")
                .reshape(64, 512) % 251).to(torch.uint8)
        sl = base[:, 100:356]  # strided 64 x 256 slice
        torch.cuda.synchronize()
        await server.asend(ep, sl, 44)
        await server.aflush_ep(ep)
        await server.aclose()

    asyncio.run(inner())


async def test_cross_process_strided_ipc(port):
    """Strided source over the cross-process hipIpc path: the pull kernel
    packs rows out of the remote allocation."""
    ctx = mp.get_context("spawn")
    ready = ctx.Event()
    p = ctx.Process(target=_strided_ipc_child, args=(port, ready))
    p.start()
    try:
        assert ready.wait(120)
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        dst = torch.zeros(64, 256, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        tag, ln = await client.arecv(dst, 0, 0)
        torch.cuda.synchronize()
        assert tag == 44 and ln == 64 * 256
        expect = (torch.arange(64 * 512, dtype=torch.int32, device="cuda")
                  .reshape(64, 512) % 251).to(torch.uint8)[:, 100:356]
        assert torch.equal(dst, expect.contiguous())
        await client.aclose()
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()


# =============================================================================
# Small-message inbox plane (push kernel + unpack/doorbell, smallmsg.hip)
# =============================================================================


@pytest.mark.parametrize("nbytes", [1, 64, 1024, 4096])
async def test_inbox_small_roundtrip(nbytes):
    async with loopback() as (server, client):
        src = torch.randint(0, 256, (nbytes,), dtype=torch.uint8,
                            device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        # Cold start: the ring's bring-up probe includes the first kernel
        # module load (~15 ms); settle so the stats assertion below sees
        # the activated plane even when this test runs first.
        await asyncio.sleep(0.3)
        fut = server.arecv(dst, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(src, 5)
        tag, length = await fut
        torch.cuda.synchronize()
        assert tag == 5 and length == nbytes
        assert torch.equal(src, dst)
        assert client._client.get_stats()["inbox_tx"] == 1
        assert server._server.get_stats()["inbox_rx"] == 1


async def test_inbox_unexpected_then_posted():
    async with loopback() as (server, client):
        src = torch.arange(1024, dtype=torch.uint8, device="cuda") % 251
        torch.cuda.synchronize()
        await client.asend(src, 42)
        await client.aflush()
        await asyncio.sleep(0.05)  # lands unmatched: parked in the slot
        dst = torch.zeros_like(src)
        tag, length = await server.arecv(dst, 0, 0)
        torch.cuda.synchronize()
        assert tag == 42 and length == 1024
        assert torch.equal(src, dst)


async def test_inbox_many_concurrent():
    # The small-messages workload shape: 64 concurrent 1 KiB sends —
    # batched through the push/unpack kernels, not one launch per message.
    async with loopback() as (server, client):
        n = 64
        srcs = [torch.full((1024,), i % 251, dtype=torch.uint8, device="cuda")
                for i in range(n)]
        dsts = [torch.zeros(1024, dtype=torch.uint8, device="cuda")
                for _ in range(n)]
        torch.cuda.synchronize()
        await asyncio.sleep(0.3)  # cold-start probe settle (stats assert)
        recvs = [server.arecv(dsts[i], 0, 0) for i in range(n)]
        await asyncio.sleep(0.01)
        await asyncio.gather(*(client.asend(srcs[i], 100 + i)
                               for i in range(n)))
        got = await asyncio.gather(*recvs)
        torch.cuda.synchronize()
        assert {t for t, _ in got} == set(range(100, 100 + n))
        total = sum(int(d[0]) == (t - 100) % 251 for d, (t, _)
                    in zip(dsts, got))
        for d, (t, _) in zip(dsts, got):
            assert torch.all(d == (t - 100) % 251), (t, d[0].item())
        assert total == n
        assert server._server.get_stats()["inbox_rx"] == n


async def test_inbox_ring_wraps_many_batches():
    # More messages than ring slots: exercises credit flow + slot reuse.
    async with loopback() as (server, client):
        rounds, n = 8, 32
        for r in range(rounds):
            srcs = [torch.full((512,), (r * n + i) % 251, dtype=torch.uint8,
                               device="cuda") for i in range(n)]
            dsts = [torch.zeros(512, dtype=torch.uint8, device="cuda")
                    for _ in range(n)]
            torch.cuda.synchronize()
            recvs = [server.arecv(dsts[i], 0, 0) for i in range(n)]
            await asyncio.sleep(0)
            await asyncio.gather(*(client.asend(srcs[i], i)
                                   for i in range(n)))
            await asyncio.gather(*recvs)
            torch.cuda.synchronize()
            for i in range(n):
                assert torch.equal(srcs[i], dsts[i])


async def test_inbox_order_with_rts_interleaved():
    # Per-sender FIFO across planes: small (inbox), big (RTS), small, with
    # wildcard recvs — delivery order must follow send order.
    async with loopback() as (server, client):
        small1 = torch.full((256,), 1, dtype=torch.uint8, device="cuda")
        big = torch.full((1 << 20,), 2, dtype=torch.uint8, device="cuda")
        small2 = torch.full((256,), 3, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        # Enqueue order defines per-sender FIFO; the RTS send (12) only
        # completes at delivery, so it cannot be awaited before the recvs.
        sends = [client.asend(small1, 11), client.asend(big, 12),
                 client.asend(small2, 13)]
        seen = []
        for _ in range(3):
            dst = torch.zeros(1 << 20, dtype=torch.uint8, device="cuda")
            tag, length = await server.arecv(dst, 0, 0)
            torch.cuda.synchronize()
            seen.append((tag, length, int(dst[0])))
        await asyncio.gather(*sends)
        await client.aflush()
        assert seen == [(11, 256, 1), (12, 1 << 20, 2), (13, 256, 3)]


async def test_inbox_small_to_host_recv():
    # Device sender, host numpy recv buffer: unpack stages through a pinned
    # bounce the engine memcpys out of.
    async with loopback() as (server, client):
        src = torch.arange(777, dtype=torch.uint8, device="cuda") % 251
        torch.cuda.synchronize()
        dst = np.zeros(777, dtype=np.uint8)
        fut = server.arecv(dst, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(src, 6)
        tag, length = await fut
        assert tag == 6 and length == 777
        np.testing.assert_array_equal(src.cpu().numpy(), dst)


def _doorbell_child(q):
    import os

    os.environ["STARWAY_DOORBELL"] = "1"  # opt-in; read at first use
    q.put(asyncio.run(_doorbell_body()))


async def _doorbell_body():
    async with loopback() as (server, client):
        ep = next(iter(server.list_clients()))
        await asyncio.sleep(0.3)  # cold-start probe settle (stats assert)
        ping = torch.full((64,), 7, dtype=torch.uint8, device="cuda")
        pong = torch.full((64,), 9, dtype=torch.uint8, device="cuda")
        rx_s = torch.zeros(64, dtype=torch.uint8, device="cuda")
        rx_c = torch.zeros(64, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        iters = 50
        for _ in range(iters):
            sfut = server.arecv(rx_s, 1, (1 << 64) - 1)
            await asyncio.sleep(0)
            await client.asend(ping, 1)
            await sfut
            cfut = client.arecv(rx_c, 2, (1 << 64) - 1)
            await asyncio.sleep(0)
            await server.asend(ep, pong, 2)
            await cfut
        torch.cuda.synchronize()
        assert torch.all(rx_s == 7) and torch.all(rx_c == 9)
        sstats = server._server.get_stats()
        cstats = client._client.get_stats()
        assert sstats["inbox_rx"] + cstats["inbox_rx"] == 2 * iters
        return sstats["doorbell_rx"] + cstats["doorbell_rx"]


async def test_doorbell_preposted_latency_path(port):
    # The doorbell is opt-in (STARWAY_DOORBELL=1, read at first engine
    # use) so it runs in a subprocess; it must deliver at least some of
    # the pre-posted pingpong iterations.
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_doorbell_child, args=(q,))
    p.start()
    try:
        doorbell_rx = q.get(timeout=180)
        assert doorbell_rx > 0
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()


def _inbox_child_client(port: int):
    import torch

    import starway_amd as sw

    async def inner():
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        await asyncio.sleep(0.3)  # let the ring's bring-up probe activate
        for i in range(16):
            src = torch.full((1024,), i, dtype=torch.uint8, device="cuda")
            torch.cuda.synchronize()
            await client.asend(src, 50 + i)
        await client.aflush()
        assert client._client.get_stats()["inbox_tx"] == 16
        await client.aclose()

    asyncio.run(inner())


async def test_inbox_cross_process(port):
    # Two processes sharing one GPU: the push kernel writes the server's
    # ring through the hipIpc mapping (the xGMI path on multi-GPU nodes).
    server = sw.Server()
    server.listen("127.0.0.1", port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_inbox_child_client, args=(port,))
    p.start()
    try:
        for i in range(16):
            dst = torch.zeros(1024, dtype=torch.uint8, device="cuda")
            torch.cuda.synchronize()
            tag, length = await server.arecv(dst, 50 + i, (1 << 64) - 1)
            torch.cuda.synchronize()
            assert length == 1024 and torch.all(dst == i)
        assert server._server.get_stats()["inbox_rx"] == 16
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
        await server.aclose()


# =============================================================================
# Calibrated perf model
# =============================================================================


async def test_calibrated_evaluate_perf(tmp_path):
    import os
    os.environ["STARWAY_CALIB_FILE"] = str(tmp_path / "perf.cal")
    vals = _core.calibrate(force=True)
    assert vals["same_gpu_gbps"] > 100  # HBM-class copy rate
    assert (tmp_path / "perf.cal").exists()
    # evaluate_perf must land within 2x of a measured 64 MiB transfer.
    async with loopback() as (server, client):
        n = 64 << 20
        src = torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda")
        dst = torch.zeros_like(src)
        torch.cuda.synchronize()
        import time as _t
        # warmup
        fut = server.arecv(dst, 0, 0)
        await client.asend(src, 1)
        await fut
        t0 = _t.perf_counter()
        reps = 5
        for _ in range(reps):
            fut = server.arecv(dst, 0, 0)
            await client.asend(src, 1)
            await fut
        measured = (_t.perf_counter() - t0) / reps
        predicted = client.evaluate_perf(n)
        assert predicted > 0
        assert measured / 2 <= predicted <= measured * 2, (
            predicted, measured)


def _inbox_flush_close_sender(port: int):
    import torch

    import starway_amd as sw

    async def inner():
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        await asyncio.sleep(0.3)  # let the ring's bring-up probe activate
        for i in range(8):
            src = torch.full((512,), 40 + i, dtype=torch.uint8,
                             device="cuda")
            torch.cuda.synchronize()
            await client.asend(src, 70 + i)
        await client.aflush()
        await client.aclose()  # sender gone before any recv is posted

    asyncio.run(inner())


async def test_inbox_flush_then_close_delivers(port):
    # Flush guarantees delivery even if the sender closes before the
    # receiver posts recvs: the payloads already live in the receiver's
    # ring and the control frames are drained before the connection dies.
    server = sw.Server()
    server.listen("127.0.0.1", port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_inbox_flush_close_sender, args=(port,))
    p.start()
    try:
        p.join(timeout=120)
        assert p.exitcode == 0
        await asyncio.sleep(0.2)  # conn death observed; slots must survive
        for i in range(8):
            dst = torch.zeros(512, dtype=torch.uint8, device="cuda")
            torch.cuda.synchronize()
            tag, ln = await asyncio.wait_for(
                server.arecv(dst, 70 + i, (1 << 64) - 1), timeout=30)
            torch.cuda.synchronize()
            assert ln == 512 and torch.all(dst == 40 + i)
    finally:
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
        await server.aclose()


def _no_inbox_child_client(port: int):
    import os

    os.environ["STARWAY_INBOX"] = "0"
    import torch

    import starway_amd as sw

    async def inner():
        client = sw.Client()
        await client.aconnect("127.0.0.1", port)
        src = torch.full((1024,), 55, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        await client.asend(src, 91)  # RTS path: completes at delivery
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_small_device_send_with_inbox_disabled(port):
    # STARWAY_INBOX=0 on either side falls back to the RTS pull plane.
    server = sw.Server()
    server.listen("127.0.0.1", port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_no_inbox_child_client, args=(port,))
    p.start()
    try:
        dst = torch.zeros(1024, dtype=torch.uint8, device="cuda")
        torch.cuda.synchronize()
        tag, ln = await asyncio.wait_for(
            server.arecv(dst, 91, (1 << 64) - 1), timeout=60)
        torch.cuda.synchronize()
        assert ln == 1024 and torch.all(dst == 55)
        # The message rode the rendezvous plane, not the ring.
        assert server._server.get_stats()["gpu_rx"] >= 1
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
        await server.aclose()
