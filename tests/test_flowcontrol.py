"""Flow-control and buffer-reuse contracts (round-2 additions).

Pins the MI355X-native engine's completion/windowing semantics:

* send completion => the engine owns the payload (buffer-reuse safety): a
  caller that overwrites its buffer the moment ``await asend`` returns must
  never corrupt the in-flight message, on both the eager-TCP path and the
  same-host CMA rendezvous path (UCX ucp_tag_send_nbx contract the
  reference inherited; see csrc/engine.cpp start_send/enqueue_eager)
* STARWAY_SEND_WINDOW bounds per-connection rendezvous bytes awaiting
  RECV_DONE; excess sends queue and drain in order (288 GB HBM3E sizing,
  BASELINE config 3 wording)
* STARWAY_UNEXP_CAP bounds unexpected-message staging per connection via
  read backpressure instead of unbounded RSS growth
* a flush covering rendezvous sends to a peer that dies fails instead of
  hanging (ADVICE round-1 finding on on_conn_dead)
"""
import asyncio
import multiprocessing as mp
import os
import time

import numpy as np
import pytest

from starway_amd import Client, Server

ADDR = "127.0.0.1"
FULL = (1 << 64) - 1


def _pattern(n, seed):
    rng = np.random.RandomState(seed)
    return rng.randint(0, 256, n, dtype=np.uint8)


# -- buffer reuse: eager path (subprocess sender, CMA disabled) --------------

def _reuse_eager_sender(port):
    os.environ["STARWAY_CMA"] = "0"
    os.environ["STARWAY_SHM"] = "0"

    async def inner():
        client = Client()
        await client.aconnect(ADDR, port)
        buf = _pattern(32 << 20, 7)
        await client.asend(buf, 2)
        # Completion fired: the engine must own the payload now. Scribble.
        buf.fill(0)
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_send_buffer_reuse_eager(port):
    server = Server()
    server.listen(ADDR, port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_reuse_eager_sender, args=(port,))
    p.start()
    recv = np.zeros(32 << 20, dtype=np.uint8)
    tag, ln = await asyncio.wait_for(server.arecv(recv, 0, 0), timeout=60)
    assert tag == 2 and ln == recv.size
    np.testing.assert_array_equal(recv, _pattern(32 << 20, 7))
    p.join()
    await server.aclose()


# -- buffer reuse: CMA rendezvous path ---------------------------------------

def _reuse_cma_sender(port):
    async def inner():
        client = Client()
        await client.aconnect(ADDR, port)
        big = _pattern(8 << 20, 11)
        await client.asend(big, 2)  # >= CMA threshold: rendezvous descriptor
        big.fill(0)                 # reuse immediately after completion
        # Marker AFTER the scribble; the receiver pulls the big message only
        # once the marker arrived, so the pull deterministically happens
        # against the overwritten user buffer — only a captured snapshot
        # can deliver the original bytes.
        await client.asend(np.ones(1, dtype=np.uint8), 1)
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_send_buffer_reuse_cma(port):
    server = Server()
    server.listen(ADDR, port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_reuse_cma_sender, args=(port,))
    p.start()
    marker = np.zeros(1, dtype=np.uint8)
    await asyncio.wait_for(server.arecv(marker, 1, FULL), timeout=30)
    recv = np.zeros(8 << 20, dtype=np.uint8)
    tag, ln = await asyncio.wait_for(server.arecv(recv, 2, FULL), timeout=30)
    assert ln == recv.size
    np.testing.assert_array_equal(recv, _pattern(8 << 20, 11))
    stats = server._server.get_stats()
    assert stats["cma_rx"] == 1  # the big message really rode CMA
    p.join()
    await server.aclose()


# -- sender-side rendezvous window -------------------------------------------

def _windowed_sender(port, q):
    os.environ["STARWAY_SEND_WINDOW"] = "1M"

    async def inner():
        client = Client()
        await client.aconnect(ADDR, port)
        for i in range(4):
            await client.asend(_pattern(2 << 20, 20 + i), 10 + i)
        # First send admitted (window admits one oversized op when idle);
        # the rest must be queued behind the window.
        q.put(client._client.get_stats()["deferred_sends"])
        await client.aflush()  # must cover the deferred sends too
        await client.aclose()

    asyncio.run(inner())


async def test_send_window_defers_and_preserves_order(port):
    server = Server()
    server.listen(ADDR, port)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_windowed_sender, args=(port, q))
    p.start()
    await asyncio.sleep(0.5)  # let the sender hit the window
    recv = np.zeros(2 << 20, dtype=np.uint8)
    for i in range(4):
        tag, ln = await asyncio.wait_for(server.arecv(recv, 0, 0), timeout=30)
        assert tag == 10 + i  # per-connection order preserved across window
        np.testing.assert_array_equal(recv, _pattern(2 << 20, 20 + i))
    deferred = q.get(timeout=10)
    assert deferred >= 3
    p.join()
    assert p.exitcode == 0  # flush + close completed over there
    await server.aclose()


# -- receiver-side unexpected staging cap ------------------------------------

def _capped_receiver(port, q):
    os.environ["STARWAY_UNEXP_CAP"] = "1M"
    os.environ["STARWAY_SHM"] = "0"

    async def inner():
        server = Server()
        server.listen(ADDR, port)
        q.put("listening")
        # Give the sender time to push as much as backpressure allows.
        await asyncio.sleep(1.5)
        staged = server._server.get_stats()["unexp_staged_bytes"]
        q.put(staged)
        recv = np.zeros(512 << 10, dtype=np.uint8)
        for i in range(16):
            tag, ln = await server.arecv(recv, 0, 0)
            assert tag == i and ln == recv.size
            assert recv[0] == i and recv[-1] == i
        q.put("done")
        await server.aclose()

    asyncio.run(inner())


async def test_unexpected_staging_cap_backpressure(port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_capped_receiver, args=(port, q))
    p.start()
    assert q.get(timeout=15) == "listening"
    client = Client()
    await client.aconnect(ADDR, port)
    # 16 x 512 KiB eager messages (below the CMA threshold), no recvs
    # posted on the other side. The receiver must stall reads near its
    # 1 MiB cap instead of staging all 8 MiB.
    msgs = []
    for i in range(16):
        m = np.full(512 << 10, i, dtype=np.uint8)
        msgs.append(m)
        await client.asend(m, i)
    staged = q.get(timeout=30)
    assert staged <= (2 << 20), f"staged {staged} bytes despite 1 MiB cap"
    assert q.get(timeout=60) == "done"  # backlog drained once recvs posted
    p.join()
    await client.aclose()


# -- flush covering rendezvous sends to a dead peer fails, never hangs --------

def _doomed_receiver(port, q):
    async def inner():
        server = Server()
        server.listen(ADDR, port)
        q.put("listening")
        await asyncio.sleep(30)  # never posts a recv; killed by the parent

    asyncio.run(inner())


async def test_flush_fails_when_peer_dies_mid_rendezvous(port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_doomed_receiver, args=(port, q))
    p.start()
    assert q.get(timeout=15) == "listening"
    client = Client()
    await client.aconnect(ADDR, port)
    await client.asend(_pattern(2 << 20, 3), 1)  # CMA rendezvous, never acked
    flush_fut = client.aflush()
    await asyncio.sleep(0.2)
    p.kill()
    p.join()
    with pytest.raises(Exception, match="reset|closed|cancel"):
        await asyncio.wait_for(flush_fut, timeout=15)
    await client.aclose()


# -- deferred-send failure on close ------------------------------------------

async def test_close_cancels_window_deferred_sends(port):
    # Same-process loopback: CMA is skipped (same uuid), so use the GPU-free
    # path only to check close() does not hang with queued commands.
    server = Server()
    server.listen(ADDR, port)
    client = Client()
    await client.aconnect(ADDR, port)
    buf = np.zeros(1024, dtype=np.uint8)
    await client.asend(buf, 1)
    await client.aclose()
    await server.aclose()


def _window_doomed_sender(port, q):
    os.environ["STARWAY_SEND_WINDOW"] = "1M"

    async def inner():
        client = Client()
        await client.aconnect(ADDR, port)
        # One admitted CMA rendezvous + two window-deferred behind it.
        # Deferred sends only complete once admitted, so do NOT await
        # them individually — their futures must FAIL at peer death.
        send_futs = [client.asend(_pattern(2 << 20, 30 + i), 20 + i)
                     for i in range(3)]
        flush_fut = client.aflush()
        q.put("sent")
        try:
            await asyncio.wait_for(flush_fut, timeout=20)
            q.put("flush-completed")
        except Exception as exc:
            q.put(f"flush-failed:{exc}")
        outcomes = await asyncio.gather(*send_futs, return_exceptions=True)
        failed = sum(isinstance(o, Exception) for o in outcomes)
        q.put(f"send-failures:{failed}")
        await client.aclose()
        q.put("closed")

    asyncio.run(inner())


async def test_deferred_sends_fail_when_peer_dies(port):
    """Peer death with window-DEFERRED rendezvous sends queued: the flush
    covering them must fail (not hang) and close must still succeed."""
    server = Server()
    server.listen(ADDR, port)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_window_doomed_sender, args=(port, q))
    p.start()
    try:
        assert q.get(timeout=30) == "sent"
        await asyncio.sleep(0.2)
        await server.aclose()  # peer disappears mid-rendezvous
        verdict = q.get(timeout=30)
        assert verdict.startswith("flush-failed"), verdict
        assert "reset" in verdict or "closed" in verdict or "cancel" in verdict
        failures = q.get(timeout=30)
        assert failures == "send-failures:2", failures  # the deferred pair
        assert q.get(timeout=30) == "closed"
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
