"""Core behavioral contract suite.

Covers the same 23 observable behaviors as the reference's integration
suite (reference tests/test_basic.py, catalogued in SURVEY.md §4) with this
repo's own structure: symmetric direction tests are parametrized, the
in-flight delivery (flush) matrix is driven by one subprocess peer runner,
and payload sizes are pinned to exact uint8 bytes. In-flight payloads are
1 GiB — far past any socket/ring buffering, so a send is genuinely in
flight when the sender closes.
"""
import asyncio
import contextlib
import gc
import multiprocessing as mp

import numpy as np
import pytest

from starway_amd import Client, Server

ADDR = "127.0.0.1"
INFLIGHT_BYTES = 1 << 30  # in-flight delivery tests: larger than any buffer
FULL_MASK = (1 << 64) - 1


@contextlib.asynccontextmanager
async def connected_pair(port):
    server = Server()
    client = Client()
    server.listen(ADDR, port)
    await client.aconnect(ADDR, port)
    try:
        yield server, client
    finally:
        await client.aclose()
        await server.aclose()


async def _roundtrip(sender, recver, payload, tag, ep=None):
    """One message sender->recver with the recv pre-posted; returns the
    (sender_tag, length) the recv reported, with content verified."""
    sink = np.zeros_like(payload)
    fut = recver.arecv(sink, 0, 0)
    await asyncio.sleep(0.01)
    if ep is not None:
        await sender.asend(ep, payload, tag)
    else:
        await sender.asend(payload, tag)
    got_tag, got_len = await fut
    np.testing.assert_array_equal(payload, sink[: len(payload)])
    return got_tag, got_len


# =============================================================================
# Lifecycle
# =============================================================================


async def test_server_listen_client_connect_close(port):
    server = Server()
    server.listen(ADDR, port)
    client = Client()
    await client.aconnect(ADDR, port)
    assert len(server.list_clients()) == 1
    await client.aclose()
    # The endpoint entry is intentionally retained after the client goes
    # away (reference keeps stale endpoints in list_clients).
    assert len(server.list_clients()) == 1
    await server.aclose()


async def test_double_connect_or_listen(port):
    server = Server()
    server.listen(ADDR, port)
    with pytest.raises(Exception):
        server.listen(ADDR, port)
    client = Client()
    await client.aconnect(ADDR, port)
    with pytest.raises(Exception):
        await client.aconnect(ADDR, port)
    await client.aclose()
    await server.aclose()


async def test_double_close(port):
    async with connected_pair(port) as (server, client):
        pass  # the context manager closed both
    with pytest.raises(RuntimeError):
        await client.aclose()
    with pytest.raises(RuntimeError):
        await server.aclose()


async def test_ops_before_connect_or_listen():
    one = np.zeros(1, dtype=np.uint8)
    client = Client()
    with pytest.raises(Exception):
        await client.asend(one, 0)
    with pytest.raises(Exception):
        await client.arecv(one, 0, 0)
    with pytest.raises(Exception):
        await client.aclose()
    server = Server()
    with pytest.raises(Exception):
        await server.arecv(one, 0, 0)
    with pytest.raises(Exception):
        await server.aclose()


async def test_connect_to_dead_server(port):
    client = Client()
    with pytest.raises(Exception) as info:
        await asyncio.wait_for(client.aconnect(ADDR, port), timeout=10)
    assert "not connected" in str(info.value)


# =============================================================================
# Worker-address (listener-less) connection mode
# =============================================================================


async def test_worker_address_connection_roundtrip():
    server = Server()
    blob = server.listen_address()
    assert isinstance(blob, bytes)
    assert server.get_worker_address() == blob

    client = Client()
    await client.aconnect_address(blob)
    for _ in range(100):
        if server.list_clients():
            break
        await asyncio.sleep(0.01)
    eps = server.list_clients()
    assert len(eps) == 1

    msg = np.arange(16, dtype=np.uint8)
    # Both directions over the reverse-established connection.
    tag, ln = await _roundtrip(server, client, msg, 1, ep=next(iter(eps)))
    assert (tag, ln) == (1, msg.size)
    tag, ln = await _roundtrip(client, server, msg, 2)
    assert (tag, ln) == (2, msg.size)

    assert isinstance(client.get_worker_address(), bytes)
    await client.aclose()
    await server.aclose()


async def test_worker_address_accept_callback_invoked():
    server = Server()
    loop = asyncio.get_running_loop()
    accepted: list = []
    fired = asyncio.Event()
    server.set_accept_cb(
        lambda ep: (accepted.append(ep),
                    loop.call_soon_threadsafe(fired.set)))
    blob = server.listen_address()

    client = Client()
    await client.aconnect_address(blob)
    await asyncio.wait_for(fired.wait(), timeout=2.0)
    assert len(accepted) == 1
    assert len(server.list_clients()) == 1
    await client.aclose()
    await server.aclose()


async def test_worker_address_multiple_clients():
    server = Server()
    blob = server.listen_address()
    clients = [Client() for _ in range(3)]
    try:
        await asyncio.gather(*(c.aconnect_address(blob) for c in clients))
        for _ in range(200):
            if len(server.list_clients()) >= 3:
                break
            await asyncio.sleep(0.01)
        assert len(server.list_clients()) >= 3
    finally:
        await asyncio.gather(*(c.aclose() for c in clients),
                             return_exceptions=True)
        await server.aclose()


# =============================================================================
# Data movement
# =============================================================================


@pytest.mark.parametrize("direction", ["c2s", "s2c"])
async def test_send_recv_reports_tag_and_length(port, direction):
    async with connected_pair(port) as (server, client):
        msg = np.arange(20, dtype=np.uint8)
        if direction == "c2s":
            tag, ln = await _roundtrip(client, server, msg, 1)
        else:
            ep = server.list_clients().pop()
            tag, ln = await _roundtrip(server, client, msg, 2, ep=ep)
        assert ln == msg.size
        assert tag in (1, 2)


@pytest.mark.parametrize("size", [1, 1024, 4096])
async def test_message_integrity_various_sizes(port, size):
    async with connected_pair(port) as (server, client):
        blob = np.random.randint(0, 256, size, dtype=np.uint8)
        ep = server.list_clients().pop()
        _, ln = await _roundtrip(client, server, blob, 3)
        assert ln == size
        _, ln = await _roundtrip(server, client, blob, 4, ep=ep)
        assert ln == size


async def test_evaluate_perf(port):
    async with connected_pair(port) as (server, client):
        for nbytes in (1, 1024, 1 << 20, 50 << 20, 1 << 30):
            assert client.evaluate_perf(nbytes) > 0
        ep = server.list_clients().pop()
        for nbytes in (1, 1024, 1 << 20):
            assert server.evaluate_perf(ep, nbytes) > 0


# =============================================================================
# Delivery guarantee: data in flight is lost at close unless flushed first
# =============================================================================


def _peer_sender(role, port, flush_mode):
    """Subprocess peer: connects (or accepts one peer), pushes INFLIGHT_BYTES
    as one tagged message, optionally flushes, then closes and exits."""

    async def inner():
        payload = np.empty(INFLIGHT_BYTES, dtype=np.uint8)
        if role == "server":
            server = Server()
            server.listen(ADDR, port)
            got_peer = asyncio.Event()
            loop = asyncio.get_running_loop()
            server.set_accept_cb(
                lambda ep: loop.call_soon_threadsafe(got_peer.set))
            await got_peer.wait()
            ep = next(iter(server.list_clients()))
            await server.asend(ep, payload, 0)
            if flush_mode == "flush":
                await server.aflush()
            elif flush_mode == "flush_ep":
                await server.aflush_ep(ep)
            await server.aclose()
        else:
            client = Client()
            await client.aconnect(ADDR, port)
            await client.asend(payload, 0)
            if flush_mode == "flush":
                await client.aflush()
            await client.aclose()

    asyncio.run(inner())


def _spawn_peer(role, port, flush_mode):
    proc = mp.get_context("spawn").Process(
        target=_peer_sender, args=(role, port, flush_mode))
    proc.start()
    return proc


@pytest.mark.parametrize("flush_mode", ["flush", "flush_ep"])
async def test_server_send_flushed_before_close_is_delivered(port, flush_mode):
    proc = _spawn_peer("server", port, flush_mode)
    await asyncio.sleep(0.5)
    client = Client()
    await client.aconnect(ADDR, port)
    sink = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    await client.arecv(sink, 0, 0)
    proc.join()
    await client.aclose()
    proc.close()


@pytest.mark.parametrize("flush_mode", ["none", "none_ep"])
async def test_server_send_unflushed_close_loses_data(port, flush_mode):
    proc = _spawn_peer("server", port, "none")
    await asyncio.sleep(0.5 if flush_mode == "none" else 0.2)
    client = Client()
    await client.aconnect(ADDR, port)
    sink = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    outcome: list = []
    client.recv(sink, 0, 0,
                lambda tag, ln: outcome.append("done"),
                lambda err: outcome.append("fail"))
    await asyncio.sleep(1.0)
    assert not outcome  # neither delivered nor failed: the data is gone
    await client.aclose()
    proc.kill()
    proc.join()
    proc.close()


async def test_client_send_with_flush_good(port):
    server = Server()
    server.listen(ADDR, port)
    loop = asyncio.get_running_loop()
    got_peer = asyncio.Event()
    server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(got_peer.set))
    proc = _spawn_peer("client", port, "flush")
    await got_peer.wait()
    sink = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    await server.arecv(sink, 0, 0)
    proc.join()
    proc.close()
    await server.aclose()


async def test_client_send_without_flush_bad(port):
    server = Server()
    server.listen(ADDR, port)
    loop = asyncio.get_running_loop()
    got_peer = asyncio.Event()
    server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(got_peer.set))
    proc = _spawn_peer("client", port, "none")
    await got_peer.wait()
    sink = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    outcome: list = []
    server.recv(sink, 0, 0,
                lambda tag, ln: outcome.append("done"),
                lambda err: outcome.append("fail"))
    await asyncio.sleep(1.0)
    assert not outcome
    proc.kill()
    proc.join()
    proc.close()
    await server.aclose()


# =============================================================================
# Concurrency
# =============================================================================


async def test_multiple_clients(port):
    server = Server()
    server.listen(ADDR, port)
    await asyncio.sleep(0.1)
    n = 5
    clients = [Client() for _ in range(n)]
    await asyncio.gather(*(c.aconnect(ADDR, port) for c in clients))
    await asyncio.sleep(0.2)
    assert len(server.list_clients()) == n

    await asyncio.gather(
        *(c.asend(np.array([i], dtype=np.uint8), i)
          for i, c in enumerate(clients)))
    sink = np.zeros(1, dtype=np.uint8)
    seen = set()
    for _ in range(n):
        tag, _ = await server.arecv(sink, 0, 0)
        seen.add(tag)
    assert seen == set(range(n))
    await asyncio.gather(*(c.aclose() for c in clients))
    await server.aclose()


async def test_concurrent_send_recv(port):
    async with connected_pair(port) as (server, client):
        n = 50
        futs = [client.asend(np.array([i], dtype=np.uint8), i)
                for i in range(n)]
        futs += [server.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
                 for _ in range(n)]
        done = await asyncio.gather(*futs)
        tags = {r[0] for r in done if isinstance(r, tuple)}
        assert tags == set(range(n))


async def test_bidirectional_traffic(port):
    async with connected_pair(port) as (server, client):
        ep = server.list_clients().pop()
        n = 2000
        down = [server.asend(ep, np.array([i % 256], dtype=np.uint8), 100 + i)
                for i in range(n)]
        down_recv = [client.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
                     for _ in range(n)]
        up = [client.asend(np.array([i % 256], dtype=np.uint8), 100 + n + i)
              for i in range(n)]
        up_recv = [server.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
                   for _ in range(n)]
        results = await asyncio.gather(*down, *down_recv, *up, *up_recv)
        client_tags = {r[0] for r in results[n:2 * n] if r is not None}
        server_tags = {r[0] for r in results[3 * n:] if r is not None}
        assert client_tags == set(range(100, 100 + n))
        assert server_tags == set(range(100 + n, 100 + 2 * n))


async def test_rapid_connect_close_client(port):
    server = Server()
    server.listen(ADDR, port)
    one = np.zeros(1, dtype=np.uint8)
    sink = np.zeros(1, dtype=np.uint8)
    cycles = 10

    async def hit_and_run():
        c = Client()
        await c.aconnect(ADDR, port)
        await c.asend(one, 1)
        await c.aclose()

    await asyncio.gather(
        *[hit_and_run() for _ in range(cycles)],
        *[server.arecv(sink, 0, 0) for _ in range(cycles)])
    await server.aclose()


# =============================================================================
# Resource management
# =============================================================================


async def test_shutdown_with_in_flight_ops(port):
    async with connected_pair(port) as (server, _):
        client = Client()
        await client.aconnect(ADDR, port)
        sink = np.ones(1 << 20, dtype=np.uint8)
        cancelled = asyncio.Event()

        async def pending():
            try:
                await client.arecv(sink, 999, FULL_MASK)
            except Exception as exc:
                assert "cancel" in str(exc)
                cancelled.set()

        task = asyncio.create_task(pending())
        await asyncio.sleep(0.01)
        await client.aclose()
        await task
        assert cancelled.is_set()


async def test_implicit_destruction_without_close(port):
    server = Server()
    server.listen(ADDR, port)
    client = Client()
    await client.aconnect(ADDR, port)
    # GC without aclose() must neither hang nor crash (dtor safety net).
    del server
    del client
    gc.collect()
    await asyncio.sleep(0.5)
