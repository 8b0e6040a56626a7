"""Concurrency stress: multiple Python threads hammering one endpoint
pair, and connect/close storms. This is the empirical race-detection
harness for the engine's cross-thread contracts (command queue, close
hand-off, completion batches) — SURVEY.md §5's "add race detection since
we own the transport now" item.
"""
import asyncio
import random
import threading

import numpy as np
import pytest

from starway_amd import Client, Server

SERVER_ADDR = "127.0.0.1"



def test_multithreaded_send_recv(port):
    """4 sender threads (raw callback API) x 500 msgs against one server;
    every message accounted for."""
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)

    connected = threading.Event()
    client._client.connect(SERVER_ADDR, port, lambda s: connected.set())
    assert connected.wait(10)

    n_threads, per_thread = 4, 500
    total = n_threads * per_thread
    recv_done = threading.Semaphore(0)
    send_done = threading.Semaphore(0)
    seen = set()
    seen_lock = threading.Lock()
    recv_bufs = [np.zeros(8, dtype=np.uint8) for _ in range(total)]

    def post_recvs():
        def make_cb():
            def cb(tag, length):
                with seen_lock:
                    seen.add(tag)
                recv_done.release()
            return cb

        for i in range(total):
            server._server.recv(recv_bufs[i], 0, 0, make_cb(),
                                lambda e: recv_done.release())

    post_recvs()

    def sender(tid):
        for i in range(per_thread):
            tag = tid * per_thread + i
            buf = np.full(8, tid, dtype=np.uint8)
            client._client.send(buf, tag, send_done.release,
                                lambda e: send_done.release())

    threads = [threading.Thread(target=sender, args=(t,))
               for t in range(n_threads)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()

    for _ in range(total):
        assert send_done.acquire(timeout=30)
    for _ in range(total):
        assert recv_done.acquire(timeout=30)
    assert seen == set(range(total))

    closed = threading.Event()
    client._client.close(lambda: closed.set())
    assert closed.wait(10)
    closed2 = threading.Event()
    server._server.close(lambda: closed2.set())
    assert closed2.wait(10)


async def test_close_storm(port):
    """Overlapped connect / traffic / close cycles from concurrent tasks."""
    server = Server()
    server.listen(SERVER_ADDR, port)

    async def churn(i):
        c = Client()
        await c.aconnect(SERVER_ADDR, port)
        buf = np.full(16, i % 256, dtype=np.uint8)
        for k in range(5):
            await c.asend(buf, i * 100 + k)
        await c.aflush()
        await c.aclose()

    recv_buf = np.zeros(16, dtype=np.uint8)
    n_clients, per_client = 8, 5
    results = await asyncio.gather(
        *[churn(i) for i in range(n_clients)],
        *[server.arecv(recv_buf, 0, 0) for _ in range(n_clients * per_client)],
    )
    tags = {r[0] for r in results if isinstance(r, tuple)}
    assert len(tags) == n_clients * per_client
    await server.aclose()


async def test_flush_under_concurrent_traffic(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)

    payload = np.ones(256 * 1024, dtype=np.uint8)
    recv_buf = np.zeros(256 * 1024, dtype=np.uint8)

    async def sender():
        for i in range(50):
            await client.asend(payload, i)
            if i % 10 == 9:
                await client.aflush()

    async def receiver():
        for _ in range(50):
            await server.arecv(recv_buf, 0, 0)

    await asyncio.gather(sender(), receiver())
    await client.aflush()
    await client.aclose()
    await server.aclose()


async def test_destructor_storm(port):
    """Create/connect/drop without close, repeatedly (GC joins threads)."""
    import gc

    server = Server()
    server.listen(SERVER_ADDR, port)
    for _ in range(10):
        c = Client()
        await c.aconnect(SERVER_ADDR, port)
        del c
    gc.collect()
    await asyncio.sleep(0.2)
    await server.aclose()
