"""Shared-memory channel: activation, introspection, opt-out, and
cross-process correctness through the ring."""
import asyncio
import multiprocessing as mp
import os
import random

import numpy as np
import pytest

from starway_amd import Client, Server

SERVER_ADDR = "127.0.0.1"



async def test_shm_activates_same_host(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    # Give the offer/ack/switch handshake a moment, then check transports.
    for _ in range(100):
        ep = next(iter(server.list_clients()))
        if ("sm", "shm_ring") in ep.view_transports():
            break
        await asyncio.sleep(0.01)
    assert ("sm", "shm_ring") in ep.view_transports()

    # Traffic after the switch flows through the ring.
    send = np.random.randint(0, 256, 1 << 16, dtype=np.uint8)
    recv = np.zeros(1 << 16, dtype=np.uint8)
    fut = server.arecv(recv, 0, 0)
    await asyncio.sleep(0.01)
    await client.asend(send, 1)
    _, ln = await fut
    assert ln == send.size
    np.testing.assert_array_equal(send, recv)
    await client.aclose()
    await server.aclose()


def _shm_child(port, n):
    async def inner():
        client = Client()
        await client.aconnect(SERVER_ADDR, port)
        buf = (np.arange(n) % 249).astype(np.uint8)
        await client.asend(buf, 11)
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_shm_cross_process_large_transfer(port):
    # 64 MiB >> ring capacity: exercises chunked ring streaming + flush.
    n = 64 << 20
    server = Server()
    server.listen(SERVER_ADDR, port)
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_shm_child, args=(port, n))
    p.start()
    try:
        recv = np.zeros(n, dtype=np.uint8)
        tag, ln = await server.arecv(recv, 0, 0)
        assert tag == 11 and ln == n
        expect = (np.arange(n) % 249).astype(np.uint8)
        np.testing.assert_array_equal(recv, expect)
    finally:
        p.join(timeout=60)
        if p.is_alive():
            p.kill()
            p.join()
        p.close()
        await server.aclose()


async def test_shm_disabled_by_env(port):
    # The offer side controls creation; with STARWAY_SHM=0 in this process
    # both offer and accept are disabled, so transports stay tcp-only.
    os.environ["STARWAY_SHM"] = "0"
    try:
        server = Server()
        client = Client()
        server.listen(SERVER_ADDR, port)
        await client.aconnect(SERVER_ADDR, port)
        await asyncio.sleep(0.1)
        ep = next(iter(server.list_clients()))
        assert ("sm", "shm_ring") not in ep.view_transports()
        send = np.arange(64, dtype=np.uint8)
        recv = np.zeros(64, dtype=np.uint8)
        fut = server.arecv(recv, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(send, 1)
        await fut
        np.testing.assert_array_equal(send, recv)
        await client.aclose()
        await server.aclose()
    finally:
        del os.environ["STARWAY_SHM"]


async def test_shm_no_segment_leak(port):
    before = set(os.listdir("/dev/shm"))
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    await asyncio.sleep(0.05)
    await client.aclose()
    await server.aclose()
    await asyncio.sleep(0.1)
    after = set(os.listdir("/dev/shm"))
    leaked = {f for f in after - before if f.startswith("sw-")}
    assert not leaked, f"leaked shm segments: {leaked}"


def _cma_child(port, n, env):
    import os

    os.environ.update(env)

    async def inner():
        client = Client()
        await client.aconnect(SERVER_ADDR, port)
        buf = (np.arange(n) % 247).astype(np.uint8)
        await client.asend(buf, 31)
        await client.aflush()
        await client.aclose()

    asyncio.run(inner())


@pytest.mark.parametrize(
    "env",
    [{}, {"STARWAY_CMA": "0"}, {"STARWAY_CMA_FORCE_EPERM": "1"}],
    ids=["cma", "ring-eager", "cma-eperm-fallback"],
)
async def test_large_cross_process_paths(port, env):
    """32 MiB cross-process via the CMA rendezvous (default) and via the
    ring/eager path (STARWAY_CMA=0 in both processes)."""
    os.environ.update(env)
    try:
        n = 32 << 20
        server = Server()
        server.listen(SERVER_ADDR, port)
        ctx = mp.get_context("spawn")
        p = ctx.Process(target=_cma_child, args=(port, n, env))
        p.start()
        try:
            recv = np.zeros(n, dtype=np.uint8)
            tag, ln = await server.arecv(recv, 0, 0)
            assert tag == 31 and ln == n
            expect = (np.arange(n) % 247).astype(np.uint8)
            np.testing.assert_array_equal(recv, expect)
        finally:
            p.join(timeout=60)
            if p.is_alive():
                p.kill()
                p.join()
            p.close()
            await server.aclose()
    finally:
        for k in env:
            os.environ.pop(k, None)
