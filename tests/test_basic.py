"""Integration suite — semantic port of the reference's 23-test contract
(reference tests/test_basic.py; catalogue in SURVEY.md §4), adapted:

* in-flight flush tests use 2 GiB payloads (vs 8 GiB) — still far past any
  socket buffering, so the send is genuinely in flight at close time
* message buffers are always uint8 (the reference leaned on nanobind dtype
  coercion for a few np.array([i]) int64 cases; we pin exact-byte semantics
  and test truncation separately in test_semantics.py)
"""
import asyncio
import contextlib
import gc
import multiprocessing as mp
import random

import numpy as np
import pytest

from starway_amd import Client, Server

SERVER_ADDR = "127.0.0.1"
INFLIGHT_BYTES = 1024 * 1024 * 1024  # large enough to be in flight



@contextlib.asynccontextmanager
async def gen_server_client(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    try:
        yield server, client
    finally:
        await client.aclose()
        await server.aclose()


# =============================================================================
# Basic functionality
# =============================================================================


async def test_server_listen_client_connect_close(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)

    assert len(server.list_clients()) == 1
    await client.aclose()
    # Stale endpoint entries persist after client close (reference behavior).
    assert len(server.list_clients()) == 1
    await server.aclose()


async def test_worker_address_connection_roundtrip():
    server = Server()
    server_address = server.listen_address()
    assert isinstance(server_address, bytes)
    assert server.get_worker_address() == server_address

    client = Client()
    await client.aconnect_address(server_address)

    for _ in range(100):
        if server.list_clients():
            break
        await asyncio.sleep(0.01)
    client_list = server.list_clients()
    assert len(client_list) == 1
    client_ep = next(iter(client_list))

    send_buf = np.arange(16, dtype=np.uint8)
    recv_buf_client = np.zeros_like(send_buf)
    recv_task_client = client.arecv(recv_buf_client, 0, 0)
    await asyncio.sleep(0.01)
    await server.asend(client_ep, send_buf, 1)
    sender_tag, length = await recv_task_client
    assert sender_tag == 1
    assert length == len(send_buf)
    np.testing.assert_array_equal(send_buf, recv_buf_client)

    recv_buf_server = np.zeros_like(send_buf)
    recv_task_server = server.arecv(recv_buf_server, 0, 0)
    await asyncio.sleep(0.01)
    await client.asend(send_buf, 2)
    sender_tag_server, length_server = await recv_task_server
    assert sender_tag_server == 2
    assert length_server == len(send_buf)
    np.testing.assert_array_equal(send_buf, recv_buf_server)

    assert isinstance(client.get_worker_address(), bytes)

    await client.aclose()
    await server.aclose()


async def test_worker_address_accept_callback_invoked():
    server = Server()
    accept_event = asyncio.Event()
    accepted_eps: list = []
    loop = asyncio.get_running_loop()

    def accept_cb(ep):
        accepted_eps.append(ep)
        loop.call_soon_threadsafe(accept_event.set)

    server.set_accept_cb(accept_cb)
    server_address = server.listen_address()
    client = Client()

    await client.aconnect_address(server_address)
    await asyncio.wait_for(accept_event.wait(), timeout=2.0)

    assert len(accepted_eps) == 1
    assert len(server.list_clients()) == 1

    await client.aclose()
    await server.aclose()


async def test_worker_address_multiple_clients():
    server = Server()
    server_address = server.listen_address()
    clients = [Client() for _ in range(3)]
    try:
        await asyncio.gather(*(c.aconnect_address(server_address) for c in clients))
        for _ in range(200):
            if len(server.list_clients()) >= len(clients):
                break
            await asyncio.sleep(0.01)
        assert len(server.list_clients()) >= len(clients)
    finally:
        await asyncio.gather(*(c.aclose() for c in clients), return_exceptions=True)
        await server.aclose()


async def test_client_to_server_send_recv(port):
    async with gen_server_client(port) as (server, client):
        send_buf = np.arange(10, dtype=np.uint8)
        recv_buf = np.zeros(10, dtype=np.uint8)

        recv_task = server.arecv(recv_buf, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(send_buf, 1)
        sender_tag, length = await recv_task

        assert sender_tag == 1
        assert length == len(send_buf)
        np.testing.assert_array_equal(send_buf, recv_buf)


async def test_server_to_client_send_recv(port):
    async with gen_server_client(port) as (server, client):
        send_buf = np.arange(20, dtype=np.uint8)
        recv_buf = np.zeros(20, dtype=np.uint8)

        client_ep = server.list_clients().pop()
        recv_task = client.arecv(recv_buf, 0, 0)
        await asyncio.sleep(0.01)
        await server.asend(client_ep, send_buf, 2)
        sender_tag, length = await recv_task

        assert sender_tag == 2
        assert length == len(send_buf)
        np.testing.assert_array_equal(send_buf, recv_buf)


# =============================================================================
# Flush / delivery-guarantee contract (subprocess peers, real process death)
# =============================================================================


def _server_send(port, with_flush=False, use_flush_ep=False):
    async def inner():
        server = Server()
        server.listen(SERVER_ADDR, port)
        connected = asyncio.Event()
        loop = asyncio.get_running_loop()
        server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(connected.set))
        await connected.wait()
        ep = next(iter(server.list_clients()))
        send_buf = np.empty(INFLIGHT_BYTES, dtype=np.uint8)
        await server.asend(ep, send_buf, 0)
        if with_flush:
            if use_flush_ep:
                await server.aflush_ep(ep)
            else:
                await server.aflush()
        await server.aclose()

    asyncio.run(inner())


def _client_send(port, with_flush=False):
    async def inner():
        client = Client()
        await client.aconnect(SERVER_ADDR, port)
        send_buf = np.empty(INFLIGHT_BYTES, dtype=np.uint8)
        await client.asend(send_buf, 0)
        if with_flush:
            await client.aflush()
        await client.aclose()

    asyncio.run(inner())


async def test_server_send_without_flush_bad(port):
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_server_send, args=(port, False))
    p.start()
    await asyncio.sleep(0.5)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    done = False

    def done_cb(sender_tag, length):
        nonlocal done
        done = True

    def fail_cb(error):
        nonlocal done
        done = True

    client.recv(recv_buf, 0, 0, done_cb, fail_cb)
    await asyncio.sleep(1.0)
    assert not done
    await client.aclose()
    p.kill()
    p.join()
    p.close()


async def test_server_send_with_flush_good(port):
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_server_send, args=(port, True))
    p.start()
    await asyncio.sleep(0.5)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    recv_future = client.arecv(recv_buf, 0, 0)
    await recv_future
    p.join()
    await client.aclose()
    p.close()


async def test_server_send_with_flush_ep_good(port):
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_server_send, args=(port, True, True))
    p.start()
    await asyncio.sleep(0.2)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    recv_future = client.arecv(recv_buf, 0, 0)
    await recv_future
    p.join()
    await client.aclose()
    p.close()


async def test_server_send_without_flush_ep_bad(port):
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_server_send, args=(port, False, True))
    p.start()
    await asyncio.sleep(0.2)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    done = False

    def done_cb(sender_tag, length):
        nonlocal done
        done = True

    def fail_cb(error):
        nonlocal done
        done = True

    client.recv(recv_buf, 0, 0, done_cb, fail_cb)
    await asyncio.sleep(1.0)
    assert not done
    await client.aclose()
    p.kill()
    p.join()
    p.close()


async def test_client_send_without_flush_bad(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    connected = asyncio.Event()
    loop = asyncio.get_running_loop()
    server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(connected.set))
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_client_send, args=(port, False))
    p.start()
    await connected.wait()
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    done = False

    def done_cb(sender_tag, length):
        nonlocal done
        done = True

    def fail_cb(error):
        nonlocal done
        done = True

    server.recv(recv_buf, 0, 0, done_cb, fail_cb)
    await asyncio.sleep(1.0)
    assert not done
    p.kill()
    p.join()
    p.close()
    await server.aclose()


async def test_client_send_with_flush_good(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    connected = asyncio.Event()
    loop = asyncio.get_running_loop()
    server.set_accept_cb(lambda ep: loop.call_soon_threadsafe(connected.set))
    ctx = mp.get_context("spawn")
    p = ctx.Process(target=_client_send, args=(port, True))
    p.start()
    await connected.wait()
    recv_buf = np.zeros(INFLIGHT_BYTES, dtype=np.uint8)
    recv_future = server.arecv(recv_buf, 0, 0)
    await recv_future
    p.join()
    p.close()
    await server.aclose()


# =============================================================================
# Integrity / perf probe
# =============================================================================


@pytest.mark.parametrize("size", [1, 1024, 4096])
async def test_message_integrity_various_sizes(port, size):
    async with gen_server_client(port) as (server, client):
        send_buf = np.random.randint(0, 256, size, dtype=np.uint8)
        recv_buf = np.zeros(size, dtype=np.uint8)
        client_ep = server.list_clients().pop()

        recv_task = server.arecv(recv_buf, 0, 0)
        await client.asend(send_buf, 3)
        _, length = await recv_task
        assert length == size
        np.testing.assert_array_equal(send_buf, recv_buf)

        recv_buf.fill(0)
        recv_task = client.arecv(recv_buf, 0, 0)
        await server.asend(client_ep, send_buf, 4)
        _, length = await recv_task
        assert length == size
        np.testing.assert_array_equal(send_buf, recv_buf)


async def test_evaluate_perf(port):
    client = Client()
    server = Server()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)

    for msg in [1, 1024, 1 << 20, 50 << 20, 1 << 30]:
        assert client.evaluate_perf(msg) > 0
    ep = server.list_clients().pop()
    for msg in [1, 1024, 1 << 20]:
        assert server.evaluate_perf(ep, msg) > 0

    await client.aclose()
    await server.aclose()


# =============================================================================
# State management and error handling
# =============================================================================


async def test_client_op_before_connect():
    client = Client()
    buf = np.zeros(1, dtype=np.uint8)
    with pytest.raises(Exception):
        await client.asend(buf, 0)
    with pytest.raises(Exception):
        await client.arecv(buf, 0, 0)
    with pytest.raises(Exception):
        await client.aclose()


async def test_server_op_before_listen():
    server = Server()
    buf = np.zeros(1, dtype=np.uint8)
    with pytest.raises(Exception):
        await server.arecv(buf, 0, 0)
    with pytest.raises(Exception):
        await server.aclose()


async def test_double_connect_or_listen(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    with pytest.raises(Exception):
        server.listen(SERVER_ADDR, port)

    client = Client()
    await client.aconnect(SERVER_ADDR, port)
    with pytest.raises(Exception):
        await client.aconnect(SERVER_ADDR, port)

    await client.aclose()
    await server.aclose()


async def test_double_close(port):
    client = Client()
    server = Server()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    await client.aclose()
    await server.aclose()
    with pytest.raises(RuntimeError):
        await client.aclose()
    with pytest.raises(RuntimeError):
        await server.aclose()


async def test_connect_to_dead_server(port):
    client = Client()
    with pytest.raises(Exception) as e_info:
        await asyncio.wait_for(client.aconnect(SERVER_ADDR, port), timeout=10)
    assert "not connected" in str(e_info.value)


# =============================================================================
# Concurrency and stress
# =============================================================================


async def test_multiple_clients(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    await asyncio.sleep(0.1)

    num_clients = 5
    clients = [Client() for _ in range(num_clients)]
    await asyncio.gather(*(c.aconnect(SERVER_ADDR, port) for c in clients))
    await asyncio.sleep(0.2)
    assert len(server.list_clients()) == num_clients

    await asyncio.gather(
        *(c.asend(np.array([i], dtype=np.uint8), i) for i, c in enumerate(clients))
    )

    recv_buf = np.zeros(1, dtype=np.uint8)
    recv_tags = set()
    for _ in range(num_clients):
        tag, _ = await server.arecv(recv_buf, 0, 0)
        recv_tags.add(tag)
    assert recv_tags == set(range(num_clients))

    await asyncio.gather(*(c.aclose() for c in clients))
    await server.aclose()


async def test_concurrent_send_recv(port):
    async with gen_server_client(port) as (server, client):
        num_messages = 50
        sends = [
            client.asend(np.array([i], dtype=np.uint8), i)
            for i in range(num_messages)
        ]
        recvs = [
            server.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
            for _ in range(num_messages)
        ]
        results = await asyncio.gather(*sends, *recvs)
        received_tags = {res[0] for res in results if isinstance(res, tuple)}
        assert received_tags == set(range(num_messages))


async def test_bidirectional_traffic(port):
    async with gen_server_client(port) as (server, client):
        client_ep = server.list_clients().pop()
        num_messages = 2000

        server_sends = [
            server.asend(client_ep, np.array([i % 256], dtype=np.uint8), 100 + i)
            for i in range(num_messages)
        ]
        client_recvs = [
            client.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
            for _ in range(num_messages)
        ]
        client_sends = [
            client.asend(np.array([i % 256], dtype=np.uint8), 100 + num_messages + i)
            for i in range(num_messages)
        ]
        server_recvs = [
            server.arecv(np.zeros(1, dtype=np.uint8), 0, 0)
            for _ in range(num_messages)
        ]

        results = await asyncio.gather(
            *server_sends, *client_recvs, *client_sends, *server_recvs
        )
        client_recv_results = results[num_messages : 2 * num_messages]
        server_recv_results = results[3 * num_messages :]
        client_tags = {r[0] for r in client_recv_results if r is not None}
        server_tags = {r[0] for r in server_recv_results if r is not None}
        assert client_tags == set(range(100, 100 + num_messages))
        assert server_tags == set(
            range(100 + num_messages, 100 + 2 * num_messages)
        )


async def test_rapid_connect_close_client(port):
    server = Server()
    server.listen(SERVER_ADDR, port)

    num_cycles = 10
    buf = np.zeros(1, dtype=np.uint8)
    buf2 = np.zeros(1, dtype=np.uint8)

    async def once():
        client = Client()
        await client.aconnect(SERVER_ADDR, port)
        await client.asend(buf, 1)
        await client.aclose()

    await asyncio.gather(
        *[once() for _ in range(num_cycles)],
        *[server.arecv(buf2, 0, 0) for _ in range(num_cycles)],
    )
    await server.aclose()


# =============================================================================
# Resource management and lifetime
# =============================================================================


async def test_shutdown_with_in_flight_ops(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)

    recv_buf = np.ones(1 << 20, dtype=np.uint8)
    failed = asyncio.Event()

    async def pending_recv():
        try:
            await client.arecv(recv_buf, 999, (1 << 64) - 1)
        except Exception as e:
            assert "cancel" in str(e)
            failed.set()

    task = asyncio.create_task(pending_recv())
    await asyncio.sleep(0.01)
    await client.aclose()
    await task
    assert failed.is_set()
    await server.aclose()


async def test_implicit_destruction_without_close(port):
    server = Server()
    server.listen(SERVER_ADDR, port)
    client = Client()
    await client.aconnect(SERVER_ADDR, port)

    del server
    del client
    gc.collect()
    await asyncio.sleep(0.5)
    assert True
