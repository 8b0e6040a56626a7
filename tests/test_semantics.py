"""Tag-matching and transport semantics beyond the reference suite:
wildcard masks, per-sender ordering, truncation errors, zero-length
messages, endpoint introspection, and a loopback run of all four benchmark
scenarios (CPU sizes) exercising the L4 control protocol end to end.
"""
import asyncio
import contextlib
import random

import numpy as np
import pytest

from starway_amd import Client, Server

SERVER_ADDR = "127.0.0.1"
FULL = (1 << 64) - 1



@contextlib.asynccontextmanager
async def pair(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    try:
        yield server, client
    finally:
        await client.aclose()
        await server.aclose()


async def test_exact_tag_mask_selects_message(port):
    async with pair(port) as (server, client):
        # Two messages with different tags; recv with full mask picks the
        # matching one even though it arrived second.
        a = np.full(4, 1, dtype=np.uint8)
        b = np.full(4, 2, dtype=np.uint8)
        await client.asend(a, 10)
        await client.asend(b, 20)
        await client.aflush()
        buf = np.zeros(4, dtype=np.uint8)
        tag, ln = await server.arecv(buf, 20, FULL)
        assert tag == 20 and buf[0] == 2
        tag, ln = await server.arecv(buf, 10, FULL)
        assert tag == 10 and buf[0] == 1


async def test_partial_mask_matching(port):
    async with pair(port) as (server, client):
        # Mask only the high byte: recv(tag=0x0500, mask=0xFF00) matches any
        # message whose tag has 0x05 in that byte.
        await client.asend(np.full(1, 9, dtype=np.uint8), 0x0563)
        buf = np.zeros(1, dtype=np.uint8)
        tag, _ = await server.arecv(buf, 0x0500, 0xFF00)
        assert tag == 0x0563 and buf[0] == 9


async def test_per_sender_fifo_ordering(port):
    async with pair(port) as (server, client):
        # Same tag, wildcard recvs: delivery order must follow send order.
        n = 200
        for i in range(n):
            await client.asend(np.array([i % 256], dtype=np.uint8), 7)
        seen = []
        buf = np.zeros(1, dtype=np.uint8)
        for i in range(n):
            await server.arecv(buf, 0, 0)
            seen.append(int(buf[0]))
        assert seen == [i % 256 for i in range(n)]


async def test_truncation_error_on_small_buffer(port):
    async with pair(port) as (server, client):
        big = np.zeros(4096, dtype=np.uint8)
        small = np.zeros(16, dtype=np.uint8)
        fut = server.arecv(small, 0, 0)
        await asyncio.sleep(0.01)  # post first => matched at header time
        await client.asend(big, 1)
        with pytest.raises(Exception, match="truncated"):
            await fut


async def test_truncation_error_unexpected_path(port):
    async with pair(port) as (server, client):
        big = np.zeros(4096, dtype=np.uint8)
        await client.asend(big, 1)
        await client.aflush()
        await asyncio.sleep(0.05)  # arrives unmatched -> staged
        small = np.zeros(16, dtype=np.uint8)
        with pytest.raises(Exception, match="truncated"):
            await server.arecv(small, 0, 0)


async def test_zero_length_message(port):
    async with pair(port) as (server, client):
        empty = np.zeros(0, dtype=np.uint8)
        buf = np.zeros(4, dtype=np.uint8)
        fut = server.arecv(buf, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(empty, 42)
        tag, ln = await fut
        assert tag == 42 and ln == 0


async def test_short_message_into_larger_buffer_reports_length(port):
    async with pair(port) as (server, client):
        msg = np.arange(10, dtype=np.uint8)
        buf = np.full(100, 0xEE, dtype=np.uint8)
        fut = server.arecv(buf, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(msg, 1)
        tag, ln = await fut
        assert ln == 10
        np.testing.assert_array_equal(buf[:10], msg)
        assert (buf[10:] == 0xEE).all()  # rest untouched


async def test_bytes_like_send_buffers(port):
    async with pair(port) as (server, client):
        # Anything with the buffer protocol works for sends.
        payload = bytes(range(32))
        buf = np.zeros(32, dtype=np.uint8)
        fut = server.arecv(buf, 0, 0)
        await asyncio.sleep(0.01)
        await client.asend(np.frombuffer(payload, dtype=np.uint8), 1)
        _, ln = await fut
        assert bytes(buf) == payload


async def test_readonly_recv_buffer_rejected(port):
    async with pair(port) as (server, client):
        ro = np.zeros(8, dtype=np.uint8)
        ro.setflags(write=False)
        with pytest.raises(Exception):
            await server.arecv(ro, 0, 0)


async def test_large_transfer_integrity(port):
    async with pair(port) as (server, client):
        n = 32 << 20
        send = np.random.randint(0, 256, n, dtype=np.uint8)
        recv = np.zeros(n, dtype=np.uint8)
        fut = server.arecv(recv, 0, 0)
        await client.asend(send, 5)
        await client.aflush()
        _, ln = await fut
        assert ln == n
        np.testing.assert_array_equal(send, recv)


async def test_large_transfer_unexpected_then_posted(port):
    # Exercises the redirect path: message starts arriving before the recv
    # is posted; the posted recv binds mid-stream.
    async with pair(port) as (server, client):
        n = 32 << 20
        send = np.random.randint(0, 256, n, dtype=np.uint8)
        await client.asend(send, 5)
        await asyncio.sleep(0.02)  # stream in progress / staged
        recv = np.zeros(n, dtype=np.uint8)
        tag, ln = await server.arecv(recv, 0, 0)
        assert tag == 5 and ln == n
        np.testing.assert_array_equal(send, recv)


async def test_endpoint_introspection(port):
    async with pair(port) as (server, client):
        ep = next(iter(server.list_clients()))
        assert ep.remote_addr == "127.0.0.1"
        assert ep.local_port == port
        assert ep.remote_port > 0
        assert isinstance(ep.name, str) and ep.name
        transports = ep.view_transports()
        assert ("tcp", "sock") in transports
        assert "ServerEndpoint" in repr(ep)


async def test_send_to_dead_endpoint_fails(port):
    server = Server()
    client = Client()
    server.listen(SERVER_ADDR, port)
    await client.aconnect(SERVER_ADDR, port)
    ep = next(iter(server.list_clients()))
    await client.aclose()
    await asyncio.sleep(0.1)  # let the server observe the close
    with pytest.raises(Exception):
        await server.asend(ep, np.zeros(4, dtype=np.uint8), 1)
    await server.aclose()


async def test_worker_address_blob_rejected_if_malformed():
    client = Client()
    with pytest.raises(Exception):
        await client.aconnect_address(b"garbage-not-an-address")


def test_benchmark_registry():
    from starway_amd import list_benchmark_scenarios
    from starway_amd.benchmarks import get_scenario

    names = list_benchmark_scenarios()
    assert set(names) == {
        "large-array",
        "small-messages",
        "pingpong-flag",
        "streaming-duplex",
    }
    for n in names:
        s = get_scenario(n)
        assert s.defaults and s.client_runner and s.server_runner
    with pytest.raises(KeyError):
        get_scenario("nope")


def test_check_sys_libs():
    from starway_amd import check_sys_libs

    assert check_sys_libs() in ("system", "wheel")


def test_bench_cli_loopback_all_scenarios(tmp_path, port):
    """End-to-end L4 check: all four scenarios over the control protocol."""
    import json

    from starway_amd.bench import main

    out = tmp_path / "report.json"
    rc = main([
        "--role", "loopback", "--port", str(port),
        "--large-bytes", "4M", "--large-iterations", "2", "--large-warmup", "1",
        "--small-iterations", "2", "--small-warmup", "1",
        "--flag-iterations", "50", "--flag-warmup", "10",
        "--stream-bytes", "1M", "--stream-iterations", "4", "--stream-warmup", "1",
        "--output", str(out), "--store-trace",
    ])
    assert rc == 0
    report = json.loads(out.read_text())
    assert len(report["scenarios"]) == 4
    by_name = {s["name"]: s for s in report["scenarios"]}
    assert by_name["large-array"]["metrics"]["avg_gbps"] > 0
    assert by_name["small-messages"]["metrics"]["messages_per_second"] > 0
    assert by_name["pingpong-flag"]["metrics"]["avg_rtt_us"] > 0
    assert by_name["streaming-duplex"]["metrics"]["aggregate_gbps"] > 0


async def test_stats_counters(port):
    async with pair(port) as (server, client):
        msg = np.arange(64, dtype=np.uint8)
        buf = np.zeros(64, dtype=np.uint8)
        for i in range(3):
            fut = server.arecv(buf, 0, 0)
            await asyncio.sleep(0.005)
            await client.asend(msg, i)
            await fut
        cs = client._client.get_stats()
        ss = server._server.get_stats()
        assert cs["msgs_sent"] == 3 and cs["bytes_sent"] == 192
        assert ss["msgs_received"] == 3 and ss["bytes_received"] == 192
        assert ss["eager_rx"] == 3
