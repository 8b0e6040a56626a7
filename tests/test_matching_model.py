"""Property-based check of the tag-matching engine against a reference
model of UCX matching semantics:

* a recv (tag, mask) matches message m iff (m.tag & mask) == (tag & mask)
* unexpected messages are kept in ARRIVAL order; a newly posted recv takes
  the FIRST matching unexpected message
* each message is delivered exactly once

The test makes arrival order deterministic by sending everything (with a
flush + settle) before posting any recv, then posts recvs one at a time
and compares each delivery against the model.
"""
import asyncio

import numpy as np
import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from starway_amd import Client, Server

SERVER_ADDR = "127.0.0.1"


def model_match(unmatched: list[int], tag: int, mask: int) -> int | None:
    for i, mtag in enumerate(unmatched):
        if (mtag & mask) == (tag & mask):
            return i
    return None


@st.composite
def scenario(draw):
    n_msgs = draw(st.integers(min_value=1, max_value=12))
    send_tags = [draw(st.integers(min_value=0, max_value=15))
                 for _ in range(n_msgs)]
    recvs = []
    for _ in range(n_msgs):
        tag = draw(st.integers(min_value=0, max_value=15))
        mask = draw(st.sampled_from([0, 0x3, 0xC, 0xF, (1 << 64) - 1]))
        recvs.append((tag, mask))
    return send_tags, recvs


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(scenario())
def test_matching_against_model(case):
    send_tags, recvs = case

    async def run():
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        server = Server()
        client = Client()
        server.listen(SERVER_ADDR, port)
        await client.aconnect(SERVER_ADDR, port)
        try:
            # Phase 1: all sends, flushed, settled => arrival order is the
            # send order and everything is in the unexpected queue.
            for i, tag in enumerate(send_tags):
                await client.asend(np.array([i], dtype=np.uint8), tag)
            await client.aflush()
            await asyncio.sleep(0.05)

            unmatched = list(send_tags)
            payloads = list(range(len(send_tags)))
            buf = np.zeros(1, dtype=np.uint8)
            for tag, mask in recvs:
                expect = model_match(unmatched, tag, mask)
                if expect is None:
                    continue  # a recv that would pend forever: skip posting
                got_tag, ln = await asyncio.wait_for(
                    server.arecv(buf, tag, mask), 10)
                assert ln == 1
                assert got_tag == unmatched[expect], (
                    f"recv({tag:#x},{mask:#x}) got tag {got_tag:#x}, model "
                    f"says {unmatched[expect]:#x} (unmatched={unmatched})")
                assert int(buf[0]) == payloads[expect]
                del unmatched[expect]
                del payloads[expect]
        finally:
            await client.aclose()
            await server.aclose()

    asyncio.run(run())


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(scenario())
def test_posted_recv_fifo_against_model(case):
    """Dual of the test above: recvs are PRE-POSTED, messages arrive one
    at a time. Model: an arriving message is claimed by the FIRST posted
    recv (in post order) whose (tag, mask) accepts it; recvs that never
    match fail with "cancel" at close."""
    send_tags, recvs = case

    async def run():
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        server = Server()
        client = Client()
        server.listen(SERVER_ADDR, port)
        await client.aconnect(SERVER_ADDR, port)
        try:
            bufs = [np.zeros(1, dtype=np.uint8) for _ in recvs]
            futs = [server.arecv(bufs[i], tag, mask)
                    for i, (tag, mask) in enumerate(recvs)]
            await asyncio.sleep(0.05)  # recvs reach the posted table

            # Model assignment: message j -> first free matching recv.
            taken: dict[int, int] = {}  # recv idx -> message idx
            for j, mtag in enumerate(send_tags):
                for i, (tag, mask) in enumerate(recvs):
                    if i in taken:
                        continue
                    if (mtag & mask) == (tag & mask):
                        taken[i] = j
                        break

            for j, tag in enumerate(send_tags):
                await client.asend(np.array([j], dtype=np.uint8), tag)
            await client.aflush()

            for i, fut in enumerate(futs):
                if i in taken:
                    got_tag, ln = await asyncio.wait_for(fut, 10)
                    j = taken[i]
                    assert got_tag == send_tags[j] and int(bufs[i][0]) == j, (
                        f"recv {i} {recvs[i]} got ({got_tag}, {bufs[i][0]}), "
                        f"model says msg {j} tag {send_tags[j]}")
        finally:
            await client.aclose()
            await server.aclose()  # cancels the never-matching recvs
            for i, fut in enumerate(futs):
                if i not in taken:
                    with pytest.raises(Exception, match="cancel"):
                        await asyncio.wait_for(fut, 10)

    asyncio.run(run())


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.lists(st.sampled_from([1, 100, 4095, 4097, 65536]),
                min_size=1, max_size=10))
def test_per_sender_order_across_eager_planes(sizes):
    """Messages from one sender with one tag must deliver in send order to
    a stream of wildcard recvs even when consecutive messages take
    different eager paths (inline <= 4096 < zero-copy/captured)."""

    async def run():
        import socket

        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()

        server = Server()
        client = Client()
        server.listen(SERVER_ADDR, port)
        await client.aconnect(SERVER_ADDR, port)
        try:
            for j, n in enumerate(sizes):
                await client.asend(np.full(n, j % 251, dtype=np.uint8), 9)
            await client.aflush()
            for j, n in enumerate(sizes):
                buf = np.zeros(max(sizes), dtype=np.uint8)
                tag, ln = await asyncio.wait_for(server.arecv(buf, 0, 0), 10)
                assert tag == 9 and ln == n, (j, n, ln)
                assert (buf[:ln] == j % 251).all(), (j, n)
        finally:
            await client.aclose()
            await server.aclose()

    asyncio.run(run())
