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
