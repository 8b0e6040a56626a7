#!/usr/bin/env python3
"""Randomized GPU messaging fuzz: mixed sizes/planes/orders on a loopback
pair. Every message is filled with (tag % 251), so any completed recv can
be validated against the sender_tag it reports regardless of matching
order. Exercises inbox/RTS/eager/h2d planes, unexpected-queue paths,
wildcard and exact masks, and interleaved flushes.
"""
from __future__ import annotations

import os

os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import asyncio
import random
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402

FULL = (1 << 64) - 1
SIZES = [1, 64, 777, 1024, 4096, 5000, 65536, 1 << 20, (1 << 21) + 13]


async def run(seed: int, rounds: int) -> None:
    import torch

    import starway_amd as sw

    rng = random.Random(seed)
    server, client = sw.Server(), sw.Client()
    await client.aconnect_address(server.listen_address())
    ep = next(iter(server.list_clients()))

    tag_counter = 1000

    async def one_round(r: int) -> None:
        nonlocal tag_counter
        k = rng.randint(1, 24)
        sizes = [rng.choice(SIZES) for _ in range(k)]
        tags = list(range(tag_counter, tag_counter + k))
        tag_counter += k + 1
        s2c = rng.random() < 0.4  # direction
        host_recv = rng.random() < 0.15
        recv_first = rng.random() < 0.5

        sender = (lambda b, t: server.asend(ep, b, t)) if s2c \
            else (lambda b, t: client.asend(b, t))
        recver = client if s2c else server

        srcs = [torch.full((sizes[i],), tags[i] % 251, dtype=torch.uint8,
                           device="cuda") for i in range(k)]
        torch.cuda.synchronize()
        if host_recv:
            dsts = [np.zeros(max(sizes), dtype=np.uint8) for _ in range(k)]
        else:
            dsts = [torch.zeros(max(sizes), dtype=torch.uint8,
                                device="cuda") for _ in range(k)]

        def post_recvs():
            order = list(range(k))
            rng.shuffle(order)
            wildcard = rng.random() < 0.5
            posted = []  # (future, the dst it fills)
            for i in order:
                if wildcard:
                    posted.append((recver.arecv(dsts[i], 0, 0), dsts[i]))
                else:
                    posted.append(
                        (recver.arecv(dsts[i], tags[i], FULL), dsts[i]))
            return posted

        if recv_first:
            posted = post_recvs()
            await asyncio.sleep(0)
            sends = [sender(srcs[i], tags[i]) for i in range(k)]
        else:
            sends = [sender(srcs[i], tags[i]) for i in range(k)]
            if rng.random() < 0.5:
                await asyncio.sleep(0.002)  # park in the unexpected queue
            posted = post_recvs()

        got = await asyncio.gather(*(f for f, _ in posted))
        await asyncio.gather(*sends)
        if rng.random() < 0.3:
            await (server.aflush() if s2c else client.aflush())

        torch.cuda.synchronize()
        seen = sorted(t for t, _ in got)
        assert seen == tags, f"round {r}: tags {seen} != {tags}"
        for (t, ln), (_, d) in zip(got, posted):
            assert ln == sizes[tags.index(t)], (r, t, ln)
            head = d[:ln]
            val = t % 251
            if isinstance(head, np.ndarray):
                assert (head == val).all(), (r, t)
            else:
                assert bool(torch.all(head == val)), (r, t)

    t0 = time.time()
    for r in range(rounds):
        await one_round(r)
    dt = time.time() - t0
    ss = server._server.get_stats()
    print(f"FUZZ PASS seed={seed} rounds={rounds} in {dt:.1f}s; server "
          f"stats: inbox={ss['inbox_rx']} gpu={ss['gpu_rx']} "
          f"eager={ss['eager_rx']} unexpected={ss['unexpected_rx']}")
    await client.aclose()
    await server.aclose()


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--seed", type=int, default=1)
    p.add_argument("--rounds", type=int, default=150)
    a = p.parse_args()
    asyncio.run(run(a.seed, a.rounds))
