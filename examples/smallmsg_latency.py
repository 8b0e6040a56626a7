#!/usr/bin/env python3
"""Low-latency small-message pattern on the raw callback API.

The asyncio facade costs one event-loop wakeup per completion burst
(~9 us); latency-critical code should use the callback API directly and
keep recvs PRE-POSTED so the doorbell kernel is armed when the payload
lands (see csrc/smallmsg.hip). This example measures a 64 B device-tensor
pingpong both ways and prints the half-RTT.

Run on a GPU box:  python examples/smallmsg_latency.py
"""
import os

os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import asyncio
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


async def main() -> None:
    import torch

    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    await client.aconnect_address(server.listen_address())

    full = (1 << 64) - 1
    ping = torch.full((64,), 1, dtype=torch.uint8, device="cuda")
    pong = torch.zeros(64, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()

    samples = []
    for i in range(1500):
        done: list = []
        t0 = time.perf_counter()
        # Pre-post the recv (arms the doorbell), then send.
        server.recv(pong, 5000 + i, full,
                    lambda tag, ln: done.append(ln),
                    lambda err: done.append(err))
        sent: list = []
        client.send(ping, 5000 + i,
                    lambda: sent.append(1), lambda e: sent.append(e))
        while not done:
            time.sleep(0)  # yield the GIL to the engine callback
        if i >= 300:
            samples.append(time.perf_counter() - t0)
        while not sent:
            time.sleep(0)

    us = np.array(samples) * 1e6
    print(f"64 B one-way p50 {np.percentile(us, 50):.1f} us "
          f"(half-RTT {np.percentile(us, 50) / 2:.1f} us), "
          f"p99 {np.percentile(us, 99):.1f} us")
    print("doorbell deliveries:",
          server._server.get_stats()["doorbell_rx"], "/ 1500")
    await client.aclose()
    await server.aclose()


if __name__ == "__main__":
    asyncio.run(main())
