#!/usr/bin/env python3
"""Pingpong-flag repro with per-iteration timing + engine stats: mirrors
the scenario exactly (same tag each iteration, bidirectional pre-posted
recvs, one event loop driving both sides)."""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import asyncio
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402

PING, PONG = 0x2B20, 0x2B21
MODE = "worker"
FULL = (1 << 64) - 1


def alloc(device, n, fill):
    if device == "cpu":
        return np.full(n, fill, dtype=np.uint8)
    import torch

    t = torch.full((n,), fill, dtype=torch.uint8, device=device)
    torch.cuda.synchronize()
    return t


async def run(device, iters, nbytes):
    import starway_amd as sw

    server = sw.Server()
    client = sw.Client()
    if MODE == "socket":
        server.listen("0.0.0.0", 18987)
        await client.aconnect("127.0.0.1", 18987)
    else:
        addr = server.listen_address()
        await client.aconnect_address(addr)
    ep = next(iter(server.list_clients()))

    ping = alloc(device, nbytes, 1)
    pong_b = alloc(device, nbytes, 2)
    rx_s = alloc(device, nbytes, 0)
    rx_c = alloc(device, nbytes, 0)

    async def server_side():
        for _ in range(iters):
            await server.arecv(rx_s, PING, FULL)
            await server.asend(ep, pong_b, PONG)

    samples = []
    send_us = []
    reply_us = []

    async def client_side():
        for _ in range(iters):
            reply = client.arecv(rx_c, PONG, FULL)
            t0 = time.perf_counter()
            await client.asend(ping, PING)
            t1 = time.perf_counter()
            await reply
            t2 = time.perf_counter()
            samples.append(t2 - t0)
            send_us.append((t1 - t0) * 1e6)
            reply_us.append((t2 - t1) * 1e6)

    await asyncio.gather(server_side(), client_side())
    ss = server._server.get_stats()
    cs = client._client.get_stats()
    await client.aclose()
    await server.aclose()
    us = np.array(samples) * 1e6
    print(json.dumps({
        "device": device, "nbytes": nbytes, "iters": iters,
        "first20_us": [round(x, 1) for x in us[:20]],
        "p50_us": round(float(np.percentile(us, 50)), 1),
        "p90_us": round(float(np.percentile(us, 90)), 1),
        "max_us": round(float(us.max()), 1),
        "send_p50_us": round(float(np.percentile(send_us, 50)), 1),
        "reply_p50_us": round(float(np.percentile(reply_us, 50)), 1),
        "server": {k: v for k, v in ss.items() if v},
        "client": {k: v for k, v in cs.items() if v},
    }))


if __name__ == "__main__":
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--iters", type=int, default=300)
    p.add_argument("--nbytes", type=int, default=1)
    p.add_argument("--mode", default="worker", choices=("worker", "socket"))
    args = p.parse_args()
    globals()["MODE"] = args.mode
    asyncio.run(run(args.device, args.iters, args.nbytes))
