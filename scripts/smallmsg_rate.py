#!/usr/bin/env python3
"""Small-message engine-rate probe: messages/second at the raw-callback
API (no asyncio), device or host buffers, same-process loopback pair.

Separates the ENGINE's small-message capacity (push/unpack batching,
completion plumbing) from the asyncio-facade cost that dominates the
`small-messages` scenario number. Prints one JSON line.
"""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import asyncio
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402


def alloc(device, n, fill):
    if device == "cpu":
        return np.full(n, fill, dtype=np.uint8)
    import torch

    t = torch.full((n,), fill, dtype=torch.uint8, device=device)
    torch.cuda.synchronize()
    return t


async def run(args):
    import starway_amd as sw

    server = sw.Server()
    client = sw.Client()
    addr = server.listen_address()
    await client.aconnect_address(addr)

    full = (1 << 64) - 1
    window = args.window
    srcs = [alloc(args.device, args.nbytes, i % 251) for i in range(window)]
    dsts = [alloc(args.device, args.nbytes, 0) for _ in range(window)]

    recv_done = [0]
    send_done = [0]

    def on_recv(tag, ln):
        recv_done[0] += 1

    def on_send():
        send_done[0] += 1

    def on_fail(err):
        raise RuntimeError(err)

    def one_round(base_tag):
        for i in range(window):
            server.recv(dsts[i], base_tag + i, full, on_recv, on_fail)
        for i in range(window):
            client.send(srcs[i], base_tag + i, on_send, on_fail)

    # Warmup
    target = 0
    for r in range(args.warmup):
        one_round(10_000 + r * window)
        target += window
        while recv_done[0] < target:
            time.sleep(0)

    t0 = time.perf_counter()
    for r in range(args.rounds):
        one_round(1_000_000 + r * window)
        target += window
        while recv_done[0] < target:
            time.sleep(0)
    dt = time.perf_counter() - t0

    total = args.rounds * window
    stats = server._server.get_stats()
    await client.aclose()
    await server.aclose()
    return {
        "probe": "smallmsg_engine_rate",
        "device": args.device,
        "nbytes": args.nbytes,
        "window": window,
        "rounds": args.rounds,
        "msgs_per_sec": round(total / dt, 1),
        "usec_per_msg": round(dt / total * 1e6, 2),
        "inbox_rx": stats["inbox_rx"],
        "gpu_rx": stats["gpu_rx"],
        "eager_rx": stats["eager_rx"],
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--nbytes", type=int, default=1024)
    p.add_argument("--window", type=int, default=64)
    p.add_argument("--rounds", type=int, default=100)
    p.add_argument("--warmup", type=int, default=10)
    args = p.parse_args()
    print(json.dumps(asyncio.run(run(args))))


if __name__ == "__main__":
    main()
