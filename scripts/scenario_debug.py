#!/usr/bin/env python3
"""Debug harness mirroring the small-messages scenario loop (asyncio
gather of N concurrent sends/recvs per batch) with per-batch timing and
engine stats, to localize where batch time goes."""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import asyncio
import gc
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

    n = args.window
    tag = 0x2B10
    srcs = [alloc(args.device, args.nbytes, i % 251) for i in range(n)]
    dsts = [alloc(args.device, args.nbytes, 0) for _ in range(n)]
    batch_ms = []
    for b in range(args.batches):
        t0 = time.perf_counter()
        recvs = [server.arecv(dsts[i], tag, (1 << 64) - 1) for i in range(n)]
        await asyncio.gather(*(client.asend(srcs[i], tag) for i in range(n)))
        t_sent = time.perf_counter()
        await client.aflush()
        t_flush = time.perf_counter()
        await asyncio.gather(*recvs)
        t1 = time.perf_counter()
        batch_ms.append((round((t_sent - t0) * 1e3, 3),
                         round((t_flush - t_sent) * 1e3, 3),
                         round((t1 - t_flush) * 1e3, 3)))
    ss = server._server.get_stats()
    cs = client._client.get_stats()
    await client.aclose()
    await server.aclose()
    total = sum(a + b + c for a, b, c in batch_ms[2:])
    msgs = n * (args.batches - 2)
    print(json.dumps({
        "device": args.device,
        "window": n,
        "batch_ms(sends,flush,recvs)": batch_ms,
        "msgs_per_sec_after_warmup": round(msgs / (total / 1e3), 1),
        "server_stats": {k: v for k, v in ss.items() if v},
        "client_stats": {k: v for k, v in cs.items() if v},
    }))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--nbytes", type=int, default=1024)
    p.add_argument("--window", type=int, default=64)
    p.add_argument("--batches", type=int, default=10)
    p.add_argument("--gc-off", action="store_true")
    args = p.parse_args()
    if args.gc_off:
        gc.disable()
        gc.freeze()
    asyncio.run(run(args))


if __name__ == "__main__":
    main()
