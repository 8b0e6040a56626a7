#!/usr/bin/env python3
"""64 B latency breakdown probe (BASELINE config 4 diagnostics).

Measures one-way tagged 64 B delivery time in a same-process loopback pair
across the independent axes that make up the measured number:

  * --device cpu|cuda      transport plane (host eager/shm vs GPU inbox)
  * --mode asyncio|raw     completion plumbing: asyncio futures (the
                           headline metric's path) vs the raw callback API
                           (also public API surface) with a GIL-yielding
                           spin — isolates the event-loop wakeup cost
  * STARWAY_INBOX / STARWAY_DOORBELL env knobs: push-ring and pre-armed
    doorbell contributions (set them before running; they are read at
    module import)

Prints one JSON line per run. Run on a GPU box:
  for m in asyncio raw; do for d in cpu cuda; do
    python scripts/latency_probe.py --device $d --mode $m; done; done
  STARWAY_INBOX=0 python scripts/latency_probe.py --device cuda --mode raw
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
import os
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np  # noqa: E402


def alloc(device, n, fill):
    if device == "cpu":
        a = np.full(n, fill, dtype=np.uint8)
        return a
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
    ping = alloc(args.device, args.nbytes, 1)
    pong = alloc(args.device, args.nbytes, 0)
    samples = []

    if args.mode == "asyncio":
        for i in range(args.warmup + args.iters):
            t0 = time.perf_counter()
            fut = server.arecv(pong, 1000 + i, full)
            await client.asend(ping, 1000 + i)
            await fut
            if i >= args.warmup:
                samples.append(time.perf_counter() - t0)
    else:
        # Raw callback API: the engine thread fires done callbacks under
        # the GIL; the main thread yields the GIL while polling, so the
        # handoff costs a context switch instead of an event-loop wakeup.
        for i in range(args.warmup + args.iters):
            done = []
            t0 = time.perf_counter()
            server.recv(pong, 1000 + i, full,
                        lambda tag, ln: done.append(1),
                        lambda err: done.append(err))
            sent = []
            client.send(ping, 1000 + i,
                        lambda: sent.append(1), lambda err: sent.append(err))
            while not done:
                time.sleep(0)  # yield the GIL to the engine's callback
            if i >= args.warmup:
                samples.append(time.perf_counter() - t0)
            while not sent:
                time.sleep(0)

    sstats = server._server.get_stats()
    cstats = client._client.get_stats()
    await client.aclose()
    await server.aclose()

    us = np.array(samples) * 1e6
    return {
        "probe": "latency_64B_oneway",
        "device": args.device,
        "mode": args.mode,
        "nbytes": args.nbytes,
        "iters": args.iters,
        "p50_us": round(float(np.percentile(us, 50)), 2),
        "p10_us": round(float(np.percentile(us, 10)), 2),
        "p99_us": round(float(np.percentile(us, 99)), 2),
        "min_us": round(float(us.min()), 2),
        "inbox_rx": sstats["inbox_rx"] + cstats["inbox_rx"],
        "doorbell_rx": sstats["doorbell_rx"] + cstats["doorbell_rx"],
        "env": {k: os.environ.get(k) for k in
                ("STARWAY_INBOX", "STARWAY_DOORBELL", "STARWAY_SHM")
                if os.environ.get(k) is not None},
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--device", default="cpu")
    p.add_argument("--mode", choices=("asyncio", "raw"), default="asyncio")
    p.add_argument("--nbytes", type=int, default=64)
    p.add_argument("--iters", type=int, default=2000)
    p.add_argument("--warmup", type=int, default=300)
    args = p.parse_args()
    out = asyncio.run(run(args))
    print(json.dumps(out))


if __name__ == "__main__":
    main()
