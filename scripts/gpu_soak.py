#!/usr/bin/env python3
"""GPU endurance soak: sustained mixed-size tagged traffic on one box,
checking for leaks (device memory, host RSS, event-pool growth) and for
counter consistency. Runs ~N seconds (default 45)."""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import asyncio
import os
import resource
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


async def main(seconds: int) -> None:
    import torch

    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    addr = server.listen_address()
    await client.aconnect_address(addr)
    ep = next(iter(server.list_clients()))

    sizes = [64, 4096, 1 << 20, 16 << 20]
    bufs = {
        n: (torch.randint(0, 256, (n,), dtype=torch.uint8, device="cuda"),
            torch.zeros(n, dtype=torch.uint8, device="cuda"))
        for n in sizes
    }
    torch.cuda.synchronize()

    free0, _ = torch.cuda.mem_get_info()
    rss0 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss

    t_end = time.time() + seconds
    it = 0
    bytes_moved = 0
    while time.time() < t_end:
        n = sizes[it % len(sizes)]
        src, dst = bufs[n]
        tag = (it % 1000) + 1
        if it % 2 == 0:
            fut = server.arecv(dst, tag, (1 << 64) - 1)
            await client.asend(src, tag)
        else:
            fut = client.arecv(dst, tag, (1 << 64) - 1)
            await server.asend(ep, src, tag)
        await fut
        bytes_moved += n
        it += 1
        if it % 500 == 0:
            torch.cuda.synchronize()
            assert torch.equal(src, dst), f"mismatch at iter {it} size {n}"
        if it % 100_000 == 0:
            rss = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
            print(f"  it={it} rss={(rss - rss0) / 1024:.1f} MB", flush=True)

    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    rss1 = resource.getrusage(resource.RUSAGE_SELF).ru_maxrss
    stats = client._client.get_stats()
    sstats = server._server.get_stats()
    print(f"iters={it} bytes={bytes_moved/1e9:.2f} GB "
          f"rate={bytes_moved/seconds/1e9:.2f} GB/s "
          f"msgs/s={it/seconds:.0f}")
    print(f"device free delta: {(free0-free1)/1e6:.1f} MB "
          f"(expect ~0); rss delta: {(rss1-rss0)/1024:.1f} MB")
    print(f"client stats: {stats}")
    print(f"server stats: {sstats}")
    assert free0 - free1 < 256e6, "device memory leak suspected"
    await client.aclose()
    await server.aclose()
    print("GPU SOAK PASS")


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=45)
    a = ap.parse_args()
    asyncio.run(main(a.seconds))
