#!/usr/bin/env python3
"""Concurrent large-transfer throughput smoke (the reference's test.py
analog): 5 concurrent 256 MiB tagged sends, loopback, CPU or GPU buffers.

Run: python examples/throughput.py [--device cuda] [--chunks 5]
"""
import argparse
import asyncio
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


async def main(device: str, chunks: int, nbytes: int) -> None:
    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    addr = server.listen_address()
    await client.aconnect_address(addr)

    def alloc(fill=None):
        if device == "cpu":
            a = np.empty(nbytes, dtype=np.uint8)
            if fill is not None:
                a.fill(fill)
            return a
        import torch

        t = torch.empty(nbytes, dtype=torch.uint8, device=device)
        if fill is not None:
            t.fill_(fill)
        return t

    sends = [alloc(fill=i) for i in range(chunks)]
    recvs = [alloc() for _ in range(chunks)]

    t0 = time.perf_counter()
    futs = [server.arecv(recvs[i], i, (1 << 64) - 1) for i in range(chunks)]
    await asyncio.gather(*(client.asend(sends[i], i) for i in range(chunks)))
    await client.aflush()
    await asyncio.gather(*futs)
    dt = time.perf_counter() - t0
    total = chunks * nbytes
    print(f"{chunks} x {nbytes / 2**20:.0f} MiB in {dt * 1e3:.1f} ms "
          f"=> {total / dt / 1e9:.2f} GB/s")

    await client.aclose()
    await server.aclose()


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--chunks", type=int, default=5)
    ap.add_argument("--mbytes", type=int, default=256)
    a = ap.parse_args()
    asyncio.run(main(a.device, a.chunks, a.mbytes << 20))
