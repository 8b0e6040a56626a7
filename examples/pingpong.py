#!/usr/bin/env python3
"""evaluate_perf-vs-measured sweep, 1 B - 1 GiB (the reference's
pingpong.py analog): compares the analytic transfer-time model against a
measured one-way transfer per size.
"""
import asyncio
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


async def main() -> None:
    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    server.listen("127.0.0.1", 0x5157)
    await client.aconnect("127.0.0.1", 0x5157)

    print(f"{'size':>12} {'model (s)':>12} {'measured (s)':>13}")
    for size in [1, 1024, 1 << 20, 50 << 20, 1 << 30]:
        est = client.evaluate_perf(size)
        send = np.empty(size, dtype=np.uint8)
        recv = np.empty(size, dtype=np.uint8)
        fut = server.arecv(recv, 1, (1 << 64) - 1)
        t0 = time.perf_counter()
        await client.asend(send, 1)
        await client.aflush()
        await fut
        dt = time.perf_counter() - t0
        print(f"{size:>12} {est:>12.6f} {dt:>13.6f}")

    await client.aclose()
    await server.aclose()


if __name__ == "__main__":
    asyncio.run(main())
