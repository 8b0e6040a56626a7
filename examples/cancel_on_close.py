#!/usr/bin/env python3
"""Cancel-on-close demo (the reference's cb.py analog): a pending recv is
failed with a "cancel" error when the endpoint closes underneath it.
"""
import asyncio
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


async def main() -> None:
    import starway_amd as sw

    server, client = sw.Server(), sw.Client()
    server.listen("127.0.0.1", 0x5158)
    await client.aconnect("127.0.0.1", 0x5158)

    buf = np.zeros(1 << 20, dtype=np.uint8)

    async def doomed_recv() -> None:
        try:
            await client.arecv(buf, 999, (1 << 64) - 1)
            print("recv completed (unexpected)")
        except Exception as e:
            print(f"recv failed as expected: {e}")

    task = asyncio.create_task(doomed_recv())
    await asyncio.sleep(0.05)
    await client.aclose()
    await task
    await server.aclose()
    print("closed cleanly")


if __name__ == "__main__":
    asyncio.run(main())
