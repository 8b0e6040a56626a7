#!/usr/bin/env python3
"""Copy-kernel microbench: bandwidth per size through sw::k_copy_b128[_nt].

Used under rocprofv3 for kernel-trace/PMC evidence:
  rocprofv3 --kernel-trace --stats -d out -- python scripts/kernel_bench.py
  rocprofv3 --pmc FETCH_SIZE WRITE_SIZE --kernel-trace -d out -- \
      python scripts/kernel_bench.py --sizes 268435456
"""
from __future__ import annotations

import os

# ROCm multiplexes streams onto GPU_MAX_HW_QUEUES hardware queues
# (default 4); oversubscription time-slices co-mapped streams at ~ms
# granularity. Must be set before the FIRST HIP init in the process
# (torch's or ours) — see ROUND2_NOTES.md "hardware-queue starvation".
os.environ.setdefault("GPU_MAX_HW_QUEUES", "16")

import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--sizes", type=int, nargs="*",
                    default=[1 << 20, 16 << 20, 256 << 20, 1 << 30])
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    import torch

    from starway_amd import _core

    torch.cuda.set_device(0)
    for size in args.sizes:
        src = torch.randint(0, 256, (size,), dtype=torch.uint8, device="cuda")
        dst = torch.empty_like(src)
        torch.cuda.synchronize()
        # warmup + correctness
        _core._copy_device_sync(dst.data_ptr(), src.data_ptr(), size, 0)
        assert torch.equal(src, dst), "copy kernel mismatch"
        t0 = time.perf_counter()
        for _ in range(args.iters):
            _core._copy_device_sync(dst.data_ptr(), src.data_ptr(), size, 0)
        dt = (time.perf_counter() - t0) / args.iters
        print(f"{size:>12} B: {size / dt / 1e12:.3f} TB/s payload "
              f"({2 * size / dt / 1e12:.3f} TB/s HBM traffic), {dt * 1e6:.1f} us",
              flush=True)


if __name__ == "__main__":
    main()
