#!/bin/bash
# On-box copy-kernel geometry sweep: blocks cap x unroll at 256 MiB / 1 GiB.
cd "$(dirname "$0")/.."
for SZ in 268435456 1073741824; do
  for BLK in 2048 4096 8192 16384; do
    for UNR in 4 8; do
      OUT=$(STARWAY_COPY_BLOCKS=$BLK STARWAY_COPY_UNROLL=$UNR ./bin/copy_bench $SZ 10 2>&1 | tail -1)
      echo "size=$SZ blocks=$BLK unroll=$UNR => $OUT"
    done
  done
done
