#!/bin/bash
# Scaling-curve runner: bench.py at N = 1, 2, 4, 8 endpoints on one node
# (SURVEY.md §7 M4). Usage: scripts/scaling_curve.sh [msg_bytes] [steps]
set -u
MSG=${1:-268435456}
STEPS=${2:-20}
WARM=5
PORT=$((29000 + RANDOM % 1000))
cd "$(dirname "$0")/.."
for N in 1 2 4 8; do
  AVAIL=$(python -c "import torch; print(torch.cuda.device_count())" 2>/dev/null || echo 0)
  if [ "$N" -gt "$AVAIL" ] && [ "$AVAIL" -gt 0 ]; then
    echo "# N=$N skipped (only $AVAIL GPUs visible)"
    continue
  fi
  if [ "$N" -eq 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARM" --msg-bytes "$MSG"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port "$PORT" \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARM" --msg-bytes "$MSG" \
      2>/dev/null | grep '"metric"'
  fi
done
