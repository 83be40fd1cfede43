#!/bin/bash
# Scaling sweep over 1/2/4/8 GPUs of one node (BASELINE.json:2 metric).
# The MI355X-native analog of the reference's cluster submission scripts
# (SURVEY.md §2.1 R7): one torch.distributed rank per GPU over RCCL/xGMI.
# Usage: bash bench/run_scaling.sh [steps] [warmup] [preset]
set -euo pipefail
STEPS=${1:-100}
WARMUP=${2:-10}
PRESET=${3:-headline}
cd "$(dirname "$0")/.."
OUT=bench/scaling_$(date +%Y%m%d_%H%M%S).jsonl
# free ephemeral rendezvous port (concurrent runs on one host must not collide)
PORT=${PCONV_MASTER_PORT:-$(python - <<'EOF'
import socket
s = socket.socket(); s.bind(("127.0.0.1", 0)); print(s.getsockname()[1]); s.close()
EOF
)}
for N in 1 2 4 8; do
  AVAIL=$(python -c "import torch; print(torch.cuda.device_count())")
  if [ "$N" -gt "$AVAIL" ]; then
    echo "skipping N=$N (only $AVAIL GPUs visible)" >&2
    continue
  fi
  if [ "$N" -eq 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP" \
      --preset "$PRESET" --json-out "$OUT"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node="$N" \
      --master-addr 127.0.0.1 --master-port "$PORT" \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP" \
      --preset "$PRESET" --json-out "$OUT"
  fi
done
echo "results in $OUT"
