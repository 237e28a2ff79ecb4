#!/bin/bash
# 1->8 GPU scaling curve in one sitting (BASELINE.json north star):
# run on a multi-GPU box as `bash tools/scale_curve.sh [steps] [warmup]`.
# Emits one JSON line per N into gpurun_out/scale_curve.jsonl; each run
# is bounded so one wedge cannot eat the lease. GPU counts beyond the
# box's device count are skipped, so the same script works on 1-GPU
# leases (N=1 only) and 8-GPU nodes (full curve).
set -u
STEPS="${1:-20}"
WARMUP="${2:-5}"
OUT=gpurun_out/scale_curve.jsonl
: > "$OUT"
export HSA_ENABLE_IPC_MODE_LEGACY=0

NGPUS=$(python -c "import torch; print(torch.cuda.device_count())")
echo "box has $NGPUS GPUs"

for n in 1 2 4 8; do
    if [ "$n" -gt "$NGPUS" ]; then
        echo "skip N=$n (only $NGPUS GPUs)"
        continue
    fi
    echo "=== N=$n"
    if [ "$n" -eq 1 ]; then
        timeout -s KILL 420 python bench.py --gpus 1 \
            --steps "$STEPS" --warmup "$WARMUP" >> "$OUT" \
            2> "gpurun_out/scale_n1.err" || echo "N=1 rc=$?"
    else
        timeout -s KILL 480 python -m torch.distributed.run --nnodes=1 \
            --nproc-per-node "$n" --master-addr 127.0.0.1 \
            --master-port 29531 bench.py --gpus "$n" \
            --steps "$STEPS" --warmup "$WARMUP" >> "$OUT" \
            2> "gpurun_out/scale_n$n.err" || echo "N=$n rc=$?"
    fi
    tail -1 "$OUT"
done
