#!/bin/bash
# Pre-scripted 1/2/4/8-GPU scaling curve: one command
# produces SCALE-style records the moment a multi-GPU lease exists.
#   bash scripts/scale_curve.sh [steps] [warmup]
set -e
STEPS=${1:-48}; WARMUP=${2:-12}
NGPU=$(python -c "import torch; print(torch.cuda.device_count())")
echo "visible GPUs: $NGPU"
mkdir -p gpurun_out
for N in 1 2 4 8; do
  [ "$N" -gt "$NGPU" ] && break
  if [ "$N" -eq 1 ]; then
    python bench.py --gpus 1 --steps "$STEPS" --warmup "$WARMUP" \
      | tee "gpurun_out/scale_n1.json"
  else
    python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
      --master-addr 127.0.0.1 --master-port 29631 \
      bench.py --gpus "$N" --steps "$STEPS" --warmup "$WARMUP" \
      | tee "gpurun_out/scale_n${N}.json"
  fi
done
echo "curve written to gpurun_out/scale_n*.json"
