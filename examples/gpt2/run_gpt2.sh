#!/usr/bin/env bash
# GPT-2 benchmark runs (the reference's examples/GPT2 launch pattern:
# fake input, stop_at_step, per-step logging). One process per GPU over
# RCCL for multi-GPU; bench.py is the measured flagship entry.
set -e
cd "$(dirname "$0")/../.."
MODEL="${MODEL:-gpt2-345m}"
GPUS="${GPUS:-1}"
if [ "$GPUS" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$GPUS" \
      --master-addr 127.0.0.1 bench.py --gpus "$GPUS" --model "$MODEL" "$@"
fi
exec python examples/gpt2/train.py --model "$MODEL" "$@"
