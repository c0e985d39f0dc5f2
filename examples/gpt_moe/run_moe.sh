#!/usr/bin/env bash
# GPT-MoE throughput run (the reference's run_moe.sh protocol:
# --fake_input, --stop_at_step=10, --log_every_step=1).
set -e
cd "$(dirname "$0")/../.."
exec python examples/gpt_moe/pretrain_moe.py --stop-at-step "${STOP:-10}" "$@"
