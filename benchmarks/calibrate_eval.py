"""Evaluator calibration: predicted step time (planner/evaluate.py, run
anywhere — it's analytic) for the configs the GPU bench measures.

  python benchmarks/calibrate_eval.py            # predictions (CPU ok)

Measured numbers come from `python bench.py --model M --micro-batch B` on
the GPU box; the predicted-vs-measured table is committed at
profiles/evaluator_calibration_r2.md.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tepdist_amd.ir import gpt2_ir
from tepdist_amd.models import GPT2_CONFIGS
from tepdist_amd.planner import AutoParallel

CONFIGS = [
    ("gpt2-345m", 64, 1024),
    ("gpt2-117m", 64, 1024),
    ("gpt2-345m", 16, 1024),
    ("gpt2-1.5b", 8, 1024),
]


def main():
    for name, batch, seq in CONFIGS:
        cfg = GPT2_CONFIGS[name]
        g = gpt2_ir(cfg, batch=batch, seq=min(seq, cfg.n_ctx))
        plan = AutoParallel(g, 1).run()
        print(json.dumps({
            "model": name, "batch": batch, "seq": min(seq, cfg.n_ctx),
            "predicted_ms": round(plan.cost.total_duration * 1e3, 1),
            "mem_gib": round(plan.cost.mem_bytes / (1 << 30), 1),
        }), flush=True)


if __name__ == "__main__":
    main()
