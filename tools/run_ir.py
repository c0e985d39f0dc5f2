"""Standalone IR runner (the reference's rpc/run_arbitary_hlo.cc debug tool,
SURVEY.md §4.4): load a serialized Graph (JSON), plan it for N devices, and
execute a few iterations through the interpreter — for kernel-level A/B and
plan debugging.

  python tools/run_ir.py graph.json --devices 8 --steps 3 --dump-dag dag.dot
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from tepdist_amd.ir.graph import Graph
from tepdist_amd.planner import AutoParallel
from tepdist_amd.rpc.service import TepdistService


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("graph")
    ap.add_argument("--devices", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--dump-dag", default="")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()

    g = Graph.from_json(open(args.graph).read())
    plan = AutoParallel(g, args.devices).run()
    print(plan.summary())
    if args.dump_dag:
        from tepdist_amd.runtime.task_graph import build_task_dag
        dag = build_task_dag(plan.pp, plan.micro_batches, dp_degree=plan.dp)
        dag.dump_dot(args.dump_dag)
        print(f"task DAG -> {args.dump_dag}")

    svc = TepdistService(ckpt_dir="/tmp/run_ir_ckpt")
    r = svc.build_execution_plan({"graph": g.to_json(),
                                  "num_devices": args.devices})
    gen = torch.Generator().manual_seed(args.seed)
    feeds = {}
    for nid in g.inputs:
        n = g.nodes[nid]
        if n.dtype == "i64":
            feeds[n.name] = torch.randint(0, 100, n.shape, generator=gen)
        else:
            feeds[n.name] = torch.randn(*n.shape, generator=gen)
    for step in range(args.steps):
        out = svc.execute_plan({"handle": r["handle"], "inputs": feeds})
        loss = float(list(out["outputs"].values())[0])
        print(f"step {step}: loss {loss:.5f} "
              f"({out['duration_ms']:.2f} ms)")


if __name__ == "__main__":
    main()
