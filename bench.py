"""Flagship benchmark: GPT-2 AUTO-PARALLEL training throughput on MI355X.

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run with one rank per
GPU over RCCL; this script reads RANK/LOCAL_RANK/WORLD_SIZE from the env.
W untimed warmup steps, then exactly K timed steps bracketed by a barrier +
torch.cuda.synchronize on both sides; elapsed time is MAX over ranks; rank 0
prints one JSON line with the whole-node aggregate tokens/sec.

Auto-parallel (default): the model's IR goes through the AutoParallel
planner (exploration mode) and the PLANNED GRAPH IS WHAT EXECUTES — the
plan's per-node DistSpec stacks drive the multi-round SpmdTransform, the
CommDevManager turns mesh rounds into RCCL process groups, and each rank
runs its transformed graph through the interpreter over the CDNA4 kernel
layer (runtime/planned.py; reference: ExecuteRPCPlan runs the planner's
compiled sub-modules, service_rt.cc:530-671). Plans with pipeline stages
run the generic stage decomposition of the planned graph through the
task-list executor (scheduled 1F1B order, gc_plan buffer release).
`--parallel dp|tp<N>|pp<N>` selects the hand-parallelized model classes
for A/B comparison.

Metric/config per BASELINE.json: tokens/sec (whole node), GPT-2
auto-parallel, synthetic data, random-init weights, bf16."""

from __future__ import annotations

import argparse
import json
import time

import torch
import torch.distributed as dist

from tepdist_amd.ir import gpt2_ir
from tepdist_amd.models import GPT2, GPT2_CONFIGS
from tepdist_amd.parallel import GradReducer, init_distributed
from tepdist_amd.train import AdamW, Trainer


def plan_parallelism(cfg, world: int, global_batch: int, seq: int,
                     override: str):
    """Returns (dp, tp, pp, micro, search_s, plan) — plan is the full
    ParallelPlan for the planned-graph path (None for hand overrides)."""
    if override != "auto":
        dp, tp, pp = world, 1, 1
        if override.startswith("tp"):
            tp = int(override[2:]); dp = world // tp
        elif override.startswith("pp"):
            pp = int(override[2:]); dp = world // pp
        elif override != "dp":
            raise ValueError(override)
        return dp, tp, pp, 1, 0.0, None
    from tepdist_amd.planner import AutoParallel
    t0 = time.time()
    g = gpt2_ir(cfg, batch=global_batch, seq=seq)
    plan = AutoParallel(g, world).run()
    search = time.time() - t0
    return plan.dp, plan.tp, plan.pp, plan.micro_batches, search, plan


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", type=str, default="gpt2-345m")
    ap.add_argument("--micro-batch", type=int, default=64,
                    help="per-GPU batch rows per step (weak scaling)")
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--parallel", type=str, default="auto",
                    help="auto | dp | tp<N> | pp<N>")
    ap.add_argument("--device", type=str, default=None,
                    help="override device (cpu for plumbing tests)")
    args = ap.parse_args()

    rank, world, local_rank = init_distributed()
    if args.device is not None:
        device = torch.device(args.device)
    else:
        device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
            else torch.device("cpu")

    cfg = GPT2_CONFIGS[args.model]
    if args.seq > cfg.n_ctx:
        # grow the position table instead of silently clamping --seq
        import dataclasses
        cfg = dataclasses.replace(cfg, n_ctx=args.seq)
    seq = args.seq
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    global_batch = args.micro_batch * world

    # ---- plan (rank 0 plans; the FULL plan object is broadcast so every
    # rank transforms the identical specs — re-planning per rank could
    # diverge under ILP time limits) ----
    search_s = 0.0
    plan = None
    if rank == 0:
        dp, tp, pp, micro, search_s, plan = plan_parallelism(
            cfg, world, global_batch, seq, args.parallel)
    if world > 1:
        obj = [plan if rank == 0 else None,
               (dp, tp, pp, micro) if rank == 0 else None]
        dist.broadcast_object_list(obj, src=0)
        plan = obj[0]
        dp, tp, pp, micro = obj[1]
    if rank == 0:
        print(f"# plan: dp={dp} tp={tp} pp={pp} micro={micro} "
              f"search={search_s:.2f}s", flush=True)
    search_b = search_s if rank == 0 else 0.0
    planned_path = args.parallel == "auto" and plan is not None

    local_batch = max(global_batch // dp, 1)
    grad_accum = max(min(micro, local_batch), 1) if pp == 1 else 1
    micro_size = max(local_batch // grad_accum, 1)

    torch.manual_seed(1234)
    # Seed the synthetic-data stream by DATA-PARALLEL rank only:
    # every rank inside a TP group and every stage of a PP chain must
    # consume the identical batch stream — tp groups are consecutive ranks
    # (dp_rank = rank // tp), pp layout is stage-major (dp_rank = rank % dp).
    data_rank = (rank % dp) if pp > 1 else (rank // tp)
    g = torch.Generator().manual_seed(4321 + data_rank)

    def make_ids(n_rows):
        return torch.randint(0, cfg.vocab_size, (n_rows, seq + 1),
                             generator=g)

    def barrier_sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)
        if world > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    # ---- build the distributed model per the plan ------------------------
    if planned_path and pp == 1:
        # THE PLANNED GRAPH EXECUTES: multi-round SpmdTransform of the
        # plan's node specs -> per-rank interpreter over the CDNA4 kernels,
        # dp rounds synced by a bucketed SUM-mode reducer, driven by the
        # standard Trainer (GA + fused AdamW + hipGraph step capture)
        from tepdist_amd.runtime.planned import PlannedModule
        model = PlannedModule(gpt2_ir(cfg, batch=global_batch, seq=seq),
                              plan, device=device, dtype=dtype)
        if rank == 0:
            print(f"# {model.describe()}", flush=True)
        opt = model.make_optimizer(lr=1e-4)   # ZeroAdamW when the plan's
        # ZeRO decision shards optimizer state over the dp group
        reducer = model.make_reducer()
        grad_accum = max(min(micro, global_batch), 1)
        trainer = Trainer(model, opt, grad_accum_steps=grad_accum,
                          reducer=reducer)
        m_rows = max(global_batch // grad_accum, 1)
        # every rank consumes the IDENTICAL global batch stream; the
        # planner-inserted dynamic_slice takes this rank's part. Generate
        # ON DEVICE (same seed + same GPU model => identical Philox
        # sequences on every rank): at N=8 the global batch is ~4 MB of
        # ids per step — CPU generation + pageable upload inside the
        # timed loop would tax weak scaling.
        if device.type == "cuda":
            gdev = torch.Generator(device=device).manual_seed(4321)
        else:
            gdev = torch.Generator().manual_seed(4321)

        def run_step():
            ids = torch.randint(0, cfg.vocab_size,
                                (m_rows * grad_accum, seq + 1),
                                generator=gdev, device=gdev.device)
            def bi(i):
                sl = ids[i * m_rows:(i + 1) * m_rows]
                return sl[:, :-1].to(device), sl[:, 1:].to(device)
            return trainer.train_step(bi)
    elif planned_path:
        # planned PIPELINE: generic stage decomposition of the planned
        # graph + per-stage mesh transform, run by the task-list executor
        # (scheduled 1F1B order, pre-posted recv queue, gc_plan release)
        from tepdist_amd.planner.stage_decomposition import decompose_stages
        from tepdist_amd.runtime.comm import CommDevManager
        from tepdist_amd.runtime.executor import build_stage_executor
        from tepdist_amd.runtime.planned import PlannedStageModule
        m_rows = max(global_batch // micro, 1)   # whole-mesh rows per micro
        g_micro = gpt2_ir(cfg, batch=m_rows, seq=seq)
        sp = decompose_stages(g_micro, plan.node_stage, pp)
        cdm = CommDevManager([n for n in plan.mesh_rounds if n > 1] or [1],
                             pp=pp)
        stage, coords = cdm.coords()
        mod = PlannedStageModule(sp, stage, g_micro, plan=plan,
                                 device=str(device), dtype=dtype, comm=cdm)
        opt = AdamW(mod.parameters(), lr=1e-4)
        reducer = mod.make_reducer()
        pp_ranks = [cdm.rank_of(s, coords) for s in range(pp)]
        ex = build_stage_executor(
            mod, stage, pp, pp_ranks, micro, act_shape=mod.act_shape,
            act_dtype=dtype, device=device, reducer=reducer, optimizer=opt,
            pp_group=cdm.pipeline_column_group())
        # rank-IDENTICAL stream: the stage graphs dp-slice the global
        # batch in-graph (dynamic_slice over the mesh round), so every
        # rank must feed the same ids
        g = torch.Generator().manual_seed(4321)

        def run_step():
            ids = make_ids(m_rows * micro)
            def bi(m):
                sl = ids[m * m_rows:(m + 1) * m_rows]
                return sl[:, :-1].to(device), sl[:, 1:].to(device)
            return ex.run_step(bi)
    elif pp == 1:
        from tepdist_amd.parallel.tp import ParallelEnv
        env = ParallelEnv.create(tp) if world > 1 else ParallelEnv.single()
        model = GPT2(cfg, dtype=dtype, env=env).to(device)
        opt = AdamW(model.parameters(), lr=1e-4)
        reducer = GradReducer(model.parameters(), env.dp_group) \
            if dp > 1 else None
        trainer = Trainer(model, opt, grad_accum_steps=grad_accum,
                          reducer=reducer)

        def run_step():
            ids = make_ids(micro_size * grad_accum)
            def bi(i):
                sl = ids[i * micro_size:(i + 1) * micro_size]
                return sl[:, :-1].to(device), sl[:, 1:].to(device)
            return trainer.train_step(bi)
    else:
        # pipeline (x dp): stage-major rank layout [pp, dp]
        from tepdist_amd.models.gpt2 import GPT2Stage, layer_ranges
        from tepdist_amd.parallel.pp import PipelineEngine
        stage = rank // dp
        dp_rank = rank % dp
        pp_ranks = [s * dp + dp_rank for s in range(pp)]
        dp_group = None
        for s in range(pp):
            ranks = [s * dp + r for r in range(dp)]
            grp = dist.new_group(ranks)
            if rank in ranks:
                dp_group = grp
        pp_group = None
        for r in range(dp):
            ranks = [s * dp + r for s in range(pp)]
            grp = dist.new_group(ranks)
            if rank in ranks:
                pp_group = grp
        lo, hi = layer_ranges(cfg.n_layer, pp)[stage]
        mod = GPT2Stage(cfg, lo, hi, stage == 0, stage == pp - 1,
                        dtype=dtype).to(device)
        opt = AdamW(mod.parameters(), lr=1e-4)
        reducer = GradReducer(mod.parameters(), dp_group) if dp > 1 else None
        m_per = max(local_batch // micro, 1)
        eng = PipelineEngine(mod, stage, pp, pp_ranks, micro,
                             act_shape=(m_per, seq, cfg.n_embd),
                             act_dtype=dtype, device=device,
                             reducer=reducer, pp_group=pp_group)

        def run_step():
            ids = make_ids(m_per * micro)
            def bi(m):
                sl = ids[m * m_per:(m + 1) * m_per]
                return sl[:, :-1].to(device), sl[:, 1:].to(device)
            loss = eng.train_step(bi)
            opt.step()
            opt.zero_grad()
            return loss

    # ---- measure ---------------------------------------------------------
    for _ in range(args.warmup):
        loss = run_step()
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        loss = run_step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    if planned_path and pp == 1:
        rows_per_step = m_rows * grad_accum          # already global
    elif planned_path:
        rows_per_step = m_rows * micro               # already global
    elif pp == 1:
        rows_per_step = micro_size * grad_accum * dp
    else:
        rows_per_step = max(local_batch // micro, 1) * micro * dp
    tokens_per_step = rows_per_step * seq
    ms_per_step = elapsed / args.steps * 1000.0
    tokens_per_sec = tokens_per_step * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec",
            "value": round(tokens_per_sec, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if dtype == torch.bfloat16 else "fp32",
            "data": "synthetic",
            "final_loss": round(float(loss), 4),
            "config": {
                "model": args.model,
                "global_batch": rows_per_step,
                "seq_len": seq,
                "parallelism": (f"auto-planned-graph:dp{dp}tp{tp}pp{pp}"
                                f"mesh{plan.mesh_rounds}micro{micro}"
                                if planned_path else
                                f"auto:dp{dp}tp{tp}pp{pp}micro{micro}"
                                if args.parallel == "auto"
                                else f"{args.parallel}:dp{dp}tp{tp}pp{pp}"),
                "plan_search_s": round(search_b, 3),
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
