"""RPC tests: in-process gRPC server + client training loop on the tiny
GPT-2 graph (the reference's grpc_client_test pattern: server + localhost
client), checkpoint RPCs, variable fetch."""

import pytest
import torch

from tests.conftest import free_port

from tepdist_amd.ir import gpt2_ir
from tepdist_amd.models.configs import GPT2_CONFIGS
from tepdist_amd.rpc.client import TepdistClient, TepdistSession
from tepdist_amd.rpc.server import serve


@pytest.fixture(scope="module")
def server(tmp_path_factory):
    port = free_port()
    ckpt = str(tmp_path_factory.mktemp("ckpt"))
    srv, svc = serve(port=port, block=False, ckpt_dir=ckpt)
    yield port, svc
    srv.stop(0)


def _batch(cfg, b, s, seed):
    g = torch.Generator().manual_seed(seed)
    ids = torch.randint(0, cfg.vocab_size, (b, s + 1), generator=g)
    return {"input_ids": ids[:, :-1].reshape(-1),
            "labels": ids[:, 1:].reshape(-1)}


@pytest.mark.timeout(300)
def test_rpc_train_loop(server):
    port, _ = server
    cfg = GPT2_CONFIGS["gpt2-test"]
    g = gpt2_ir(cfg, batch=4, seq=16)
    sess = TepdistSession(TepdistClient(f"127.0.0.1:{port}"))
    info = sess.compile_graph(g, num_devices=1)
    assert info["handle"] >= 1
    assert "search_time_s" in info

    losses = [sess.step(_batch(cfg, 4, 16, 42)) for _ in range(8)]
    assert losses[-1] < losses[0], losses

    # variable fetch (FetchResourceVars)
    vars_ = sess.client.fetch_resource_vars(["wte"])
    assert "wte" in vars_ and vars_["wte"].shape[0] == cfg.padded_vocab


@pytest.mark.timeout(300)
def test_rpc_checkpoint_roundtrip(server):
    port, svc = server
    cfg = GPT2_CONFIGS["gpt2-test"]
    client = TepdistClient(f"127.0.0.1:{port}")
    r = client.do_remote_save(max_to_keep=3, global_step=100)
    assert r["ok"]
    before = client.fetch_resource_vars(["wte"])["wte"].clone()
    # perturb then restore
    svc.vars["wte"].tensor.data.add_(1.0)
    client.do_remote_restore(100)
    g = gpt2_ir(cfg, batch=4, seq=16)
    sess = TepdistSession(client)
    sess.compile_graph(g, num_devices=1)
    after = client.fetch_resource_vars(["wte"])["wte"]
    torch.testing.assert_close(after, before, rtol=1e-3, atol=1e-4)


# -- master -> slave coordination (the reference's slave lifecycle,
# TransferModuleAndDefCtx -> DispatchPlan ->
# ExecuteRemotePlan fan-out, one thread per worker) ------------------------


@pytest.fixture(scope="module")
def two_workers(tmp_path_factory):
    from tepdist_amd.rpc.server import serve
    servers = []
    workers = []
    for i in range(2):
        port = free_port()
        ckpt = str(tmp_path_factory.mktemp(f"wck{i}"))
        srv, svc = serve(port=port, task_index=i, block=False, ckpt_dir=ckpt)
        servers.append(srv)
        workers.append({"ip": "127.0.0.1", "port": port, "gpu_ids": [i]})
    yield workers
    for s in servers:
        s.stop(0)


@pytest.mark.timeout(300)
def test_coordinator_dispatch_and_remote_execute(two_workers):
    from tepdist_amd.rpc.coordinator import ExecutionCoordinator

    cfg = GPT2_CONFIGS["gpt2-test"]
    g = gpt2_ir(cfg, batch=2, seq=16)
    coord = ExecutionCoordinator({"master": {"ip": "127.0.0.1"},
                                  "workers": two_workers}).init()
    rs = coord.transfer_module_and_defctx(g.to_json())
    assert all(r["ok"] for r in rs)
    rs = coord.init_remote_comm("127.0.0.1", 29617)
    assert all(r["ok"] for r in rs)
    rs = coord.dispatch_plan({"dp": 2, "tp": 1, "pp": 1})
    assert all(r["ok"] for r in rs)
    handles = [r["handle"] for r in rs]

    feeds = _batch(cfg, 2, 16, seed=5)
    outs = coord.execute_remote_plan(handles, feeds)
    losses = [float(list(o["outputs"].values())[0]) for o in outs]
    assert all(0 < l < 20 for l in losses)
    # identical graph + sharded-initializer determinism => identical losses
    assert abs(losses[0] - losses[1]) < 1e-4

    rs = coord.do_remote_save(global_step=1)
    assert all(r["ok"] for r in rs)
    fetched = coord.fetch_resource_vars()  # client unwraps to {name: tensor}
    assert len(fetched) == 2 and len(fetched[0]) > 0


# -- dispatched SPMD plans: master plans, workers execute the transformed
# sharded graph with real collectives between their processes -------------


def _spmd_worker_server(port, ckpt_dir):
    from tepdist_amd.rpc.server import serve
    serve(port=port, block=True, ckpt_dir=ckpt_dir)


@pytest.mark.timeout(300)
def test_dispatched_sharded_plan_matches_single(tmp_path):
    import multiprocessing as pmp
    import time as _t

    from tepdist_amd.ir.graph import Graph
    from tepdist_amd.ir.interpreter import GraphInterpreter
    from tepdist_amd.planner.spmd import CostSpmdStrategy
    from tepdist_amd.rpc.coordinator import ExecutionCoordinator
    from tepdist_amd.runtime.initializers import init_shard, InitSpec

    cfg = GPT2_CONFIGS["gpt2-test"]
    g = gpt2_ir(cfg, batch=4, seq=16)
    plan = CostSpmdStrategy(g, 2).run()
    # multi-round wire format: per-node LIST of per-round triples + mesh
    # (the hybrid-capable DispatchPlan surface; single round here)
    node_specs = {str(k): [[v.kind, v.partition_dim, v.num_shards]]
                  for k, v in plan.node_specs.items()}

    ctx = pmp.get_context("spawn")
    ports = [free_port() for i in range(2)]
    gloo_port = free_port()
    procs = [ctx.Process(target=_spmd_worker_server,
                         args=(p, str(tmp_path / f"w{i}")), daemon=True)
             for i, p in enumerate(ports)]
    for p in procs:
        p.start()
    _t.sleep(3)
    try:
        coord = ExecutionCoordinator(
            {"workers": [{"ip": "127.0.0.1", "port": p} for p in ports]}
        ).init()
        assert all(r["ok"] for r in
                   coord.transfer_module_and_defctx(g.to_json()))
        coord.init_remote_comm("127.0.0.1", gloo_port, join=True)
        # ship the scheduled runtime task lists too (the reference's
        # ComputeTask protos): workers rebuild the LocalPlan from them
        from tepdist_amd.runtime.scheduler import TaskScheduler
        from tepdist_amd.runtime.task_graph import build_task_dag
        dag = build_task_dag(1, 1)
        sched = TaskScheduler(dag, mem_cap_bytes=float("inf")).schedule()
        rs = coord.dispatch_plan({"node_specs": node_specs, "mesh": [2],
                                  "task_dag": dag.to_wire(),
                                  "sched_order": {str(k): v for k, v
                                                  in sched.order.items()}})
        assert all(r["ok"] for r in rs), rs
        handles = [r["handle"] for r in rs]
        feeds = _batch(cfg, 4, 16, seed=11)
        outs = coord.execute_remote_plan(handles, feeds)
        losses = [float(list(o["outputs"].values())[0]) for o in outs]

        # single-device reference with the same deterministic init
        variables = {}
        for name, nid in g.params.items():
            shape = g.nodes[nid].shape
            if name.endswith("_g"):
                spec = InitSpec("ones")
            elif name.endswith("_b"):
                spec = InitSpec("zeros")
            elif len(shape) >= 2:
                spec = InitSpec("random_normal", std=0.02)
            else:
                spec = InitSpec("zeros")
            variables[name] = init_shard(name, shape, spec,
                                         dtype=torch.float32)
        ref = list(GraphInterpreter(g).run(
            feeds, variables).values())[0].item()
        for l in losses:
            assert abs(l - ref) < 5e-3 * max(abs(ref), 1.0), (losses, ref)
    finally:
        for p in procs:
            p.terminate()


@pytest.mark.timeout(300)
def test_async_client_pipelining(server, monkeypatch):
    """NUM_PARALLEL_RPC_STEPS client-side step pipelining + periodic lazy
    variable fetch (reference jit/xla_ops.cc:617-648): in-flight steps
    overlap the RPC round-trips; the server's execute lock serializes
    them (the reference's execute_plan_mutex_)."""
    import math

    port, _ = server
    monkeypatch.setenv("NUM_PARALLEL_RPC_STEPS", "2")
    monkeypatch.setenv("FETCH_RESOURCE_VAR_STEPS", "3")
    cfg = GPT2_CONFIGS["gpt2-test"]
    g = gpt2_ir(cfg, batch=4, seq=16)
    sess = TepdistSession(TepdistClient(f"127.0.0.1:{port}"))
    sess.compile_graph(g, num_devices=1)
    losses = [sess.step(_batch(cfg, 4, 16, 42)) for _ in range(6)]
    losses += sess.drain()
    real = [l for l in losses if not math.isnan(l)]
    # warm-up returns nan until the pipeline fills; 6 submitted + drain
    # must yield 6 results total
    assert len(real) == 6, losses
    assert real[-1] < real[0], real
    # lazy fetch ran (every 3rd step)
    assert sess.last_vars is not None and "wte" in sess.last_vars
