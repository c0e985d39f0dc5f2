"""Llama IR capture: the interpreter must reproduce the module loss, and
the auto-parallel planner + SpmdTransform must run on the llama graph
(auto-parallel beyond the GPT-2 family)."""

import torch

from tepdist_amd.ir.capture import llama_ir
from tepdist_amd.ir.interpreter import GraphInterpreter
from tepdist_amd.models.llama import LLAMA_CONFIGS, Llama


def _feeds_and_vars(cfg, batch, seq, model):
    gen = torch.Generator().manual_seed(5)
    ids = torch.randint(0, cfg.vocab_size, (batch, seq + 1), generator=gen)
    feeds = {"input_ids": ids[:, :-1].reshape(-1),
             "labels": ids[:, 1:].reshape(-1)}
    named = dict(model.named_parameters())
    variables = {"wte": named["wte"], "lnf_g": named["ln_f_g"],
                 "lm_head": named["lm_head"]}
    for l, blk in enumerate(model.blocks):
        variables.update({
            f"h{l}.ln1_g": blk.ln1_g, f"h{l}.ln2_g": blk.ln2_g,
            f"h{l}.w_qkv": blk.w_qkv, f"h{l}.w_o": blk.w_o,
            f"h{l}.w_gate": blk.w_gate, f"h{l}.w_up": blk.w_up,
            f"h{l}.w_down": blk.w_down})
    return feeds, variables, ids


def test_llama_ir_matches_module():
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-test"]
    model = Llama(cfg)
    g = llama_ir(cfg, batch=2, seq=32)
    assert set(g.params) == {n for n in g.params}
    feeds, variables, ids = _feeds_and_vars(cfg, 2, 32, model)
    out = list(GraphInterpreter(g).run(feeds, variables).values())[0]
    ref = model(ids[:, :-1], labels=ids[:, 1:])
    assert torch.allclose(out, ref, rtol=1e-4, atol=1e-5), \
        (out.item(), ref.item())


def test_llama_ir_plans():
    from tepdist_amd.planner.auto_parallel import AutoParallel
    cfg = LLAMA_CONFIGS["llama-test"]
    g = llama_ir(cfg, batch=8, seq=32)
    plan = AutoParallel(g, 4).run()
    assert plan.dp * plan.tp * plan.pp == 4
    assert plan.dp >= 1


def test_llama_ir_transform_consistent():
    from tepdist_amd.planner.spmd import CostSpmdStrategy
    from tepdist_amd.planner.transform import SpmdTransform
    cfg = LLAMA_CONFIGS["llama-test"]
    g = llama_ir(cfg, batch=4, seq=16)
    r = CostSpmdStrategy(g, 2).run()
    res = SpmdTransform(g, r.node_specs, 2).run()
    assert len(res.graph.nodes) >= len(g.nodes)
