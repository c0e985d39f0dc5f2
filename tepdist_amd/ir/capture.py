"""IR capture: build planner graphs from models.

Two paths, mirroring the reference's client (which produces whole-graph HLO
from TF):
  - from_fx: generic torch.fx symbolic trace + shape propagation for simple
    models whose forward is built from tepdist_amd.ops calls (the
    smoke_testing MLP class of clients);
  - gpt2_ir / llama_ir: explicit exporters for the benchmark model families
    (the client KNOWS its graph; op_group tags = layer ids, which is what
    the reference's dapple_scope provides).

Activations are kept in the flattened (B*S, hidden) form with a `batch`
attr carrying the true batch size, so dim-0 splits are sequence-preserving
batch splits."""

from __future__ import annotations

import torch

from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.models.configs import GPT2Config


def gpt2_ir(cfg: GPT2Config, batch: int, seq: int) -> Graph:
    g = Graph()
    V, d, H, L = cfg.padded_vocab, cfg.n_embd, cfg.n_head, cfg.n_layer
    BS = batch * seq
    ids = g.add_input("input_ids", (BS,), "i64")
    ids.attrs["batch"] = batch
    labels = g.add_input("labels", (BS,), "i64")
    labels.attrs["batch"] = batch

    wte = g.add_param("wte", (V, d))
    wpe = g.add_param("wpe", (cfg.n_ctx, d))
    pos = g.add("data", [], (BS,), "i64", attrs={"batch": batch}, name="pos")

    xe = g.add("embedding", [ids, wte], (BS, d), attrs={"batch": batch})
    xp = g.add("embedding", [pos, wpe], (BS, d), attrs={"batch": batch})
    x = g.add("add", [xe, xp], (BS, d), attrs={"batch": batch})

    for l in range(L):
        a = {"batch": batch, "heads": H, "seq": seq}
        ln1g = g.add_param(f"h{l}.ln1_g", (d,), op_group=l)
        ln1b = g.add_param(f"h{l}.ln1_b", (d,), op_group=l)
        h = g.add("layernorm", [x, ln1g, ln1b], (BS, d), attrs=a, op_group=l)
        wqkv = g.add_param(f"h{l}.w_qkv", (3 * d, d), op_group=l)
        bqkv = g.add_param(f"h{l}.b_qkv", (3 * d,), op_group=l)
        qkv = g.add("linear", [h, wqkv, bqkv], (BS, 3 * d), attrs=a,
                    op_group=l)
        # fused packed-qkv attention: the CDNA4 flash kernels consume the
        # packed [B,S,3d] projection directly (no q/k/v transpose copies —
        # the round-1 profile showed 9.7% of GPU time in transposes)
        att = g.add("attention_qkv", [qkv], (BS, d), attrs=a, op_group=l)
        wproj = g.add_param(f"h{l}.w_proj", (d, d), op_group=l)
        bproj = g.add_param(f"h{l}.b_proj", (d,), op_group=l)
        pr = g.add("linear", [att, wproj, bproj], (BS, d), attrs=a,
                   op_group=l)
        x = g.add("add", [x, pr], (BS, d), attrs=a, op_group=l)
        ln2g = g.add_param(f"h{l}.ln2_g", (d,), op_group=l)
        ln2b = g.add_param(f"h{l}.ln2_b", (d,), op_group=l)
        h2 = g.add("layernorm", [x, ln2g, ln2b], (BS, d), attrs=a, op_group=l)
        wfc = g.add_param(f"h{l}.w_fc", (4 * d, d), op_group=l)
        bfc = g.add_param(f"h{l}.b_fc", (4 * d,), op_group=l)
        f = g.add("linear", [h2, wfc, bfc], (BS, 4 * d),
                  attrs={**a, "act": "gelu"}, op_group=l)
        wout = g.add_param(f"h{l}.w_out", (d, 4 * d), op_group=l)
        bout = g.add_param(f"h{l}.b_out", (d,), op_group=l)
        o = g.add("linear", [f, wout, bout], (BS, d), attrs=a, op_group=l)
        x = g.add("add", [x, o], (BS, d), attrs=a, op_group=l)

    lnfg = g.add_param("lnf_g", (d,), op_group=L - 1)
    lnfb = g.add_param("lnf_b", (d,), op_group=L - 1)
    x = g.add("layernorm", [x, lnfg, lnfb], (BS, d),
              attrs={"batch": batch}, op_group=L - 1)
    logits = g.add("linear", [x, wte], (BS, V), attrs={"batch": batch},
                   op_group=L - 1)
    loss = g.add("cross_entropy", [logits, labels], (),
                 attrs={"batch": batch}, op_group=L - 1)
    g.outputs = [loss.id]
    return g


def llama_ir(cfg, batch: int, seq: int) -> Graph:
    """Llama-family exporter (models/llama.py): RMSNorm + rotary + SwiGLU
    decoder in the same flattened (B*S, hidden) form as gpt2_ir, so the
    auto-parallel planner and SpmdTransform apply unchanged."""
    g = Graph()
    V, d, H, L = cfg.vocab_size, cfg.n_embd, cfg.n_head, cfg.n_layer
    hd = cfg.ffn_mult * d
    BS = batch * seq
    ids = g.add_input("input_ids", (BS,), "i64")
    ids.attrs["batch"] = batch
    labels = g.add_input("labels", (BS,), "i64")
    labels.attrs["batch"] = batch

    wte = g.add_param("wte", (V, d))
    x = g.add("embedding", [ids, wte], (BS, d), attrs={"batch": batch})

    for l in range(L):
        a = {"batch": batch, "heads": H, "seq": seq}
        ln1 = g.add_param(f"h{l}.ln1_g", (d,), op_group=l)
        h = g.add("rmsnorm", [x, ln1], (BS, d), attrs=a, op_group=l)
        wqkv = g.add_param(f"h{l}.w_qkv", (3 * d, d), op_group=l)
        qkv = g.add("linear", [h, wqkv], (BS, 3 * d), attrs=a, op_group=l)
        q = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 0},
                  op_group=l)
        k = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 1},
                  op_group=l)
        v = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 2},
                  op_group=l)
        qr = g.add("rope", [q], (BS, d), attrs=a, op_group=l)
        kr = g.add("rope", [k], (BS, d), attrs=a, op_group=l)
        att = g.add("attention", [qr, kr, v], (BS, d), attrs=a, op_group=l)
        wo = g.add_param(f"h{l}.w_o", (d, d), op_group=l)
        pr = g.add("linear", [att, wo], (BS, d), attrs=a, op_group=l)
        x = g.add("add", [x, pr], (BS, d), attrs=a, op_group=l)
        ln2 = g.add_param(f"h{l}.ln2_g", (d,), op_group=l)
        h2 = g.add("rmsnorm", [x, ln2], (BS, d), attrs=a, op_group=l)
        wg = g.add_param(f"h{l}.w_gate", (hd, d), op_group=l)
        wu = g.add_param(f"h{l}.w_up", (hd, d), op_group=l)
        gate = g.add("linear", [h2, wg], (BS, hd), attrs=a, op_group=l)
        up = g.add("linear", [h2, wu], (BS, hd), attrs=a, op_group=l)
        sw = g.add("swiglu", [gate, up], (BS, hd), attrs=a, op_group=l)
        wd = g.add_param(f"h{l}.w_down", (d, hd), op_group=l)
        o = g.add("linear", [sw, wd], (BS, d), attrs=a, op_group=l)
        x = g.add("add", [x, o], (BS, d), attrs=a, op_group=l)

    lnf = g.add_param("lnf_g", (d,), op_group=L - 1)
    x = g.add("rmsnorm", [x, lnf], (BS, d), attrs={"batch": batch},
              op_group=L - 1)
    head = g.add_param("lm_head", (V, d), op_group=L - 1)
    logits = g.add("linear", [x, head], (BS, V), attrs={"batch": batch},
                   op_group=L - 1)
    loss = g.add("cross_entropy", [logits, labels], (),
                 attrs={"batch": batch}, op_group=L - 1)
    g.outputs = [loss.id]
    return g


# --------------------------------------------------------------------------
# generic fx capture (simple clients)
# --------------------------------------------------------------------------

_OP_MAP = {
    "linear": "linear",
    "matmul": "matmul",
    "layernorm": "layernorm",
    "softmax": "softmax",
    "attention": "attention",
    "embedding": "embedding",
    "cross_entropy": "cross_entropy",
    "dropout": "dropout",
}


def from_fx(model: torch.nn.Module, *example_args) -> Graph:
    """Capture a model built from tepdist_amd.ops calls via torch.fx.
    Works for static-shape forward graphs (the smoke-test MLP class)."""
    import torch.fx as fx
    from torch.fx.passes.shape_prop import ShapeProp

    traced = fx.symbolic_trace(model)
    ShapeProp(traced).propagate(*example_args)

    g = Graph()
    env = {}
    pidx = 0
    for n in traced.graph.nodes:
        meta = n.meta.get("tensor_meta")
        shape = tuple(meta.shape) if meta is not None else ()
        dtype = "bf16"
        if meta is not None and meta.dtype in (torch.int64, torch.int32):
            dtype = "i64"
        if n.op == "placeholder":
            env[n] = g.add_input(n.name, shape, dtype)
        elif n.op == "get_attr":
            try:
                t = traced.get_parameter(n.target)
            except AttributeError:
                t = traced.get_buffer(n.target)
            env[n] = g.add_param(n.target, tuple(t.shape))
        elif n.op == "call_function":
            fname = getattr(n.target, "__name__", str(n.target))
            op = _OP_MAP.get(fname)
            ins = [env[a] for a in n.args if isinstance(a, fx.Node)]
            if op is None:
                if fname in ("add", "mul"):
                    op = fname
                else:
                    op = "elementwise"
            env[n] = g.add(op, ins, shape, dtype)
        elif n.op == "call_method":
            ins = [env[a] for a in ([n.args[0]] if n.args else [])
                   if isinstance(a, fx.Node)]
            env[n] = g.add("elementwise", ins, shape, dtype)
        elif n.op == "output":
            args = n.args[0]
            outs = args if isinstance(args, (tuple, list)) else [args]
            g.outputs = [env[a].id for a in outs if isinstance(a, fx.Node)]
    return g


def moe_ir(cfg, batch: int, seq: int,
           capacity_factor: float = None) -> Graph:
    """GPT-MoE exporter (models/moe.py semantics; reference
    examples/gpt_moe): GShard-style MoE FFN layers expressed STATICALLY —
    `moe_dispatch` builds the group-blocked [E, capacity, d] tensor,
    the experts run as batched matmuls over the expert dim, and
    `moe_combine` returns tokens. Expert parallelism is then a planner
    RESHARD: dispatch is capacity-split (each token shard fills its
    capacity block), the expert matmuls want expert-dim splits, and the
    SpmdTransform inserts the all-to-all on the mismatch (the reference's
    kDAPPLEAllToAll path, SURVEY.md §2.7 EP). Auxiliary load-balancing
    loss is a training-loop concern and not part of the planned graph."""
    from tepdist_amd.models.configs import MoEConfig  # noqa: F401
    g = Graph()
    V, d, H, L = cfg.padded_vocab, cfg.n_embd, cfg.n_head, cfg.n_layer
    E, K = cfg.num_experts, cfg.top_k
    cf = capacity_factor if capacity_factor is not None \
        else cfg.capacity_factor
    BS = batch * seq
    C = max(int(cf * BS * K / E), 4)
    ids = g.add_input("input_ids", (BS,), "i64")
    ids.attrs["batch"] = batch
    labels = g.add_input("labels", (BS,), "i64")
    labels.attrs["batch"] = batch
    wte = g.add_param("wte", (V, d))
    wpe = g.add_param("wpe", (cfg.n_ctx, d))
    pos = g.add("data", [], (BS,), "i64", attrs={"batch": batch}, name="pos")
    xe = g.add("embedding", [ids, wte], (BS, d), attrs={"batch": batch})
    xp = g.add("embedding", [pos, wpe], (BS, d), attrs={"batch": batch})
    x = g.add("add", [xe, xp], (BS, d), attrs={"batch": batch})

    for l in range(L):
        a = {"batch": batch, "heads": H, "seq": seq}
        ln1g = g.add_param(f"h{l}.ln1_g", (d,), op_group=l)
        ln1b = g.add_param(f"h{l}.ln1_b", (d,), op_group=l)
        h = g.add("layernorm", [x, ln1g, ln1b], (BS, d), attrs=a, op_group=l)
        wqkv = g.add_param(f"h{l}.w_qkv", (3 * d, d), op_group=l)
        bqkv = g.add_param(f"h{l}.b_qkv", (3 * d,), op_group=l)
        qkv = g.add("linear", [h, wqkv, bqkv], (BS, 3 * d), attrs=a,
                    op_group=l)
        att = g.add("attention_qkv", [qkv], (BS, d), attrs=a, op_group=l)
        wproj = g.add_param(f"h{l}.w_proj", (d, d), op_group=l)
        bproj = g.add_param(f"h{l}.b_proj", (d,), op_group=l)
        pr = g.add("linear", [att, wproj, bproj], (BS, d), attrs=a,
                   op_group=l)
        x = g.add("add", [x, pr], (BS, d), attrs=a, op_group=l)
        ln2g = g.add_param(f"h{l}.ln2_g", (d,), op_group=l)
        ln2b = g.add_param(f"h{l}.ln2_b", (d,), op_group=l)
        h2 = g.add("layernorm", [x, ln2g, ln2b], (BS, d), attrs=a,
                   op_group=l)
        if (l + 1) % cfg.moe_every == 0:
            # MoE FFN: gate -> dispatch -> expert batched matmuls -> combine
            wgate = g.add_param(f"h{l}.moe_gate", (E, d), op_group=l)
            glog = g.add("linear", [h2, wgate], (BS, E), attrs=a, op_group=l)
            disp = g.add("moe_dispatch", [h2, glog], (E, C, d),
                         attrs={"batch": batch, "k": K}, op_group=l)
            w1 = g.add_param(f"h{l}.moe_w1", (E, d, 4 * d), op_group=l)
            hh = g.add("matmul", [disp, w1], (E, C, 4 * d), op_group=l)
            hg = g.add("gelu", [hh], (E, C, 4 * d), op_group=l)
            w2 = g.add_param(f"h{l}.moe_w2", (E, 4 * d, d), op_group=l)
            ye = g.add("matmul", [hg, w2], (E, C, d), op_group=l)
            o = g.add("moe_combine", [ye, h2, glog], (BS, d),
                      attrs={"batch": batch, "k": K}, op_group=l)
        else:
            wfc = g.add_param(f"h{l}.w_fc", (4 * d, d), op_group=l)
            bfc = g.add_param(f"h{l}.b_fc", (4 * d,), op_group=l)
            f = g.add("linear", [h2, wfc, bfc], (BS, 4 * d),
                      attrs={**a, "act": "gelu"}, op_group=l)
            wout = g.add_param(f"h{l}.w_out", (d, 4 * d), op_group=l)
            bout = g.add_param(f"h{l}.b_out", (d,), op_group=l)
            o = g.add("linear", [f, wout, bout], (BS, d), attrs=a,
                      op_group=l)
        x = g.add("add", [x, o], (BS, d), attrs=a, op_group=l)

    lnfg = g.add_param("lnf_g", (d,), op_group=L - 1)
    lnfb = g.add_param("lnf_b", (d,), op_group=L - 1)
    x = g.add("layernorm", [x, lnfg, lnfb], (BS, d),
              attrs={"batch": batch}, op_group=L - 1)
    logits = g.add("linear", [x, wte], (BS, V), attrs={"batch": batch},
                   op_group=L - 1)
    loss = g.add("cross_entropy", [logits, labels], (),
                 attrs={"batch": batch}, op_group=L - 1)
    g.outputs = [loss.id]
    return g


def wrn_ir(cfg, batch: int) -> Graph:
    """Wide-ResNet PLANNER graph (models/wide_resnet.py structure;
    reference examples/wide_resnet): conv2d/add/linear/cross_entropy
    nodes with real shapes and flops so AutoParallel can search the
    image path (batch vs out-channel splits, stage cuts, memory). The
    execution engine for WRN remains the module path (conv-as-GEMM
    kernels + Trainer); this graph is the planning surface."""
    from tepdist_amd.models.wide_resnet import _DEPTH
    g = Graph()
    blocks = _DEPTH[cfg.n_layer]
    base = 64 * cfg.width_factor
    r = cfg.image_size // 4          # stem 7x7/2 + maxpool/2
    x = g.add_input("images", (batch, 3, cfg.image_size, cfg.image_size))
    x.attrs["batch"] = batch
    labels = g.add_input("labels", (batch,), "i64")
    w = g.add_param("stem.w", (base, 3, 7, 7))
    x = g.add("conv2d", [x, w], (batch, base, r, r),
              attrs={"batch": batch, "kernel": (7, 7)})
    cin = base
    for si, nb in enumerate(blocks):
        planes = base * (2 ** si)
        cout = planes * 4
        stride = 1 if si == 0 else 2
        if si > 0:
            r //= 2
        for b in range(nb):
            grp = si * 10 + b
            a = {"batch": batch}
            w1 = g.add_param(f"s{si}.b{b}.w1", (planes, cin, 1, 1),
                             op_group=grp)
            h = g.add("conv2d", [x, w1], (batch, planes, r, r),
                      attrs={**a, "kernel": (1, 1)}, op_group=grp)
            w2 = g.add_param(f"s{si}.b{b}.w2", (planes, planes, 3, 3),
                             op_group=grp)
            h = g.add("conv2d", [h, w2], (batch, planes, r, r),
                      attrs={**a, "kernel": (3, 3)}, op_group=grp)
            w3 = g.add_param(f"s{si}.b{b}.w3", (cout, planes, 1, 1),
                             op_group=grp)
            h = g.add("conv2d", [h, w3], (batch, cout, r, r),
                      attrs={**a, "kernel": (1, 1)}, op_group=grp)
            if cin != cout:
                wd = g.add_param(f"s{si}.b{b}.down", (cout, cin, 1, 1),
                                 op_group=grp)
                x = g.add("conv2d", [x, wd], (batch, cout, r, r),
                          attrs={**a, "kernel": (1, 1)}, op_group=grp)
            x = g.add("add", [x, h], (batch, cout, r, r), attrs=a,
                      op_group=grp)
            cin = cout
    x = g.add("elementwise", [x], (batch, cin), attrs={"batch": batch})
    wfc = g.add_param("fc.w", (cfg.num_classes, cin))
    logits = g.add("linear", [x, wfc], (batch, cfg.num_classes),
                   attrs={"batch": batch})
    loss = g.add("cross_entropy", [logits, labels], (),
                 attrs={"batch": batch})
    g.outputs = [loss.id]
    return g
