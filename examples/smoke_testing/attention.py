"""Self-attention block client (counterpart of the reference's
examples/smoke_testing/attention.py): one attention layer trained via the
TepDist server."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.ir.graph import Graph
from tepdist_amd.rpc.client import TepdistClient, TepdistSession


def attn_graph(batch=4, seq=32, d=64, heads=4):
    g = Graph()
    BS = batch * seq
    a = {"batch": batch, "heads": heads, "seq": seq}
    x = g.add_input("x", (BS, d))
    x.attrs.update(a)
    y = g.add_input("labels", (BS,), "i64")
    wqkv = g.add_param("w_qkv", (3 * d, d))
    bqkv = g.add_param("b_qkv", (3 * d,))
    wo = g.add_param("w_o", (10, d))
    bo = g.add_param("b_o", (10,))
    qkv = g.add("linear", [x, wqkv, bqkv], (BS, 3 * d), attrs=a)
    q = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 0})
    k = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 1})
    v = g.add("split", [qkv], (BS, d), attrs={**a, "dim": 1, "index": 2})
    att = g.add("attention", [q, k, v], (BS, d), attrs=a)
    logits = g.add("linear", [att, wo, bo], (BS, 10), attrs=a)
    loss = g.add("cross_entropy", [logits, y], ())
    g.outputs = [loss.id]
    return g


def main():
    sess = TepdistSession(TepdistClient())
    info = sess.compile_graph(attn_graph(), num_devices=1)
    print("plan:", info["summary"])
    gen = torch.Generator().manual_seed(0)
    x = torch.randn(128, 64, generator=gen)
    labels = torch.randint(0, 10, (128,), generator=gen)
    for step in range(5):
        print(f"step {step} loss {sess.step({'x': x, 'labels': labels}):.4f}")


if __name__ == "__main__":
    main()
