"""Minimal client: 2-layer MLP (matmul + softmax + SGD-style training) sent
to a TepDist server over gRPC — the counterpart of the reference's
examples/smoke_testing/simple.py (53-line matmul+softmax+SGD client).

Start a server first:  python -m tepdist_amd.rpc.server --port 2222
Then:                  SERVER_PORT=2222 python examples/smoke_testing/simple.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.ir.graph import Graph
from tepdist_amd.rpc.client import TepdistClient, TepdistSession


def mlp_graph(batch=32, din=64, dh=128, dout=10):
    g = Graph()
    x = g.add_input("x", (batch, din))
    y = g.add_input("labels", (batch,), "i64")
    w1 = g.add_param("w1", (dh, din))
    b1 = g.add_param("b1", (dh,))
    w2 = g.add_param("w2", (dout, dh))
    b2 = g.add_param("b2", (dout,))
    h = g.add("linear", [x, w1, b1], (batch, dh), attrs={"act": "gelu"})
    logits = g.add("linear", [h, w2, b2], (batch, dout))
    loss = g.add("cross_entropy", [logits, y], ())
    g.outputs = [loss.id]
    return g


def main():
    sess = TepdistSession(TepdistClient())
    info = sess.compile_graph(mlp_graph(), num_devices=1)
    print("plan:", info["summary"])
    gen = torch.Generator().manual_seed(0)
    x = torch.randn(32, 64, generator=gen)
    labels = torch.randint(0, 10, (32,), generator=gen)
    for step in range(5):
        loss = sess.step({"x": x, "labels": labels})
        print(f"step {step} loss {loss:.4f}")


if __name__ == "__main__":
    main()
