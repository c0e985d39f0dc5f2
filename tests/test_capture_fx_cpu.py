"""Generic torch.fx capture (the smoke-test MLP client class): trace a
model built from tepdist_amd.ops calls, interpret the captured graph and
compare against the module output."""

import torch

from tepdist_amd import ops
from tepdist_amd.ir.capture import from_fx
from tepdist_amd.ir.interpreter import GraphInterpreter


class TinyMLP(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.w1 = torch.nn.Parameter(torch.randn(32, 16) * 0.3)
        self.w2 = torch.nn.Parameter(torch.randn(8, 32) * 0.3)

    def forward(self, x):
        h = ops.linear(x, self.w1)
        h = ops.softmax(h)
        return ops.linear(h, self.w2)


def test_from_fx_capture_and_interpret():
    torch.manual_seed(0)
    m = TinyMLP()
    x = torch.randn(4, 16)
    g = from_fx(m, x)
    ops_seen = [n.op for n in g.topo()]
    assert ops_seen.count("linear") == 2 and "softmax" in ops_seen
    assert len(g.params) == 2

    feeds = {n.name: x for n in [g.nodes[i] for i in g.inputs]}
    variables = {name: dict(m.named_parameters())[name]
                 for name in g.params}
    out = list(GraphInterpreter(g).run(feeds, variables).values())[0]
    ref = m(x)
    assert torch.allclose(out, ref, atol=1e-5)
