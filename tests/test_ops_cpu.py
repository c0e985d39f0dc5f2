"""CPU tests of the op layer: autograd correctness of every planner-sharded
op against plain torch (fp32) autograd."""


import torch

from tepdist_amd import ops


def _randn(*shape, dtype=torch.float32, seed=0):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(*shape, generator=g, dtype=torch.float32).to(dtype)


def test_linear_matches_torch():
    x = _randn(32, 48, seed=1).requires_grad_()
    w = _randn(24, 48, seed=2).requires_grad_()
    b = _randn(24, seed=3).requires_grad_()
    y = ops.linear(x, w, b)
    y.sum().backward()

    x2 = x.detach().clone().requires_grad_()
    w2 = w.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    y2 = torch.nn.functional.linear(x2, w2, b2)
    y2.sum().backward()

    torch.testing.assert_close(y, y2, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(w.grad, w2.grad, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(b.grad, b2.grad, rtol=1e-5, atol=1e-5)


def test_linear_gelu_fused():
    x = _randn(16, 32, seed=4).requires_grad_()
    w = _randn(64, 32, seed=5).requires_grad_()
    b = _randn(64, seed=6).requires_grad_()
    y = ops.linear(x, w, b, act="gelu")
    y.sum().backward()

    x2 = x.detach().clone().requires_grad_()
    w2 = w.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    y2 = torch.nn.functional.gelu(
        torch.nn.functional.linear(x2, w2, b2), approximate="tanh")
    y2.sum().backward()

    torch.testing.assert_close(y, y2, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-4)
    torch.testing.assert_close(w.grad, w2.grad, rtol=1e-4, atol=1e-4)


def test_layernorm_matches_torch():
    x = _randn(8, 10, 32, seed=7).requires_grad_()
    g = _randn(32, seed=8).requires_grad_()
    b = _randn(32, seed=9).requires_grad_()
    y = ops.layernorm(x, g, b)
    (y * _randn(8, 10, 32, seed=10)).sum().backward()

    x2 = x.detach().clone().requires_grad_()
    g2 = g.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    y2 = torch.nn.functional.layer_norm(x2, (32,), g2, b2, eps=1e-5)
    (y2 * _randn(8, 10, 32, seed=10)).sum().backward()

    torch.testing.assert_close(y, y2, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(x.grad, x2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g.grad, g2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(b.grad, b2.grad, rtol=1e-4, atol=1e-5)


def test_softmax_causal():
    s = _randn(2, 3, 8, 8, seed=11).requires_grad_()
    p = ops.softmax(s, scale=0.5, causal=True)
    (p * _randn(2, 3, 8, 8, seed=12)).sum().backward()

    s2 = s.detach().clone().requires_grad_()
    mask = torch.ones(8, 8, dtype=torch.bool).tril()
    p2 = torch.softmax((s2 * 0.5).masked_fill(~mask, float("-inf")), dim=-1)
    (p2 * _randn(2, 3, 8, 8, seed=12)).sum().backward()

    torch.testing.assert_close(p, p2, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(s.grad, s2.grad, rtol=1e-4, atol=1e-6)


def test_attention_matches_sdpa():
    q = _randn(2, 4, 16, 8, seed=13).requires_grad_()
    k = _randn(2, 4, 16, 8, seed=14).requires_grad_()
    v = _randn(2, 4, 16, 8, seed=15).requires_grad_()
    out = ops.attention(q, k, v, causal=True)
    out.sum().backward()

    q2, k2, v2 = (t.detach().clone().requires_grad_() for t in (q, k, v))
    out2 = torch.nn.functional.scaled_dot_product_attention(
        q2, k2, v2, is_causal=True)
    out2.sum().backward()

    torch.testing.assert_close(out, out2, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(q.grad, q2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(k.grad, k2.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(v.grad, v2.grad, rtol=1e-4, atol=1e-5)


def test_embedding_matches_torch():
    table = _randn(100, 16, seed=16).requires_grad_()
    ids = torch.randint(0, 100, (4, 7), generator=torch.Generator().manual_seed(17))
    y = ops.embedding(ids, table)
    (y * _randn(4, 7, 16, seed=18)).sum().backward()

    t2 = table.detach().clone().requires_grad_()
    y2 = torch.nn.functional.embedding(ids, t2)
    (y2 * _randn(4, 7, 16, seed=18)).sum().backward()

    torch.testing.assert_close(y, y2)
    torch.testing.assert_close(table.grad, t2.grad, rtol=1e-5, atol=1e-6)


def test_cross_entropy_matches_torch():
    logits = _randn(20, 37, seed=19).requires_grad_()
    targets = torch.randint(0, 37, (20,), generator=torch.Generator().manual_seed(20))
    targets[3] = -1  # ignore
    loss = ops.cross_entropy(logits, targets, ignore_index=-1)
    loss.backward()

    l2 = logits.detach().clone().requires_grad_()
    loss2 = torch.nn.functional.cross_entropy(l2, targets, ignore_index=-1)
    loss2.backward()

    torch.testing.assert_close(loss, loss2, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(logits.grad, l2.grad, rtol=1e-4, atol=1e-6)


def test_dropout_scale_and_determinism():
    x = torch.ones(1000)
    y1 = ops.dropout(x, 0.3, seed=5, offset=7)
    y2 = ops.dropout(x, 0.3, seed=5, offset=7)
    torch.testing.assert_close(y1, y2)
    kept = (y1 != 0)
    assert abs(kept.float().mean().item() - 0.7) < 0.08
    torch.testing.assert_close(y1[kept], torch.full_like(y1[kept], 1 / 0.7))


def test_adamw_matches_torch():
    torch.manual_seed(0)
    p_ref = torch.nn.Parameter(torch.randn(50, 20))
    p_ours = torch.nn.Parameter(p_ref.detach().clone())
    opt_ref = torch.optim.AdamW([p_ref], lr=1e-2, betas=(0.9, 0.999),
                                eps=1e-8, weight_decay=0.01)
    from tepdist_amd.train import AdamW
    opt_ours = AdamW([p_ours], lr=1e-2, weight_decay=0.01, no_decay_1d=False)
    for i in range(5):
        g = torch.randn(50, 20, generator=torch.Generator().manual_seed(i))
        p_ref.grad = g.clone()
        p_ours.grad = g.clone()
        opt_ref.step()
        opt_ours.step()
    torch.testing.assert_close(p_ours.data, p_ref.data, rtol=1e-5, atol=1e-6)


def test_mlp_op_matches_composed():
    """ops.mlp (fused on GPU) composes to exactly linear(gelu)+linear on
    CPU — values and all five grads."""
    import torch

    from tepdist_amd import ops
    torch.manual_seed(0)
    T, d = 16, 8
    x = torch.randn(T, d, requires_grad=True)
    w1 = torch.randn(4 * d, d, requires_grad=True)
    b1 = torch.randn(4 * d, requires_grad=True)
    w2 = torch.randn(d, 4 * d, requires_grad=True)
    b2 = torch.randn(d, requires_grad=True)
    y = ops.mlp(x, w1, b1, w2, b2)
    y.sum().backward()
    x2, w12, b12, w22, b22 = (t.detach().clone().requires_grad_()
                              for t in (x, w1, b1, w2, b2))
    ref = ops.linear(ops.linear(x2, w12, b12, act="gelu"), w22, b22)
    ref.sum().backward()
    torch.testing.assert_close(y, ref)
    for a, b in ((x, x2), (w1, w12), (b1, b12), (w2, w22), (b2, b22)):
        torch.testing.assert_close(a.grad, b.grad)


def test_add_layernorm_op_cpu():
    """ops.add_layernorm returns (sum, normed) with correct grads to both
    consumers on the CPU composed path."""
    import torch

    from tepdist_amd import ops
    torch.manual_seed(1)
    x = torch.randn(8, 16, requires_grad=True)
    r = torch.randn(8, 16, requires_grad=True)
    g = torch.ones(16, requires_grad=True)
    b = torch.zeros(16, requires_grad=True)
    s, y = ops.add_layernorm(x, r, g, b)
    (s.pow(2).sum() + y.sum()).backward()
    x2, r2, g2, b2 = (t.detach().clone().requires_grad_()
                      for t in (x, r, g, b))
    s2 = x2 + r2
    y2 = ops.layernorm(s2, g2, b2)
    (s2.pow(2).sum() + y2.sum()).backward()
    torch.testing.assert_close(s, s2)
    torch.testing.assert_close(y, y2)
    torch.testing.assert_close(x.grad, x2.grad)
    torch.testing.assert_close(g.grad, g2.grad)
