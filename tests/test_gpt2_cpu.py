"""End-to-end CPU plumbing test: tiny GPT-2 trains and the loss decreases
(the reference's smoke_testing acceptance = loss trajectory)."""

import torch

from tepdist_amd.models import GPT2, GPT2_CONFIGS
from tepdist_amd.train import AdamW, Trainer


def test_gpt2_tiny_loss_decreases():
    cfg = GPT2_CONFIGS["gpt2-test"]
    torch.manual_seed(0)
    model = GPT2(cfg, dtype=torch.float32)
    opt = AdamW(model.parameters(), lr=1e-3)
    trainer = Trainer(model, opt)

    g = torch.Generator().manual_seed(42)
    ids = torch.randint(0, cfg.vocab_size, (4, 33), generator=g)
    batch = (ids[:, :-1], ids[:, 1:])

    losses = []
    for _ in range(12):
        losses.append(trainer.train_step(lambda i: batch))
    assert losses[-1] < losses[0] * 0.9, losses


def test_gpt2_bf16_forward_finite():
    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.bfloat16)
    ids = torch.randint(0, cfg.vocab_size, (2, 17))
    loss = model(ids, labels=ids.clone())
    assert torch.isfinite(loss)
    loss.backward()
    for p in model.parameters():
        assert p.grad is None or torch.isfinite(p.grad).all()


def test_grad_accum_equals_big_batch():
    cfg = GPT2_CONFIGS["gpt2-test"]
    torch.manual_seed(0)
    m1 = GPT2(cfg, dtype=torch.float32)
    m2 = GPT2(cfg, dtype=torch.float32)
    m2.load_state_dict(m1.state_dict())

    g = torch.Generator().manual_seed(7)
    ids = torch.randint(0, cfg.vocab_size, (4, 17), generator=g)
    micro = [(ids[i:i + 2, :-1], ids[i:i + 2, 1:]) for i in (0, 2)]

    # accumulate over 2 micro-batches
    loss1 = sum(m1(x, labels=y) for x, y in micro) / 2
    loss1.backward()
    # single big batch
    loss2 = m2(ids[:, :-1], labels=ids[:, 1:])
    loss2.backward()

    # same mean loss and (since micro-batches are equal-sized and no ignored
    # labels) matching grads
    torch.testing.assert_close(loss1, loss2, rtol=1e-5, atol=1e-6)
    for (n1, p1), (_, p2) in zip(m1.named_parameters(), m2.named_parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, rtol=1e-4, atol=1e-5)
