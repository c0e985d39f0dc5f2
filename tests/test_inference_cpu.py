"""KV-cache generation engine: greedy decode with the cache must match
the argmax chain of repeated full forwards exactly."""

import torch

from tepdist_amd.inference import Generator
from tepdist_amd.models import GPT2, GPT2_CONFIGS


def _full_forward_greedy(model, ids, n):
    cfg = model.cfg
    out = ids.clone()
    for _ in range(n):
        logits = model(out)[:, -1, :cfg.vocab_size].float()
        nxt = logits.argmax(-1, keepdim=True)
        out = torch.cat([out, nxt], dim=1)
    return out


def test_cached_greedy_matches_full_forward():
    torch.manual_seed(0)
    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.float32).eval()
    ids = torch.randint(0, cfg.vocab_size, (2, 12))
    gen = Generator(model)
    with torch.no_grad():
        ref = _full_forward_greedy(model, ids, 6)
    got = gen.generate(ids, max_new_tokens=6)
    assert torch.equal(got, ref), (got, ref)


def test_sampled_generation_is_reproducible():
    torch.manual_seed(0)
    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.float32).eval()
    ids = torch.randint(0, cfg.vocab_size, (1, 8))
    gen = Generator(model)
    a = gen.generate(ids, 5, temperature=0.8, top_k=20, seed=7)
    b = gen.generate(ids, 5, temperature=0.8, top_k=20, seed=7)
    assert torch.equal(a, b)
    assert a.shape == (1, 13)


def test_chunked_prefill_matches_full():
    """Feeding the prompt in two chunks through the cache must give the
    same next-token logits as one full forward."""
    torch.manual_seed(0)
    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.float32).eval()
    gen = Generator(model)
    ids = torch.randint(0, cfg.vocab_size, (1, 12))

    H, d = cfg.n_head, cfg.n_embd
    D = d // H
    kc = [torch.zeros(1, H, 12, D) for _ in range(cfg.n_layer)]
    vc = [torch.zeros(1, H, 12, D) for _ in range(cfg.n_layer)]
    with torch.no_grad():
        from tepdist_amd import ops
        pos0 = torch.arange(8)
        x = ops.embedding(ids[:, :8], model.wte) + \
            ops.embedding(pos0, model.wpe)
        gen._stack(x, kc, vc, 0)
        pos1 = torch.arange(8, 12)
        x2 = ops.embedding(ids[:, 8:], model.wte) + \
            ops.embedding(pos1, model.wpe)
        h = gen._stack(x2, kc, vc, 8)          # chunked continuation
        logits = gen._logits(h)[:, -1, :cfg.vocab_size]
        ref = model(ids)[:, -1, :cfg.vocab_size]
    assert torch.allclose(logits, ref, atol=1e-4), \
        (logits - ref).abs().max()
