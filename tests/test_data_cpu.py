"""Synthetic input pipeline (SURVEY.md §2.8 inputs.py parity):
determinism, rank sharding, CPU passthrough of the prefetcher."""

import itertools

import torch

from tepdist_amd.data import SyntheticImages, SyntheticTokens, \
    device_prefetcher


def test_tokens_rank_sharding_disjoint_and_deterministic():
    full = next(iter(SyntheticTokens(100, 8, 16, seed=3)))
    r0 = next(iter(SyntheticTokens(100, 8, 16, seed=3, rank=0, world=2)))
    r1 = next(iter(SyntheticTokens(100, 8, 16, seed=3, rank=1, world=2)))
    assert torch.equal(torch.cat([r0[0], r1[0]]), full[0])
    again = next(iter(SyntheticTokens(100, 8, 16, seed=3, rank=1, world=2)))
    assert torch.equal(again[1], r1[1])


def test_tokens_labels_are_shifted_inputs():
    x, y = next(iter(SyntheticTokens(50, 2, 8, seed=1)))
    assert torch.equal(x[:, 1:], y[:, :-1])


def test_images_shapes():
    x, y = next(iter(SyntheticImages(4, (3, 8, 8), classes=10, seed=2)))
    assert x.shape == (4, 3, 8, 8) and y.shape == (4,)
    assert int(y.max()) < 10


def test_prefetcher_cpu_passthrough():
    it = SyntheticTokens(10, 2, 4, seed=0)
    pre = device_prefetcher(iter(it), "cpu")
    batches = list(itertools.islice(pre, 3))
    ref = list(itertools.islice(iter(it), 3))
    for (x, y), (rx, ry) in zip(batches, ref):
        assert torch.equal(x, rx) and torch.equal(y, ry)
