"""Synthetic input pipelines (the reference's examples feed fake data
through tf.data — examples/gpt2/inputs.py, wide_resnet fake-data mode;
SURVEY.md §2.8). Deterministic, rank-sharded token/image streams plus a
device prefetcher that overlaps H2D copies with compute on a side HIP
stream (the reference's input tasks run sharded H2D off the compute
stream, DAPPLEBufferUtils::H2D)."""

from __future__ import annotations

from typing import Iterator, Optional, Tuple

import torch


class SyntheticTokens:
    """Seeded token batches: rank r of `world` sees shard r of every
    global batch, so DP runs consume disjoint slices of the same stream
    (matches the initializers' global-index determinism)."""

    def __init__(self, vocab_size: int, batch: int, seq: int, seed: int = 0,
                 rank: int = 0, world: int = 1):
        assert batch % world == 0
        self.vocab, self.batch, self.seq = vocab_size, batch, seq
        self.seed, self.rank, self.world = seed, rank, world

    def __iter__(self) -> Iterator[Tuple[torch.Tensor, torch.Tensor]]:
        step = 0
        local = self.batch // self.world
        while True:
            g = torch.Generator().manual_seed(self.seed + step)
            ids = torch.randint(0, self.vocab, (self.batch, self.seq + 1),
                                generator=g)
            mine = ids[self.rank * local:(self.rank + 1) * local]
            yield mine[:, :-1].contiguous(), mine[:, 1:].contiguous()
            step += 1


class SyntheticImages:
    def __init__(self, batch: int, shape=(3, 224, 224), classes: int = 1000,
                 seed: int = 0, rank: int = 0, world: int = 1):
        assert batch % world == 0
        self.batch, self.shape, self.classes = batch, shape, classes
        self.seed, self.rank, self.world = seed, rank, world

    def __iter__(self):
        step = 0
        local = self.batch // self.world
        while True:
            g = torch.Generator().manual_seed(self.seed + step)
            x = torch.randn(self.batch, *self.shape, generator=g)
            y = torch.randint(0, self.classes, (self.batch,), generator=g)
            s = self.rank * local
            yield x[s:s + local].contiguous(), y[s:s + local].contiguous()
            step += 1


def device_prefetcher(it, device: str, dtype: Optional[torch.dtype] = None,
                      depth: int = 2):
    """Wraps a (x, y) iterator: stages H2D copies on a dedicated stream
    `depth` batches ahead; the consumer's stream waits only on the batch
    it takes (copy/compute overlap; no-op passthrough on CPU)."""
    if not device.startswith("cuda"):
        for x, y in it:
            yield (x.to(dtype) if dtype and x.is_floating_point() else x), y
        return
    stream = torch.cuda.Stream(device=device)
    queue = []
    src = iter(it)

    def stage():
        x, y = next(src)
        with torch.cuda.stream(stream):
            xd = x.pin_memory().to(device, non_blocking=True)
            if dtype is not None and xd.is_floating_point():
                xd = xd.to(dtype)
            yd = y.pin_memory().to(device, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record(stream)
        queue.append((xd, yd, ev))

    for _ in range(depth):
        stage()
    while True:
        xd, yd, ev = queue.pop(0)
        ev.wait()          # consumer stream waits; host does not block
        stage()
        yield xd, yd
