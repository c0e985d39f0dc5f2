"""Training-step driver: micro-batch gradient accumulation + DP all-reduce.

This is the executed counterpart of the planner's SyncFree decomposition
(CG per micro-batch -> GA accumulate -> AR all-reduce -> AG
apply): each micro-batch runs forward+backward (CG), grads accumulate in the
autograd .grad buffers (GA), a bucketed RCCL all-reduce averages them across
data-parallel ranks (AR, overlapped with backward via the GradReducer), and
the fused AdamW applies them (AG).
"""

from __future__ import annotations

import time
from typing import Callable, Optional

import torch

from tepdist_amd.config import get_env

from tepdist_amd.train.optim import AdamW
from tepdist_amd.parallel.dp import GradReducer


class Trainer:
    def __init__(self, model: torch.nn.Module, optimizer: AdamW,
                 grad_accum_steps: int = 1,
                 reducer: Optional[GradReducer] = None):
        self.model = model
        self.opt = optimizer
        self.grad_accum_steps = grad_accum_steps
        self.reducer = reducer

    def train_step(self, batch_iter: Callable[[int], tuple]) -> float:
        """Runs one optimizer step = grad_accum_steps micro-batches.
        batch_iter(i) returns (input_ids, labels) for micro-batch i.
        Returns the mean loss (host float)."""
        debug = get_env().debug
        t0 = time.perf_counter() if debug else 0.0
        self.opt.zero_grad()
        if self.reducer is not None:
            self.reducer.reset()
        total = 0.0
        n = self.grad_accum_steps
        sync = get_env().sync_mode  # blocking-sync debug
        from tepdist_amd.utils.tracing import trace_span
        for i in range(n):
            with trace_span(f"micro_batch/{i}"):
                input_ids, labels = batch_iter(i)
                loss = self.model(input_ids, labels=labels)
                scaled = loss / n
                if self.reducer is not None and i == n - 1:
                    # overlap the all-reduce of each bucket with the
                    # remaining backward of the LAST micro-batch only
                    # (earlier micros accumulate locally - sync-free)
                    self.reducer.arm()
                scaled.backward()
                total += loss.item()
            if sync and torch.cuda.is_available():
                torch.cuda.synchronize()
        if self.reducer is not None:
            self.reducer.finalize()
        self.opt.step()
        if debug:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            print(f"[ExecutePlan Duration] "
                  f"{(time.perf_counter() - t0) * 1e3:.2f} ms "
                  f"loss={total / n:.4f}", flush=True)
        return total / n
