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
                 reducer: Optional[GradReducer] = None,
                 hip_graph: Optional[bool] = None):
        self.model = model
        self.opt = optimizer
        self.grad_accum_steps = grad_accum_steps
        self.reducer = reducer
        # hipGraph step capture: after one eager warm-up step (gradients +
        # MT-optimizer tables exist; graph allocations use their own pool)
        # the whole step — grad zero, every
        # micro-batch forward+backward, the bucketed all-reduce, the fused
        # optimizer — is captured once and replayed per step, removing the
        # ~2.5k per-step kernel-launch round trips. Kill-switch
        # TEPDIST_HIP_GRAPH=0; any capture failure falls back to eager.
        if hip_graph is None:
            import os as _os

            import torch.distributed as _dist
            multi = _dist.is_initialized() and _dist.get_world_size() > 1
            if multi and "TEPDIST_HIP_GRAPH" not in _os.environ:
                # multi-rank default: eager. Capturing RCCL collectives
                # inside a hipGraph is unverified on this pool (1-GPU
                # leases); a capture segfault would kill the whole job.
                # TEPDIST_HIP_GRAPH=1 opts in explicitly.
                hip_graph = False
            else:
                hip_graph = get_env().hip_graph
        self.hip_graph = hip_graph
        self._graph = None
        self._static = None
        self._graph_losses = None
        self._eager_steps = 0

    # -- hipGraph capture/replay ------------------------------------------

    def _step_body(self, batches):
        n = self.grad_accum_steps
        torch._foreach_zero_([p.grad for p in self.opt.params
                              if p.grad is not None])
        if self.reducer is not None:
            self.reducer.reset()
        losses = []
        for i, (input_ids, labels) in enumerate(batches):
            loss = self.model(input_ids, labels=labels)
            if self.reducer is not None and i == n - 1:
                self.reducer.arm()
            (loss / n).backward()
            losses.append(loss.detach())
        if self.reducer is not None:
            self.reducer.finalize()
        self.opt.step()
        return losses

    def _try_capture(self, batches):
        self._static = [(i.clone(), l.clone()) for i, l in batches]
        self.opt.prepare_graph()
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._graph_losses = self._step_body(self._static)
        self._graph = g

    def _replay(self, batch_iter) -> float:
        for i in range(self.grad_accum_steps):
            ids, lab = batch_iter(i)
            self._static[i][0].copy_(ids, non_blocking=True)
            self._static[i][1].copy_(lab, non_blocking=True)
        self.opt.refresh_hyper()
        self._graph.replay()
        return sum(l.item() for l in self._graph_losses) / \
            self.grad_accum_steps

    def train_step(self, batch_iter: Callable[[int], tuple]) -> float:
        """Runs one optimizer step = grad_accum_steps micro-batches.
        batch_iter(i) returns (input_ids, labels) for micro-batch i.
        Returns the mean loss (host float)."""
        debug = get_env().debug
        t0 = time.perf_counter() if debug else 0.0
        if self._graph is not None:
            loss = self._replay(batch_iter)
            if debug:
                torch.cuda.synchronize()
                print(f"[ExecutePlan Duration] "
                      f"{(time.perf_counter() - t0) * 1e3:.2f} ms "
                      f"loss={loss:.4f} (graph)", flush=True)
            return loss
        if (self.hip_graph and torch.cuda.is_available()
                and self._eager_steps >= 1):
            batches = [batch_iter(i) for i in range(self.grad_accum_steps)]
            try:
                self._try_capture(batches)
                return self._replay(lambda i: batches[i])
            except Exception as e:  # fall back to eager for good
                import warnings
                warnings.warn(f"hipGraph capture failed, running eager: {e}")
                self.hip_graph = False
                self._graph = None
                if getattr(self.opt, "graph_mode", False):
                    self.opt.graph_mode = False
        self._eager_steps += 1
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
