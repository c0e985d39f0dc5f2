"""AdamW with fp32 master weights over bf16 model params.

The per-tensor update is the fused HIP kernel (ops.adamw_step) on GPU and
the torch reference on CPU. Plays the role of the reference's server-side
apply-gradients (AG) module (SURVEY.md §2.3 SyncFreeDecomposition: the AG
DefContext runs the optimizer on the server).
"""

from __future__ import annotations

from typing import Iterable, List

import torch

from tepdist_amd import ops


class AdamW:
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 1e-4,
                 betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.01,
                 no_decay_1d: bool = True):
        self.params: List[torch.nn.Parameter] = [p for p in params if p.requires_grad]
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.no_decay_1d = no_decay_1d
        self.step_count = 0
        self.state = {}
        for p in self.params:
            self.state[p] = {
                "master": p.detach().float().clone(),
                "exp_avg": torch.zeros_like(p, dtype=torch.float32),
                "exp_avg_sq": torch.zeros_like(p, dtype=torch.float32),
            }

    def _try_mt(self):
        """Multi-tensor fused path (single kernel launch) on GPU with bf16
        grads for every param."""
        import torch as _t
        ps = [p for p in self.params if p.grad is not None]

        def _ok(p):
            # bf16 param+grad or fp32 param+grad (Wide-ResNet batch-norm
            # affines), contiguous — mixed sets run in ONE mt launch via
            # the per-tensor ptypes table
            return p.dtype in (_t.bfloat16, _t.float32) and \
                p.grad.dtype == p.dtype and p.grad.is_contiguous()

        if len(ps) < 4 or not ps[0].is_cuda or not all(_ok(p) for p in ps):
            return None
        from tepdist_amd.ops import hip as be
        key = tuple(id(p) for p in ps)
        if getattr(self, "_mt_key", None) != key:
            self._mt_key = key
            wds = [0.0 if (self.no_decay_1d and p.dim() == 1)
                   else self.weight_decay for p in ps]
            self._mt = be.AdamWMT(ps, [self.state[p] for p in ps], wds)
        return ps

    # -- hipGraph capture support (trainer.py): the MT kernel switches to
    # a device hyper-parameter buffer so the captured launch stays valid
    # as the step count advances -----------------------------------------

    def prepare_graph(self):
        ps = self._try_mt()
        if ps is None:
            raise RuntimeError("graph capture needs the multi-tensor path")
        self._graph_ps = ps
        self._mt.prepare_graph([p.grad for p in ps])
        self.graph_mode = True

    def refresh_hyper(self):
        self.step_count += 1
        self._mt.set_hyper(self.lr, self.beta1, self.beta2, self.step_count)

    @torch.no_grad()
    def step(self):
        if getattr(self, "graph_mode", False):
            # inside capture (or replayed): hyper buffer carries lr/bc
            self._mt.step_graph(self.beta1, self.beta2, self.eps)
            return
        self.step_count += 1
        ps = self._try_mt()
        if ps is not None:
            self._mt.step([p.grad for p in ps], self.lr, self.beta1,
                          self.beta2, self.eps, self.step_count)
            return
        for p in self.params:
            if p.grad is None:
                continue
            st = self.state[p]
            wd = 0.0 if (self.no_decay_1d and p.dim() == 1) else self.weight_decay
            ops.adamw_step(p.data, st["master"], p.grad, st["exp_avg"],
                           st["exp_avg_sq"], lr=self.lr, beta1=self.beta1,
                           beta2=self.beta2, eps=self.eps, weight_decay=wd,
                           step=self.step_count)

    def zero_grad(self):
        for p in self.params:
            p.grad = None

    def state_dict(self):
        return {
            "step": self.step_count,
            "state": [
                {k: v for k, v in self.state[p].items()} for p in self.params
            ],
        }

    def load_state_dict(self, sd):
        self.step_count = sd["step"]
        for p, st in zip(self.params, sd["state"]):
            for k in ("master", "exp_avg", "exp_avg_sq"):
                self.state[p][k].copy_(st[k])


class ZeroAdamW(AdamW):
    """ZeRO-1: optimizer states (fp32 master + Adam moments) sharded across
    the data-parallel group — the executor side of the planner's
    SplitPlanByMemCost decision (planner/zero.py; reference
    cost_spmd_strategy.cc:1487 under VAR_MEM_LIMIT). Gradients arrive
    full (bucketed all-reduce); each rank updates only the parameters it
    owns and broadcasts the refreshed values."""

    def __init__(self, params, group=None, **kw):
        import torch.distributed as dist
        self.group = group
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        self.world = dist.get_world_size(group) if dist.is_initialized() else 1
        plist = [p for p in params if p.requires_grad]
        # greedy balanced partition by numel
        loads = [0] * self.world
        self.owner = {}
        for p in sorted(plist, key=lambda q: -q.numel()):
            o = loads.index(min(loads))
            self.owner[id(p)] = o
            loads[o] += p.numel()
        self._owned = [p for p in plist if self.owner[id(p)] == self.rank]
        super().__init__(self._owned, **kw)
        self.all_params = plist

    @torch.no_grad()
    def step(self):
        import torch.distributed as dist
        super().step()  # update owned shard only
        if self.world == 1:
            return
        ranks = dist.get_process_group_ranks(self.group) if self.group \
            is not None else list(range(dist.get_world_size()))
        works = []
        for p in self.all_params:
            src = ranks[self.owner[id(p)]]
            works.append(dist.broadcast(p.data, src, group=self.group,
                                        async_op=True))
        for w in works:
            w.wait()

    def zero_grad(self):
        for p in self.all_params:
            p.grad = None

    def state_bytes(self) -> int:
        return sum(p.numel() * 12 for p in self._owned)
