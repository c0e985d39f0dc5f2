"""Pipeline-parallel execution engine.

The runtime counterpart of the planner's stage cut: executes the TaskDAG's
per-device scheduled order (runtime/scheduler.py produces a 1F1B-flavored
list bounded by micro_num_limit, like the reference's TaskScheduler +
ExecuteTaskList loop, virtual_client.cc:1662-1810). Cross-stage activation
and gradient tensors move with torch.distributed P2P — RCCL send/recv over
xGMI on the GPU node, gloo in CPU tests — with async sends (the reference
uses dedicated send/recv streams with event barriers; ASYNC_SEND/RECV
kill-switches fall back to synchronous)."""

from __future__ import annotations

from typing import Callable, List, Optional, Tuple

import torch
import torch.distributed as dist

from tepdist_amd.config import get_env
from tepdist_amd.runtime.scheduler import TaskScheduler
from tepdist_amd.runtime.task_graph import TaskType, build_task_dag


def make_1f1b_order(num_stages: int, num_micro: int, stage: int,
                    stage_flops=None) -> List[Tuple[str, int]]:
    """Builds this stage's (fw|bw, micro) order by scheduling the task DAG
    (micro_num_limit = in-flight cap = num_stages - stage, the 1F1B bound)."""
    dag = build_task_dag(num_stages, num_micro,
                         stage_flops=stage_flops or [1.0] * num_stages,
                         act_bytes_per_micro=1.0)
    sched = TaskScheduler(dag, micro_num_limit=num_stages - stage,
                          mem_cap_bytes=float("inf"))
    res = sched.schedule()
    order = []
    for tid in res.order.get(stage, []):
        t = dag.tasks[tid]
        if t.type == TaskType.COMPUTE_FW:
            order.append(("fw", t.split.micro))
        elif t.type == TaskType.COMPUTE_BW:
            order.append(("bw", t.split.micro))
    return order


class PipelineEngine:
    """Runs one optimizer step of a staged model over a pipeline group."""

    def __init__(self, stage_module: torch.nn.Module, stage: int,
                 num_stages: int, pp_ranks: List[int], micro_batches: int,
                 act_shape, act_dtype=torch.bfloat16, device="cpu",
                 reducer=None, order: Optional[List[Tuple[str, int]]] = None,
                 pp_group=None):
        self.pp_group = pp_group
        self.mod = stage_module
        self.stage = stage
        self.S = num_stages
        self.ranks = pp_ranks              # global rank per stage
        self.M = micro_batches
        self.act_shape = tuple(act_shape)  # per-micro activation shape
        self.act_dtype = act_dtype
        self.device = device
        self.reducer = reducer
        self.order = order or make_1f1b_order(num_stages, micro_batches,
                                              stage)
        self.async_send = get_env().async_send
        self.async_recv = get_env().async_recv
        self._posted = None   # (key, work, tensor): one pre-posted irecv

    # -- p2p ----------------------------------------------------------------

    def _send(self, tensor: torch.Tensor, to_stage: int, pending: list):
        t = tensor.contiguous()
        if self.async_send:
            pending.append((dist.isend(t, self.ranks[to_stage]), t))
        else:
            dist.send(t, self.ranks[to_stage])

    def _recv_key(self, entry):
        """(from_stage,) the entry will receive from, or None."""
        kind, m = entry
        if kind == "fw" and self.stage > 0:
            return ("fw", m, self.stage - 1)
        if kind == "bw" and self.stage < self.S - 1:
            return ("bw", m, self.stage + 1)
        return None

    def _prepost(self, entry):
        """ASYNC_RECV: post the next entry's irecv so the transfer runs
        under the current entry's compute (the reference's dedicated recv
        stream; kill-switch falls back to blocking recv)."""
        key = self._recv_key(entry)
        if key is None or self._posted is not None:
            return
        t = torch.empty(self.act_shape, dtype=self.act_dtype,
                        device=self.device)
        self._posted = (key, dist.irecv(t, self.ranks[key[2]]), t)

    def _recv(self, from_stage: int, key=None) -> torch.Tensor:
        if self._posted is not None and self._posted[0] == key:
            _, work, t = self._posted
            self._posted = None
            work.wait()
            return t
        t = torch.empty(self.act_shape, dtype=self.act_dtype,
                        device=self.device)
        dist.recv(t, self.ranks[from_stage])
        return t

    # -- step ---------------------------------------------------------------

    def train_step(self, batch_iter: Callable[[int], tuple]) -> float:
        """batch_iter(m) -> (inputs, labels) for micro-batch m. Every stage
        gets the same batch stream; stage 0 consumes inputs, the last stage
        consumes labels. Returns the mean loss (valid on every rank)."""
        is_first = self.stage == 0
        is_last = self.stage == self.S - 1
        fw_in = {}    # micro -> input tensor (requires_grad for bw)
        fw_out = {}   # micro -> output tensor
        pending = []
        total_loss = torch.zeros((), device=self.device)

        if self.reducer is not None:
            self.reducer.reset()
        bw_done = 0
        for oi, (kind, m) in enumerate(self.order):
            if kind == "fw":
                inputs, labels = batch_iter(m)
                if is_first:
                    x = inputs
                else:
                    x = self._recv(self.stage - 1,
                                   ("fw", m, self.stage - 1)).requires_grad_()
                fw_in[m] = x
                if is_last:
                    loss = self.mod(x, labels=labels)
                    fw_out[m] = loss
                    total_loss = total_loss + loss.detach()
                else:
                    y = self.mod(x)
                    fw_out[m] = y
                    self._send(y.detach(), self.stage + 1, pending)
            else:  # bw
                bw_done += 1
                if self.reducer is not None and bw_done == self.M:
                    self.reducer.arm()
                if is_last:
                    (fw_out.pop(m) / self.M).backward()
                else:
                    grad = self._recv(self.stage + 1, ("bw", m, self.stage + 1))
                    fw_out.pop(m).backward(grad)
                if not is_first:
                    g = fw_in[m].grad
                    self._send(g, self.stage - 1, pending)
                fw_in.pop(m)
            if self.async_recv and oi + 1 < len(self.order):
                self._prepost(self.order[oi + 1])
        for w, _ in pending:
            w.wait()
        if self.reducer is not None:
            self.reducer.finalize()
        # everyone learns the loss (reference returns the result literal to
        # the client from the merge task)
        total_loss = total_loss / self.M
        dist.broadcast(total_loss, self.ranks[-1], group=self.pp_group)
        return total_loss.item()
