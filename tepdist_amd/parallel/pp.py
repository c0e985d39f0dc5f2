"""Pipeline-parallel execution: a thin shell over the runtime task-list
executor.

The planner's stage cut becomes a TaskDAG (runtime/task_graph.py), the
TaskScheduler orders it (1F1B-flavored, bounded by micro_num_limit), and
runtime/executor.TaskListExecutor WALKS that order — typed task dispatch,
pre-posted recv queue, async sends, gc_plan-driven activation release
(the reference's ExecuteTaskList loop, virtual_client.cc:1662-1810).
Cross-stage tensors move with torch.distributed P2P — RCCL send/recv over
xGMI on the GPU node, gloo in CPU tests; ASYNC_SEND/ASYNC_RECV and
TEPDIST_SYNC_MODE kill-switches fall back to synchronous execution
(SURVEY.md §5.2)."""

from __future__ import annotations

from typing import Callable, List, Optional, Tuple

import torch

from tepdist_amd.runtime.executor import build_stage_executor
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
    """Runs one optimizer step of a staged model over a pipeline group
    (shell over runtime.executor.TaskListExecutor)."""

    def __init__(self, stage_module: torch.nn.Module, stage: int,
                 num_stages: int, pp_ranks: List[int], micro_batches: int,
                 act_shape, act_dtype=torch.bfloat16, device="cpu",
                 reducer=None, order: Optional[List[Tuple[str, int]]] = None,
                 pp_group=None):
        self.mod = stage_module
        self.stage = stage
        self.S = num_stages
        self.M = micro_batches
        self.exec = build_stage_executor(
            stage_module, stage, num_stages, pp_ranks, micro_batches,
            act_shape, act_dtype, device, reducer=reducer,
            pp_group=pp_group)

    def train_step(self, batch_iter: Callable[[int], tuple]) -> float:
        """batch_iter(m) -> (inputs, labels) for micro-batch m. Every stage
        gets the same batch stream; stage 0 consumes inputs, the last stage
        consumes labels. Returns the mean loss (valid on every rank)."""
        return self.exec.run_step(batch_iter)
