"""Runtime task DAG.

Mirrors the reference's TaskDAG/TaskNode design (pjrt/task_graph.h:40-399,
SURVEY.md §2.6): the compiled plan becomes a graph of typed tasks — per
(micro-batch, stage) forward/backward Compute tasks, Input/Output, Send/Recv
pairs on cross-stage edges (inserted like SourceCalibration/
CrossDeviceCalibration, execution_plan.cc:409-560), GAInit/GA accumulation
and the AG optimizer task — each addressed by a SplitId (micro ordinal x
stage x spmd shard) and carrying schedule order + lifetime info. The
executor (pipeline_exec.py) walks each device's scheduled task list; the
scheduler (scheduler.py) simulates it for cost and memory."""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Dict, List, Optional


class TaskType(Enum):
    SPLIT = "split"      # source
    INPUT = "input"
    COMPUTE_FW = "fw"
    COMPUTE_BW = "bw"
    SEND = "send"
    RECV = "recv"
    GA_INIT = "ga_init"
    GA = "ga"
    AR = "ar"            # gradient all-reduce (DP)
    AG = "apply"         # optimizer
    OUTPUT = "output"
    MERGE = "merge"      # sink


@dataclass
class SplitId:
    """Multi-dim split address (dev_id_util.h:94-189): micro ordinal,
    pipeline stage, spmd shard index."""
    micro: int = 0
    stage: int = 0
    shard: int = 0


@dataclass
class TaskNode:
    id: int
    type: TaskType
    split: SplitId
    device: int = 0                  # stage-group device (pp rank)
    def_id: int = -1                 # DefContext this task executes
    parents: List[int] = field(default_factory=list)
    children: List[int] = field(default_factory=list)
    sched_idx: int = -1              # order within its device's list
    flops: float = 0.0
    out_bytes: float = 0.0           # bytes produced (held until released)
    release_at: Optional[int] = None  # task id whose completion frees out
    peer: Optional[int] = None       # for SEND/RECV: the matching task

    @property
    def key(self) -> str:
        return f"{self.type.value}_m{self.split.micro}_s{self.split.stage}"


class TaskDAG:
    def __init__(self):
        self.tasks: Dict[int, TaskNode] = {}
        self._next = 0
        self.source: Optional[int] = None
        self.sink: Optional[int] = None

    def new(self, type_: TaskType, micro=0, stage=0, shard=0, device=0,
            def_id=-1, flops=0.0, out_bytes=0.0) -> TaskNode:
        t = TaskNode(self._next, type_, SplitId(micro, stage, shard), device,
                     def_id, flops=flops, out_bytes=out_bytes)
        self.tasks[t.id] = t
        self._next += 1
        return t

    def edge(self, a: TaskNode, b: TaskNode):
        if b.id not in a.children:
            a.children.append(b.id)
            b.parents.append(a.id)

    def by_device(self) -> Dict[int, List[TaskNode]]:
        out: Dict[int, List[TaskNode]] = {}
        for t in self.tasks.values():
            out.setdefault(t.device, []).append(t)
        return out

    def topo(self) -> List[TaskNode]:
        indeg = {i: len(t.parents) for i, t in self.tasks.items()}
        ready = [i for i, d in indeg.items() if d == 0]
        order = []
        while ready:
            i = ready.pop()
            order.append(self.tasks[i])
            for c in self.tasks[i].children:
                indeg[c] -= 1
                if indeg[c] == 0:
                    ready.append(c)
        assert len(order) == len(self.tasks), "cycle in TaskDAG"
        return order

    # -- wire form (the reference serializes TaskNodes as ComputeTask
    # protos for DispatchPlan, xla.proto:491-508) -------------------------

    def to_wire(self) -> dict:
        """JSON-able form of the full DAG: one record per task with type,
        split address, device, def id, schedule index, edges and peer."""
        return {
            "source": self.source, "sink": self.sink,
            "tasks": [{
                "id": t.id, "type": t.type.value,
                "micro": t.split.micro, "stage": t.split.stage,
                "shard": t.split.shard, "device": t.device,
                "def_id": t.def_id, "sched_idx": t.sched_idx,
                "parents": list(t.parents), "children": list(t.children),
                "flops": t.flops, "out_bytes": t.out_bytes,
                "release_at": t.release_at, "peer": t.peer,
            } for t in self.tasks.values()],
        }

    @staticmethod
    def from_wire(d: dict) -> "TaskDAG":
        dag = TaskDAG()
        for r in d["tasks"]:
            t = TaskNode(r["id"], TaskType(r["type"]),
                         SplitId(r["micro"], r["stage"], r["shard"]),
                         r["device"], r["def_id"],
                         list(r["parents"]), list(r["children"]),
                         r["sched_idx"], r["flops"], r["out_bytes"],
                         r["release_at"], r["peer"])
            dag.tasks[t.id] = t
            dag._next = max(dag._next, t.id + 1)
        dag.source = d.get("source")
        dag.sink = d.get("sink")
        return dag

    def dump_dot(self, path: str):
        """Graphviz dump (the reference's TaskDAG::Dump dag.dot,
        task_graph.cc:377)."""
        colors = {TaskType.COMPUTE_FW: "lightblue",
                  TaskType.COMPUTE_BW: "lightsalmon",
                  TaskType.SEND: "gold", TaskType.RECV: "khaki",
                  TaskType.GA: "palegreen", TaskType.AG: "orchid"}
        with open(path, "w") as f:
            f.write("digraph tasks {\nrankdir=LR;\n")
            for t in self.tasks.values():
                c = colors.get(t.type, "white")
                f.write(f'  t{t.id} [label="{t.key}\\ndev{t.device}" '
                        f'style=filled fillcolor={c}];\n')
            for t in self.tasks.values():
                for ch in t.children:
                    f.write(f"  t{t.id} -> t{ch};\n")
            f.write("}\n")


def build_task_dag(num_stages: int, num_micro: int, stage_flops=None,
                   act_bytes_per_micro: float = 0.0,
                   grad_bytes_per_stage: float = 0.0,
                   dp_degree: int = 1) -> TaskDAG:
    """Builds the iteration TaskDAG for a pipeline x micro-batch plan
    (the reference's CompileTaskDAG, virtual_client.cc:601-773; device here
    = pipeline stage rank; the dp/tp shards run the same per-device list)."""
    dag = TaskDAG()
    sf = stage_flops or [1.0] * num_stages
    src = dag.new(TaskType.SPLIT)
    dag.source = src.id
    ga_init = dag.new(TaskType.GA_INIT)
    dag.edge(src, ga_init)

    fw: Dict[tuple, TaskNode] = {}
    bw: Dict[tuple, TaskNode] = {}
    for m in range(num_micro):
        inp = dag.new(TaskType.INPUT, micro=m, stage=0, device=0)
        dag.edge(src, inp)
        prev = inp
        for s in range(num_stages):
            t = dag.new(TaskType.COMPUTE_FW, micro=m, stage=s, device=s,
                        flops=sf[s], out_bytes=act_bytes_per_micro)
            fw[(m, s)] = t
            if s == 0:
                dag.edge(prev, t)
            else:
                snd = dag.new(TaskType.SEND, micro=m, stage=s - 1,
                              device=s - 1, out_bytes=act_bytes_per_micro)
                rcv = dag.new(TaskType.RECV, micro=m, stage=s, device=s)
                snd.peer, rcv.peer = rcv.id, snd.id
                dag.edge(fw[(m, s - 1)], snd)
                dag.edge(snd, rcv)
                dag.edge(rcv, t)
        for s in reversed(range(num_stages)):
            t = dag.new(TaskType.COMPUTE_BW, micro=m, stage=s, device=s,
                        flops=2.0 * sf[s])
            bw[(m, s)] = t
            if s == num_stages - 1:
                dag.edge(fw[(m, s)], t)
            else:
                snd = dag.new(TaskType.SEND, micro=m, stage=s + 1,
                              device=s + 1, out_bytes=act_bytes_per_micro)
                rcv = dag.new(TaskType.RECV, micro=m, stage=s, device=s)
                snd.peer, rcv.peer = rcv.id, snd.id
                dag.edge(bw[(m, s + 1)], snd)
                dag.edge(snd, rcv)
                dag.edge(rcv, t)
                dag.edge(fw[(m, s)], t)
            # activation of fw(m,s) is freed when bw(m,s) completes
            fw[(m, s)].release_at = t.id

    sink = dag.new(TaskType.MERGE)
    dag.sink = sink.id
    for s in range(num_stages):
        ga = dag.new(TaskType.GA, stage=s, device=s,
                     out_bytes=grad_bytes_per_stage)
        dag.edge(ga_init, ga)
        for m in range(num_micro):
            dag.edge(bw[(m, s)], ga)
        last = ga
        if dp_degree > 1:
            ar = dag.new(TaskType.AR, stage=s, device=s,
                         out_bytes=grad_bytes_per_stage)
            dag.edge(ga, ar)
            last = ar
        ag = dag.new(TaskType.AG, stage=s, device=s)
        dag.edge(last, ag)
        out = dag.new(TaskType.OUTPUT, stage=s, device=s)
        dag.edge(ag, out)
        dag.edge(out, sink)
    return dag


def idoms(dag: "TaskDAG") -> dict:
    """Immediate dominators of every task (virtual root = -1): the
    reference's TaskDAG dominance tree (Cooper-Harvey-Kennedy,
    task_graph.h:643-717), used by the GC planner. Native C++ core with a
    Python fallback."""
    ids = sorted(dag.tasks)
    idx = {tid: i for i, tid in enumerate(ids)}
    p_off, p_ids = [0], []
    for tid in ids:
        p_ids.extend(idx[p] for p in dag.tasks[tid].parents)
        p_off.append(len(p_ids))
    try:
        from tepdist_amd.runtime import _tepdist_rt as _rt
        dom = _rt.idom_tree(p_off, p_ids)
        return {ids[i]: (ids[d] if d >= 0 else -1)
                for i, d in enumerate(dom)}
    except ImportError:
        pass
    # Python fallback: in a DAG processed in topo order every predecessor
    # is finalized first, so one Cooper-style pass suffices
    order = [idx[t.id] for t in dag.topo()]
    rpo = {u: i for i, u in enumerate(order)}
    idom = {}

    def intersect(a, b):
        while a != b:
            if a == -1 or b == -1:
                return -1
            while a != -1 and rpo[a] > rpo[b]:
                a = idom[a]
            while b != -1 and a != -1 and rpo[b] > rpo[a]:
                b = idom[b]
            if a == -1 or b == -1:
                return -1
        return a

    for u in order:
        preds = [idx[p] for p in dag.tasks[ids[u]].parents]
        if not preds:
            idom[u] = -1
            continue
        nd = preds[0]
        for p in preds[1:]:
            nd = intersect(nd, p)
        idom[u] = nd
    return {ids[u]: (ids[d] if d != -1 else -1)
            for u, d in idom.items()}
