"""Analytic cost model with MI355X constants.

Replaces the reference Evaluator/PerfUtils V100 constants
(service/parallel/evaluator.h:48-57: 15 TFLOPS, 32 GB, intra 300 GB/s,
inter 3.125 GB/s) with MI355X numbers measured in this repo (profiles/):
bf16 MFMA-kernel effective throughput, 288 GB HBM3E, xGMI 7 p2p links of
~153 GB/s per GPU. Collective formulas model RING algorithms, which on the
fully-connected xGMI mesh are bound by ONE link's bandwidth per step —
hence per-link bandwidth, not aggregate."""

from __future__ import annotations

from dataclasses import dataclass

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import COMPUTE_SENSITIVE, Graph, Node


@dataclass
class HardwareModel:
    # effective sustained rates (measured, not peak; see profiles/)
    # measured (profiles/, benchmarks/shapes_bench.py): the 256-tile GEMM
    # sustains 750-930 TF/s at the GPT-2 K=1024 shapes and ~1.05 PF/s at
    # 8k^3; 780 is the training-mix average the evaluator should price with
    bf16_tflops: float = 780.0
    hbm_gbps: float = 6300.0
    hbm_bytes: int = 288 << 30
    xgmi_link_gbps: float = 153.0       # per p2p link
    xgmi_links: int = 7
    internode_gbps: float = 25.0
    kernel_launch_us: float = 8.0

    @staticmethod
    def from_env() -> "HardwareModel":
        e = get_env()
        return HardwareModel(
            bf16_tflops=min(e.gpu_bf16_tflops, 780.0),
            hbm_gbps=e.hbm_bw_gbps, hbm_bytes=e.hbm_bytes,
            xgmi_link_gbps=e.xgmi_link_gbps, xgmi_links=e.xgmi_links,
            internode_gbps=e.internode_gbps)


@dataclass
class Cost:
    """Mirrors the reference Evaluator output (evaluator.h:27-44)."""
    total_duration: float = 0.0       # seconds per iteration
    gpu_efficiency: float = 0.0
    coll_ratio: float = 0.0
    bubble_ratio: float = 0.0
    mem_bytes: float = 0.0            # peak per-device state+activation

    def __lt__(self, other):
        return self.total_duration < other.total_duration


class CostModel:
    def __init__(self, hw: HardwareModel = None):
        self.hw = hw or HardwareModel.from_env()

    # -- compute ------------------------------------------------------------

    def compute_time(self, g: Graph, n: Node, shards: int = 1,
                     training: bool = True) -> float:
        """Seconds for one (sharded) execution of the op, fw+bw."""
        fl = g.flops(n) * (3.0 if training else 1.0) / max(shards, 1)
        by = g.bytes_of(n) * (3.0 if training else 1.0) / max(shards, 1)
        if n.op in COMPUTE_SENSITIVE:
            t = fl / (self.hw.bf16_tflops * 1e12)
        else:
            t = by * 3.0 / (self.hw.hbm_gbps * 1e9)  # r+w+grad traffic
        return t + self.hw.kernel_launch_us * 1e-6

    # -- collectives (reference PerfUtils AllReduceCost etc.,
    #    performance_utils.h:25-35, re-fit to xGMI rings) -------------------

    def _bw(self, nd: int, internode: bool) -> float:
        return (self.hw.internode_gbps if internode
                else self.hw.xgmi_link_gbps) * 1e9

    def all_reduce(self, nbytes: float, nd: int, internode=False) -> float:
        if nd <= 1:
            return 0.0
        return 2.0 * (nd - 1) / nd * nbytes / self._bw(nd, internode) + 20e-6

    def all_gather(self, nbytes: float, nd: int, internode=False) -> float:
        if nd <= 1:
            return 0.0
        return (nd - 1) / nd * nbytes / self._bw(nd, internode) + 15e-6

    reduce_scatter = all_gather

    def all_to_all(self, nbytes: float, nd: int, internode=False) -> float:
        if nd <= 1:
            return 0.0
        # fully-connected xGMI: each pair exchanges directly over its link
        return (nd - 1) / nd * nbytes / self._bw(nd, internode) + 15e-6

    def p2p(self, nbytes: float, internode=False) -> float:
        return nbytes / self._bw(2, internode) + 10e-6

    def collective(self, kind: str, nbytes: float, nd: int,
                   internode=False) -> float:
        if kind is None:
            return 0.0
        if kind == "all_reduce":
            return self.all_reduce(nbytes, nd, internode)
        if kind in ("all_gather", "reduce_scatter"):
            return self.all_gather(nbytes, nd, internode)
        if kind == "all_to_all":
            return self.all_to_all(nbytes, nd, internode)
        if kind == "dynamic_slice":
            return 5e-6
        return self.p2p(nbytes, internode)


class Evaluator:
    """Whole-plan cost: compute + collectives + pipeline bubble + memory
    check (the reference Evaluator)."""

    def __init__(self, cm: CostModel = None):
        self.cm = cm or CostModel()

    def run(self, compute_s: float, coll_s: float, stages: int,
            micro_batches: int, mem_bytes_per_dev: float) -> Cost:
        mb = max(micro_batches, 1)
        bubble = (stages - 1) / (mb + stages - 1) if stages > 1 else 0.0
        work = compute_s + coll_s
        total = work / (1.0 - bubble) if bubble < 1 else float("inf")
        if mem_bytes_per_dev > self.cm.hw.hbm_bytes:
            total = float("inf")  # infeasible plan
        eff = compute_s / total if total > 0 and total != float("inf") else 0
        return Cost(total_duration=total, gpu_efficiency=eff,
                    coll_ratio=coll_s / max(work, 1e-12),
                    bubble_ratio=bubble, mem_bytes=mem_bytes_per_dev)
