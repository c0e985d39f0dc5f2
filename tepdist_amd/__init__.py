"""TePDist-AMD: MI355X-native automatic distributed training system.

A brand-new implementation of the capabilities of alibaba/TePDist (the Rhino
system): client/server auto-parallel planning over a whole-graph tensor IR,
SPMD strategy search (cone partitioning + ILP + inter-subgraph DP), ZeRO-style
variable sharding, micro-batch ("sync-free") gradient accumulation, ILP pipeline
stage cutting, a task-DAG runtime with multi-stream execution, sharded
server-side initialization and sharded checkpointing.

The execution engine is MI355X-first: PyTorch-ROCm host orchestration, one
process per GPU over RCCL/xGMI, and hand-written CDNA4 HIP kernels (MFMA/LDS)
for the compute ops the planner shards.

Reference parity map: see SURVEY.md (reference = alibaba/TePDist at
/root/reference, structure documented per file:line there).
"""

__version__ = "0.1.0"

from tepdist_amd.config import ServiceEnv  # noqa: F401
