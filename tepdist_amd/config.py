"""Typed env/config table.

Plays the role of the reference's ServiceEnv declarative env-var table
(tf_tepdist/tensorflow/compiler/xla/service/service_env.h:46-108): one place
declaring every tunable with name, env var, type and default, plus optional
config-file override, and a PrintAllEnvs-style dump at startup.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field, fields
from typing import Any, Optional


def _env(name: str, typ, default):
    raw = os.environ.get(name)
    if raw is None:
        return default
    if typ is bool:
        return raw.lower() in ("1", "true", "yes", "on")
    if typ is int:
        return int(raw)
    if typ is float:
        return float(raw)
    return raw


@dataclass
class ServiceEnv:
    """All server-side tunables. Mirrors the reference env table where the
    concept carries over; MI355X-specific knobs are new."""

    # --- debugging / logging ---
    debug: bool = field(default_factory=lambda: _env("TEPDIST_DEBUG", bool, False))
    dump_dir: str = field(default_factory=lambda: _env("TEPDIST_DUMP_DIR", str, ""))

    # --- planner mode (reference: auto_parallel.cc:395-409) ---
    rule_mode: bool = field(default_factory=lambda: _env("RULE_MODE", bool, False))
    ignore_annotation: bool = field(default_factory=lambda: _env("IGNORE_ANNOTATION", bool, False))
    aux_affinity: bool = field(default_factory=lambda: _env("AUX_AFFINITY", bool, True))
    opt_level: int = field(default_factory=lambda: _env("OPT_LEVEL", int, 3))

    # --- pipeline / micro-batching (reference: service_env.h) ---
    num_stages: int = field(default_factory=lambda: _env("NUM_STAGES", int, 0))  # 0 = auto
    num_micro_batches: int = field(default_factory=lambda: _env("NUM_MICRO_BATCHES", int, 0))
    micro_num_limit: int = field(default_factory=lambda: _env("MICRO_NUM_LIMIT", int, 0))
    unbalanced_ratio: int = field(default_factory=lambda: _env("UNBALANCED_RATIO", int, 8))
    pp_bandwidth_gbps: float = field(default_factory=lambda: _env("PP_BANDWIDTH", float, 153.0))

    # --- ILP (reference: ILP_TIME_LIMIT default 5s) ---
    ilp_time_limit_s: float = field(default_factory=lambda: _env("ILP_TIME_LIMIT", float, 5.0))

    # --- ZeRO-style variable sharding (reference: VAR_MEM_LIMIT 20GB) ---
    var_mem_limit_bytes: int = field(
        default_factory=lambda: _env("VAR_MEM_LIMIT", int, 96 * (1 << 30))
    )

    # --- scheduling ---
    group_sched_count: int = field(default_factory=lambda: _env("GROUP_SCHED_COUNT", int, 2))
    async_send: bool = field(default_factory=lambda: _env("ASYNC_SEND", bool, True))
    async_recv: bool = field(default_factory=lambda: _env("ASYNC_RECV", bool, True))

    # --- runtime / numerics ---
    fake_input: bool = field(default_factory=lambda: _env("FAKE_INPUT", bool, False))
    disable_buffer_alias: bool = field(
        default_factory=lambda: _env("DISABLE_BUFFER_ALIAS", bool, False)
    )
    fp16_comm: bool = field(default_factory=lambda: _env("FP16_COMM", bool, False))
    # Blocking-sync debug mode: hipStreamSynchronize after every task
    # (SURVEY.md §5.2 recommends keeping such a kill-switch).
    sync_mode: bool = field(default_factory=lambda: _env("TEPDIST_SYNC_MODE", bool, False))
    hip_graph: bool = field(default_factory=lambda: _env("TEPDIST_HIP_GRAPH", bool, True))

    # --- cluster ---
    cluster_spec: str = field(default_factory=lambda: _env("CLUSTER_SPEC", str, ""))
    server_ip: str = field(default_factory=lambda: _env("SERVER_IP", str, "127.0.0.1"))
    server_port: int = field(default_factory=lambda: _env("SERVER_PORT", int, 2222))

    # --- MI355X hardware model (planner cost constants; replaces the
    #     reference's V100 numbers in evaluator.h:48-57). Values refined from
    #     measurements as they land (profiles/). ---
    gpu_bf16_tflops: float = field(default_factory=lambda: _env("TEPDIST_GPU_TFLOPS", float, 1300.0))
    hbm_bytes: int = field(default_factory=lambda: _env("TEPDIST_HBM_BYTES", int, 288 * (1 << 30)))
    hbm_bw_gbps: float = field(default_factory=lambda: _env("TEPDIST_HBM_BW", float, 6300.0))
    xgmi_link_gbps: float = field(default_factory=lambda: _env("TEPDIST_XGMI_LINK_BW", float, 153.0))
    xgmi_links: int = field(default_factory=lambda: _env("TEPDIST_XGMI_LINKS", int, 7))
    internode_gbps: float = field(default_factory=lambda: _env("TEPDIST_INTERNODE_BW", float, 25.0))

    # --- kernels ---
    # 'hip' = our CDNA4 kernels (default on GPU); 'torch' = rocBLAS/eager
    # fallback for A/B comparison only.
    gemm_backend: str = field(default_factory=lambda: _env("TEPDIST_GEMM", str, "hip"))

    @classmethod
    def from_file(cls, path: str) -> "ServiceEnv":
        cfg = cls()
        with open(path) as f:
            data = json.load(f)
        for k, v in data.items():
            if hasattr(cfg, k):
                setattr(cfg, k, v)
        return cfg

    def dump(self) -> str:
        lines = ["[ServiceEnv] effective configuration:"]
        for f in fields(self):
            lines.append(f"  {f.name} = {getattr(self, f.name)}")
        return "\n".join(lines)


_GLOBAL_ENV: Optional[ServiceEnv] = None


def get_env() -> ServiceEnv:
    global _GLOBAL_ENV
    if _GLOBAL_ENV is None:
        _GLOBAL_ENV = ServiceEnv()
    return _GLOBAL_ENV


def set_env(env: ServiceEnv) -> None:
    global _GLOBAL_ENV
    _GLOBAL_ENV = env


def reset_env() -> None:
    """Drops the cached ServiceEnv so the next get_env() re-reads the
    process environment (tests toggling kill-switches)."""
    global _GLOBAL_ENV
    _GLOBAL_ENV = None
