from tepdist_amd.parallel.dp import GradReducer, init_distributed  # noqa: F401
from tepdist_amd.parallel.ring_attention import ring_attention  # noqa: F401


def __getattr__(name):  # PEP 562 lazy import: cp imports models.gpt2,
    # which imports this package — a module-level import would be circular
    if name in ("ContextParallelGPT2", "cp_shard"):
        from tepdist_amd.parallel import cp
        return getattr(cp, name)
    raise AttributeError(name)
