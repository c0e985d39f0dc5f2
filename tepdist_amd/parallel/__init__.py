from tepdist_amd.parallel.dp import GradReducer, init_distributed  # noqa: F401
from tepdist_amd.parallel.ring_attention import ring_attention  # noqa: F401
from tepdist_amd.parallel.cp import ContextParallelGPT2, cp_shard  # noqa: F401
