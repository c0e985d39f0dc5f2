from tepdist_amd.parallel.dp import GradReducer, init_distributed  # noqa: F401
