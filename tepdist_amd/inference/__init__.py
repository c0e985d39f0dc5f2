from tepdist_amd.inference.engine import Generator  # noqa: F401
