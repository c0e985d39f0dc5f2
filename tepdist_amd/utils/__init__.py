from tepdist_amd.utils.tracing import Tracer, get_tracer, trace_span  # noqa: F401
