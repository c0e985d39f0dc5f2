from tepdist_amd.ir.graph import Graph, Node  # noqa: F401
from tepdist_amd.ir.capture import from_fx, gpt2_ir  # noqa: F401
