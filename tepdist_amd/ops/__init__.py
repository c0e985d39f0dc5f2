from tepdist_amd.ops.interface import (  # noqa: F401
    gelu,
    linear,
    mlp,
    matmul,
    layernorm,
    add_layernorm,
    softmax,
    attention,
    attention_qkv,
    embedding,
    cross_entropy,
    dropout,
    adamw_step,
    rmsnorm,
    rope,
    swiglu,
)

# torch.fx: leaf-wrap the public ops IN THIS NAMESPACE (clients call
# ops.<name>), so generic capture (ir/capture.from_fx) records op nodes
# instead of tracing into the backend dispatch.
import torch.fx as _fx  # noqa: E402

for _name in ("linear", "matmul", "layernorm", "softmax", "attention",
              "attention_qkv", "embedding", "cross_entropy", "dropout",
              "rmsnorm", "rope", "swiglu"):
    _fx.wrap(_name)
del _fx, _name
