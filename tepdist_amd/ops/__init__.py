from tepdist_amd.ops.interface import (  # noqa: F401
    linear,
    matmul,
    layernorm,
    softmax,
    attention,
    attention_qkv,
    embedding,
    cross_entropy,
    dropout,
    adamw_step,
)
