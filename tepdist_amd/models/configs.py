"""Model configurations mirroring the reference's benchmark set.

GPT-2 sizes follow /root/reference/examples/GPT2/{117M,345M,1.5B,175B}.json
(vocab 50257, n_ctx 1024); wide-resnet sizes follow
/root/reference/examples/wide_resnet/README.md (250M..13B); the MoE config
follows /root/reference/examples/gpt_moe/pretrain_moe.json (768 hidden,
8 layers, 8 experts, top-2 gating, seq 1024).
"""

from __future__ import annotations

from dataclasses import dataclass


def _pad_to(x: int, mult: int) -> int:
    return (x + mult - 1) // mult * mult


@dataclass
class GPT2Config:
    name: str
    n_layer: int
    n_embd: int
    n_head: int
    n_ctx: int = 1024
    vocab_size: int = 50257
    embd_pdrop: float = 0.0
    attn_pdrop: float = 0.0
    resid_pdrop: float = 0.0
    ln_eps: float = 1e-5

    @property
    def padded_vocab(self) -> int:
        # Pad the embedding/logits dimension to a multiple of 256 so sharded
        # kernels and TP splits see friendly shapes; loss masks the padding.
        return _pad_to(self.vocab_size, 256)

    @property
    def n_params(self) -> int:
        d, L, V = self.n_embd, self.n_layer, self.padded_vocab
        per_layer = 12 * d * d + 13 * d
        return V * d + self.n_ctx * d + L * per_layer + 2 * d


GPT2_CONFIGS = {
    "gpt2-117m": GPT2Config("gpt2-117m", n_layer=12, n_embd=768, n_head=12),
    "gpt2-345m": GPT2Config("gpt2-345m", n_layer=24, n_embd=1024, n_head=16),
    "gpt2-762m": GPT2Config("gpt2-762m", n_layer=36, n_embd=1280, n_head=20),
    "gpt2-1.5b": GPT2Config("gpt2-1.5b", n_layer=48, n_embd=1600, n_head=25),
    "gpt2-175b": GPT2Config("gpt2-175b", n_layer=96, n_embd=12288, n_head=96, n_ctx=2048),
    # tiny config for CPU tests
    "gpt2-test": GPT2Config("gpt2-test", n_layer=2, n_embd=64, n_head=4, n_ctx=64,
                            vocab_size=503),
}


@dataclass
class MoEConfig:
    name: str = "gpt-moe-base"
    n_layer: int = 8
    n_embd: int = 768
    n_head: int = 12
    n_ctx: int = 1024
    vocab_size: int = 50257
    num_experts: int = 8
    top_k: int = 2
    moe_every: int = 2        # every 2nd layer is an MoE layer
    capacity_factor: float = 1.25
    ln_eps: float = 1e-5

    @property
    def padded_vocab(self) -> int:
        return _pad_to(self.vocab_size, 256)


MOE_CONFIGS = {
    "gpt-moe-base": MoEConfig(),
    "gpt-moe-test": MoEConfig(name="gpt-moe-test", n_layer=2, n_embd=64, n_head=4,
                              n_ctx=64, vocab_size=503, num_experts=4, top_k=2,
                              moe_every=1),
}


@dataclass
class WideResNetConfig:
    name: str
    n_layer: int           # resnet depth (50 or 101)
    width_factor: int      # channel multiplier
    num_classes: int = 1000
    image_size: int = 224


# width factors chosen to land near the reference's published parameter
# counts (examples/wide_resnet/README.md:20-31)
WIDE_RESNET_CONFIGS = {
    "wrn-250m": WideResNetConfig("wrn-250m", n_layer=50, width_factor=3),
    "wrn-500m": WideResNetConfig("wrn-500m", n_layer=50, width_factor=4),
    "wrn-1b": WideResNetConfig("wrn-1b", n_layer=50, width_factor=6),
    "wrn-2b": WideResNetConfig("wrn-2b", n_layer=50, width_factor=8),
    "wrn-4b": WideResNetConfig("wrn-4b", n_layer=50, width_factor=12),
    "wrn-7b": WideResNetConfig("wrn-7b", n_layer=50, width_factor=16),
    "wrn-13b": WideResNetConfig("wrn-13b", n_layer=101, width_factor=16),
    "wrn-test": WideResNetConfig("wrn-test", n_layer=50, width_factor=1,
                                 num_classes=10, image_size=32),
}
