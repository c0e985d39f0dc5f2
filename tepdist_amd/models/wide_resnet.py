"""Wide-ResNet benchmark family (250M..13B).

Mirrors the reference's wide_resnet example
(/root/reference/examples/wide_resnet/resnet.py: ResNet-50/101 bottleneck
with a channel width multiplier, fake-data benchmark protocol in its
README). Convolutions run as im2col + the MFMA GEMM kernel (the classic
lowering: torch unfold produces the column matrix, our batched bf16 GEMM
does the flops), which keeps the planner's conv2d sharding strategies
(batch / out-channel splits) executable with the same kernel set."""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from tepdist_amd import ops
from tepdist_amd.models.configs import WideResNetConfig


class Conv2d(nn.Module):
    """Conv via im2col + MFMA GEMM. weight [Cout, Cin, kh, kw] bf16."""

    def __init__(self, cin: int, cout: int, k: int = 3, stride: int = 1,
                 padding: Optional[int] = None, dtype=torch.bfloat16):
        super().__init__()
        self.k, self.stride = k, stride
        self.padding = padding if padding is not None else k // 2
        w = torch.randn(cout, cin, k, k) * math.sqrt(2.0 / (cin * k * k))
        self.weight = nn.Parameter(w.to(dtype))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, C, H, W = x.shape
        cout = self.weight.shape[0]
        ho = (H + 2 * self.padding - self.k) // self.stride + 1
        wo = (W + 2 * self.padding - self.k) // self.stride + 1
        if self.k == 1 and self.stride == 1 and self.padding == 0:
            # pointwise: plain GEMM on the channel dim
            xm = x.permute(0, 2, 3, 1).reshape(-1, C)
            y = ops.linear(xm, self.weight.reshape(cout, C))
            return y.reshape(B, H, W, cout).permute(0, 3, 1, 2).contiguous()
        cols = F.unfold(x, self.k, padding=self.padding,
                        stride=self.stride)          # [B, C*k*k, L]
        w2 = self.weight.reshape(cout, -1)           # [Cout, C*k*k]
        y = ops.matmul(cols.transpose(1, 2).contiguous(),
                       w2.t())                       # [B, L, Cout]
        return y.transpose(1, 2).reshape(B, cout, ho, wo).contiguous()


class BatchNorm2d(nn.Module):
    """fp32 batch norm over bf16 activations (running stats)."""

    def __init__(self, c: int, eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(c))
        self.bias = nn.Parameter(torch.zeros(c))
        self.register_buffer("running_mean", torch.zeros(c))
        self.register_buffer("running_var", torch.ones(c))
        self.eps, self.momentum = eps, momentum

    def forward(self, x):
        y = F.batch_norm(x.float(), self.running_mean, self.running_var,
                         self.weight, self.bias, self.training,
                         self.momentum, self.eps)
        return y.to(x.dtype)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, dtype=torch.bfloat16):
        super().__init__()
        cout = planes * self.expansion
        self.conv1 = Conv2d(cin, planes, 1, 1, 0, dtype)
        self.bn1 = BatchNorm2d(planes)
        self.conv2 = Conv2d(planes, planes, 3, stride, 1, dtype)
        self.bn2 = BatchNorm2d(planes)
        self.conv3 = Conv2d(planes, cout, 1, 1, 0, dtype)
        self.bn3 = BatchNorm2d(cout)
        self.down = None
        if stride != 1 or cin != cout:
            self.down = nn.Sequential(Conv2d(cin, cout, 1, stride, 0, dtype),
                                      BatchNorm2d(cout))

    def forward(self, x):
        idn = x if self.down is None else self.down(x)
        y = F.relu(self.bn1(self.conv1(x)))
        y = F.relu(self.bn2(self.conv2(y)))
        y = self.bn3(self.conv3(y))
        return F.relu(y + idn)


_DEPTH = {50: (3, 4, 6, 3), 101: (3, 4, 23, 3)}


class WideResNet(nn.Module):
    def __init__(self, cfg: WideResNetConfig, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        blocks = _DEPTH[cfg.n_layer]
        w = cfg.width_factor
        base = 64 * w
        self.stem = nn.Sequential(Conv2d(3, base, 7, 2, 3, dtype),
                                  BatchNorm2d(base), nn.ReLU())
        layers = []
        cin = base
        for i, n in enumerate(blocks):
            planes = base * (2 ** i)
            for j in range(n):
                stride = 2 if (i > 0 and j == 0) else 1
                layers.append(Bottleneck(cin, planes, stride, dtype))
                cin = planes * Bottleneck.expansion
        self.layers = nn.Sequential(*layers)
        self.fc_w = nn.Parameter(torch.empty(cfg.num_classes, cin,
                                             dtype=dtype))
        self.fc_b = nn.Parameter(torch.zeros(cfg.num_classes, dtype=dtype))
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        g = torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                fan_in = p[0].numel()
                p.copy_(torch.randn(p.shape, generator=g) *
                        math.sqrt(2.0 / max(fan_in, 1)))

    def forward(self, x, labels=None):
        x = self.stem(x)
        x = F.max_pool2d(x, 3, 2, 1)
        x = self.layers(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        logits = ops.linear(x, self.fc_w, self.fc_b)
        if labels is None:
            return logits
        return ops.cross_entropy(logits.float().to(x.dtype), labels,
                                 ignore_index=-1)
