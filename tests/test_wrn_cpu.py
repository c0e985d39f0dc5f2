"""Wide-ResNet CPU tests: conv-as-GEMM numerics vs torch conv2d, and a tiny
training run."""

import torch
import torch.nn.functional as F

from tepdist_amd.models.configs import WIDE_RESNET_CONFIGS
from tepdist_amd.models.wide_resnet import Conv2d, WideResNet


def test_conv_matches_torch():
    torch.manual_seed(0)
    for k, stride in ((3, 1), (3, 2), (1, 1), (7, 2)):
        conv = Conv2d(5, 8, k, stride, dtype=torch.float32)
        x = torch.randn(2, 5, 13, 13, requires_grad=True)
        y = conv(x)
        ref = F.conv2d(x, conv.weight, stride=stride, padding=conv.padding)
        torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-5)
        y.sum().backward()
        assert torch.isfinite(conv.weight.grad).all()
        conv.weight.grad = None


def test_wrn_tiny_trains():
    cfg = WIDE_RESNET_CONFIGS["wrn-test"]
    torch.manual_seed(0)
    model = WideResNet(cfg, dtype=torch.float32)
    opt = torch.optim.AdamW(model.parameters(), lr=5e-4)
    x = torch.randn(4, 3, 32, 32)
    y = torch.randint(0, cfg.num_classes, (4,))
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = model(x, labels=y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses
