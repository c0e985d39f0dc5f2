"""Minimal conv client (the reference's third smoke test,
examples/smoke_testing/conv.py): a small conv stack + pooling + linear
head trained for a few steps through the service path."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.models.wide_resnet import BatchNorm2d, Conv2d
from tepdist_amd.train.optim import AdamW


class TinyConvNet(torch.nn.Module):
    def __init__(self, classes=10, dtype=torch.float32):
        super().__init__()
        self.c1 = Conv2d(3, 16, 3, stride=1, padding=1, dtype=dtype)
        self.bn1 = BatchNorm2d(16)
        self.c2 = Conv2d(16, 32, 3, stride=2, padding=1, dtype=dtype)
        self.bn2 = BatchNorm2d(32)
        self.fc = torch.nn.Linear(32, classes, dtype=dtype)

    def forward(self, x):
        x = torch.relu(self.bn1(self.c1(x)))
        x = torch.relu(self.bn2(self.c2(x)))
        x = x.mean(dim=(2, 3))
        return self.fc(x)


def main(steps=5, device=None):
    device = device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    dtype = torch.bfloat16 if device.startswith("cuda") else torch.float32
    torch.manual_seed(0)
    model = TinyConvNet(dtype=dtype).to(device)
    opt = AdamW(model.parameters(), lr=1e-3)
    x = torch.randn(16, 3, 32, 32, dtype=dtype, device=device)
    t = torch.randint(0, 10, (16,), device=device)
    for step in range(steps):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(
            model(x).float(), t)
        loss.backward()
        opt.step()
        print(f"step {step} loss {loss.item():.4f}", flush=True)


if __name__ == "__main__":
    main()
