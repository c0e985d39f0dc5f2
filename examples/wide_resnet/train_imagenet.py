"""Wide-ResNet fake-data benchmark (the reference's
examples/wide_resnet/train_imagenet.py counterpart)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.models.configs import WIDE_RESNET_CONFIGS
from tepdist_amd.models.wide_resnet import WideResNet
from tepdist_amd.parallel import GradReducer, init_distributed
from tepdist_amd.train import AdamW, Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="wrn-250m")
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--stop-at-step", type=int, default=10)
    args = ap.parse_args()

    rank, world, local = init_distributed()
    dev = torch.device("cuda", local) if torch.cuda.is_available() else "cpu"
    cfg = WIDE_RESNET_CONFIGS[args.config]
    dtype = torch.bfloat16 if dev != "cpu" else torch.float32
    model = WideResNet(cfg, dtype=dtype).to(dev)
    opt = AdamW(model.parameters(), lr=1e-4)
    reducer = GradReducer(model.parameters()) if world > 1 else None
    trainer = Trainer(model, opt, reducer=reducer)

    g = torch.Generator().manual_seed(11 + rank)
    for step in range(args.stop_at_step):
        x = torch.randn(args.batch, 3, cfg.image_size, cfg.image_size,
                        generator=g).to(dtype)
        y = torch.randint(0, cfg.num_classes, (args.batch,), generator=g)
        t0 = time.time()
        loss = trainer.train_step(lambda i: (x.to(dev), y.to(dev)))
        if rank == 0:
            ips = args.batch * world / (time.time() - t0)
            print(f"step {step} loss {loss:.4f} images/s {ips:.1f}")


if __name__ == "__main__":
    main()
