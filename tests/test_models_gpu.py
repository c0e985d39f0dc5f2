"""GPU end-to-end tests for the non-GPT model families (MoE, Wide-ResNet):
the HIP kernel path must run them (no eager fallback)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpt_moe_gpu_trains():
    from tepdist_amd.models.configs import MOE_CONFIGS
    from tepdist_amd.models.moe import GPTMoE
    from tepdist_amd.train import AdamW, Trainer
    cfg = MOE_CONFIGS["gpt-moe-test"]
    torch.manual_seed(0)
    model = GPTMoE(cfg, dtype=torch.bfloat16).cuda()
    opt = AdamW(model.parameters(), lr=1e-3)
    trainer = Trainer(model, opt)
    ids = torch.randint(0, cfg.vocab_size, (4, 33),
                        generator=torch.Generator().manual_seed(1)).cuda()
    losses = [trainer.train_step(lambda i: (ids[:, :-1], ids[:, 1:]))
              for _ in range(8)]
    assert losses[-1] < losses[0], losses


def test_wide_resnet_gpu_trains():
    from tepdist_amd.models.configs import WIDE_RESNET_CONFIGS
    from tepdist_amd.models.wide_resnet import WideResNet
    from tepdist_amd.train import AdamW, Trainer
    cfg = WIDE_RESNET_CONFIGS["wrn-test"]
    torch.manual_seed(0)
    model = WideResNet(cfg, dtype=torch.bfloat16).cuda()
    opt = AdamW(model.parameters(), lr=5e-4)
    trainer = Trainer(model, opt)
    x = torch.randn(8, 3, 32, 32).bfloat16().cuda()
    y = torch.randint(0, cfg.num_classes, (8,)).cuda()
    losses = [trainer.train_step(lambda i: (x, y)) for _ in range(8)]
    assert losses[-1] < losses[0] * 1.05, losses


def test_rpc_service_gpu_step():
    """Server-side IR interpreter executes on the GPU through the kernels."""
    from tepdist_amd.ir import gpt2_ir
    from tepdist_amd.models.configs import GPT2_CONFIGS
    from tepdist_amd.rpc.service import TepdistService
    cfg = GPT2_CONFIGS["gpt2-test"]
    svc = TepdistService(ckpt_dir="/tmp/tepdist_gpu_ckpt")
    assert svc.device.startswith("cuda")
    g = gpt2_ir(cfg, batch=4, seq=16)
    r = svc.build_execution_plan({"graph": g.to_json(), "num_devices": 1})
    ids = torch.randint(0, cfg.vocab_size, (4, 17),
                        generator=torch.Generator().manual_seed(2))
    feeds = {"input_ids": ids[:, :-1].reshape(-1),
             "labels": ids[:, 1:].reshape(-1)}
    losses = []
    for _ in range(6):
        out = svc.execute_plan({"handle": r["handle"], "inputs": feeds})
        losses.append(float(list(out["outputs"].values())[0]))
    assert losses[-1] < losses[0], losses
