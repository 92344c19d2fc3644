"""GPU model tests: the HIP-kernel path agrees with the CPU fp32 reference
path, and a training loop on the fused pipeline learns."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from tosem2021_amd.data.synthetic import synthetic_batch
from tosem2021_amd.models.classifier import CONFIGS, MLTC
from tosem2021_amd.train import TrainConfig, Trainer


def test_forward_parity_cpu_reference():
    cfg = CONFIGS["mltc-tiny"]
    torch.manual_seed(0)
    model = MLTC(cfg).to(torch.bfloat16)
    tokens, mask, _ = synthetic_batch(cfg, 4, 64, seed=7)
    out_cpu = {k: v.float() for k, v in model(tokens, mask).items()}
    gm = model.cuda()
    out_gpu = gm(tokens.cuda(), mask.cuda())
    for k in out_cpu:
        a, b = out_gpu[k].float().cpu(), out_cpu[k]
        assert torch.allclose(a, b, atol=0.05, rtol=0.05), \
            f"{k}: max diff {(a-b).abs().max()}"


def test_training_reduces_loss():
    torch.manual_seed(0)
    cfg = CONFIGS["mltc-tiny"]
    trainer = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0),
                      device=torch.device("cuda"))
    tokens, mask, labels = synthetic_batch(cfg, 16, 64, device="cuda", seed=3)
    losses = [trainer.step(tokens, mask, labels) for _ in range(30)]
    assert losses[-1] < losses[0] * 0.8, losses[::10]
    assert all(l == l for l in losses), "NaN loss"


def test_flat_param_views_alias_storage():
    trainer = Trainer(TrainConfig(model="mltc-tiny", warmup_steps=0),
                      device=torch.device("cuda"))
    p0 = trainer.flat.params[0]
    trainer.flat.flat.zero_()
    assert float(p0.data.abs().sum()) == 0.0, "params must alias the flat buffer"
