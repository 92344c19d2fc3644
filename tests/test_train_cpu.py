"""CPU trainer tests: flat packing, learning, checkpoint/resume."""
import os

import torch

from tosem2021_amd.data.synthetic import synthetic_batch
from tosem2021_amd.models.classifier import CONFIGS
from tosem2021_amd.train import FlatParams, TrainConfig, Trainer


def _tiny_trainer(**kw):
    return Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0, **kw),
                   device=torch.device("cpu"))


def test_flat_params_alias_and_pad():
    trainer = _tiny_trainer()
    fp = trainer.flat
    assert fp.padded % 512 == 0
    assert fp.padded >= fp.numel
    # mutating flat mutates the params
    fp.flat.fill_(1.0)
    for p in fp.params:
        assert float(p.data.float().mean()) == 1.0


def test_training_reduces_loss_cpu():
    torch.manual_seed(0)
    trainer = _tiny_trainer()
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=5)
    losses = [trainer.step(tokens, mask, labels) for _ in range(25)]
    assert losses[-1] < losses[0] * 0.9, losses[::8]


def test_checkpoint_resume(tmp_path):
    torch.manual_seed(0)
    ck = str(tmp_path / "ckpts")
    t1 = _tiny_trainer(ckpt_dir=ck)
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 4, 32, seed=9)
    for _ in range(3):
        t1.step(tokens, mask, labels)
    path = t1.save()
    assert os.path.exists(path)
    l_next = t1.step(tokens, mask, labels)

    torch.manual_seed(123)  # resume must not depend on init RNG
    t2 = _tiny_trainer(ckpt_dir=ck)
    assert t2.load_or_init()
    assert t2.step_num == 3
    assert torch.equal(t2.flat.flat, t1.flat.flat) is False  # t1 stepped once more
    l_resumed = t2.step(tokens, mask, labels)
    assert abs(l_resumed - l_next) < 1e-5, (l_resumed, l_next)


def test_latest_checkpoint_picks_newest(tmp_path):
    ck = str(tmp_path)
    open(os.path.join(ck, "ckpt_00000001.pt"), "w").close()
    open(os.path.join(ck, "ckpt_00000010.pt"), "w").close()
    assert Trainer.latest_checkpoint(ck).endswith("ckpt_00000010.pt")


def test_gradient_accumulation_matches_big_batch():
    torch.manual_seed(7)
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=21)

    torch.manual_seed(50)
    big = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                              dtype="f32"), device=torch.device("cpu"))
    big.step(tokens, mask, labels)

    torch.manual_seed(50)
    acc = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                              dtype="f32"), device=torch.device("cpu"))
    halves = [(tokens[:4], mask[:4], {k: v[:4] for k, v in labels.items()}),
              (tokens[4:], mask[4:], {k: v[4:] for k, v in labels.items()})]
    acc.step_accum(halves)
    diff = (big.flat.flat - acc.flat.flat).abs().max()
    assert float(diff) < 5e-5, float(diff)


def test_checkpoint_retention(tmp_path):
    ck = str(tmp_path / "cks")
    t = Trainer(TrainConfig(model="mltc-tiny", warmup_steps=0, ckpt_dir=ck,
                            ckpt_keep=2), device=torch.device("cpu"))
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 2, 16, seed=3)
    for _ in range(4):
        t.step(tokens, mask, labels)
        t.save()
    kept = sorted(f for f in os.listdir(ck) if f.endswith(".pt"))
    assert len(kept) == 2
    assert kept[-1] == "ckpt_00000004.pt"
