"""Multi-process CPU tests of the bucketed-allreduce DP path (gloo, ws=2).

Validates the distributed construction the driver's 8-GPU RCCL bench relies
on: after one step, all ranks hold identical parameters, and those equal a
single-process run on the concatenated global batch.
"""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tosem2021_amd.data.synthetic import synthetic_batch
from tosem2021_amd.models.classifier import CONFIGS
from tosem2021_amd.train import TrainConfig, Trainer

PORT = 29871


def _worker(rank, world, port, out, accum=False, ckpt_dir=None):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.manual_seed(100 + rank)  # different init per rank; bcast fixes it
        trainer = Trainer(
            TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                        dtype="f32", bucket_mb=1, ckpt_dir=ckpt_dir),
            device=torch.device("cpu"))
        if ckpt_dir:
            trainer.load_or_init()   # crash-resume path: all ranks load the
                                     # same checkpoint -> consistent state
        cfg = CONFIGS["mltc-tiny"]
        # global batch 8 split across ranks
        per = 8 // world
        tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=42)
        sl = slice(rank * per, rank * per + per)
        for _ in range(2):
            if accum:
                # two micro-batches per optimizer step: collectives must
                # fire only on the last micro-backward (ddp.sync flag)
                h = per // 2
                micros = [
                    (tokens[sl][:h], mask[sl][:h],
                     {k: v[sl][:h] for k, v in labels.items()}),
                    (tokens[sl][h:], mask[sl][h:],
                     {k: v[sl][h:] for k, v in labels.items()}),
                ]
                trainer.step_accum(micros)
            else:
                trainer.step(tokens[sl], mask[sl],
                             {k: v[sl] for k, v in labels.items()})
        if ckpt_dir and rank == 0:
            trainer.save()
        out[rank] = trainer.flat.flat.clone()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_ddp_matches_single_process():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.start_processes(_worker, args=(world, PORT, out),
                           nprocs=world, join=True, start_method="spawn")
        flats = {r: out[r] for r in range(world)}

    # ranks agree bitwise
    assert torch.equal(flats[0], flats[1])

    # equals a single-process full-batch run with the rank-0 init
    torch.manual_seed(100)
    solo = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                               dtype="f32"), device=torch.device("cpu"))
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=42)
    for _ in range(2):
        solo.step(tokens, mask, labels)
    diff = (solo.flat.flat - flats[0]).abs().max()
    assert float(diff) < 5e-5, float(diff)


@pytest.mark.timeout(300)
def test_ddp_grad_accum_matches_single_process():
    """ws=2 x 2 micro-batches: the sync-flag gating and the 1/(world*n)
    grad-scale folding must reproduce the full-batch single-process step."""
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.start_processes(_worker, args=(world, PORT + 2, out, True),
                           nprocs=world, join=True, start_method="spawn")
        flats = {r: out[r] for r in range(world)}
    assert torch.equal(flats[0], flats[1])
    torch.manual_seed(100)
    solo = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                               dtype="f32"), device=torch.device("cpu"))
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=42)
    for _ in range(2):
        solo.step(tokens, mask, labels)
    diff = (solo.flat.flat - flats[0]).abs().max()
    assert float(diff) < 5e-5, float(diff)


@pytest.mark.timeout(300)
def test_ddp_world4_matches_single_process():
    """The driver's scale bench runs up to 8 ranks; rehearse ws=4 here."""
    world = 4
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.start_processes(_worker, args=(world, PORT + 4, out),
                           nprocs=world, join=True, start_method="spawn")
        flats = {r: out[r] for r in range(world)}
    for r in range(1, world):
        assert torch.equal(flats[0], flats[r])
    torch.manual_seed(100)
    solo = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                               dtype="f32"), device=torch.device("cpu"))
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=42)
    for _ in range(2):
        solo.step(tokens, mask, labels)
    diff = (solo.flat.flat - flats[0]).abs().max()
    assert float(diff) < 5e-5, float(diff)


@pytest.mark.timeout(300)
def test_ddp_checkpoint_resume(tmp_path):
    """Kill-and-resume across a checkpoint (SURVEY.md §5 checkpoint/resume
    + failure recovery): 2 DDP steps, rank-0 checkpoint, fresh processes
    resume and take 2 more steps == 4 uninterrupted steps."""
    world = 2
    ck = str(tmp_path / "ck")
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        mp.start_processes(_worker, args=(world, PORT + 6, out, False, ck),
                           nprocs=world, join=True, start_method="spawn")
        out2 = mgr.dict()
        mp.start_processes(_worker, args=(world, PORT + 8, out2, False, ck),
                           nprocs=world, join=True, start_method="spawn")
        resumed = out2[0].clone()
        assert torch.equal(out2[0], out2[1])

    torch.manual_seed(100)
    solo = Trainer(TrainConfig(model="mltc-tiny", lr=1e-3, warmup_steps=0,
                               dtype="f32"), device=torch.device("cpu"))
    cfg = CONFIGS["mltc-tiny"]
    tokens, mask, labels = synthetic_batch(cfg, 8, 32, seed=42)
    for _ in range(4):
        solo.step(tokens, mask, labels)
    diff = (solo.flat.flat - resumed).abs().max()
    assert float(diff) < 1e-4, float(diff)
