#!/usr/bin/env python3
"""Flagship training benchmark: MLTC (mltc-base) classifier training step.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
For N > 1 the driver launches this under torch.distributed.run, one rank per
GPU over RCCL.  W untimed warmup steps, then exactly K timed steps bracketed
by barrier + torch.cuda.synchronize() on both sides; wall time is the MAX
over ranks; rank 0 prints one JSON line.

Metric: whole-job training throughput in tokens/s, bf16, synthetic data
(random token ids + random taxonomy labels, random-init weights — there is no
network for datasets).  The reference publishes no performance numbers
(BASELINE.json: metric N/A, published {}), so vs_baseline is null.
"""
from __future__ import annotations

import argparse
import json
import os
import time

# GEMM algorithm selection (TunableOp) — must be configured before the
# first torch import.  The committed table artifacts/tunableop_gfx9500.csv
# (round 2, tuned on an MI355X at the bench shapes) fixes the wgrad-family
# algo picks the default heuristic misses (qkv wgrad -23%, outp wgrad -19%,
# ffn-up wgrad -17%; whole step 88.7 -> 85.8 ms measured back-to-back on
# one box).  TunableOp inserts the device ordinal before ".csv", so the
# FILENAME below resolves to the committed ...gfx9500.csv on device 0.
# TOSEM_TUNE=1 re-tunes and rewrites the table; TOSEM_NOTUNE=1 disables.
_TUNE_FILE = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "artifacts", "tunableop_gfx950.csv")
if os.environ.get("TOSEM_TUNE", "0") == "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE_FILE)
elif os.path.exists(_TUNE_FILE.replace(".csv", "0.csv")) and \
        os.environ.get("TOSEM_NOTUNE") != "1":
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE_FILE)
    # TunableOp resolves the table per DEVICE ordinal (…gfx950<K>.csv); on
    # a multi-GPU launch ranks 1..7 would otherwise find no table and run
    # the untuned heuristics (dragging the max-over-ranks step time) —
    # replicate the tuned table for every local ordinal.
    try:
        _src = _TUNE_FILE.replace(".csv", "0.csv")
        for _k in range(1, 8):
            _dst = _TUNE_FILE.replace(".csv", f"{_k}.csv")
            if not os.path.exists(_dst):
                import shutil
                shutil.copyfile(_src, _dst)
    except OSError:
        pass   # read-only checkout: ordinal-0 rank still gets the table

import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=24)
    ap.add_argument("--warmup", type=int, default=6)
    ap.add_argument("--batch", type=int, default=128, help="per-GPU batch")
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--model", type=str, default="mltc-base")
    ap.add_argument("--bucket-mb", type=int, default=64)
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    if world > 1:
        import torch.distributed as dist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(
            backend="nccl" if use_gpu else "gloo", rank=rank, world_size=world)

    from tosem2021_amd.data.synthetic import synthetic_batch
    from tosem2021_amd.models.classifier import CONFIGS
    from tosem2021_amd.train import TrainConfig, Trainer

    cfg = CONFIGS[args.model]
    if args.seq > cfg.max_seq:
        # extend the position table for long-sequence runs
        from tosem2021_amd.models.classifier import MLTCConfig
        cfg = MLTCConfig(**{**cfg.__dict__, "max_seq": args.seq})
    tcfg = TrainConfig(model=args.model, warmup_steps=0,
                       total_steps=max(args.steps * 100, 1000),
                       bucket_mb=args.bucket_mb)
    trainer = Trainer(tcfg, device=device, model_cfg=cfg)

    tokens, mask, labels = synthetic_batch(
        cfg, args.batch, args.seq, device=device, seed=1234 + rank)

    def barrier_sync():
        if world > 1:
            import torch.distributed as dist
            dist.barrier()
        if use_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        trainer.step(tokens, mask, labels)
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        trainer.step(tokens, mask, labels)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    n_gpus = world if use_gpu else args.gpus
    tokens_per_step = args.batch * args.seq * world
    value = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1e3
    if rank == 0:
        print(json.dumps({
            "metric": "train_tokens_per_s",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * world,
                "seq_len": args.seq,
                "parallelism": f"dp{world}",
            },
        }))
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
