#!/usr/bin/env python3
"""Prove RCCL executes on MI355X.

Two ranks share one GPU (RCCL supports multi-rank-per-device), nccl(=RCCL)
backend:
  1. allreduce correctness: sum of per-rank constants;
  2. allreduce timing sweep (1 MiB .. 256 MiB bf16) — same-device, so the
     numbers measure RCCL's kernel path, not xGMI wires;
  3. three DDP training steps of the mltc model through parallel/ddp.py's
     bucketed async allreduce on the nccl backend, losses compared across
     ranks (they see identical data -> identical loss).

Launch (single MI355X box):
    python -m torch.distributed.run --nnodes 1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --standalone scripts/rccl_proof.py
Writes rank-0 results JSON to gpurun_out/rccl_proof.json (or stdout).
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                ".."))

import torch
import torch.distributed as dist


def main() -> None:
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    # both ranks pin the same physical GPU
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=rank, world_size=world)
    dev = torch.device("cuda", 0)
    res = {"backend": dist.get_backend(), "world_size": world,
           "nccl_version": list(torch.cuda.nccl.version()),
           "device": torch.cuda.get_device_name(0)}

    # 1. correctness
    t = torch.full((1024,), float(rank + 1), device=dev)
    dist.all_reduce(t)
    expect = sum(range(1, world + 1))
    assert torch.all(t == expect), t[:4]
    res["allreduce_correct"] = True

    # 2. timing sweep (bf16, sizes in MiB)
    sweep = {}
    for mib in (1, 16, 64, 256):
        n = mib * (1 << 20) // 2
        x = torch.ones(n, device=dev, dtype=torch.bfloat16)
        for _ in range(3):
            dist.all_reduce(x)
        torch.cuda.synchronize()
        dist.barrier()
        t0 = time.perf_counter()
        iters = 10
        for _ in range(iters):
            dist.all_reduce(x)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / iters
        # algbw convention: bytes / time
        sweep[f"{mib}MiB"] = {"us": round(dt * 1e6, 1),
                              "algbw_GBps": round(mib / 1024 / dt, 1)}
    res["allreduce_sweep"] = sweep

    # 3. DDP training steps through parallel/ddp.py on RCCL
    from tosem2021_amd.data.synthetic import synthetic_batch
    from tosem2021_amd.models.classifier import CONFIGS
    from tosem2021_amd.train import TrainConfig, Trainer

    torch.manual_seed(7)
    trainer = Trainer(TrainConfig(model="mltc-base", warmup_steps=0),
                      device=dev)
    assert trainer.ddp.enabled and trainer.ddp.world_size == world
    cfg = CONFIGS["mltc-base"]
    losses = []
    for _ in range(3):
        tokens, mask, labels = synthetic_batch(cfg, 8, 256, device=dev,
                                               seed=11)
        losses.append(trainer.step(tokens, mask, labels))
    # identical data on both ranks -> allreduced grads == local grads,
    # losses must match across ranks
    lt = torch.tensor(losses, device=dev)
    gathered = [torch.empty_like(lt) for _ in range(world)]
    dist.all_gather(gathered, lt)
    max_dev = max(float((g - gathered[0]).abs().max()) for g in gathered)
    res["ddp_steps_losses"] = [round(x, 4) for x in losses]
    res["cross_rank_loss_max_dev"] = max_dev
    assert max_dev < 1e-6, max_dev
    res["ddp_buckets"] = len(trainer.ddp._buckets)

    if rank == 0:
        os.makedirs("gpurun_out", exist_ok=True)
        with open("gpurun_out/rccl_proof.json", "w") as f:
            json.dump(res, f, indent=1)
        print(json.dumps(res))
    dist.destroy_process_group()


if __name__ == "__main__":
    try:
        main()
    except Exception:
        import traceback
        traceback.print_exc()
        sys.stdout.flush()
        raise
