"""Bucketed gradient all-reduce data parallelism over RCCL/xGMI.

MI355X-first design (SURVEY.md §5 "Distributed communication backend"):
the only collective the reference corpus actually uses is the DDP-style
gradient all-reduce (ray util/sgd/torch/distributed_torch_runner.py:37-61,
rllib ddppo.py:157-203); here it is rebuilt natively for xGMI rather than
wrapped: gradients live in ONE flat bf16 buffer (see train.py), partitioned
into large contiguous buckets sized for xGMI's per-link ring bound
(7 p2p links x ~153 GB/s -> latency amortizes at tens of MB, so default
64 MiB vs torch-DDP's 25 MiB), reduced asynchronously as backward produces
them, overlapping communication with the remaining backward compute.

The averaging divide (1/world) is folded into the fused AdamW kernel's
grad_scale, so no extra pass over the gradient buffer is ever made.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist


@dataclass
class _Bucket:
    view: torch.Tensor          # contiguous slice of the flat grad buffer
    param_ids: set              # ids of params whose grads land in this bucket
    pending: int = 0
    work: Optional[object] = None


class BucketedAllReduce:
    """Overlapped gradient all-reduce over a flat grad buffer.

    Parameters are registered in reverse creation order (backward produces
    late-layer grads first), so the tail bucket fires while early layers are
    still differentiating.
    """

    def __init__(self, params: List[torch.nn.Parameter], grad_flat: torch.Tensor,
                 offsets: List[int], bucket_bytes: int = 64 << 20,
                 process_group=None):
        self.group = process_group
        self.enabled = dist.is_available() and dist.is_initialized() and \
            dist.get_world_size(process_group) > 1
        self.world_size = dist.get_world_size(process_group) if self.enabled else 1
        self.grad_flat = grad_flat
        self.sync = True     # False during gradient-accumulation micro-steps
        self._buckets: List[_Bucket] = []
        self._param_bucket = {}
        self._hooks = []
        if not self.enabled:
            return
        elem = grad_flat.element_size()
        bucket_elems = max(bucket_bytes // elem, 1)
        # Build buckets over the flat buffer walking params in REVERSE order.
        order = sorted(range(len(params)), key=lambda i: -offsets[i])
        cur_ids: set = set()
        cur_lo, cur_hi, cur_n = None, None, 0
        def flush():
            nonlocal cur_ids, cur_lo, cur_hi, cur_n
            if cur_ids:
                b = _Bucket(view=grad_flat[cur_lo:cur_hi], param_ids=cur_ids)
                self._buckets.append(b)
                for pid in cur_ids:
                    self._param_bucket[pid] = b
            cur_ids, cur_lo, cur_hi, cur_n = set(), None, None, 0
        for i in order:
            p = params[i]
            n = p.numel()
            lo = offsets[i]
            if cur_lo is None:
                cur_lo, cur_hi = lo, lo + n
            else:
                cur_lo = min(cur_lo, lo)
            cur_ids.add(id(p))
            cur_n += n
            if cur_n >= bucket_elems:
                flush()
        flush()
        # install per-param hooks
        for p in params:
            if not p.requires_grad:
                continue
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready)
            )
        self.reset()

    def reset(self):
        for b in self._buckets:
            b.pending = len(b.param_ids)
            b.work = None

    def _on_grad_ready(self, param):
        if not self.sync:
            return
        b = self._param_bucket.get(id(param))
        if b is None:
            return
        b.pending -= 1
        if b.pending == 0:
            b.work = dist.all_reduce(b.view, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def finalize(self):
        """Wait for all in-flight reductions; call after backward()."""
        if not self.enabled:
            return
        for b in self._buckets:
            if b.work is not None:
                b.work.wait()
            elif b.pending > 0:
                # params in this bucket got no grad this step (unused head):
                # reduce anyway so ranks stay consistent.
                dist.all_reduce(b.view, op=dist.ReduceOp.SUM, group=self.group)
        self.reset()

    @property
    def grad_scale(self) -> float:
        """Fold the DP average into the optimizer's grad scale."""
        return 1.0 / self.world_size
