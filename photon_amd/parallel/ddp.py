"""Bucketed gradient all-reduce — the inner data-parallel degree.

Replaces Composer's DDP wrapper (ddp_sync_strategy=FORCED_SYNC,
photon/clients/trainer_utils.py:1714): gradients are synchronized ONCE per
optimization batch, after the last microbatch's backward (exactly the
FORCED_SYNC semantics — no autograd-hook overlap complexity, no sync per
microbatch), as a sequence of bucketed all-reduces.

Bucket sizing is xGMI-aware: each of the 7 links runs ≈153 GB/s and ring
all-reduce is per-link bound, so buckets are large (default 64 MiB) to
amortize latency — fewer, larger collectives (SURVEY.md §2.4 mapping).
"""

from __future__ import annotations

import torch
import torch.distributed as dist


class BucketedGradSync:
    """grad_sync_hook for photon_amd.train.Trainer."""

    def __init__(self, process_group=None, bucket_bytes: int = 64 << 20,
                 world_size: int | None = None):
        self.group = process_group
        self.bucket_bytes = bucket_bytes
        self.world_size = world_size or (
            dist.get_world_size(process_group) if dist.is_initialized() else 1
        )

    @torch.no_grad()
    def __call__(self, model: torch.nn.Module) -> None:
        if self.world_size <= 1 or not dist.is_initialized():
            return
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        if not grads:
            return
        bucket: list[torch.Tensor] = []
        size = 0
        for g in grads:
            bucket.append(g)
            size += g.numel() * g.element_size()
            if size >= self.bucket_bytes:
                self._reduce(bucket)
                bucket, size = [], 0
        if bucket:
            self._reduce(bucket)

    def _reduce(self, bucket: list[torch.Tensor]) -> None:
        flat = torch.cat([g.reshape(-1) for g in bucket])
        dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.group)
        flat.div_(self.world_size)
        off = 0
        for g in bucket:
            n = g.numel()
            g.copy_(flat[off : off + n].view_as(g))
            off += n
