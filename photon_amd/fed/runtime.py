"""RCCL/xGMI distributed choreography — the comm-stack replacement.

The reference stacks five transports (gRPC, NCCL, POSIX SHM, Ray, S3 —
SURVEY.md §5.8). The MI355X-native design collapses the data plane into ONE
torch.distributed process group over RCCL (backend "nccl" on ROCm; "gloo"
for CPU tests): each federated client is one rank that owns one MI355X.

* broadcast of global parameters  -> ncclBroadcast of the flat HBM buffer
  (replaces broadcast_parameters_to_nodes + SHM/Ray/S3,
  photon/server/broadcast_utils.py:60-201), needed only at init/restore —
  per-round the server-opt update is applied redundantly on every rank, so
  steady-state rounds cost exactly ONE all-reduce;
* weighted aggregation            -> pre-scale by n_i/sum(n) then
  ncclAllReduce(SUM) (replaces the gRPC + NumPy streaming aggregation,
  photon/strategy/aggregation.py:19-87);
* partial participation           -> zero-contribution masking in the same
  all-reduce (no communicator churn per round);
* control plane (client sampling) -> deterministic seeded RNG replicated on
  every rank; no messages at all.

xGMI note: each GPU has 7 point-to-point links (~153 GB/s each); a single
ring all-reduce is per-link bound, so the flat buffer is reduced as one
large collective letting RCCL pick multi-ring/tree algorithms
(SURVEY.md §2.4 mapping).
"""

from __future__ import annotations

import os
import random
from dataclasses import dataclass

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None, timeout_sec: int = 600) -> tuple[int, int]:
    """Initialize torch.distributed from torchrun-style env vars.

    Returns (rank, world_size); (0, 1) without dist env (single process).
    """
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    import datetime

    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_sec)
    )
    rank, world = dist.get_rank(), dist.get_world_size()
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank % torch.cuda.device_count())))
    return rank, world


@dataclass
class Comm:
    rank: int
    world_size: int

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1 and dist.is_initialized()

    def barrier(self) -> None:
        if self.is_distributed:
            dist.barrier()

    def broadcast_flat(self, flat: torch.Tensor, src: int = 0) -> None:
        if self.is_distributed:
            dist.broadcast(flat, src=src)

    def all_gather_scalars(self, value: float) -> list[float]:
        if not self.is_distributed:
            return [value]
        t = torch.tensor([value], dtype=torch.float64)
        device = None
        if dist.get_backend() == "nccl":
            device = torch.device("cuda", torch.cuda.current_device())
            t = t.to(device)
        out = [torch.zeros_like(t) for _ in range(self.world_size)]
        dist.all_gather(out, t)
        return [float(x.item()) for x in out]

    def all_gather_obj(self, obj):
        """All-gather an arbitrary picklable object (per-cid eval rows);
        returns a list of world_size objects."""
        if not self.is_distributed:
            return [obj]
        out = [None] * self.world_size
        dist.all_gather_object(out, obj)
        return out

    def all_reduce_(self, flat: torch.Tensor) -> torch.Tensor:
        if self.is_distributed:
            dist.all_reduce(flat, op=dist.ReduceOp.SUM)
        return flat

    def weighted_average_(
        self, local_weighted_sum: torch.Tensor, local_weight: float
    ) -> tuple[torch.Tensor, float]:
        """All-reduce a locally pre-scaled contribution.

        Caller passes sum_i(n_i * x_i) over its local clients (zeros if none)
        and sum_i(n_i). Returns (global weighted mean, total weight), leaving
        the mean in local_weighted_sum.
        """
        weights = self.all_gather_scalars(local_weight)
        total = sum(weights)
        self.all_reduce_(local_weighted_sum)
        if total > 0:
            local_weighted_sum.div_(total)
        return local_weighted_sum, total


# Incremental sampler cache: one Random stream per (seed, n_total,
# n_per_round), advanced round by round — O(1) per round instead of
# replaying from round 1 on every call (O(n_rounds^2) total).
_SAMPLER_CACHE: dict[tuple, tuple[random.Random, int, list[int]]] = {}


def sample_clients(
    seed: int, current_round: int, n_total: int, n_per_round: int
) -> list[int]:
    """Deterministic per-round client sampling, replicated on every rank.

    Mirrors the reference's server-side PRNG fast-forward semantics
    (photon/server_app.py:188-192,295): one Random(seed) stream advanced
    round by round so resume reproduces the same schedule. The stream is
    cached and advanced incrementally; asking for an earlier round than
    the cache has reached replays from scratch (resume path, happens once).
    """
    key = (seed, n_total, n_per_round)
    rng, done, last = _SAMPLER_CACHE.get(key, (None, 0, []))
    if rng is None or current_round < done:
        rng, done, last = random.Random(seed), 0, []
    while done < current_round:
        last = rng.sample(range(n_total), n_per_round)
        done += 1
    _SAMPLER_CACHE[key] = (rng, done, last)
    return sorted(last)


def assign_clients_to_ranks(sampled: list[int], world_size: int) -> dict[int, list[int]]:
    """Round-robin assignment of sampled client ids to ranks — the analogue
    of the reference's work-queue scheduler (server_util.py:65-202)."""
    out: dict[int, list[int]] = {r: [] for r in range(world_size)}
    for i, cid in enumerate(sampled):
        out[i % world_size].append(cid)
    return out
