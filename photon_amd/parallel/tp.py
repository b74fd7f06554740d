"""Tensor parallelism — column/row-parallel linears over RCCL.

The reference plumbs a ``tp_config`` into Composer's
``parallelism_config={"fsdp","tp"}`` (trainer_utils.py:1376-1398,1641-1648)
and reverts to DDP on one GPU. The MI355X-native equivalent is explicit:

* ``ColumnParallelLinear`` — weight rows sharded; forward is local, the
  output is the rank's column shard (gather_output optionally all-gathers).
* ``RowParallelLinear``    — weight columns sharded; forward ends in ONE
  all-reduce over xGMI.
* ``apply_tensor_parallel(model, plan)`` — swaps nn.Linear modules per the
  MPT layer plan (Wqkv/up_proj column, out_proj/down_proj row) in place.

TP degree is bounded by the xGMI point-to-point topology: every all-reduce
in the forward runs at the ~153 GB/s per-link rate, so TP pays off only for
layers whose GEMM time exceeds d_model*seq*2B / 153GB/s per microbatch —
for MPT-125M..7B on ONE node the federated outer loop (1 GPU per client)
remains the default; TP is a capability hook, exactly like the reference.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F


class _AllReduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group):
        if dist.is_initialized() and dist.get_world_size(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, g):
        return g, None


class _CopyToParallel(torch.autograd.Function):
    """Identity forward; all-reduce in backward (column-parallel input)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, g):
        if dist.is_initialized() and dist.get_world_size(ctx.group) > 1:
            g = g.contiguous()
            dist.all_reduce(g, group=ctx.group)
        return g, None


class ColumnParallelLinear(nn.Module):
    def __init__(self, linear: nn.Linear, rank: int, world: int, group=None,
                 gather_output: bool = False):
        super().__init__()
        out_f, in_f = linear.weight.shape
        assert out_f % world == 0, f"out_features {out_f} % tp {world} != 0"
        self.shard = out_f // world
        self.group = group
        self.world = world
        self.gather_output = gather_output
        w = linear.weight.detach()[rank * self.shard : (rank + 1) * self.shard]
        self.weight = nn.Parameter(w.clone())
        if linear.bias is not None:
            b = linear.bias.detach()[rank * self.shard : (rank + 1) * self.shard]
            self.bias = nn.Parameter(b.clone())
        else:
            self.bias = None

    def forward(self, x):
        x = _CopyToParallel.apply(x, self.group)
        y = F.linear(x, self.weight, self.bias)
        if self.gather_output and self.world > 1:
            ys = [torch.empty_like(y) for _ in range(self.world)]
            dist.all_gather(ys, y.contiguous(), group=self.group)
            y = torch.cat(ys, dim=-1)
        return y


class RowParallelLinear(nn.Module):
    def __init__(self, linear: nn.Linear, rank: int, world: int, group=None,
                 input_is_parallel: bool = True):
        super().__init__()
        out_f, in_f = linear.weight.shape
        assert in_f % world == 0, f"in_features {in_f} % tp {world} != 0"
        self.shard = in_f // world
        self.group = group
        self.world = world
        self.input_is_parallel = input_is_parallel
        w = linear.weight.detach()[:, rank * self.shard : (rank + 1) * self.shard]
        self.weight = nn.Parameter(w.clone())
        # bias applied once (after the all-reduce), not per shard
        self.bias = nn.Parameter(linear.bias.detach().clone()) if linear.bias is not None else None
        self.rank = rank

    def forward(self, x):
        if not self.input_is_parallel:
            x = x[..., self.rank * self.shard : (self.rank + 1) * self.shard]
        y = F.linear(x, self.weight)
        y = _AllReduce.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


class QKVColumnParallelLinear(ColumnParallelLinear):
    """Column sharding for the FUSED [q|k|v] projection: each of the three
    D-row sections is sharded separately so every rank gets
    [q_shard | k_shard | v_shard] (a contiguous row slice would hand rank 0
    all of q plus part of k)."""

    def __init__(self, linear: nn.Linear, rank: int, world: int, group=None):
        nn.Module.__init__(self)
        out_f, in_f = linear.weight.shape
        assert out_f % (3 * world) == 0
        d = out_f // 3
        sh = d // world
        self.shard = 3 * sh
        self.group = group
        self.world = world
        self.gather_output = False
        rows = []
        for sec in range(3):
            lo = sec * d + rank * sh
            rows.append(linear.weight.detach()[lo : lo + sh])
        self.weight = nn.Parameter(torch.cat(rows).clone())
        if linear.bias is not None:
            bs = [linear.bias.detach()[sec * d + rank * sh : sec * d + rank * sh + sh]
                  for sec in range(3)]
            self.bias = nn.Parameter(torch.cat(bs).clone())
        else:
            self.bias = None


# The MPT layer plan (reference build_tp_strategies layer_plan).
MPT_TP_PLAN = {
    "attn.Wqkv": "qkv_column",
    "attn.out_proj": "row",
    "ffn.up_proj": "column",
    "ffn.down_proj": "row",
}


def apply_tensor_parallel(model: nn.Module, rank: int, world: int,
                          group=None, plan: dict | None = None) -> list[str]:
    """Swap linears per the plan. world==1 leaves the model untouched
    (the reference's revert-to-DDP-on-1-GPU, trainer_utils.py:1379-1393).

    Attention modules whose Wqkv gets column-sharded also get their head
    count and ALiBi slopes sliced to the rank's shard so the per-rank
    attention runs on local heads only."""
    if world <= 1:
        return []
    plan = plan or MPT_TP_PLAN
    replaced = []
    for name, module in list(model.named_modules()):
        for suffix, kind in plan.items():
            if name.endswith(suffix):
                parent = model.get_submodule(name.rsplit(".", 1)[0])
                attr = name.rsplit(".", 1)[1]
                lin = getattr(parent, attr)
                if kind == "qkv_column":
                    new = QKVColumnParallelLinear(lin, rank, world, group)
                elif kind == "column":
                    new = ColumnParallelLinear(lin, rank, world, group)
                else:
                    new = RowParallelLinear(lin, rank, world, group)
                setattr(parent, attr, new)
                replaced.append(name)
                # head-shard the owning attention module
                if attr == "Wqkv" and hasattr(parent, "n_heads"):
                    assert parent.n_heads % world == 0, (
                        f"n_heads {parent.n_heads} % tp {world} != 0"
                    )
                    hl = parent.n_heads // world
                    parent.n_heads = hl
                    parent.slopes = parent.slopes[rank * hl : (rank + 1) * hl]
    return replaced
