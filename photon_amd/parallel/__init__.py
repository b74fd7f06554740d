"""Parallelism strategies — MI355X-native over RCCL/xGMI.

* ``ddp``  — bucketed gradient all-reduce for the inner data-parallel degree
             (the reference's Composer DDP with FORCED_SYNC,
             trainer_utils.py:1714); per-link-bounded bucket sizing for xGMI.
* ``tp``   — column/row-parallel linears + the MPT layer plan (the
             reference's tp_config -> build_tp_strategies path,
             trainer_utils.py:1376-1398); reverts to plain modules at
             world_size 1 exactly like the reference.
* ``fsdp`` — FULL_SHARD/SHARD_GRAD_OP wrap of the MPT blocks
             (fsdp_config, mpt-125m.yaml:84-91). On MI355X the 288 GB HBM
             holds MPT-7B + AdamW state unsharded, so FSDP is an option,
             not a necessity.
"""

from .ddp import BucketedGradSync
from .tp import (
    ColumnParallelLinear,
    RowParallelLinear,
    apply_tensor_parallel,
)
from .fsdp import apply_fsdp

__all__ = [
    "BucketedGradSync",
    "ColumnParallelLinear",
    "RowParallelLinear",
    "apply_tensor_parallel",
    "apply_fsdp",
]
