"""FSDP wrap — the reference's fsdp_config capability checkbox.

The reference defaults to FULL_SHARD with PURE mixed precision
(mpt-125m.yaml:84-91) and offers SHARD_GRAD_OP (photon_llm_125M.sh:114).
On MI355X the 288 GB HBM holds MPT-7B + full AdamW state unsharded, so the
federated default stays unsharded; apply_fsdp keeps the capability for
multi-GPU-per-client configs.
"""

from __future__ import annotations

import torch
import torch.nn as nn


def apply_fsdp(model: nn.Module, fsdp_config: dict | None, device=None):
    """Wrap MPT blocks with torch FSDP per the reference's fsdp_config.

    Returns the wrapped model; a falsy config returns the model untouched
    (the ~llm_config.fsdp_config DDP path, photon_llm_125M.sh:112).
    """
    if not fsdp_config:
        return model
    import torch.distributed as dist
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP
    from torch.distributed.fsdp import MixedPrecision, ShardingStrategy
    from torch.distributed.fsdp.wrap import ModuleWrapPolicy

    if not dist.is_initialized() or dist.get_world_size() <= 1:
        return model

    strategy = {
        "FULL_SHARD": ShardingStrategy.FULL_SHARD,
        "SHARD_GRAD_OP": ShardingStrategy.SHARD_GRAD_OP,
        "NO_SHARD": ShardingStrategy.NO_SHARD,
    }[str(fsdp_config.get("sharding_strategy", "FULL_SHARD")).upper()]
    mp = None
    if str(fsdp_config.get("mixed_precision", "PURE")).upper() == "PURE":
        mp = MixedPrecision(
            param_dtype=torch.bfloat16,
            reduce_dtype=torch.bfloat16,
            buffer_dtype=torch.bfloat16,
        )
    from ..models.mpt import MPTBlock

    return FSDP(
        model,
        sharding_strategy=strategy,
        mixed_precision=mp,
        auto_wrap_policy=ModuleWrapPolicy({MPTBlock}),
        device_id=device if device is not None and str(device) != "cpu" else None,
        use_orig_params=True,
    )
