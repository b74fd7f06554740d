"""RCCL environment tuning for the MI355X xGMI topology.

The 8-GPU node is point-to-point xGMI: 7 links x ~153 GB/s per GPU, no
switch. Ring all-reduce is per-link bound, so the wins come from (a) big
flat buffers (photon_amd aggregates the WHOLE model in one collective —
fed/flat.py) and (b) letting RCCL spread channels across links.

``apply_rccl_env(cfg)`` sets conservative defaults (only when unset) and
passes through any ``rccl.env`` mapping from the config — the reference's
"expose RCCL_* tuning in config" requirement.

Must run BEFORE the first collective (ideally before init_process_group).
"""

from __future__ import annotations

import os

# Defaults chosen for few, large collectives on a 7-link point-to-point
# topology; every one can be overridden by the environment or cfg.
_DEFAULTS = {
    # keep the dmabuf IPC mode the pool requires
    "HSA_ENABLE_IPC_MODE_LEGACY": "0",
    # larger per-channel buffers help the single ~0.5-26 GB all-reduce
    "NCCL_BUFFSIZE": str(8 << 20),
    # NCCL_MIN_NCHANNELS is deliberately NOT defaulted: RCCL's own channel
    # heuristics are already xGMI-aware on MI3xx; force it per-run via
    # cfg rccl.env when measurement says so.
}


def apply_rccl_env(cfg: dict | None = None) -> dict[str, str]:
    applied = {}
    for k, v in _DEFAULTS.items():
        if k not in os.environ:
            os.environ[k] = v
            applied[k] = v
    user = ((cfg or {}).get("rccl") or {}).get("env") or {}
    for k, v in user.items():
        os.environ[str(k)] = str(v)
        applied[str(k)] = str(v)
    return applied
