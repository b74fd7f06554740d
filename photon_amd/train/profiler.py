"""Trace profiler — the reference's Composer Profiler + JSONTraceHandler
(trainer_utils.py:1456-1482) on torch.profiler-ROCm.

Config surface parity: ``train_cfg.profiler`` with a cyclic schedule
{wait, warmup, active, repeat} and a trace folder; produces Chrome traces
viewable in chrome://tracing / perfetto. Kernel-level numbers come from
rocprofv3 (profiles/ in this repo); this profiler covers the op/python
timeline the reference got from Composer.
"""

from __future__ import annotations

import contextlib
from pathlib import Path

import torch


def build_profiler(profiler_cfg: dict | None, folder: str | Path = "traces"):
    """Returns a torch.profiler.profile (or None) from the config subtree."""
    if not profiler_cfg:
        return None
    sched = profiler_cfg.get("schedule", {}) or {}
    folder = Path(profiler_cfg.get("folder", folder))
    folder.mkdir(parents=True, exist_ok=True)
    activities = [torch.profiler.ProfilerActivity.CPU]
    if torch.cuda.is_available():
        activities.append(torch.profiler.ProfilerActivity.CUDA)
    return torch.profiler.profile(
        activities=activities,
        schedule=torch.profiler.schedule(
            wait=int(sched.get("wait", 1)),
            warmup=int(sched.get("warmup", 1)),
            active=int(sched.get("active", 3)),
            repeat=int(sched.get("repeat", 1)),
        ),
        on_trace_ready=torch.profiler.tensorboard_trace_handler(str(folder)),
        record_shapes=bool(profiler_cfg.get("record_shapes", False)),
        profile_memory=bool(profiler_cfg.get("profile_memory", False)),
        with_stack=bool(profiler_cfg.get("with_stack", False)),
    )


@contextlib.contextmanager
def maybe_profile(profiler_cfg: dict | None, folder: str | Path = "traces"):
    prof = build_profiler(profiler_cfg, folder)
    if prof is None:
        yield None
        return
    with prof:
        yield prof
