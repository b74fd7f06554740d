"""Hung-rank detection and communicator rebuild — SURVEY §7 hard-part 3.

The reference handles hung/dead training processes by NodeManager-side
monitoring: a failing worker re-enqueues its task and suicides
(photon/worker/worker.py:437-448), and dead workers are restarted before
each task (photon/node_manager/node_manager_app.py:326-351). In the
RCCL-symmetric design there is no manager process — a rank hung inside its
local fit would deadlock every other rank at the round's all-reduce.

MI355X-native equivalent, built from two pieces:

1. **Fit-phase watchdog** (:class:`RoundWatchdog`): a tiny TCPStore control
   plane beside the RCCL data plane. Each rank posts ``fit.{round}.{rank}``
   when its local fits finish; ranks then poll for everyone's key up to
   ``fit_timeout_s``. The first rank to win an atomic ``store.add`` becomes
   the decider and publishes the agreed-alive set; a rank NOT in the set
   must self-terminate (the reference's auto_terminate suicide). Survivors
   rebuild the torch.distributed process group *without* the dead ranks and
   finish the round — dead ranks' clients count as failures against
   ``accept_failures_cnt`` (fit_utils.py:198-288 semantics).

2. **Local suicide timer** (:func:`fit_suicide_timer`): a rank whose OWN fit
   hangs (stuck kernel) cannot run recovery code on the hung thread; a
   monitor thread hard-exits the process after the timeout so peers see a
   clean death instead of a zombie (worker.py:437-448 analogue).

The store master is hosted by rank 0; rank-0 death is unrecoverable (as is
server death in the reference). Single-node 127.0.0.1 topology means any
surviving rank can host the REBUILT group's store.
"""

from __future__ import annotations

import datetime
import os
import threading
import time

import torch.distributed as dist


class RankDeclaredDeadError(RuntimeError):
    """This rank was declared dead by the survivors' agreement (it posted
    its fit result after the decider sealed the alive set). The only safe
    action is to exit: the survivors' rebuilt communicator excludes us."""


def fit_suicide_timer(timeout_s: float, rank: int):
    """Arm a hard-exit timer around a local fit; returns a cancel handle.

    threading.Timer (not signal-based) so it works off the main thread and
    under torchrun. os._exit skips atexit/finalizers on purpose — the
    process state after a hung HIP kernel is not worth unwinding.
    """
    def _boom():
        print(f"[watchdog] rank {rank}: local fit exceeded {timeout_s}s — "
              "self-terminating (reference worker auto_terminate)",
              flush=True)
        os._exit(3)

    t = threading.Timer(timeout_s, _boom)
    t.daemon = True
    t.start()
    return t


class RoundWatchdog:
    """Fit-phase liveness agreement over a TCPStore control plane."""

    def __init__(self, rank: int, world_size: int,
                 master_addr: str | None = None, port: int | None = None,
                 store: "dist.TCPStore | None" = None):
        self.rank = rank
        self.world_size = world_size
        self.addr = master_addr or os.environ.get("MASTER_ADDR", "127.0.0.1")
        if store is not None:
            self.store = store
        else:
            if port is None:
                port = int(os.environ.get("MASTER_PORT", "29500")) + 7
            self.port = port
            self.store = dist.TCPStore(
                self.addr, port, world_size, is_master=(rank == 0),
                timeout=datetime.timedelta(seconds=300),
            )

    # -- per-round protocol --------------------------------------------------
    def report_fit_done(self, server_round: int) -> None:
        self.store.set(f"fit.{server_round}.{self.rank}", "done")

    def agree_alive(self, server_round: int, timeout_s: float,
                    poll_s: float = 0.2) -> list[int]:
        """Poll for every rank's fit key until all present or timeout; then
        the first rank to win the atomic counter publishes the final set.

        Returns the agreed sorted alive ranks. Raises RankDeclaredDeadError
        if this rank is not in the set.
        """
        deadline = time.monotonic() + timeout_s
        pending = set(range(self.world_size))
        pending.discard(self.rank)  # we are here
        while pending and time.monotonic() < deadline:
            # store.check is non-blocking (get would block for the store
            # timeout on a missing key)
            for r in sorted(pending):
                if self.store.check([f"fit.{server_round}.{r}"]):
                    pending.discard(r)
            if pending:
                time.sleep(poll_s)
        observed = sorted(set(range(self.world_size)) - pending)
        if len(observed) == self.world_size:
            return observed
        # somebody is missing: first claimer seals the set
        claim = self.store.add(f"decider.{server_round}", 1)
        key = f"alive.{server_round}"
        if claim == 1:
            self.store.set(key, ",".join(str(r) for r in observed))
            final = observed
        else:
            raw = self.store.get(key).decode()
            final = sorted(int(x) for x in raw.split(",") if x != "")
        if self.rank not in final:
            raise RankDeclaredDeadError(
                f"rank {self.rank} excluded from round {server_round} "
                f"alive set {final}"
            )
        return final


def rebuild_process_group(alive: list[int], old_rank: int, backend: str,
                          base_port: int | None = None,
                          generation: int = 1,
                          timeout_s: float = 300.0) -> tuple[int, int]:
    """Re-form torch.distributed with only the surviving ranks.

    All survivors call this with the same agreed ``alive`` list. The old
    group is destroyed; a fresh TCPStore-backed group of size len(alive) is
    initialized with ranks renumbered by position. The store is hosted by
    the lowest surviving rank (single-node: every rank can bind 127.0.0.1).
    Returns (new_rank, new_world_size).
    """
    new_rank = alive.index(old_rank)
    new_world = len(alive)
    addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    if base_port is None:
        base_port = int(os.environ.get("MASTER_PORT", "29500"))
    port = base_port + 100 + generation  # deterministic fresh port per rebuild
    if dist.is_initialized():
        dist.destroy_process_group()
    store = dist.TCPStore(
        addr, port, new_world, is_master=(new_rank == 0),
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    dist.init_process_group(
        backend, store=store, rank=new_rank, world_size=new_world,
        timeout=datetime.timedelta(seconds=timeout_s),
    )
    return new_rank, new_world
