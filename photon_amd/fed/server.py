"""The federated round loop — reference server_app.main (photon/server_app.py:85-424)
re-designed for the RCCL-symmetric topology.

Every rank executes the same loop; there is no separate server process.
Rank 0 additionally owns checkpointing and history. Per round:

  1. sample clients       — replicated seeded RNG (no control message),
  2. local fit            — each rank trains its assigned client ids for
                            local_steps (the NodeManager work-queue collapses
                            to a per-rank loop),
  3. aggregation          — ONE RCCL all-reduce of the n_i/sum(n)-scaled flat
                            buffers (photon/server/fit_utils.py:41-217 +
                            strategy/aggregation.py in a single collective),
  4. server-opt update    — the FedOpt strategy applied redundantly on every
                            rank (deterministic => replicas stay bit-identical;
                            replaces the post-update re-broadcast at
                            server_app.py:327 with zero communication),
  5. eval every eval_period — weighted_loss_avg over ranks,
  6. rank-0 server checkpoint + retention cleanup.

Failure semantics parity (fit_utils.py:198-288): a client whose local fit
raises contributes zero weight; failure counts are all-reduced so every rank
agrees; more than accept_failures_cnt failures raises TooManyFailuresError
unless ignore_failed_rounds.
"""

from __future__ import annotations

import time
from pathlib import Path

import torch

from ..conf.schema import duration_to_batches
from ..history import History
from ..models import build_model
from .client import FedClient
from .flat import FlatParams
from .noise_scale import FedSimpleNoiseScale
from .runtime import Comm, assign_clients_to_ranks, sample_clients
from .params_ops import join_payload, split_payload
from .server_ckpt import (
    copy_old_checkpoints_to_new_run,
    delete_rounds,
    interpret_resume_round,
    load_client_momenta,
    obtain_sorted_rounds,
    resume_from_round,
    upload_server_checkpoint,
)
from .strategies import dispatch_strategy


class TooManyFailuresError(RuntimeError):
    pass


def weighted_loss_avg(losses_and_weights: list[tuple[float, float]]) -> float:
    total = sum(w for _, w in losses_and_weights)
    if total == 0:
        return float("nan")
    return sum(l * w for l, w in losses_and_weights) / total


class FedServer:
    """Holds the symmetric round-loop state for one rank."""

    def __init__(self, cfg, comm: Comm, device):
        self.cfg = cfg
        self.comm = comm
        self.device = torch.device(device)
        fl = cfg["fl"]
        self.n_total = int(fl["n_total_clients"])
        self.n_per_round = int(fl["n_clients_per_round"])
        self.n_rounds = int(fl["n_rounds"])
        self.eval_period = int(fl.get("eval_period", 1))
        self.accept_failures_cnt = int(fl.get("accept_failures_cnt", 0))
        self.ignore_failed_rounds = bool(fl.get("ignore_failed_rounds", False))
        self.seed = int(cfg.get("seed", 1337))
        self.run_uuid = str(cfg.get("run_uuid", "run"))
        self.saving_path = Path(
            cfg["photon"].get("saving_path") or "checkpoints"
        )
        self.checkpoint_enabled = bool(cfg["photon"].get("checkpoint", False))

        self.client = FedClient(cfg, self.device, rank=comm.rank)
        self.layout = FlatParams(
            self.client.model,
            filter_key=(
                str(fl.get("set_trainer_key_to_filter", "transformer"))
                if fl.get("set_trainer_params_filter_keys", True)
                else None
            ),
            device=self.device,
        )
        self.strategy = dispatch_strategy(
            fl.get("strategy_name", "NESTOROV"), self.layout, fl.get("strategy_kwargs")
        )
        self.noise_scale = (
            FedSimpleNoiseScale(beta=float(fl.get("noise_scale_beta", 0.99)))
            if fl.get("use_noise_scale_metric", False)
            else None
        )
        self.history = History(
            run_dir=self.saving_path / self.run_uuid if comm.rank == 0 else None,
            use_wandb=bool(cfg.get("use_wandb", False)),
            wandb_setup=dict(cfg.get("wandb", {}).get("setup", {})) if cfg.get("use_wandb") else None,
            suffix="_server",
        )
        self.server_steps_cumulative = 0
        self.start_round = 1
        # cumulative wall time across resumes (reference state.bin
        # time_offset, s3_utils.py:374-389)
        self.time_offset = 0.0
        self._t_started = time.time()
        # aggregated client momenta when fl.aggregate_momenta
        self.aggregate_momenta = bool(fl.get("aggregate_momenta", False))
        self.client_m1 = self.layout.like() if self.aggregate_momenta else None
        self.client_m2 = self.layout.like() if self.aggregate_momenta else None
        # hung-rank watchdog (photon.fit_timeout_s; SURVEY §7 hard-part 3):
        # opt-in TCPStore control plane beside the RCCL data plane
        self.fit_timeout_s = float(cfg["photon"].get("fit_timeout_s", 0) or 0)
        self.watchdog = None
        self._rebuild_gen = 0
        if self.fit_timeout_s > 0 and comm.is_distributed:
            from .watchdog import RoundWatchdog

            self.watchdog = RoundWatchdog(comm.rank, comm.world_size)

    # -- initialization / resume -------------------------------------------
    def initialize(self) -> None:
        """initialize_round / resume_from_round / restore
        (init_utils.py:128-287, server_app.py:159-219)."""
        resumed = None
        photon_cfg = self.cfg["photon"]
        # cross-run restore: copy an old run's latest complete server round
        # (+ client checkpoints) under this run's prefix before resuming
        restore_uuid = photon_cfg.get("restore_run_uuid")
        if restore_uuid and self.comm.rank == 0:
            copy_old_checkpoints_to_new_run(
                self.saving_path, str(restore_uuid), self.run_uuid,
                self.strategy.state_keys,
            )
        if restore_uuid:
            self.comm.barrier()
        resume_round = self.cfg["photon"].get("resume_round", -1)
        rounds = obtain_sorted_rounds(
            self.saving_path, self.run_uuid, self.strategy.state_keys
        )
        target = interpret_resume_round(resume_round, rounds)
        if target is not None:
            state = resume_from_round(
                self.saving_path, self.run_uuid, target, self.strategy, self.layout
            )
            self.server_steps_cumulative = int(state.get("server_steps_cumulative", 0))
            self.time_offset = float(state.get("time_offset", 0.0))
            self.history.load_state(state.get("history", {}))
            # client_state round-trips as str() in state.bin (reference
            # s3_utils.py:374-389); restore steps_done per client so lr
            # schedules and skip-and-load continue where they left off
            import ast

            from .client import ClientState

            raw = state.get("client_state")
            if raw:
                try:
                    parsed = ast.literal_eval(raw) if isinstance(raw, str) else raw
                    for cid, st in parsed.items():
                        self.client.client_states[int(cid)] = ClientState(
                            cid=int(cid),
                            steps_done=int(st.get("steps_done", 0)),
                            metrics=dict(st.get("metrics", {})),
                        )
                except (ValueError, SyntaxError):
                    pass
            # restore aggregated client momenta (fl.aggregate_momenta):
            # without this the first post-resume round would broadcast zero
            # momenta straight into the clients' Adam state
            if self.aggregate_momenta:
                cm = load_client_momenta(
                    self.saving_path, self.run_uuid, target, self.layout
                )
                if cm is not None:
                    self.client_m1.copy_(cm[0])
                    self.client_m2.copy_(cm[1])
            self.start_round = target + 1
            resumed = target
        else:
            # fresh init: rank 0's model init is the global model; broadcast
            # the flat buffer once (the only full-parameter broadcast).
            # restore_cent_run_uuid bootstraps from a centralized run's
            # final_parameters.npz (init_utils.py:43-125).
            cent_uuid = photon_cfg.get("restore_cent_run_uuid")
            if cent_uuid:
                # Composer-format .pt (ep{e}-ba{b}-rank0.pt) or the run's
                # final_parameters.npz (reference init_utils.py:43-125)
                from .server_ckpt import get_centralized_run_parameters

                flat = get_centralized_run_parameters(
                    self.saving_path, str(cent_uuid), self.layout,
                    photon_cfg.get("restore_cent_run_batches"),
                )
                self.layout.flat.copy_(flat)
            else:
                self.layout.copy_from_model(self.client.model)
            self.comm.broadcast_flat(self.layout.flat, src=0)
            self.strategy.initialize(self.layout.flat)
        # all ranks start from identical global params
        self.comm.broadcast_flat(self.strategy.params, src=0)
        if resumed is not None and self.comm.rank == 0:
            print(f"[fed] resumed from round {resumed}")

    # -- one round -----------------------------------------------------------
    def run_round(self, server_round: int) -> dict:
        t_round = time.time()
        fl = self.cfg["fl"]
        sampled = sample_clients(self.seed, server_round, self.n_total, self.n_per_round)
        assignment = assign_clients_to_ranks(sampled, self.comm.world_size)
        my_cids = assignment[self.comm.rank]

        outgoing = join_payload(self.strategy.params, self.client_m1, self.client_m2)
        local_sum = torch.zeros_like(outgoing)
        local_weight = 0.0
        failures = 0
        steps_done_max = 0
        fit_metrics: dict = {}
        per_client_sq_norms: list[tuple[float, float]] = []  # (n_i, ||g_i||^2)

        t_fit = time.time()
        suicide = None
        if self.watchdog is not None:
            # a rank whose own fit hangs must die so peers can proceed
            # (reference worker auto_terminate, worker.py:437-448)
            from .watchdog import fit_suicide_timer

            suicide = fit_suicide_timer(
                self.fit_timeout_s * max(len(my_cids), 1) + 30.0,
                self.comm.rank,
            )
        for cid in my_cids:
            try:
                local_payload, n_samples, metrics = self.client.fit(
                    cid,
                    outgoing,
                    self.layout,
                    server_round,
                    reset_optimizer=bool(fl.get("reset_optimizer", True))
                    and not self.aggregate_momenta,
                )
                local_sum.add_(local_payload, alpha=n_samples)
                local_weight += n_samples
                steps_done_max = max(steps_done_max, int(metrics.get("steps_done", 0)))
                fit_metrics = metrics
                if self.noise_scale is not None:
                    g = self.strategy.params - local_payload[: self.layout.total]
                    per_client_sq_norms.append(
                        (n_samples, float(torch.dot(g, g)))
                    )
            except Exception as e:  # failure budget semantics
                failures += 1
                if self.comm.rank == 0 or True:
                    print(f"[fed] client {cid} fit failed: {e!r}")
        fit_time = time.time() - t_fit
        if suicide is not None:
            suicide.cancel()

        # hung-rank detection BEFORE any collective: a dead rank must not
        # deadlock the all-reduce. Survivors agree on the alive set via the
        # TCPStore, rebuild the process group without the dead ranks, and
        # count their clients as failures.
        dead_clients = 0
        if self.watchdog is not None:
            import torch.distributed as dist

            from .watchdog import rebuild_process_group

            self.watchdog.report_fit_done(server_round)
            alive = self.watchdog.agree_alive(server_round, self.fit_timeout_s)
            if len(alive) < self.comm.world_size:
                dead = sorted(set(range(self.comm.world_size)) - set(alive))
                dead_clients = sum(len(assignment[r]) for r in dead)
                backend = dist.get_backend()
                self._rebuild_gen += 1
                new_rank, new_world = rebuild_process_group(
                    alive, self.comm.rank, backend,
                    generation=self._rebuild_gen,
                )
                print(f"[watchdog] round {server_round}: ranks {dead} dead; "
                      f"rebuilt group as rank {new_rank}/{new_world}",
                      flush=True)
                self.comm.rank, self.comm.world_size = new_rank, new_world
                # the watchdog itself keeps OLD rank numbering for its store
                # keys; re-key it to the new group
                self.watchdog.rank = new_rank
                self.watchdog.world_size = new_world
                # REQUEUE the dead ranks' clients onto the survivors
                # (reference node_manager_app.py:574-579: a failed worker's
                # cid goes back on the work list). Deterministic from the
                # agreed alive set, so every survivor runs the same split.
                requeue = [cid for r in dead for cid in assignment[r]]
                my_extra = [
                    cid for i, cid in enumerate(requeue)
                    if i % new_world == new_rank
                ]
                for cid in my_extra:
                    try:
                        local_payload, n_samples, metrics = self.client.fit(
                            cid, outgoing, self.layout, server_round,
                            reset_optimizer=bool(fl.get("reset_optimizer",
                                                        True))
                            and not self.aggregate_momenta,
                        )
                        local_sum.add_(local_payload, alpha=n_samples)
                        local_weight += n_samples
                        steps_done_max = max(
                            steps_done_max, int(metrics.get("steps_done", 0))
                        )
                        # recovered: the all-gather below subtracts it from
                        # the (globally identical) dead_clients count
                        failures -= 1
                    except Exception as e:
                        print(f"[fed] requeued client {cid} failed: {e!r}")

        # agree on failures across ranks: local failures minus local
        # recoveries, plus the dead ranks' client count (identical on every
        # survivor so added after the sum)
        fail_total = sum(self.comm.all_gather_scalars(float(failures)))
        fail_total += dead_clients
        if fail_total > self.accept_failures_cnt and not self.ignore_failed_rounds:
            raise TooManyFailuresError(
                f"round {server_round}: {int(fail_total)} client failures "
                f"(accept_failures_cnt={self.accept_failures_cnt})"
            )

        # ONE weighted all-reduce over xGMI (params and momenta together)
        t_agg = time.time()
        fedavg_payload, total_weight = self.comm.weighted_average_(local_sum, local_weight)
        agg_time = time.time() - t_agg
        if total_weight == 0:
            raise TooManyFailuresError(f"round {server_round}: no successful clients")
        fedavg_flat, m1_avg, m2_avg = split_payload(
            fedavg_payload, self.layout.total, self.aggregate_momenta
        )
        if self.aggregate_momenta:
            self.client_m1.copy_(m1_avg)
            self.client_m2.copy_(m2_avg)

        # debug mode (photon.debug_checks): runtime invariants in the spirit
        # of the reference's parameters_checker asserts —
        # replicated state must stay bit-identical across ranks.
        if self.cfg["photon"].get("debug_checks", False):
            check = self.strategy.params.clone()
            self.comm.broadcast_flat(check, src=0)
            from .params_ops import parameters_checker

            parameters_checker(check, self.strategy.params, equal=True)
            assert torch.isfinite(fedavg_flat).all(), "non-finite aggregate"

        # replicated server-opt update
        strat_metrics = self.strategy.update(
            fedavg_flat, server_round, len(sampled)
        )
        if self.noise_scale is not None:
            g_big_sq = float(strat_metrics.get("l2_norm_pseudo_gradient", 0.0)) ** 2
            ns = self.noise_scale.update(per_client_sq_norms, g_big_sq, self.comm)
            strat_metrics.update(ns)

        # steps bookkeeping: server_steps_cumulative += max(steps_done)
        steps_max_global = max(
            self.comm.all_gather_scalars(float(steps_done_max))
        )
        # steps_done is cumulative per client, so the global max IS the
        # cumulative server step count (fit_utils.py:179-183 semantics).
        self.server_steps_cumulative = int(steps_max_global)

        round_metrics = {
            "server/fit_round_time": fit_time,
            "server/aggregate_time": agg_time,
            "server/round_time": time.time() - t_round,
            "server/sampled_clients": len(sampled),
            "server/failures": fail_total,
            "server_steps_cumulative": self.server_steps_cumulative,
            **{k: v for k, v in strat_metrics.items() if not isinstance(v, list)},
            **{k: v for k, v in fit_metrics.items() if isinstance(v, (int, float))},
        }
        if self.comm.rank == 0:
            self.history.add_metrics_distributed(server_round, round_metrics)
        return round_metrics

    def evaluate_round(self, server_round: int) -> float:
        """Reference evaluate_round (evaluate_utils.py:232 +
        node_manager_app.py:594-725): EVERY sampled client is evaluated —
        each rank loops over its round-robin shard of the sampled cids,
        per-cid (loss, n, metrics) rows are all-gathered and aggregated with
        weighted_loss_avg. ``eval_subset_num_batches: -1`` means the full
        eval split (trainer semantics), not a coerced window."""
        subset = int(self.cfg["llm_config"].get("eval_subset_num_batches", -1))
        sampled = sample_clients(
            self.seed, server_round, self.n_total, self.n_per_round
        )
        assignment = assign_clients_to_ranks(sampled, self.comm.world_size)
        local_rows = []
        for cid in assignment[self.comm.rank]:
            loss, n, metrics = self.client.evaluate(
                cid, self.strategy.params, self.layout, subset
            )
            local_rows.append(
                (cid, float(loss), float(n),
                 {k: v for k, v in metrics.items()
                  if isinstance(v, (int, float))})
            )
        rows = [r for per_rank in self.comm.all_gather_obj(local_rows)
                for r in per_rank]
        avg = weighted_loss_avg([(l, n) for _, l, n, _ in rows])
        if self.comm.rank == 0:
            self.history.add_loss_distributed(server_round, avg)
            extra = {}
            if self.cfg["fl"].get("split_eval", False):
                extra = {
                    f"metrics/eval/LanguageCrossEntropy_client_{cid}": l
                    for cid, l, _, _ in rows
                }
            # unigram metric aggregation across clients (the gather above is
            # the RCCL analogue of torchmetrics dist_reduce_fx="sum")
            for key in (
                "metrics/eval/PureUnigramCrossEntropy",
                "metrics/eval/UnigramNormalizedLanguageCrossEntropy",
                "metrics/eval/UnigramNormalizedLanguagePerplexity",
            ):
                vals = [(m[key], n) for _, _, n, m in rows if key in m]
                if vals:
                    extra[key] = weighted_loss_avg(vals)
            # config-gated ICL / gauntlet evaluation — explicitly on the
            # CURRENT global params (strategy.params), never layout.flat
            # which is only written at fresh init
            icl_cfg = self.cfg.get("icl_tasks_config") or {}
            if icl_cfg.get("icl_tasks"):
                from ..centralised_train import run_icl_eval

                views = self.layout.layer_views_of(self.strategy.params)
                params = dict(self.client.model.named_parameters())
                with torch.no_grad():
                    for n_, v in zip(self.layout.names, views):
                        params[n_].data.copy_(v.to(params[n_].dtype))
                extra.update(
                    run_icl_eval(self.cfg, self.client.model, self.device)
                )
            self.history.add_metrics_distributed(
                server_round,
                {"metrics/eval/LanguageCrossEntropy_avg": avg, **extra},
            )
        return avg

    # -- full loop -----------------------------------------------------------
    def run(self, n_rounds: int | None = None) -> History:
        self.initialize()
        last = self.start_round + (n_rounds or self.n_rounds) - 1
        refresh = int(self.cfg["photon"].get("refresh_period", 0) or 0)
        for r in range(self.start_round, last + 1):
            if refresh > 0 and r > self.start_round and (r - 1) % refresh == 0:
                # prophylactic trainer refresh (reference worker respawn
                # every photon.refresh_period rounds, client_app.py:175-177)
                self.client.trainer = None
            self.run_round(r)
            if self.eval_period > 0 and r % self.eval_period == 0:
                self.evaluate_round(r)
            if self.checkpoint_enabled and self.comm.rank == 0:
                upload_server_checkpoint(
                    self.saving_path,
                    self.run_uuid,
                    r,
                    self.strategy,
                    self.layout,
                    self.history.state(),
                    {cid: vars(st) for cid, st in self.client.client_states.items()},
                    self.server_steps_cumulative,
                    time_offset=self.time_offset + (time.time() - self._t_started),
                    client_momenta=(
                        (self.client_m1, self.client_m2)
                        if self.aggregate_momenta
                        else None
                    ),
                )
                # per-round retention (reference cleanup_checkpoints_per_round,
                # server_app.py:403-405): keep the newest N complete rounds
                keep = int(self.cfg["photon"].get("save_num_rounds_to_keep", 0) or 0)
                if keep > 0:
                    rounds = obtain_sorted_rounds(
                        self.saving_path, self.run_uuid, self.strategy.state_keys
                    )
                    delete_rounds(self.saving_path, self.run_uuid, rounds[-keep:])
            self.comm.barrier()
        return self.history
