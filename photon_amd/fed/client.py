"""Client-side local training/eval — the reference's llm_fit / llm_eval
(photon/clients/llm_client_functions.py:53-353) re-shaped for one process
per GPU (no NodeManager/Worker/SHM mailboxes — SURVEY.md §3.2 collapses to
a function call)."""

from __future__ import annotations

import time
from dataclasses import dataclass, field

import torch

from ..conf.schema import duration_to_batches
from ..data import build_eval_loader, build_train_loader
from ..models import build_model
from ..models.mpt import resize_vocab
from ..train import Trainer
from .flat import FlatParams
from .params_ops import (
    freeze_blocks,
    manipulate_pre_training,
    post_process_client_result,
    set_optimizer_state,
)


@dataclass
class ClientState:
    """Reference photon/utils.py:41-53."""

    cid: int
    steps_done: int = 0
    metrics: dict = field(default_factory=dict)


class FedClient:
    """A persistent per-rank client: owns the model + trainer across rounds
    (the reference's persistent-Trainer reuse path, trainer_utils.py:330-653),
    re-seeded per cid for multi-client-per-rank rounds."""

    def __init__(self, cfg, device, rank: int = 0, world_size_inner: int = 1):
        self.cfg = cfg
        self.device = torch.device(device)
        llm = cfg["llm_config"]
        torch.manual_seed(int(llm.get("seed", 17)))
        self.model = build_model(llm)
        rv = int(cfg.get("fl", {}).get("resize_vocab") or 0)
        if rv:
            resize_vocab(self.model, rv)
        self.trainer: Trainer | None = None
        self.rank = rank
        from pathlib import Path

        photon = cfg.get("photon", {}) or {}
        self.save_root = (
            Path(photon.get("saving_path") or "checkpoints")
            / str(cfg.get("run_uuid", "run"))
        )
        self.save_client_checkpoints = bool(photon.get("checkpoint", False))
        self.client_states: dict[int, ClientState] = {}
        # Per-client persistent trainer state (the reference restores the
        # client's timestamp + dataset_state at every fit,
        # llm_client_functions.py:163-175) so results are identical whether a
        # client runs on its own rank or shares one.
        self._timestamps: dict[int, dict] = {}
        self._loader_states: dict[int, dict] = {}
        # last local params per cid, kept only when personalized_layers is on
        self._personal: dict[int, torch.Tensor] = {}

    def _ensure_trainer(self, cid: int) -> Trainer:
        llm = self.cfg["llm_config"]
        if self.trainer is None:
            fl = self.cfg.get("fl", {})
            if fl.get("frozen_layers"):
                # fl.frozen/unfrozen_layers (photon/utils.py:322-387)
                freeze_blocks(self.model, fl.get("frozen_layers"),
                              fl.get("unfrozen_layers"))
            self.trainer = Trainer(
                self.model,
                llm,
                train_loader=build_train_loader(self.cfg, client_id=cid),
                eval_loader=build_eval_loader(self.cfg, client_id=cid),
                device=self.device,
                run_name=str(self.cfg.get("run_uuid", "run")),
                rank=self.rank,
            )
        else:
            # switch streams when this rank picks up a different client id
            self.trainer.train_loader = build_train_loader(self.cfg, client_id=cid)
            self.trainer.eval_loader = build_eval_loader(self.cfg, client_id=cid)
        return self.trainer

    def fit(
        self,
        cid: int,
        payload: torch.Tensor,
        layout: FlatParams,
        server_round: int,
        local_steps=None,
        reset_optimizer: bool = True,
    ) -> tuple[torch.Tensor, float, dict]:
        """Run local_steps batches from the incoming payload.

        payload is the flat global params, or [params | m1 | m2] when
        fl.aggregate_momenta. Returns (outgoing payload fp32, n_samples,
        metrics); n_samples = local_steps * global_train_batch_size
        (reference post_process_client_result, clients/utils.py:514-652).
        """
        llm = self.cfg["llm_config"]
        fl = self.cfg.get("fl", {})
        momenta = bool(fl.get("aggregate_momenta", False))
        # duration precedence: explicit arg > fl.n_local_steps > llm_config
        # (reference n_local_steps/local_steps duality, base_schema.py FL)
        if local_steps is None and int(fl.get("n_local_steps", 0) or 0) > 0:
            local_steps = int(fl["n_local_steps"])
        steps = duration_to_batches(
            local_steps if local_steps is not None else llm.get("local_steps", "500ba")
        )
        trainer = self._ensure_trainer(cid)
        # restore per-client timestamp + dataset state
        from ..train.timestamp import Timestamp

        trainer.timestamp = Timestamp()
        if cid in self._timestamps and not fl.get("reset_timestamp", False):
            trainer.timestamp.load_state_dict(self._timestamps[cid])
        if cid in self._loader_states and not fl.get("reset_dataset_state", False):
            trainer.train_loader.load_state_dict(self._loader_states[cid])

        # Mid-round resume: if a client checkpoint already holds exactly
        # steps_done + local_steps batches, load it and SKIP the fit
        # (reference llm_config_functions.py:642-764, clients/utils.py:217-228).
        ckpt_dir = self.save_root / f"client_{cid}"
        expect = trainer.timestamp.batch + steps
        # epoch-aware probe (not hard-coded ep0): the restored per-client
        # timestamp carries the epoch the checkpoint was written at
        ep = trainer.timestamp.epoch
        skip_path = ckpt_dir / f"ep{ep}-ba{expect}-rank{self.rank}.pt"
        if self.save_client_checkpoints and skip_path.exists():
            trainer.save_folder = ckpt_dir
            trainer.load_checkpoint(
                skip_path, load_ignore_keys=["*scheduler*"]
            )
            local_flat = torch.zeros_like(layout.flat)
            out_views = layout.layer_views_of(local_flat)
            params = dict(self.model.named_parameters())
            with torch.no_grad():
                for n, v in zip(layout.names, out_views):
                    v.copy_(params[n].detach().to(torch.float32))
            n_samples = float(steps * int(llm.get("global_train_batch_size", 256)))
            st = self.client_states.setdefault(cid, ClientState(cid))
            st.steps_done = trainer.timestamp.batch
            self._timestamps[cid] = trainer.timestamp.state_dict()
            momenta_out = bool(fl.get("aggregate_momenta", False))
            out_payload, pp_metrics = post_process_client_result(
                layout, local_flat, local_flat, n_samples,
                trainer=trainer, aggregate_momenta=momenta_out,
                report_layer_norms=False,
            )
            return out_payload, n_samples, {
                "steps_done": st.steps_done,
                "client/fit_skipped_from_checkpoint": 1.0,
                **pp_metrics,
            }

        t0 = time.time()
        global_flat, m1_in, m2_in = manipulate_pre_training(
            payload, layout, fl, cid, local_params=self._personal.get(cid),
            server_round=server_round,
        )
        # Optimizer reset FIRST: state.clear() would otherwise wipe the
        # exact-fp32 masters that sync_masters injects below (the round's
        # parameter set must not round-trip through bf16).
        if reset_optimizer:
            trainer.optimizer.state.clear()
        # set params from the global buffer (HBM->HBM copies, no host hop);
        # with bf16 weights the fp32 optimizer masters get the EXACT global
        # values (no bf16 round-trip)
        views = layout.layer_views_of(global_flat)
        params = dict(self.model.named_parameters())
        with torch.no_grad():
            if getattr(trainer, "master_weights", False):
                order = [params[n] for n in layout.names]
                trainer.optimizer.sync_masters(order, views)
            else:
                for n, v in zip(layout.names, views):
                    params[n].data.copy_(v.to(params[n].dtype))
        set_params_time = time.time() - t0

        if momenta and m1_in is not None:
            # import aggregated momenta + step for bias correction
            st_prev = self.client_states.get(cid)
            step = st_prev.steps_done if st_prev else self._max_steps_done()
            set_optimizer_state(trainer, layout, m1_in, m2_in, step=step)

        t1 = time.time()
        fit_metrics = trainer.fit(steps)
        fit_time = time.time() - t1

        t2 = time.time()
        local_flat = torch.zeros_like(global_flat)
        out_views = layout.layer_views_of(local_flat)
        params = dict(self.model.named_parameters())
        with torch.no_grad():
            for n, v in zip(layout.names, out_views):
                v.copy_(params[n].detach().to(torch.float32))
        get_params_time = time.time() - t2

        n_samples = float(steps * int(llm.get("global_train_batch_size", 256)))
        st = self.client_states.setdefault(cid, ClientState(cid))
        st.steps_done += steps
        self._timestamps[cid] = trainer.timestamp.state_dict()
        self._loader_states[cid] = trainer.train_loader.state_dict()
        if fl.get("personalized_layers"):
            self._personal[cid] = local_flat.clone()
        if self.save_client_checkpoints:
            # Composer-format client checkpoint (client_{cid}/ep{e}-ba{b}-
            # rank{r}.pt) — what the skip-and-load path above reads.
            trainer.save_checkpoint(ckpt_dir)

        out_payload, pp_metrics = post_process_client_result(
            layout, global_flat, local_flat, n_samples,
            trainer=trainer, aggregate_momenta=momenta,
            report_layer_norms=bool(fl.get("report_layer_norms", False)),
        )
        metrics = {
            "client/fit_set_parameters_time": set_params_time,
            "client/fit_time": fit_time,
            "client/fit_get_parameters_time": get_params_time,
            "loss/train/total": fit_metrics.get("loss/train/total", float("nan")),
            "steps_done": st.steps_done,
            **pp_metrics,
        }
        return out_payload, n_samples, metrics

    def _max_steps_done(self) -> int:
        return max((s.steps_done for s in self.client_states.values()), default=0)

    @torch.no_grad()
    def evaluate(
        self, cid: int, global_flat: torch.Tensor, layout: FlatParams,
        subset_num_batches: int = 8,
    ) -> tuple[float, float, dict]:
        """Returns (eval_loss, n_samples, metrics) — reference llm_eval."""
        trainer = self._ensure_trainer(cid)
        views = layout.layer_views_of(global_flat)
        params = dict(self.model.named_parameters())
        for n, v in zip(layout.names, views):
            params[n].data.copy_(v.to(params[n].dtype))
        metrics = trainer.eval(subset_num_batches)
        loss = metrics.get("metrics/eval/LanguageCrossEntropy", float("nan"))
        n = float(metrics.get("eval_samples", 0))
        if self.cfg.get("fl", {}).get("use_unigram_metrics", False):
            metrics.update(self._unigram_metrics(cid, trainer, loss,
                                                 subset_num_batches))
        return loss, n, metrics

    @torch.no_grad()
    def _unigram_metrics(self, cid: int, trainer, model_ce: float,
                         subset_num_batches: int) -> dict:
        """Unigram-normalized eval CE (reference unigram metrics path,
        SURVEY.md §5.5): model CE minus the unigram CE of the SAME labels,
        using the client's 1_gram.json written at dataset conversion.
        Failures are tolerated per fl.allow_unigram_metrics_failures."""
        import math
        from pathlib import Path

        from ..metrics import (
            PureUnigramCrossEntropy,
            load_freq_map,
            unigram_tensor_from_freq,
        )

        try:
            split_cfg = self.cfg["dataset"]["train"]
            root = Path(split_cfg.get("root_local") or "")
            freq_path = root / f"client_{cid}" / "1_gram.json"
            vocab = int(self.cfg["llm_config"]["model"].get("vocab_size", 50368))
            probs = unigram_tensor_from_freq(load_freq_map(freq_path), vocab)
            metric = PureUnigramCrossEntropy(probs)
            # same window the eval loop used
            loader = trainer.eval_loader
            state = loader.state_dict()
            nb = subset_num_batches if subset_num_batches > 0 else 8
            loader.load_state_dict(
                {"samples_consumed":
                 max(0, state["samples_consumed"] - nb * loader.batch_size)}
            )
            for _ in range(nb):
                metric.update(loader.next_batch()["input_ids"][:, 1:])
            unigram_ce = metric.compute()
            return {
                "metrics/eval/PureUnigramCrossEntropy": unigram_ce,
                "metrics/eval/UnigramNormalizedLanguageCrossEntropy":
                    model_ce - unigram_ce,
                "metrics/eval/UnigramNormalizedLanguagePerplexity":
                    math.exp(model_ce - unigram_ce),
            }
        except (OSError, KeyError, ValueError) as e:
            if self.cfg["fl"].get("allow_unigram_metrics_failures", True):
                return {}
            raise
