"""Parameter-payload operations for the federated pipeline.

Re-implements (MI355X-native, flat-tensor based) the reference's client
parameter/state utilities:

* ``manipulate_pre_training`` — split an incoming [params | m1 | m2]
  payload when ``fl.aggregate_momenta``, personalize / randomize layers
  (photon/clients/utils.py:405-511);
* ``set_optimizer_state`` — inject aggregated momenta + step into the local
  optimizer (photon/clients/utils.py:257-402);
* ``post_process_client_result`` — per-layer pseudo-gradient norms, momenta
  append (photon/clients/utils.py:514-652);
* ``freeze_blocks`` (photon/utils.py:322-387), ``randomize_layers``
  (photon/clients/utils.py:871), ``personalize_layers`` (:950);
* ``parameters_checker`` equality/inequality assertions
  (photon/utils.py:147-224) and the l2-norm helpers (photon/utils.py:819-908).

All operations act on the HBM-resident flat fp32 buffers of
:class:`photon_amd.fed.flat.FlatParams` — no host ndarray hops.
"""

from __future__ import annotations

import fnmatch
import math

import torch

from .flat import FlatParams


# ---------------------------------------------------------------------------
# Norms (photon/utils.py:819-908)
# ---------------------------------------------------------------------------
def l2_norm(flat: torch.Tensor) -> float:
    return float(torch.linalg.vector_norm(flat))


def sum_of_squares(flat: torch.Tensor) -> float:
    return float(torch.dot(flat.reshape(-1), flat.reshape(-1)))


def layer_l2_norms(layout: FlatParams, flat: torch.Tensor) -> dict[str, float]:
    """Per-layer L2 norms of a flat buffer (per-layer pseudo-gradient norm
    reporting, clients/utils.py:599-619 / fedavg_eff.py:308-364)."""
    return {
        name: float(torch.linalg.vector_norm(view))
        for name, view in zip(layout.names, layout.layer_views_of(flat))
    }


def l2_norm_of_momenta(m1: torch.Tensor, m2: torch.Tensor) -> dict[str, float]:
    return {
        "l2_norm_first_momentum": l2_norm(m1),
        "l2_norm_second_momentum": l2_norm(m2),
    }


# ---------------------------------------------------------------------------
# parameters_checker (photon/utils.py:147-224)
# ---------------------------------------------------------------------------
def parameters_checker(
    a: torch.Tensor, b: torch.Tensor, equal: bool = True, atol: float = 0.0
) -> None:
    """Assert two flat buffers are (not) equal — the reference's runtime
    invariant checks around every parameter set."""
    if a.shape != b.shape:
        raise AssertionError(f"shape mismatch: {a.shape} vs {b.shape}")
    same = torch.allclose(a, b, atol=atol, rtol=0.0)
    if equal and not same:
        diff = float((a - b).abs().max())
        raise AssertionError(f"parameters differ (max abs diff {diff:.3e})")
    if not equal and same:
        raise AssertionError("parameters unexpectedly identical")


# ---------------------------------------------------------------------------
# Payload split/join for aggregate_momenta
# ---------------------------------------------------------------------------
def join_payload(params: torch.Tensor, m1: torch.Tensor | None,
                 m2: torch.Tensor | None) -> torch.Tensor:
    """[params | m1 | m2] concatenation (the reference appends two momenta
    copies to the ndarray list when aggregate_momenta)."""
    if m1 is None:
        return params
    return torch.cat([params, m1, m2])


def split_payload(payload: torch.Tensor, total: int, momenta: bool):
    """Inverse of join_payload: (params, m1|None, m2|None)."""
    if not momenta:
        return payload, None, None
    assert payload.numel() == 3 * total, (
        f"momenta payload expected {3 * total} elements, got {payload.numel()}"
    )
    return payload[:total], payload[total : 2 * total], payload[2 * total :]


# ---------------------------------------------------------------------------
# Optimizer momenta import/export on flat buffers
# ---------------------------------------------------------------------------
@torch.no_grad()
def set_optimizer_state(
    trainer, layout: FlatParams, m1_flat: torch.Tensor, m2_flat: torch.Tensor,
    step: int | None = None,
) -> None:
    """Inject aggregated momenta into the optimizer, aligned to the wire
    order, with the step counter for bias correction
    (photon/clients/utils.py:257-402)."""
    params = dict(trainer.model.named_parameters())
    order = [params[n] for n in layout.names]
    m1_views = layout.layer_views_of(m1_flat)
    m2_views = layout.layer_views_of(m2_flat)
    trainer.optimizer.import_momenta(order, m1_views, m2_views, step=step)


@torch.no_grad()
def get_optimizer_momenta(
    trainer, layout: FlatParams
) -> tuple[torch.Tensor, torch.Tensor]:
    """Export optimizer momenta as flat fp32 buffers in the wire order."""
    params = dict(trainer.model.named_parameters())
    order = [params[n] for n in layout.names]
    m1_list, m2_list = trainer.optimizer.export_momenta(order)
    m1 = torch.zeros_like(layout.flat)
    m2 = torch.zeros_like(layout.flat)
    for view, src in zip(layout.layer_views_of(m1), m1_list):
        view.copy_(src.view_as(view).to(torch.float32))
    for view, src in zip(layout.layer_views_of(m2), m2_list):
        view.copy_(src.view_as(view).to(torch.float32))
    return m1, m2


# ---------------------------------------------------------------------------
# Layer selection: freeze / personalize / randomize
# ---------------------------------------------------------------------------
def _pattern_matches(name: str, p: str) -> bool:
    """Anchored pattern match: exact name, glob, or ``.``-boundary
    substring. Boundary anchoring means 'blocks.1' matches
    'transformer.blocks.1.attn.Wqkv.weight' but NOT 'blocks.10.…', and a
    bare index like '1' only matches a whole dotted component."""
    if p == name:
        return True
    if any(ch in p for ch in "*?["):
        return fnmatch.fnmatch(name, p)
    start = 0
    while True:
        i = name.find(p, start)
        if i < 0:
            return False
        before_ok = i == 0 or name[i - 1] == "."
        j = i + len(p)
        after_ok = j == len(name) or name[j] == "."
        if before_ok and after_ok:
            return True
        start = i + 1


def _match_names(names: list[str], patterns) -> list[str]:
    """Names matching any pattern (reference passes exact names,
    substrings, or block indices — anchored here, see _pattern_matches)."""
    pats = [str(p) for p in (patterns or [])]
    return [n for n in names if any(_pattern_matches(n, p) for p in pats)]


def freeze_blocks(model: torch.nn.Module, frozen, unfrozen=None) -> list[str]:
    """Set requires_grad by name patterns with the reference's semantics
    (photon/utils.py:368-387): a param is frozen when it matches ``frozen``
    OR when ``unfrozen`` is given and it does NOT match ``unfrozen``
    (i.e. providing unfrozen_layers freezes the complement)."""
    names = [n for n, _ in model.named_parameters()]
    to_freeze = set(_match_names(names, frozen)) if frozen else set()
    keep = set(_match_names(names, unfrozen)) if unfrozen else None
    touched = []
    for n, p in model.named_parameters():
        if not p.requires_grad:
            continue
        if n in to_freeze or (keep is not None and n not in keep):
            p.requires_grad_(False)
            touched.append(n)
    return touched


@torch.no_grad()
def randomize_layers(
    layout: FlatParams, flat: torch.Tensor, patterns, seed: int, std: float = 0.02
) -> list[str]:
    """Re-initialize selected layers of an incoming payload
    (photon/clients/utils.py:871): deterministic per seed."""
    chosen = _match_names(layout.names, patterns)
    gen = torch.Generator(device="cpu").manual_seed(seed)
    for name, view in zip(layout.names, layout.layer_views_of(flat)):
        if name in chosen:
            r = torch.randn(view.shape, generator=gen) * std
            view.copy_(r.to(view.device))
    return chosen


@torch.no_grad()
def personalize_layers(
    layout: FlatParams, incoming: torch.Tensor, local: torch.Tensor, patterns
) -> list[str]:
    """Keep the client's OWN weights for selected layers instead of the
    broadcast global ones (photon/clients/utils.py:950)."""
    chosen = _match_names(layout.names, patterns)
    in_views = layout.layer_views_of(incoming)
    local_views = layout.layer_views_of(local)
    for name, iv, lv in zip(layout.names, in_views, local_views):
        if name in chosen:
            iv.copy_(lv)
    return chosen


# ---------------------------------------------------------------------------
# manipulate_pre_training (photon/clients/utils.py:405-511)
# ---------------------------------------------------------------------------
def manipulate_pre_training(
    payload: torch.Tensor,
    layout: FlatParams,
    fl_cfg: dict,
    cid: int,
    local_params: torch.Tensor | None = None,
    server_round: int | None = None,
):
    """Split + transform the incoming payload before local training.

    Returns (params, m1|None, m2|None). Applies personalized/random layer
    substitution per the fl config.
    """
    momenta = bool(fl_cfg.get("aggregate_momenta", False))
    params, m1, m2 = split_payload(payload, layout.total, momenta)
    params = params.clone()
    rand = fl_cfg.get("random_layers")
    freq = int(fl_cfg.get("random_init_freq", 0) or 0)
    due = freq <= 0 or (server_round is not None and server_round % freq == 0)
    if rand and due:
        # truly_random_init: fresh randomness each round; else the same
        # deterministic per-client init every time (reference flag pair)
        seed = int(fl_cfg.get("seed", 0)) + cid
        if fl_cfg.get("truly_random_init", True) and server_round is not None:
            seed += server_round * 10007
        randomize_layers(layout, params, rand, seed=seed)
    pers = fl_cfg.get("personalized_layers")
    if pers and local_params is not None:
        personalize_layers(layout, params, local_params, pers)
    return params, m1, m2


# ---------------------------------------------------------------------------
# post_process_client_result (photon/clients/utils.py:514-652)
# ---------------------------------------------------------------------------
def post_process_client_result(
    layout: FlatParams,
    global_params: torch.Tensor,
    local_params: torch.Tensor,
    n_samples: float,
    trainer=None,
    aggregate_momenta: bool = False,
    report_layer_norms: bool = True,
) -> tuple[torch.Tensor, dict]:
    """Build the outgoing payload + metrics: per-layer pseudo-gradient
    norms, momenta appended when aggregate_momenta."""
    metrics: dict = {"n_samples": n_samples}
    pseudo_grad = global_params - local_params
    metrics["l2_norm_pseudo_gradient_client"] = l2_norm(pseudo_grad)
    if report_layer_norms:
        metrics["layer_pseudo_grad_norms"] = layer_l2_norms(layout, pseudo_grad)
    payload = local_params
    if aggregate_momenta:
        assert trainer is not None
        m1, m2 = get_optimizer_momenta(trainer, layout)
        metrics.update(l2_norm_of_momenta(m1, m2))
        payload = join_payload(local_params, m1, m2)
    return payload, metrics
