"""Server round checkpoints — bit-compatible with the reference layout.

Layout per round (photon/server/s3_utils.py:348-548; key names
photon/strategy/constants.py:4-6):

    {saving_path}/{run_uuid}/server/{round}/state.bin
        pickle: {server_round, history, time_offset, client_state (as str),
                 server_steps_cumulative}
    {saving_path}/{run_uuid}/server/{round}/current_server_parameters.npz
    {saving_path}/{run_uuid}/server/{round}/current_momentum_vector.npz
    {saving_path}/{run_uuid}/server/{round}/current_second_momentum_vector.npz

Resume: ``photon.resume_round`` (negative = index from the latest), rounds
enumerated by requiring all of the strategy's state_keys present
(s3_utils.py:215-272, 1261-1318). The store is the local FS here; the
``list_objects`` local-or-S3 abstraction survives in fed/store.py.
"""

from __future__ import annotations

import pickle
from pathlib import Path

import numpy as np
import torch

from .flat import FlatParams
from .strategies import STATE_M1, STATE_M2, STATE_PARAMS, Strategy

_KEY_FILES = {
    STATE_PARAMS: "current_server_parameters.npz",
    STATE_M1: "current_momentum_vector.npz",
    STATE_M2: "current_second_momentum_vector.npz",
}

# Aggregated CLIENT momenta (fl.aggregate_momenta): in the reference these
# ride inside the parameters ndarray list and are therefore saved in
# current_server_parameters.npz; the flat-buffer design splits them out into
# their own files. Optional — not part of state_keys, so old checkpoints
# stay resumable.
_CLIENT_MOMENTA_FILES = (
    "aggregated_client_momentum_vector.npz",
    "aggregated_client_second_momentum_vector.npz",
)


def server_dir(saving_path: str | Path, run_uuid: str) -> Path:
    return Path(saving_path) / str(run_uuid) / "server"


def upload_server_checkpoint(
    saving_path,
    run_uuid: str,
    server_round: int,
    strategy: Strategy,
    layout: FlatParams,
    history: dict,
    client_state: dict,
    server_steps_cumulative: int,
    time_offset: float = 0.0,
    client_momenta: tuple[torch.Tensor, torch.Tensor] | None = None,
) -> Path:
    rd = server_dir(saving_path, run_uuid) / str(server_round)
    rd.mkdir(parents=True, exist_ok=True)
    state = {
        "server_round": int(server_round),
        "history": history,
        "time_offset": float(time_offset),
        "client_state": str(client_state),
        "server_steps_cumulative": int(server_steps_cumulative),
    }
    with open(rd / "state.bin", "wb") as f:
        pickle.dump(state, f)
    for key, tensor in strategy.state_tensors().items():
        layout.save_npz(rd / _KEY_FILES[key], tensor)
    if client_momenta is not None:
        for fname, tensor in zip(_CLIENT_MOMENTA_FILES, client_momenta):
            layout.save_npz(rd / fname, tensor)
    return rd


def load_client_momenta(
    saving_path, run_uuid: str, server_round: int, layout: FlatParams,
) -> tuple[torch.Tensor, torch.Tensor] | None:
    """Restore aggregated client momenta saved by upload_server_checkpoint;
    None when the round predates aggregate_momenta persistence."""
    rd = server_dir(saving_path, run_uuid) / str(server_round)
    paths = [rd / f for f in _CLIENT_MOMENTA_FILES]
    if not all(p.exists() for p in paths):
        return None
    out = []
    for p in paths:
        arrays = layout.load_npz(p)
        out.append(
            torch.cat(
                [torch.from_numpy(np.ascontiguousarray(a, dtype=np.float32)).reshape(-1)
                 for a in arrays]
            ).to(layout.flat.device)
        )
    return out[0], out[1]


def obtain_sorted_rounds(saving_path, run_uuid: str, state_keys) -> list[int]:
    """Rounds with a complete checkpoint (all state_keys + state.bin),
    ascending (s3_utils.py:1261-1318)."""
    base = server_dir(saving_path, run_uuid)
    if not base.exists():
        return []
    rounds = []
    for d in base.iterdir():
        if not d.name.isdigit():
            continue
        need = [d / "state.bin"] + [d / _KEY_FILES[k] for k in state_keys]
        if all(p.exists() for p in need):
            rounds.append(int(d.name))
    return sorted(rounds)


def interpret_resume_round(resume_round: int | None, rounds: list[int]) -> int | None:
    """Reference semantics (s3_utils.py:215-272): positive = that round,
    negative = index from the latest (-1 = latest), None/0 or no rounds = no resume."""
    if not rounds or resume_round in (None, 0):
        return None
    if resume_round > 0:
        return resume_round if resume_round in rounds else None
    try:
        return rounds[resume_round]
    except IndexError:
        return None


def resume_from_round(
    saving_path, run_uuid: str, server_round: int, strategy: Strategy, layout: FlatParams
) -> dict:
    """Load strategy state + server state from a round dir; returns state.bin."""
    rd = server_dir(saving_path, run_uuid) / str(server_round)
    with open(rd / "state.bin", "rb") as f:
        state = pickle.load(f)
    tensors = {}
    for key in strategy.state_keys:
        arrays = layout.load_npz(rd / _KEY_FILES[key])
        flat = torch.cat(
            [torch.from_numpy(np.ascontiguousarray(a, dtype=np.float32)).reshape(-1) for a in arrays]
        ).to(strategy.params.device)
        tensors[key] = flat
    strategy.load_state_tensors(tensors)
    return state


def delete_rounds(saving_path, run_uuid: str, keep_rounds: list[int]) -> None:
    """Retention cleanup (s3_utils.py:1321-1641): remove round dirs not in keep."""
    import shutil

    base = server_dir(saving_path, run_uuid)
    if not base.exists():
        return
    for d in base.iterdir():
        if d.name.isdigit() and int(d.name) not in keep_rounds:
            shutil.rmtree(d)


def copy_old_checkpoints_to_new_run(
    saving_path, old_run_uuid: str, new_run_uuid: str, state_keys,
    copy_client_checkpoints: bool = True,
) -> int | None:
    """Cross-run restore (s3_utils.py:275-345,1478-1608): copy the latest
    complete server round (and client checkpoint dirs) into the new run.
    Returns the restored round or None."""
    import shutil

    rounds = obtain_sorted_rounds(saving_path, old_run_uuid, state_keys)
    if not rounds:
        return None
    latest = rounds[-1]
    src = server_dir(saving_path, old_run_uuid) / str(latest)
    dst = server_dir(saving_path, new_run_uuid) / str(latest)
    dst.mkdir(parents=True, exist_ok=True)
    for f in src.iterdir():
        shutil.copy2(f, dst / f.name)
    if copy_client_checkpoints:
        old_base = Path(saving_path) / str(old_run_uuid)
        new_base = Path(saving_path) / str(new_run_uuid)
        for cdir in old_base.glob("client_*"):
            shutil.copytree(cdir, new_base / cdir.name, dirs_exist_ok=True)
    return latest


def get_centralized_run_parameters(
    saving_path, cent_run_uuid: str, layout: FlatParams,
    desired_batches: int | None = None,
) -> torch.Tensor:
    """Bootstrap federated params from a CENTRALIZED run's Composer-format
    checkpoint (reference get_centralized_run_parameters,
    photon/server/init_utils.py:43-125): enumerate ep{e}-ba{b}-rank0.pt
    files under the run dir, pick the one whose batch count matches
    ``photon.restore_cent_run_batches`` (or the latest when unset), load the
    model state dict ignoring optimizer/scheduler/dataset state, and return
    the flat fp32 buffer in the wire order. Falls back to the run's
    final_parameters.npz when no .pt checkpoint exists."""
    import re

    run_dir = Path(saving_path) / str(cent_run_uuid)
    pairs = []
    for p in run_dir.rglob("ep*-ba*-rank0.pt"):
        m = re.search(r"ep(\d+)-ba(\d+)-rank0\.pt$", p.name)
        if m:
            pairs.append((int(m.group(2)), int(m.group(1)), p))
    if not pairs:
        npz = run_dir / "final_parameters.npz"
        if npz.exists():
            arrays = layout.load_npz(npz)
            flat = torch.cat([
                torch.from_numpy(np.ascontiguousarray(a, dtype=np.float32))
                .reshape(-1) for a in arrays
            ]).to(layout.flat.device)
            return flat
        raise FileNotFoundError(
            f"no Composer checkpoint or final_parameters.npz under {run_dir}"
        )
    pairs.sort()
    if desired_batches is not None:
        match = [p for p in pairs if p[0] == int(desired_batches)]
        if not match:
            raise ValueError(
                f"no checkpoint with {desired_batches} batches in {run_dir} "
                f"(have {[b for b, _, _ in pairs]})"
            )
        _, _, path = match[0]
    else:
        _, _, path = pairs[-1]
    ckpt = torch.load(path, map_location="cpu", weights_only=False)
    model_sd = ckpt["state"]["model"]
    flat = torch.zeros_like(layout.flat)
    views = layout.layer_views_of(flat)
    for name, view in zip(layout.names, views):
        if name not in model_sd:
            raise KeyError(f"parameter {name} missing from {path}")
        view.copy_(model_sd[name].detach().to(torch.float32).reshape(view.shape))
    return flat
