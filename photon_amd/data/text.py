"""Dataloader construction from the dataset config subtree.

The analogue of the reference's stream-config munging
(photon/clients/llm_config_functions.py:388-529): each federated client gets
its own stream (``dataset/streams/8_clients.yaml``); ``client_id=None`` (the
centralized path) concatenates all streams. Falls back to synthetic
C4-shaped tokens when the stream's ``local`` shard directory does not exist
or the dataset preset sets ``synthetic: true`` — the BASELINE bench path.
"""

from __future__ import annotations

from pathlib import Path

from .shards import StatefulLoader, TokenShardDataset
from .synthetic import SyntheticTokenDataset


def _stream_local(stream_group_entry: dict) -> str | None:
    (stream_cfg,) = stream_group_entry["client_streams"].values()
    return stream_cfg.get("local")


def _build_dataset(ds_cfg: dict, split_cfg: dict, seq_len: int, client_id, split, seed,
                   vocab_size: int = 50368):
    streams = split_cfg.get("streams")
    synthetic = bool(split_cfg.get("synthetic", False))
    local = None
    if streams and client_id is not None and client_id < len(streams):
        local = _stream_local(streams[client_id])
    elif streams and client_id is None:
        local = _stream_local(streams[0])
    root = split_cfg.get("root_local") or ""
    if local is not None and not synthetic:
        path = Path(root) / local / str(split_cfg.get("split", split))
        if not path.exists():
            path = Path(local)
        if (path / "index.json").exists():
            return TokenShardDataset(
                path,
                seq_len,
                shuffle=bool(split_cfg.get("shuffle", False)),
                shuffle_seed=int(split_cfg.get("shuffle_seed", 9176) or 9176),
            )
    return SyntheticTokenDataset(
        seq_len,
        vocab_size=vocab_size,
        seed=seed,
        client_id=client_id if client_id is not None else 0,
        split=str(split_cfg.get("split", split)),
    )


def build_train_loader(cfg, client_id=None, batch_size=None) -> StatefulLoader:
    llm = cfg["llm_config"]
    seq_len = int(llm["max_seq_len"])
    split_cfg = cfg["dataset"]["train"]
    bs = batch_size or int(llm.get("device_train_microbatch_size", 8))
    ds = _build_dataset(cfg["dataset"], split_cfg, seq_len, client_id, "train",
                        int(cfg.get("seed", 1337)),
                        vocab_size=int(llm["model"].get("vocab_size", 50368)))
    return StatefulLoader(ds, bs)


def build_eval_loader(cfg, client_id=None, batch_size=None) -> StatefulLoader:
    llm = cfg["llm_config"]
    seq_len = int(llm["max_seq_len"])
    split_cfg = cfg["dataset"]["val"]
    bs = batch_size or int(llm.get("device_eval_batch_size", 8))
    ds = _build_dataset(cfg["dataset"], split_cfg, seq_len, client_id, "validation",
                        int(cfg.get("seed", 1337)),
                        vocab_size=int(llm["model"].get("vocab_size", 50368)))
    return StatefulLoader(ds, bs)
