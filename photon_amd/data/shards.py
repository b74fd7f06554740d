"""Pre-tokenized token-shard datasets with deterministic resumable state.

Replaces the reference's mosaicml-streaming MDS reader (conf/dataset/fed-c4.yaml,
SURVEY.md §2.3): shards are flat little-endian uint32 token files plus an
``index.json``, written per client by photon_amd.data.convert (the analogue of
photon/dataset/convert_dataset_hf.py). Samples are consecutive
``max_seq_len``-token windows in deterministic shuffled order; the loader's
``state_dict()/load_state_dict()`` reproduces the reference's
``dataset_state`` resume semantics (SURVEY.md §7 hard part 5).
"""

from __future__ import annotations

import json
from pathlib import Path

import numpy as np
import torch


class TokenShardWriter:
    def __init__(self, out_dir: str | Path, tokens_per_shard: int = 1 << 24):
        self.out_dir = Path(out_dir)
        self.out_dir.mkdir(parents=True, exist_ok=True)
        self.tokens_per_shard = tokens_per_shard
        self._buf: list[np.ndarray] = []
        self._buf_len = 0
        self._shards: list[dict] = []

    def write(self, tokens: np.ndarray) -> None:
        tokens = np.asarray(tokens, dtype=np.uint32)
        self._buf.append(tokens)
        self._buf_len += len(tokens)
        while self._buf_len >= self.tokens_per_shard:
            self._flush_shard()

    def _flush_shard(self) -> None:
        take = min(self._buf_len, self.tokens_per_shard)
        chunks, got = [], 0
        while got < take and self._buf:
            c = self._buf[0]
            need = take - got
            if len(c) <= need:
                chunks.append(c)
                got += len(c)
                self._buf.pop(0)
            else:
                chunks.append(c[:need])
                self._buf[0] = c[need:]
                got += need
        self._buf_len -= got
        data = np.concatenate(chunks) if len(chunks) > 1 else chunks[0]
        name = f"shard_{len(self._shards):05d}.bin"
        data.tofile(self.out_dir / name)
        self._shards.append({"file": name, "num_tokens": int(len(data))})

    def close(self) -> None:
        while self._buf_len > 0:
            self._flush_shard()
        index = {
            "format": "photon_amd_tokens_v1",
            "dtype": "uint32",
            "num_tokens": int(sum(s["num_tokens"] for s in self._shards)),
            "shards": self._shards,
        }
        with open(self.out_dir / "index.json", "w") as f:
            json.dump(index, f, indent=1)


class TokenShardDataset:
    """Map-style dataset of [seq_len] uint32 windows over the shard stream."""

    def __init__(self, local: str | Path, seq_len: int, shuffle: bool = False,
                 shuffle_seed: int = 9176):
        self.dir = Path(local)
        with open(self.dir / "index.json") as f:
            self.index = json.load(f)
        self.seq_len = seq_len
        self._maps = [
            np.memmap(self.dir / s["file"], dtype=np.uint32, mode="r")
            for s in self.index["shards"]
        ]
        self._offsets = np.cumsum([0] + [s["num_tokens"] for s in self.index["shards"]])
        self.num_samples = self.index["num_tokens"] // seq_len
        self.shuffle = shuffle
        self.shuffle_seed = shuffle_seed
        self._order_epoch = -1
        self._order = np.arange(self.num_samples)

    def _order_for(self, epoch: int) -> np.ndarray:
        # Deterministic PER-EPOCH reshuffle (mosaicml-streaming's
        # epoch-seeded shuffle semantics): order = perm(seed + epoch),
        # reproducible from any resume point.
        if not self.shuffle:
            return self._order
        if epoch != self._order_epoch:
            rng = np.random.default_rng(self.shuffle_seed + epoch)
            self._order = rng.permutation(self.num_samples)
            self._order_epoch = epoch
        return self._order

    def __len__(self) -> int:
        return self.num_samples

    def __getitem__(self, i: int) -> torch.Tensor:
        epoch = int(i) // self.num_samples
        sample = int(self._order_for(epoch)[int(i) % self.num_samples])
        start = sample * self.seq_len
        end = start + self.seq_len
        # locate shard(s); windows can straddle shard boundaries
        out = np.empty(self.seq_len, dtype=np.uint32)
        got = 0
        while got < self.seq_len:
            pos = start + got
            si = int(np.searchsorted(self._offsets, pos, side="right")) - 1
            local = pos - self._offsets[si]
            take = min(self.seq_len - got, len(self._maps[si]) - local)
            out[got : got + take] = self._maps[si][local : local + take]
            got += take
        return torch.from_numpy(out.astype(np.int64))


class StatefulLoader:
    """Deterministic, resumable batch loader (the ``dataset_state`` contract).

    Yields {"input_ids": LongTensor[B, S]} forever (wrapping the dataset),
    tracking ``samples_consumed`` for checkpoint/resume.
    """

    def __init__(self, dataset, batch_size: int, drop_last: bool = True):
        self.dataset = dataset
        self.batch_size = batch_size
        self.samples_consumed = 0

    def state_dict(self) -> dict:
        return {"samples_consumed": self.samples_consumed}

    def load_state_dict(self, state: dict) -> None:
        self.samples_consumed = int(state.get("samples_consumed", 0))

    def next_batch(self, device=None) -> dict[str, torch.Tensor]:
        idx = range(self.samples_consumed, self.samples_consumed + self.batch_size)
        batch = torch.stack([self.dataset[i] for i in idx])
        self.samples_consumed += self.batch_size
        if device is not None:
            batch = batch.to(device, non_blocking=True)
        return {"input_ids": batch}

    def __iter__(self):
        while True:
            yield self.next_batch()
