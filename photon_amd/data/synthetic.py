"""Synthetic C4-shaped token data (no-network BASELINE runs).

Deterministic per-(client, split, seed) random token windows with the same
loader interface and resume semantics as TokenShardDataset. Token ids are
drawn from a Zipf-ish distribution over the 50368 vocab so that CE losses
and unigram metrics behave like natural text rather than uniform noise.
"""

from __future__ import annotations

import numpy as np
import torch


class SyntheticTokenDataset:
    def __init__(
        self,
        seq_len: int,
        vocab_size: int = 50368,
        num_samples: int = 1 << 20,
        seed: int = 1337,
        client_id: int = 0,
        split: str = "train",
    ):
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.num_samples = num_samples
        split_salt = {"train": 0, "validation": 1, "val": 1}.get(split, 2)
        self.base_seed = (seed * 1_000_003 + client_id * 997 + split_salt) & 0x7FFFFFFF
        # Zipf-ish token distribution shared across clients (one "language"):
        # p(i) ∝ 1/(i+10); sampled via inverse CDF on per-sample uniforms.
        ranks = np.arange(vocab_size, dtype=np.float64)
        p = 1.0 / (ranks + 10.0)
        self._cdf = np.cumsum(p / p.sum())

    def __len__(self) -> int:
        return self.num_samples

    def __getitem__(self, i: int) -> torch.Tensor:
        rng = np.random.default_rng((self.base_seed, int(i) % self.num_samples))
        u = rng.random(self.seq_len)
        toks = np.searchsorted(self._cdf, u).astype(np.int64)
        return torch.from_numpy(np.minimum(toks, self.vocab_size - 1))
