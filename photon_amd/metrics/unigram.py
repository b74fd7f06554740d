"""Unigram-normalized language-modeling metrics.

Re-implements the reference's torchmetrics classes
(photon/metrics/unigram_normalized_metrics.py:12-264): cross-entropy and
perplexity of the *unigram* distribution over the labels, and the
unigram-NORMALIZED language CE/perplexity (model CE minus unigram CE),
which makes losses comparable across vocabularies/tokenizers.

The unigram distribution comes from per-client ``1_gram.json`` frequency
maps written at dataset conversion (photon/dataset/convert_dataset_hf.py:
313-363); ``merge_freq_dicts`` + ``unigram_tensor_from_freq`` mirror
photon/utils.py:1017-1063.

Metric classes follow the torchmetrics update()/compute()/reset() protocol
with plain tensors (sum-reducible state for distributed sync — the
reference relies on ``dist_reduce_fx="sum"``; here the caller all-reduces
``_state()`` when running multi-rank).
"""

from __future__ import annotations

import json
from pathlib import Path

import torch


# ---------------------------------------------------------------------------
# Frequency maps (photon/utils.py:1017-1063)
# ---------------------------------------------------------------------------
def merge_freq_dicts(dicts: list[dict]) -> dict:
    out: dict[str, int] = {}
    for d in dicts:
        for k, v in d.items():
            out[k] = out.get(k, 0) + int(v)
    return out


def unigram_tensor_from_freq(freq: dict, vocab_size: int,
                             smoothing: float = 1.0) -> torch.Tensor:
    """Unigram probabilities over the vocab with add-k smoothing."""
    counts = torch.full((vocab_size,), float(smoothing))
    for k, v in freq.items():
        idx = int(k)
        if 0 <= idx < vocab_size:
            counts[idx] += float(v)
    return counts / counts.sum()


def load_freq_map(path: str | Path) -> dict:
    with open(path) as f:
        return json.load(f)


# ---------------------------------------------------------------------------
# Metrics
# ---------------------------------------------------------------------------
class _MeanMetric:
    """Sum/count state, torchmetrics-style."""

    def __init__(self):
        self.total = torch.zeros((), dtype=torch.float64)
        self.count = torch.zeros((), dtype=torch.float64)

    def reset(self) -> None:
        self.total.zero_()
        self.count.zero_()

    def _state(self) -> tuple[torch.Tensor, torch.Tensor]:
        return self.total, self.count

    def compute(self) -> float:
        return float(self.total / self.count) if float(self.count) > 0 else float("nan")


class PureUnigramCrossEntropy(_MeanMetric):
    """-E[log p_unigram(label)] — the loss of the unigram LM on the labels
    (reference :12-61)."""

    def __init__(self, unigram_probs: torch.Tensor, ignore_index: int = -100):
        super().__init__()
        self.log_p = torch.log(unigram_probs.double().clamp_min(1e-12))
        self.ignore_index = ignore_index

    def update(self, labels: torch.Tensor) -> None:
        labels = labels.reshape(-1)
        mask = labels != self.ignore_index
        lab = labels[mask]
        self.total += -self.log_p.to(lab.device)[lab].sum().cpu()
        self.count += float(mask.sum())


class PureUnigramPerplexity(PureUnigramCrossEntropy):
    def compute(self) -> float:
        import math

        return math.exp(super().compute())


class UnigramNormalizedLanguageCrossEntropy(_MeanMetric):
    """Model CE minus unigram CE per token (reference :111-230):
    vocab-independent comparability across tokenizers."""

    def __init__(self, unigram_probs: torch.Tensor, ignore_index: int = -100):
        super().__init__()
        self.log_p = torch.log(unigram_probs.double().clamp_min(1e-12))
        self.ignore_index = ignore_index

    def update(self, token_losses: torch.Tensor, labels: torch.Tensor) -> None:
        """token_losses: per-token model CE (same shape as labels)."""
        labels = labels.reshape(-1)
        token_losses = token_losses.reshape(-1).double()
        mask = labels != self.ignore_index
        lab = labels[mask]
        model_ce = token_losses[mask].sum().cpu()
        unigram_ce = -self.log_p.to(lab.device)[lab].sum().cpu()
        self.total += model_ce - unigram_ce
        self.count += float(mask.sum())


class UnigramNormalizedLanguagePerplexity(UnigramNormalizedLanguageCrossEntropy):
    def compute(self) -> float:
        import math

        return math.exp(super().compute())


# Registry map (reference :259-264)
UNIGRAM_METRICS = {
    "PureUnigramCrossEntropy": PureUnigramCrossEntropy,
    "PureUnigramPerplexity": PureUnigramPerplexity,
    "UnigramNormalizedLanguageCrossEntropy": UnigramNormalizedLanguageCrossEntropy,
    "UnigramNormalizedLanguagePerplexity": UnigramNormalizedLanguagePerplexity,
}
