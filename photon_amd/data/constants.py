"""Dataset split constants — the analogue of photon/dataset/constants/
(mc4.py: per-language C4/mC4 split names with truncation sizes, and the
split-spec types).

The reference enumerates C4/mC4 language subsets and defines truncated
split variants (e.g. ``train_small`` = 100k samples) used to build
per-client streams. Facts (language codes, HF dataset names, sizes) are
public dataset metadata.
"""

from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class SplitSpec:
    """A named split with an optional sample truncation
    (photon/dataset/constants/types.py)."""

    hf_split: str
    truncated_samples: int | None = None


# Languages the reference supports for mC4 (photon/dataset/constants/mc4.py:15-28).
MC4_LANGUAGES = [
    "en", "sr", "la", "sw", "ur", "ms", "zh", "it", "es", "de", "el", "ru", "hi",
]

# Split table: full splits plus the truncated variants the reference uses
# for small-scale runs (train_small=100000, val_small=10000).
SPLITS = {
    "train": SplitSpec("train"),
    "validation": SplitSpec("validation"),
    "train_small": SplitSpec("train", truncated_samples=100_000),
    "val_small": SplitSpec("validation", truncated_samples=10_000),
}


def dataset_name(language: str) -> str:
    """HF dataset path for a language subset (c4 for en, mc4 otherwise)."""
    return "allenai/c4" if language == "en" else "mc4"
