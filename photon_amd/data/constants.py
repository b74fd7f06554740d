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

# English split table: full splits plus every truncated variant the
# reference defines (mc4.py:31-75: train_small=100000, val_small=10000,
# val_xsmall=3000, val_xxsmall=100).
SPLITS = {
    "train": SplitSpec("train"),
    "validation": SplitSpec("validation"),
    "train_small": SplitSpec("train", truncated_samples=100_000),
    "val_small": SplitSpec("validation", truncated_samples=10_000),
    "val_xsmall": SplitSpec("validation", truncated_samples=3_000),
    "val_xxsmall": SplitSpec("validation", truncated_samples=100),
}

# Non-English languages carry plain train/validation splits in the
# reference's table (mc4.py:78-339, no truncated variants).
_NON_EN_SPLITS = {
    "train": SplitSpec("train"),
    "validation": SplitSpec("validation"),
}

# Per-language split tables — the reference's full DatasetConstants map
# (photon/dataset/constants/mc4.py:15-339).
LANGUAGE_SPLITS: dict[str, dict[str, SplitSpec]] = {
    lang: (SPLITS if lang == "en" else dict(_NON_EN_SPLITS))
    for lang in MC4_LANGUAGES
}


def dataset_name(language: str) -> str:
    """HF dataset path for a language subset (c4 for en, mc4 otherwise)."""
    return "allenai/c4" if language == "en" else "mc4"


def split_spec(language: str, split: str) -> SplitSpec:
    """Lookup with the reference's error semantics: unknown language or
    split raises a ValueError naming the table."""
    try:
        return LANGUAGE_SPLITS[language][split]
    except KeyError as e:
        raise ValueError(
            f"unknown language/split {language!r}/{split!r} "
            f"(languages: {MC4_LANGUAGES})"
        ) from e
