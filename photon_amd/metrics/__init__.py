from .unigram import (
    PureUnigramCrossEntropy,
    PureUnigramPerplexity,
    UnigramNormalizedLanguageCrossEntropy,
    UnigramNormalizedLanguagePerplexity,
    UNIGRAM_METRICS,
    merge_freq_dicts,
    unigram_tensor_from_freq,
)

__all__ = [
    "PureUnigramCrossEntropy",
    "PureUnigramPerplexity",
    "UnigramNormalizedLanguageCrossEntropy",
    "UnigramNormalizedLanguagePerplexity",
    "UNIGRAM_METRICS",
    "merge_freq_dicts",
    "unigram_tensor_from_freq",
]
