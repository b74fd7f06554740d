from .unigram import (
    load_freq_map,
    PureUnigramCrossEntropy,
    PureUnigramPerplexity,
    UnigramNormalizedLanguageCrossEntropy,
    UnigramNormalizedLanguagePerplexity,
    UNIGRAM_METRICS,
    merge_freq_dicts,
    unigram_tensor_from_freq,
)

__all__ = [
    "load_freq_map",
    "PureUnigramCrossEntropy",
    "PureUnigramPerplexity",
    "UnigramNormalizedLanguageCrossEntropy",
    "UnigramNormalizedLanguagePerplexity",
    "UNIGRAM_METRICS",
    "merge_freq_dicts",
    "unigram_tensor_from_freq",
]
