from .shards import StatefulLoader, TokenShardDataset, TokenShardWriter
from .synthetic import SyntheticTokenDataset
from .text import build_train_loader, build_eval_loader

__all__ = [
    "StatefulLoader",
    "TokenShardDataset",
    "TokenShardWriter",
    "SyntheticTokenDataset",
    "build_train_loader",
    "build_eval_loader",
]
