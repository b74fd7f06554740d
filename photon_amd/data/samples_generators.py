"""Sample generators: iterate, untokenize and RE-tokenize existing shard
data — the reference's photon/dataset/samples_generators.py surface
(generate_samples_from_dataloader, stream_and_untokenize,
generate_samples_retokenized_streaming_text_dataset) mapped onto the
token-shard reader instead of mosaicml-streaming.

Use case (reference): convert a corpus tokenized with tokenizer A into
shards for tokenizer B without re-downloading — decode each sample back to
text and re-encode.
"""

from __future__ import annotations

from collections.abc import Iterable, Iterator

import numpy as np
import torch


def generate_samples_from_dataloader(
    loader, truncate_num_samples: int | None = None
) -> Iterable[dict[str, np.ndarray]]:
    """Flatten a dataloader's batches into per-sample dicts with optional
    truncation (reference samples_generators.py:23-61)."""
    remaining = -1 if truncate_num_samples is None else truncate_num_samples
    for batch in loader:
        keys = list(batch.keys())
        n = len(batch[keys[0]])
        for i in range(n):
            if remaining == 0:
                return
            remaining -= 1
            yield {
                k: (v[i].numpy() if isinstance(v[i], torch.Tensor) else v[i])
                for k, v in batch.items()
            }


def stream_and_untokenize(
    loader, tokenizer, truncate_num_batches: int | None = None
) -> Iterator[list[str]]:
    """Yield batches of DECODED text from a pre-tokenized loader
    (reference samples_generators.py:63-128)."""
    remaining = -1 if truncate_num_batches is None else truncate_num_batches
    for batch in loader:
        if remaining == 0:
            return
        remaining -= 1
        ids = batch["input_ids"] if isinstance(batch, dict) else batch
        if hasattr(tokenizer, "batch_decode"):
            yield tokenizer.batch_decode(ids, skip_special_tokens=True)
        else:
            yield [tokenizer.decode(row.tolist()) for row in ids]


def generate_retokenized_samples(
    shard_dir, src_tokenizer, dst_tokenizer, seq_len: int,
    truncate_num_samples: int | None = None,
) -> Iterator[np.ndarray]:
    """Re-tokenization stream (reference
    generate_samples_retokenized_streaming_text_dataset): read token shards
    written under ``src_tokenizer``, decode to text, re-encode with
    ``dst_tokenizer``, and emit fixed-length concat samples."""
    from .shards import StatefulLoader, TokenShardDataset

    ds = TokenShardDataset(shard_dir, seq_len)
    loader = StatefulLoader(ds, batch_size=1)
    eos = getattr(dst_tokenizer, "eos_token_id", None)
    buf: list[int] = []
    emitted = 0
    for _ in range(len(ds)):
        batch = loader.next_batch()
        for text in stream_and_untokenize(
            iter([batch]), src_tokenizer
        ):
            for t in text:
                ids = dst_tokenizer.encode(t)
                if eos is not None:
                    ids = ids + [eos]
                buf.extend(ids)
        while len(buf) >= seq_len:
            if truncate_num_samples is not None and emitted >= truncate_num_samples:
                return
            yield np.asarray(buf[:seq_len], dtype=np.int64)
            buf = buf[seq_len:]
            emitted += 1
