"""Dataset conversion CLI — text/HF corpus -> per-client token shards.

The analogue of the reference's photon/dataset/convert_dataset_hf.py
(HF -> MDS shards per client, 1_gram.json token-frequency map + tokenizer
dir, :313-363). Output layout mirrors the reference's
``client_{i}/{split}`` directories, but shards are the photon_amd token
format (data/shards.py) instead of MDS:

    out_root/client_{i}/{split}/shard_*.bin + index.json + 1_gram.json
    out_root/tokenizer/      (saved tokenizer, when one was used)

Sources (this environment has no network):
  * ``--source text:<path>``      newline-delimited text / .jsonl ("text" field)
  * ``--source hf_disk:<path>``   a datasets.load_from_disk dataset
  * ``--source synthetic:<n>``    n Zipf-sampled documents (bench/data-free CI)

Tokenizer: ``--tokenizer <local dir>`` (transformers AutoTokenizer), or the
built-in byte tokenizer (vocab 256 + BOS/EOS) when none is given.

Usage:
    python -m photon_amd.data.convert --source synthetic:1000 \
        --out /tmp/shards --num-clients 8 --concat-tokens 2048
"""

from __future__ import annotations

import argparse
import json
from collections import Counter
from pathlib import Path

import numpy as np

from .shards import TokenShardWriter

BYTE_EOS = 256
BYTE_VOCAB = 258  # 256 bytes + EOS + pad


class ByteTokenizer:
    """Dependency-free fallback: UTF-8 bytes + EOS (vocab 258)."""

    vocab_size = BYTE_VOCAB
    eos_token_id = BYTE_EOS

    def encode(self, text: str) -> list[int]:
        return list(text.encode("utf-8"))

    def decode(self, ids) -> str:
        return bytes(int(i) & 0xFF for i in ids).decode("utf-8", "ignore")

    def save_pretrained(self, path):
        Path(path).mkdir(parents=True, exist_ok=True)
        (Path(path) / "tokenizer_config.json").write_text(
            json.dumps({"tokenizer_class": "photon_amd.ByteTokenizer",
                        "vocab_size": BYTE_VOCAB})
        )


def load_tokenizer(spec: str | None, bos_eos_workaround: bool = True):
    if spec is None:
        return ByteTokenizer()
    from transformers import AutoTokenizer

    tok = AutoTokenizer.from_pretrained(spec)
    if bos_eos_workaround:
        apply_bos_eos_workaround(tok)
    return tok


def apply_bos_eos_workaround(tok) -> None:
    """gpt-neox/GPT2-family tokenizers do not insert BOS/EOS even with
    add_special_tokens — install a TemplateProcessing post-processor
    (reference photon/dataset/utils.py:66-102). No-op for tokenizers that
    already insert them or have no bos/eos ids."""
    bos, eos = getattr(tok, "bos_token_id", None), getattr(tok, "eos_token_id", None)
    if bos is None or eos is None or not hasattr(tok, "_tokenizer"):
        return
    test = tok("test")["input_ids"]
    if test and (test[0] == bos or test[-1] == eos):
        return
    try:
        from tokenizers.processors import TemplateProcessing
    except ImportError:
        return
    tok._tokenizer.post_processor = TemplateProcessing(
        single=tok.bos_token + " $A " + tok.eos_token,
        # no special token between $A and $B: the concat writer already
        # EOS-joins distinct sequences
        pair=tok.bos_token + " $A $B " + tok.eos_token,
        special_tokens=[(tok.eos_token, eos), (tok.bos_token, bos)],
    )
    test = tok("test")["input_ids"]
    if not (test and (test[0] == bos or test[-1] == eos)):
        raise ValueError(
            "tokenizer inserts neither BOS nor EOS even after the "
            "TemplateProcessing workaround; concatenation would attach "
            "sequences without separators"
        )


def iter_documents(source: str, seed: int = 1337):
    kind, _, arg = source.partition(":")
    if kind == "text":
        path = Path(arg)
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                if line.startswith("{"):
                    try:
                        yield json.loads(line).get("text", "")
                        continue
                    except json.JSONDecodeError:
                        pass
                yield line
    elif kind == "hf_disk":
        import datasets

        ds = datasets.load_from_disk(arg)
        if hasattr(ds, "values"):  # DatasetDict: use train split
            ds = ds.get("train") or next(iter(ds.values()))
        for row in ds:
            yield row.get("text", "")
    elif kind == "hf_stream":
        # streaming-HF ingestion (reference build_hf_dataset streaming
        # path): hf_stream:path[:name[:split[:truncate]]] — e.g.
        # hf_stream:allenai/c4:en:train_small applies the reference split
        # table truncation (data/constants.py).
        import datasets

        from .constants import split_spec

        parts = arg.split(":")
        path = parts[0]
        name = parts[1] if len(parts) > 1 and parts[1] else None
        split = parts[2] if len(parts) > 2 else "train"
        truncate = None
        try:
            spec = split_spec(name or "en", split)
            split_hf, truncate = spec.hf_split, spec.truncated_samples
        except ValueError:
            split_hf = split
        if len(parts) > 3:
            truncate = int(parts[3])
        ds = datasets.load_dataset(path, name, split=split_hf, streaming=True)
        for i, row in enumerate(ds):
            if truncate is not None and i >= truncate:
                break
            yield row.get("text", "")
    elif kind == "synthetic":
        rng = np.random.default_rng(seed)
        words = [f"w{i}" for i in range(2000)]
        for _ in range(int(arg)):
            n = int(rng.integers(20, 200))
            idx = rng.zipf(1.5, size=n).clip(1, len(words)) - 1
            yield " ".join(words[i] for i in idx)
    else:
        raise ValueError(f"unknown source kind {kind!r}")


def convert(
    source: str,
    out_root: str | Path,
    num_clients: int = 8,
    concat_tokens: int = 2048,
    split: str = "train",
    tokenizer_spec: str | None = None,
    val_fraction: float = 0.0,
    seed: int = 1337,
) -> dict:
    """Tokenize + concat documents (EOS-joined) and round-robin them across
    ``num_clients`` shard dirs, tracking per-client unigram counts."""
    out_root = Path(out_root)
    tok = load_tokenizer(tokenizer_spec)
    eos = getattr(tok, "eos_token_id", None)
    writers = [
        TokenShardWriter(out_root / f"client_{i}" / split) for i in range(num_clients)
    ]
    freqs = [Counter() for _ in range(num_clients)]
    totals = [0] * num_clients
    buf: list[list[int]] = [[] for _ in range(num_clients)]
    n_docs = 0
    for doc in iter_documents(source, seed=seed):
        cid = n_docs % num_clients
        ids = tok.encode(doc)
        if eos is not None:
            ids = ids + [eos]
        buf[cid].extend(ids)
        freqs[cid].update(ids)
        totals[cid] += len(ids)
        n_docs += 1
        # flush in concat_tokens blocks (sample granularity of the reader)
        while len(buf[cid]) >= concat_tokens:
            writers[cid].write(np.asarray(buf[cid][:concat_tokens]))
            buf[cid] = buf[cid][concat_tokens:]
    for cid in range(num_clients):
        if buf[cid]:
            writers[cid].write(np.asarray(buf[cid]))
        writers[cid].close()
        fdir = out_root / f"client_{cid}"
        with open(fdir / "1_gram.json", "w") as f:
            json.dump({str(k): int(v) for k, v in freqs[cid].items()}, f)
    if tokenizer_spec is not None or True:
        tok.save_pretrained(out_root / "tokenizer")
    manifest = {
        "num_clients": num_clients,
        "n_documents": n_docs,
        "tokens_per_client": totals,
        "concat_tokens": concat_tokens,
        "split": split,
        "vocab_size": getattr(tok, "vocab_size", None),
    }
    with open(out_root / "manifest.json", "w") as f:
        json.dump(manifest, f, indent=1)
    return manifest


def main() -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--source", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--num-clients", type=int, default=8)
    ap.add_argument("--concat-tokens", type=int, default=2048)
    ap.add_argument("--split", default="train")
    ap.add_argument("--tokenizer", default=None)
    ap.add_argument("--seed", type=int, default=1337)
    args = ap.parse_args()
    m = convert(
        args.source, args.out, args.num_clients, args.concat_tokens,
        args.split, args.tokenizer, seed=args.seed,
    )
    print(json.dumps(m))


if __name__ == "__main__":
    main()
