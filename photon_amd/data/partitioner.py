"""IID stream partitioner — photon/dataset/stream_partitioner.py (58 LoC)
re-done for token shards: split one converted shard dir into N client dirs
by round-robin over fixed-size token blocks.

    python -m photon_amd.data.partitioner --src <dir> --out <root> \
        --num-clients 8 [--block-tokens 2048]
"""

from __future__ import annotations

import argparse
import json
from pathlib import Path

import numpy as np

from .shards import TokenShardDataset, TokenShardWriter


def partition(src: str | Path, out_root: str | Path, num_clients: int,
              block_tokens: int = 2048, split: str = "train") -> list[int]:
    ds = TokenShardDataset(src, seq_len=block_tokens)
    writers = [
        TokenShardWriter(Path(out_root) / f"client_{i}" / split)
        for i in range(num_clients)
    ]
    counts = [0] * num_clients
    for i in range(len(ds)):
        cid = i % num_clients
        writers[cid].write(ds[i].numpy().astype(np.uint32))
        counts[cid] += block_tokens
    for w in writers:
        w.close()
    return counts


def main() -> None:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--src", required=True)
    ap.add_argument("--out", required=True)
    ap.add_argument("--num-clients", type=int, default=8)
    ap.add_argument("--block-tokens", type=int, default=2048)
    ap.add_argument("--split", default="train")
    args = ap.parse_args()
    counts = partition(args.src, args.out, args.num_clients, args.block_tokens, args.split)
    print(json.dumps({"tokens_per_client": counts}))


if __name__ == "__main__":
    main()
