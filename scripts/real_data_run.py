"""Real-text federated quality run, fully offline (VERDICT r01 missing #4
"quality evidence on real tokens").

The reference's artifact run trains MPT-125M on C4-small and expects
perplexity in the low 40s; this container has no network, so this script
exercises the SAME real-data pipeline end-to-end on text that exists
locally (CPython stdlib sources + /usr/share/doc):

  1. collect a real-text corpus (~40 MB),
  2. train a BPE tokenizer on it (HF `tokenizers`, offline),
  3. convert to per-client token shards + 1_gram.json (photon_amd.data.convert
     — the same path a C4 download would take),
  4. run N federated rounds with eval every round, printing the
     per-round eval cross-entropy / perplexity trajectory.

Scores are NOT C4-comparable (different corpus, small); the trajectory is
evidence that the real-data path (tokenizer training, BOS/EOS handling,
sharding, unigram maps, fed loop) learns on real text.

Usage:  python scripts/real_data_run.py --out /tmp/realrun \
            [--rounds 8] [--local-steps 50] [--vocab 8192] [--gpu]
"""

from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def build_corpus(out: Path, max_mb: int = 48) -> Path:
    """Concatenate local real text into one jsonl-ish text file."""
    corpus = out / "corpus.txt"
    budget = max_mb * 1024 * 1024
    n = 0
    with open(corpus, "w", errors="ignore") as f:
        sources = sorted(Path("/usr/lib/python3.10").rglob("*.py"))
        sources += sorted(Path("/usr/share/doc").rglob("*.txt"))
        for p in sources:
            try:
                text = p.read_text(errors="ignore").strip()
            except OSError:
                continue
            if len(text) < 256:
                continue
            f.write(json.dumps({"text": text}) + "\n")
            n += len(text)
            if n > budget:
                break
    print(f"[corpus] {n / 1e6:.1f} MB of real text -> {corpus}")
    return corpus


def train_tokenizer(corpus: Path, out: Path, vocab: int) -> Path:
    """Train a byte-level BPE on the corpus (offline; the reference uses
    the pretrained gpt-neox-20b tokenizer, which needs the network)."""
    from tokenizers import ByteLevelBPETokenizer

    tok = ByteLevelBPETokenizer()
    tok.train([str(corpus)], vocab_size=vocab,
              special_tokens=["<|endoftext|>"])
    tok_dir = out / "tokenizer"
    tok_dir.mkdir(parents=True, exist_ok=True)
    tok.save(str(tok_dir / "tokenizer.json"))
    # minimal transformers-loadable layout
    (tok_dir / "tokenizer_config.json").write_text(json.dumps({
        "tokenizer_class": "PreTrainedTokenizerFast",
        "eos_token": "<|endoftext|>", "bos_token": "<|endoftext|>",
    }))
    print(f"[tokenizer] BPE vocab {vocab} -> {tok_dir}")
    return tok_dir


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="/tmp/realrun")
    ap.add_argument("--rounds", type=int, default=8)
    ap.add_argument("--local-steps", type=int, default=50)
    ap.add_argument("--vocab", type=int, default=8192)
    ap.add_argument("--clients", type=int, default=8)
    ap.add_argument("--global-batch", type=int, default=64)
    ap.add_argument("--microbatch", type=int, default=8)
    ap.add_argument("--seq-len", type=int, default=1024)
    ap.add_argument("--d-model", type=int, default=768)
    ap.add_argument("--n-layers", type=int, default=12)
    args = ap.parse_args()

    import torch

    from photon_amd.data.convert import convert
    from photon_amd.fed.runtime import Comm, init_distributed
    from photon_amd.fed.server import FedServer

    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    shards = out / "shards"
    if not (shards / "manifest.json").exists():
        corpus = build_corpus(out)
        tok_dir = train_tokenizer(corpus, out, args.vocab)
        m = convert(f"text:{corpus}", shards, num_clients=args.clients,
                    concat_tokens=args.seq_len, tokenizer_spec=str(tok_dir))
        print(f"[convert] {m['tokens_per_client']} tokens/client")

    cfg = {
        "run_uuid": "real_text_quality",
        "seed": 1337,
        "use_wandb": False,
        "photon": {"saving_path": str(out / "ck"), "checkpoint": False,
                   "resume_round": 0},
        "fl": {
            "n_total_clients": args.clients,
            "n_clients_per_round": args.clients,
            "n_rounds": args.rounds,
            "eval_period": 1,
            "strategy_name": "NESTOROV",
            "strategy_kwargs": {"server_learning_rate": 0.7,
                                "server_momentum": 0.7},
            "reset_optimizer": False,
            "accept_failures_cnt": 0,
            "ignore_failed_rounds": False,
            "set_trainer_params_filter_keys": True,
            "set_trainer_key_to_filter": "transformer",
            "use_noise_scale_metric": False,
            "use_unigram_metrics": True,
            "split_eval": False,
        },
        "llm_config": {
            "model": {
                "d_model": args.d_model, "n_heads": args.d_model // 64,
                "n_layers": args.n_layers, "expansion_ratio": 4,
                "max_seq_len": args.seq_len, "vocab_size": args.vocab,
                "attn_config": {
                    "attn_impl": "flash" if torch.cuda.is_available()
                    else "torch"},
            },
            "optimizer": {"name": "decoupled_adamw", "lr": 6e-4,
                          "betas": [0.9, 0.95], "eps": 1e-8,
                          "weight_decay": 0.0},
            "scheduler": {"schedulers": {"lr": {
                "name": "cosine_with_warmup", "t_warmup": "20ba",
                "alpha_f": 0.1, "t_max": f"{args.rounds * args.local_steps}ba",
            }}},
            "precision": "amp_bf16" if torch.cuda.is_available() else "fp32",
            "master_weights": torch.cuda.is_available(),
            "device_train_microbatch_size": args.microbatch,
            "global_train_batch_size": args.global_batch,
            "device_eval_batch_size": args.microbatch,
            "eval_subset_num_batches": 8,
            "max_duration": "1000000ba",
            "local_steps": f"{args.local_steps}ba",
            "seed": 17,
            "max_seq_len": args.seq_len,
            "algorithms": {"gradient_clipping": {
                "clipping_type": "norm", "clipping_threshold": 1.0}},
        },
        "dataset": {
            "train": {
                "root_local": str(shards), "split": "train", "shuffle": True,
                "streams": [
                    {"client_streams": {f"c{i}": {"local": f"client_{i}"}}}
                    for i in range(args.clients)
                ],
            },
            # eval over the same real-text shards (no separate val split in
            # the local corpus; the trajectory, not the absolute, matters)
            "val": {
                "root_local": str(shards), "split": "train",
                "streams": [
                    {"client_streams": {f"c{i}": {"local": f"client_{i}"}}}
                    for i in range(args.clients)
                ],
            },
        },
    }

    rank, world = init_distributed()
    device = ("cuda" if torch.cuda.is_available() else "cpu")
    srv = FedServer(cfg, Comm(rank, world), device)
    srv.initialize()
    import math
    traj = []
    for r in range(1, args.rounds + 1):
        m = srv.run_round(r)
        ce = srv.evaluate_round(r)
        traj.append(ce)
        if rank == 0:
            print(f"[round {r:2d}] train_loss="
                  f"{m.get('loss/train/total', float('nan')):.4f} "
                  f"eval_CE={ce:.4f} ppl={math.exp(min(ce, 20)):.1f}",
                  flush=True)
    if rank == 0:
        print(json.dumps({"eval_ce_trajectory": [round(x, 4) for x in traj],
                          "ppl_trajectory":
                          [round(math.exp(min(x, 20)), 1) for x in traj]}))


if __name__ == "__main__":
    main()
