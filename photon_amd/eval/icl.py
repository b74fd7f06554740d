"""ICL task evaluation + Eval Gauntlet composite.

Re-implements the slice of llm-foundry's ICL harness the reference drives
through ``icl_tasks_config`` / ``eval_gauntlet_config``
(conf/{icl_tasks,eval_gauntlet}_config/*; both default to
``empty`` and are config-gated — photon/conf/base.yaml defaults block).

Task format: jsonl files with {"context": ..., "continuation": ...} for
``language_modeling`` tasks, or {"query": ..., "choices": [...],
"gold": idx} for ``multiple_choice`` — the same fields llm-foundry's ICL
datasets carry. Scoring:

* language_modeling  -> per-token greedy accuracy over the continuation
* multiple_choice    -> accuracy of argmin over length-normalized CE

The gauntlet composite is the category-weighted mean of task scores
(llm-foundry eval_gauntlet semantics).
"""

from __future__ import annotations

import json
from pathlib import Path

import torch
import torch.nn.functional as F


def _resolve_task_path(path: str | Path) -> Path:
    """Relative task paths resolve against the photon_amd package dir
    (bundled demo tasks), falling back to the cwd."""
    p = Path(path)
    if not p.is_absolute() and not p.exists():
        pkg = Path(__file__).resolve().parent.parent / p
        if pkg.exists():
            return pkg
    return p


def load_jsonl_task(path: str | Path) -> list[dict]:
    out = []
    with open(_resolve_task_path(path)) as f:
        for line in f:
            line = line.strip()
            if line:
                out.append(json.loads(line))
    return out


@torch.no_grad()
def _continuation_stats(model, tok, context: str, continuation: str, device,
                        max_len: int):
    ctx = tok.encode(context)
    cont = tok.encode(continuation)
    ids = (ctx + cont)[-max_len:]
    n_cont = min(len(cont), len(ids) - 1)
    if n_cont <= 0:
        return 0.0, 0, 0.0
    x = torch.tensor([ids], dtype=torch.long, device=device)
    logits = model(x)["logits"][0]  # [T, V]
    # positions predicting the continuation tokens
    tgt = x[0, -n_cont:]
    pred_logits = logits[-n_cont - 1 : -1]
    ce = float(F.cross_entropy(pred_logits.float(), tgt, reduction="sum"))
    correct = int((pred_logits.argmax(-1) == tgt).sum())
    return ce, n_cont, correct


@torch.no_grad()
def evaluate_icl_tasks(model, tasks_cfg, tokenizer, device="cpu",
                       max_seq_len: int = 2048,
                       limit_examples: int | None = None) -> dict[str, float]:
    """Returns {"metrics/icl/{label}/accuracy": v, ...} per task."""
    model.eval()
    results: dict[str, float] = {}
    for task in tasks_cfg or []:
        label = str(task.get("label", "task"))
        kind = str(task.get("icl_task_type", "language_modeling"))
        delim = str(task.get("continuation_delimiter", " "))
        examples = load_jsonl_task(task["dataset_uri"])
        if limit_examples:
            examples = examples[:limit_examples]
        correct, total = 0, 0
        for ex in examples:
            if kind == "language_modeling":
                ce, n, c = _continuation_stats(
                    model, tokenizer, ex["context"] + delim,
                    ex["continuation"], device, max_seq_len,
                )
                correct += c
                total += n
            elif kind == "multiple_choice":
                ces = []
                for choice in ex["choices"]:
                    ce, n, _ = _continuation_stats(
                        model, tokenizer, ex["query"] + delim, choice,
                        device, max_seq_len,
                    )
                    ces.append(ce / max(n, 1))
                correct += int(min(range(len(ces)), key=ces.__getitem__)
                               == int(ex["gold"]))
                total += 1
            else:
                raise ValueError(f"unknown icl_task_type {kind!r}")
        results[f"metrics/icl/{label}/accuracy"] = (
            correct / total if total else float("nan")
        )
    model.train()
    return results


def gauntlet_composite(task_results: dict[str, float],
                       gauntlet_cfg: dict | None) -> dict[str, float]:
    """Category-weighted composite (eval_gauntlet semantics): categories are
    {name, benchmarks: [{name, weight}]}; composite = mean over categories
    of the weighted benchmark mean."""
    if not gauntlet_cfg:
        return {}
    out = {}
    cat_scores = []
    for cat in gauntlet_cfg.get("categories", []):
        num, den = 0.0, 0.0
        for b in cat.get("benchmarks", []):
            key = f"metrics/icl/{b['name']}/accuracy"
            if key in task_results:
                w = float(b.get("weight", 1.0))
                num += w * task_results[key]
                den += w
        if den > 0:
            score = num / den
            out[f"metrics/eval_gauntlet/{cat['name']}"] = score
            cat_scores.append(score)
    if cat_scores:
        out["metrics/eval_gauntlet/average"] = sum(cat_scores) / len(cat_scores)
    return out
