"""Generate bundled OFFLINE stand-in datasets for every task in the
reference's tasks_v0.3.yaml (photon/conf/icl_tasks_config/tasks_v0.3.yaml).

The reference downloads its ICL datasets (eval/local_data/*) from the
network; this container has none, so each dataset_uri gets a small
deterministic synthetic file with the correct SCHEMA for its task type.
These exercise the full harness + gauntlet composite end-to-end; scores on
them are plumbing checks, not benchmark results.

Run from the repo root:  python scripts/make_gauntlet_local_data.py
"""

from __future__ import annotations

import json
import random
from pathlib import Path

import yaml

ROOT = Path(__file__).resolve().parent.parent
TASKS = ROOT / "photon_amd/conf/yaml/icl_tasks_config/tasks_v0.3.yaml"
OUT_BASE = ROOT / "photon_amd"

WORDS = ("red green blue stone river mountain cloud paper candle garden "
         "window bottle copper silver market castle").split()


def make_examples(kind: str, rng: random.Random, n: int = 8) -> list[dict]:
    out = []
    for i in range(n):
        a, b, c = rng.sample(WORDS, 3)
        if kind == "language_modeling":
            out.append({
                "context": f"The {a} is next to the {b}. The {a} is next to",
                "continuation": f" the {b}.",
            })
        elif kind == "multiple_choice":
            gold = rng.randrange(3)
            choices = rng.sample(WORDS, 3)
            out.append({
                "query": f"Q: which word was listed first: "
                         f"{', '.join(choices)}? ",
                "choices": choices,
                "gold": 0,
            })
        elif kind == "schema":
            out.append({
                "context_options": [
                    f"The {a} was too big, so it replaced",
                    f"The {b} was too big, so it replaced",
                ],
                "continuation": f" the {c}.",
                "gold": i % 2,
            })
        elif kind == "generation_task_with_answers":
            out.append({
                "context": f"Q: repeat the word '{a}'.",
                "answer": a,
                "aliases": [a.upper()],
            })
        else:
            raise ValueError(kind)
    return out


def main() -> None:
    cfg = yaml.safe_load(TASKS.read_text())
    rng = random.Random(1337)
    n_written = 0
    for task in cfg["icl_tasks"]:
        uri = task["dataset_uri"]
        kind = task.get("icl_task_type", "language_modeling")
        path = OUT_BASE / uri
        path.parent.mkdir(parents=True, exist_ok=True)
        with open(path, "w") as f:
            for ex in make_examples(kind, rng):
                f.write(json.dumps(ex) + "\n")
        n_written += 1
    print(f"wrote {n_written} stand-in datasets under "
          f"{OUT_BASE / 'eval/local_data'}")


if __name__ == "__main__":
    main()
