#!/usr/bin/env bash
# Prepare the gauntlet ICL datasets — reference prepare_eval_dataset.sh
# downloads llm-foundry's eval/local_data from the network; this offline
# equivalent materializes the bundled deterministic stand-in datasets for
# every task in tasks_v0.3.yaml (same jsonl schemas). Point dataset_uri at
# real downloads when network data is available.
set -euo pipefail
cd "$(dirname "$0")/../.."
python scripts/make_gauntlet_local_data.py
