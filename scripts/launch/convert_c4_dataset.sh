#!/usr/bin/env bash
# Dataset conversion — reference convert_c4_dataset.sh (HF C4 -> 8-client
# MDS shards). Here: any local text/HF-disk corpus -> 8-client token shards
# + per-client 1_gram.json. With no corpus argument, generates a synthetic
# one (the data-free CI/bench path).
set -euo pipefail
SRC=${1:-synthetic:100000}
OUT=${2:-data/c4_8clients}
python -m photon_amd.data.convert --source "$SRC" --out "$OUT" \
    --num-clients 8 --concat-tokens 2048 --split train
python -m photon_amd.data.convert --source "$SRC" --out "$OUT" \
    --num-clients 8 --concat-tokens 2048 --split val --seed 7331
