#!/usr/bin/env bash
# MoE Llama-124M (8 experts, top-2) — single MI355X; 288 GB HBM holds large
# expert counts without expert parallelism.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m core.training --config configs/model-config-moe-124m.yaml --overwrite "$@"
