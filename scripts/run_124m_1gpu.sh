#!/usr/bin/env bash
# Llama-124M bf16, FlashAttention + fused AdamW, 1 MI355X.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m core.training --config configs/model-config-124m.yaml --overwrite "$@"
