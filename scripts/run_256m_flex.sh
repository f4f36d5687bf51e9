#!/usr/bin/env bash
# Llama-256M with the FlexAttention sliding-window path + Shampoo.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m core.training --config configs/model-config-256m-flex.yaml --overwrite "$@"
