#!/usr/bin/env bash
# 2M-param sample config — plumbing check, runs without a GPU.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m core.training --config configs/model-config-sample.yaml --overwrite "$@"
