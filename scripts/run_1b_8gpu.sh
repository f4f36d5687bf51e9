#!/usr/bin/env bash
# Llama-1B bf16 DP=8 over RCCL/xGMI (BASELINE.json headline config).
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "${NGPUS:-8}" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
    -m core.training --config configs/model-config-1b.yaml --overwrite "$@"
