#!/usr/bin/env bash
# Llama-400M with the Muon optimizer (Newton-Schulz), DP=8 over xGMI.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "${NGPUS:-8}" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29501}" \
    -m core.training --config configs/model-config-400m-muon.yaml --overwrite "$@"
