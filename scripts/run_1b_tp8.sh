#!/usr/bin/env bash
# Llama-1B tensor-parallel over all 8 GPUs (TP=8 + sequence parallelism —
# the reference's model_parallel placeholder made real, parallel/tp.py).
# Lower model_parallel_size in the config for a TP x DP mesh.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "${NGPUS:-8}" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
    -m core.training --config configs/model-config-1b-tp8.yaml --overwrite "$@"
