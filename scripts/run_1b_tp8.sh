#!/usr/bin/env bash
# Llama-1B with tensor parallelism over all 8 GPUs (TP=8, the reference's
# model_parallel placeholder made real — parallel/tp.py). Add DP by lowering
# model_parallel_size (e.g. 2 -> TP=2 x DP=4 mesh); add sequence parallelism
# with system.sequence_parallel: true.
set -euo pipefail
cd "$(dirname "$0")/.."
exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "${NGPUS:-8}" \
    --master-addr 127.0.0.1 --master-port "${MASTER_PORT:-29500}" \
    -m core.training --config configs/model-config-1b.yaml --overwrite \
    --auto-resume "$@" \
    2>&1 | sed 's/^/[tp8] /'
# NOTE: enable TP in the config (system.model_parallel: true,
#       model_parallel_size: 8) or ship a dedicated YAML; this script is the
#       launch shape. Checkpoints are per-shard (merge with
#       tools/merge_tp_checkpoint.py).
