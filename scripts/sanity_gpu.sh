#!/usr/bin/env bash
# CI-lite GPU sanity: run the GPU kernel tests under
# serialized kernel execution (surfaces async launch failures at the
# offending kernel) and with deterministic workspace config.
set -euo pipefail
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3       # serialize launches: fault at the culprit
export AMD_SERIALIZE_COPY=3
export HSA_ENABLE_IPC_MODE_LEGACY=${HSA_ENABLE_IPC_MODE_LEGACY:-0}
python -m pytest tests -m gpu -q "$@"
