# Compatibility shim: the reference exposes `python -m core.training` /
# `python -m core.generation` from the repo root; keep those entrypoints
# working against the real package.
