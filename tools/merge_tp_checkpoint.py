#!/usr/bin/env python3
"""Merge a tensor-parallel checkpoint (per-shard ``step_<N>_tp<r>_*`` files,
written by TP runs — parallel/tp.py) back into a single full-model
checkpoint loadable by any tp=1 run, evaluation, or the HF export tool.

Inverse of ``apply_tensor_parallel``'s sharding:
  wqkv      : q/k/v row sections concatenated across ranks back into
              [q_all | k_all | v_all]
  wo        : column-concat; bias comes from tp-rank 0 (full there, frozen
              zero elsewhere)
  w_gate_up : gate/up row sections concatenated back into [gate_all | up_all]
  w_down    : column-concat; bias from tp-rank 0
  everything else (embeddings, norms, lm head, routers): replicated — taken
  from tp-rank 0.

Optimizer state is NOT merged (moments live per shard); a resumed tp=1 run
starts with a fresh optimizer. The merged triple is written as
``step_<tag>_{model,state}`` next to the shards.
"""
from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path
from typing import Dict

import torch

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def merge_tp_state_dicts(shards, num_heads: int, num_kv_heads: int, head_dim: int,
                         intermediate_size: int,
                         vocab_parallel: bool = False) -> Dict[str, torch.Tensor]:
    """shards: list of state dicts ordered by tp_rank. vocab_parallel: the
    lm head (and, for tied models, the embedding) rows are vocab shards."""
    world = len(shards)
    lq, lkv = num_heads // world * head_dim, num_kv_heads // world * head_dim
    li = intermediate_size // world
    out: Dict[str, torch.Tensor] = {}
    for name, t0 in shards[0].items():
        parts = [sd[name] for sd in shards]
        if t0.dim() == 3:
            # stacked MoE expert weights: expert-parallel shards concat on
            # the expert dim (parallel/tp.py _shard_experts)
            out[name] = torch.cat(parts, dim=0)
        elif "wqkv" in name:
            q = torch.cat([p[:lq] for p in parts], dim=0)
            k = torch.cat([p[lq:lq + lkv] for p in parts], dim=0)
            v = torch.cat([p[lq + lkv:] for p in parts], dim=0)
            out[name] = torch.cat([q, k, v], dim=0)
        elif "wo.weight" in name or "w_down.weight" in name:
            out[name] = torch.cat(parts, dim=1)
        elif vocab_parallel and name in ("output.weight", "tok_embeddings.weight"):
            # vocab-parallel head / tied embedding: row-concat back to the
            # full vocab. Untied replicated embeddings are caught below: when
            # output.weight exists the embedding was NOT sharded.
            if name == "tok_embeddings.weight" and "output.weight" in shards[0]:
                out[name] = t0  # untied: embedding stayed replicated
            else:
                out[name] = torch.cat(parts, dim=0)
        elif "w_gate_up" in name:
            g = torch.cat([p[:li] for p in parts], dim=0)
            u = torch.cat([p[li:] for p in parts], dim=0)
            out[name] = torch.cat([g, u], dim=0)
        else:
            # replicated params AND the rank-0-local row-parallel biases
            out[name] = t0
    return out


def merge_checkpoint(base: str) -> str:
    """base: .../checkpoints/step_<tag> (the plain marker path)."""
    from safetensors.torch import load_file, save_file

    state = json.loads(Path(f"{base}_state.json").read_text())
    world = int(state.get("tp_world", 0))
    vp = bool(state.get("vocab_parallel", False))
    if world < 2:
        raise ValueError(f"{base}_state.json has no tp_world — not a TP checkpoint")
    shards = [load_file(f"{base}_tp{r}_model.safetensors") for r in range(world)]

    from mlx_cuda_distributed_pretraining_amd.core.config import Config
    from mlx_cuda_distributed_pretraining_amd.models.llama import ModelArgs

    run_dir = Path(base).parent.parent
    cfg = Config.from_yaml(run_dir / "config.yaml")
    vocab = shards[0]["tok_embeddings.weight"].shape[0]
    if vp and "output.weight" not in shards[0]:
        vocab *= world  # tied sharded embedding: shard rows = V / tp
    args = ModelArgs.from_config(cfg.model, vocab)

    merged = merge_tp_state_dicts(shards, args.num_heads, args.num_kv_heads,
                                  args.head_dim, args.intermediate_size,
                                  vocab_parallel=vp)
    save_file(merged, f"{base}_model.safetensors", metadata={"format": "pt"})
    state.pop("tp_world", None)
    Path(f"{base}_state.json").write_text(json.dumps(state))
    return f"{base}_model.safetensors"


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Merge TP checkpoint shards")
    p.add_argument("--checkpoint", required=True,
                   help="checkpoint base path (.../checkpoints/step_<tag>)")
    a = p.parse_args(argv)
    out = merge_checkpoint(a.checkpoint)
    print(f"merged -> {out}")


if __name__ == "__main__":
    main()
