#!/usr/bin/env python3
"""Re-shard a tensor-parallel checkpoint to a DIFFERENT tp degree.

ROADMAP r1 #13 follow-up: ``merge_tp_checkpoint.py`` already inverts the
sharding; this tool composes merge (tp=A -> full) with the pure state-dict
shard below (full -> tp=B), so a run saved at one TP degree restarts at
another (``step_<tag>_tp<r>_model.safetensors`` files + updated state json).
Optimizer moments stay per-shard and are not converted (the resumed run
starts with a fresh optimizer, same as the merge tool).

The shard function mirrors apply_tensor_parallel's slicing rules
(parallel/tp.py) on raw tensors:
  wqkv      rows [q | k | v] -> per-rank head sections
  wo        columns          -> per-rank sections (bias: full on rank 0,
                                zeros elsewhere — matches the live sharding)
  w_gate_up rows [gate | up] -> per-rank intermediate sections
  w_down    columns
  3-D expert stacks          -> expert-dim sections (EP)
  lm head / tied embedding   -> vocab rows when vocab_parallel
  everything else            -> replicated
"""
from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path
from typing import Dict, List

import torch

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from tools.merge_tp_checkpoint import merge_tp_state_dicts  # noqa: E402


def shard_tp_state_dict(full: Dict[str, torch.Tensor], rank: int, world: int,
                        num_heads: int, num_kv_heads: int, head_dim: int,
                        intermediate_size: int,
                        vocab_parallel: bool = False) -> Dict[str, torch.Tensor]:
    if num_heads % world or num_kv_heads % world or intermediate_size % world:
        raise ValueError(f"tp={world} must divide heads/kv_heads/intermediate")
    lq = num_heads // world * head_dim
    lkv = num_kv_heads // world * head_dim
    li = intermediate_size // world
    hq, hkv = num_heads * head_dim, num_kv_heads * head_dim
    untied = "output.weight" in full
    out: Dict[str, torch.Tensor] = {}
    for name, t in full.items():
        if t.dim() == 3:  # stacked experts (EP)
            le = t.shape[0] // world
            out[name] = t[rank * le:(rank + 1) * le].clone()
        elif "wqkv" in name:
            q = t[rank * lq:(rank + 1) * lq]
            k = t[hq + rank * lkv:hq + (rank + 1) * lkv]
            v = t[hq + hkv + rank * lkv:hq + hkv + (rank + 1) * lkv]
            out[name] = torch.cat([q, k, v], dim=0)
        elif "wo.weight" in name or "w_down.weight" in name:
            lc = t.shape[1] // world
            out[name] = t[:, rank * lc:(rank + 1) * lc].clone()
        elif ("wo.bias" in name or "w_down.bias" in name):
            out[name] = t.clone() if rank == 0 else torch.zeros_like(t)
        elif "w_gate_up" in name:
            inter = intermediate_size
            g = t[rank * li:(rank + 1) * li]
            u = t[inter + rank * li:inter + (rank + 1) * li]
            out[name] = torch.cat([g, u], dim=0)
        elif vocab_parallel and name == "output.weight":
            lv = t.shape[0] // world
            out[name] = t[rank * lv:(rank + 1) * lv].clone()
        elif vocab_parallel and name == "tok_embeddings.weight" and not untied:
            lv = t.shape[0] // world
            out[name] = t[rank * lv:(rank + 1) * lv].clone()
        else:
            out[name] = t.clone()
    return out


def reshard_checkpoint(base: str, tp: int) -> List[str]:
    from safetensors.torch import load_file, save_file

    from mlx_cuda_distributed_pretraining_amd.core.config import Config
    from mlx_cuda_distributed_pretraining_amd.models.llama import ModelArgs

    state = json.loads(Path(f"{base}_state.json").read_text())
    old_world = int(state.get("tp_world", 0))
    vp = bool(state.get("vocab_parallel", False))
    run_dir = Path(base).parent.parent
    cfg = Config.from_yaml(run_dir / "config.yaml")

    if old_world >= 2:
        shards = [load_file(f"{base}_tp{r}_model.safetensors") for r in range(old_world)]
        vocab = shards[0]["tok_embeddings.weight"].shape[0]
        if vp and "output.weight" not in shards[0]:
            vocab *= old_world
        args = ModelArgs.from_config(cfg.model, vocab)
        full = merge_tp_state_dicts(shards, args.num_heads, args.num_kv_heads,
                                    args.head_dim, args.intermediate_size,
                                    vocab_parallel=vp)
    else:
        full = load_file(f"{base}_model.safetensors")
        args = ModelArgs.from_config(cfg.model, full["tok_embeddings.weight"].shape[0])

    written: List[str] = []
    if tp <= 1:
        save_file(full, f"{base}_model.safetensors", metadata={"format": "pt"})
        state.pop("tp_world", None)
        written.append(f"{base}_model.safetensors")
    else:
        for r in range(tp):
            sd = shard_tp_state_dict(full, r, tp, args.num_heads, args.num_kv_heads,
                                     args.head_dim, args.intermediate_size,
                                     vocab_parallel=vp)
            path = f"{base}_tp{r}_model.safetensors"
            save_file(sd, path, metadata={"format": "pt"})
            written.append(path)
        state["tp_world"] = tp
    Path(f"{base}_state.json").write_text(json.dumps(state))
    return written


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Re-shard a TP checkpoint")
    p.add_argument("--checkpoint", required=True,
                   help="checkpoint base path (.../checkpoints/step_<tag>)")
    p.add_argument("--tp", type=int, required=True, help="target TP degree")
    a = p.parse_args(argv)
    for f in reshard_checkpoint(a.checkpoint, a.tp):
        print(f"wrote {f}")


if __name__ == "__main__":
    main()
