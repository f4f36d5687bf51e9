#!/usr/bin/env python3
"""Print a model architecture summary for a config or a trained run.

Parity surface: /root/reference/tools/visualize_model.py (per-layer table,
parameter counts, memory estimate).
"""
from __future__ import annotations

import argparse
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from mlx_cuda_distributed_pretraining_amd.core.config import Config  # noqa: E402
from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs  # noqa: E402


def summarize(model: Model, bytes_per_param: int = 2) -> str:
    lines = []
    total = 0
    lines.append(f"{'module':<44s} {'shape':<20s} {'params':>12s}")
    lines.append("-" * 78)
    for name, p in model.named_parameters():
        n = p.numel()
        total += n
        lines.append(f"{name:<44s} {str(tuple(p.shape)):<20s} {n:>12,d}")
    lines.append("-" * 78)
    args = model.args
    attn_per_layer = sum(p.numel() for n, p in model.named_parameters()
                         if n.startswith("layers.0.attention"))
    mlp_per_layer = sum(p.numel() for n, p in model.named_parameters()
                        if n.startswith("layers.0.mlp."))
    lines.append(f"layers: {args.num_layers}  hidden: {args.hidden_size}  "
                 f"heads: {args.num_heads}/{args.num_kv_heads}kv  head_dim: {args.head_dim}")
    lines.append(f"per-layer params: attention {attn_per_layer:,d}  mlp {mlp_per_layer:,d}")
    lines.append(f"total params: {total:,d} ({total/1e6:.2f}M)")
    lines.append(f"weights (bf16): {total * bytes_per_param / 2**30:.2f} GiB  |  "
                 f"AdamW step state (fp32 m+v+master): {total * 12 / 2**30:.2f} GiB")
    return "\n".join(lines)


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Visualize model architecture")
    p.add_argument("--config", type=str, default=None, help="YAML config path")
    p.add_argument("--run", type=str, default=None, help="run dir (uses its config.yaml)")
    a = p.parse_args(argv)
    cfg_path = a.config or (str(Path(a.run) / "config.yaml") if a.run else None)
    if cfg_path is None:
        p.error("need --config or --run")
    cfg = Config.from_yaml(cfg_path)
    tk = cfg.data.tokenizer
    vocab = int(tk.get("normal_vocab_size", 256)) + len(tk.get("special_tokens", {}))
    model = Model(ModelArgs.from_config(cfg.model, vocab))
    print(summarize(model))


if __name__ == "__main__":
    main()
