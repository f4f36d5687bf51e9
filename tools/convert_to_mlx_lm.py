#!/usr/bin/env python3
"""Export a trained run to an mlx-lm / HF-Llama-compatible directory.

Parity surface: /root/reference/tools/convert-to-mlx-lm.py:13-181 (copies the
final safetensors + tokenizer, writes an HF-style config.json and
tokenizer_config.json with a BOS post-processor patch).

Our in-memory model uses MI355X-friendly FUSED projections (one QKV GEMM, one
gate+up GEMM — models/llama.py); export therefore SPLITS those back into the
standard HF Llama parameter layout (q_proj/k_proj/v_proj, gate_proj/up_proj)
so any HF/mlx-lm Llama loader can consume the result.
"""
from __future__ import annotations

import argparse
import json
import shutil
import sys
from pathlib import Path
from typing import Dict

import torch

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from mlx_cuda_distributed_pretraining_amd.core.checkpoint import (  # noqa: E402
    CheckpointManager,
)
from mlx_cuda_distributed_pretraining_amd.core.config import Config  # noqa: E402
from mlx_cuda_distributed_pretraining_amd.models.llama import ModelArgs  # noqa: E402


def fused_to_hf_state(state: Dict[str, torch.Tensor], args: ModelArgs) -> Dict[str, torch.Tensor]:
    """Map our fused-parameter state_dict to HF Llama names, splitting the
    fused QKV and gate+up weights."""
    q_dim = args.num_heads * args.head_dim
    kv_dim = args.num_kv_heads * args.head_dim
    inter = args.intermediate_size
    out: Dict[str, torch.Tensor] = {}
    for name, t in state.items():
        if name == "tok_embeddings.weight":
            out["model.embed_tokens.weight"] = t
        elif name == "norm.weight":
            out["model.norm.weight"] = t
        elif name == "output.weight":
            out["lm_head.weight"] = t
        elif name.startswith("layers."):
            parts = name.split(".")
            n = parts[1]
            rest = ".".join(parts[2:])
            pre = f"model.layers.{n}"
            if rest == "attention_norm.weight":
                out[f"{pre}.input_layernorm.weight"] = t
            elif rest == "mlp_norm.weight":
                out[f"{pre}.post_attention_layernorm.weight"] = t
            elif rest in ("attention.wqkv.weight", "attention.wqkv.bias"):
                kind = rest.rsplit(".", 1)[1]
                q, k, v = t.split([q_dim, kv_dim, kv_dim], dim=0)
                out[f"{pre}.self_attn.q_proj.{kind}"] = q.contiguous()
                out[f"{pre}.self_attn.k_proj.{kind}"] = k.contiguous()
                out[f"{pre}.self_attn.v_proj.{kind}"] = v.contiguous()
            elif rest in ("attention.wo.weight", "attention.wo.bias"):
                kind = rest.rsplit(".", 1)[1]
                out[f"{pre}.self_attn.o_proj.{kind}"] = t
            elif rest in ("mlp.w_gate_up.weight", "mlp.w_gate_up.bias"):
                kind = rest.rsplit(".", 1)[1]
                g, u = t.split([inter, inter], dim=0)
                out[f"{pre}.mlp.gate_proj.{kind}"] = g.contiguous()
                out[f"{pre}.mlp.up_proj.{kind}"] = u.contiguous()
            elif rest in ("mlp.w_down.weight", "mlp.w_down.bias"):
                kind = rest.rsplit(".", 1)[1]
                out[f"{pre}.mlp.down_proj.{kind}"] = t
            # rope tables / alibi slopes are non-persistent buffers -> skipped
    return out


def hf_config_from(config: Config, args: ModelArgs) -> Dict:
    return {
        "architectures": ["LlamaForCausalLM"],
        "model_type": "llama",
        "hidden_size": args.hidden_size,
        "intermediate_size": args.intermediate_size,
        "num_hidden_layers": args.num_layers,
        "num_attention_heads": args.num_heads,
        "num_key_value_heads": args.num_kv_heads,
        "head_dim": args.head_dim,
        "vocab_size": args.vocab_size,
        "max_position_embeddings": int(args.max_position_embeddings or 2048),
        "rms_norm_eps": args.rms_norm_eps,
        "rope_theta": args.rope_theta,
        # linear rope scaling survives the export (HF "rope_scaling" dict)
        **({"rope_scaling": {"type": "linear", "factor": float(args.rope_scaling)}}
           if args.rope_scaling else {}),
        "tie_word_embeddings": args.tie_word_embeddings,
        "hidden_act": "silu",
        "attention_bias": args.attention_bias,
        "mlp_bias": args.mlp_bias,
        "bos_token_id": 1,
        "eos_token_id": 2,
        "pad_token_id": 0,
        "torch_dtype": "bfloat16",
    }


def convert_run(run_dir: str | Path, out_path: str | Path, checkpoint: str = "final") -> Path:
    run_dir = Path(run_dir)
    out = Path(out_path)
    out.mkdir(parents=True, exist_ok=True)

    config = Config.from_yaml(run_dir / "config.yaml")
    # vocab size: prefer the tokenizer actually saved with the run
    tok_json = run_dir / "tokenizer" / "tokenizer.json"
    vocab_size = None
    if tok_json.exists():
        tok_data = json.loads(tok_json.read_text())
        vocab_size = len(tok_data.get("model", {}).get("vocab", {})) or None
    if vocab_size is None:
        tk = config.data.tokenizer
        vocab_size = int(tk.get("normal_vocab_size", 256)) + len(tk.get("special_tokens", {}))
    args = ModelArgs.from_config(config.model, vocab_size)
    if args.num_local_experts:
        raise NotImplementedError(
            "convert_to_mlx_lm: MoE runs have no HF-Llama equivalent layout"
        )

    model_path, _opt, _state = CheckpointManager.get_checkpoint_paths(
        str(run_dir / "checkpoints" / f"step_{checkpoint}")
    )
    from safetensors.torch import load_file, save_file

    state = load_file(model_path)
    hf_state = fused_to_hf_state(state, args)
    save_file(hf_state, str(out / "model.safetensors"),
              metadata={"format": "pt"})

    (out / "config.json").write_text(json.dumps(hf_config_from(config, args), indent=2))

    if tok_json.exists():
        shutil.copy(tok_json, out / "tokenizer.json")
    special = config.data.tokenizer.get("special_tokens", {})
    tokenizer_config = {
        "tokenizer_class": "PreTrainedTokenizerFast",
        "bos_token": special.get("bos", "<bos>"),
        "eos_token": special.get("eos", "<eos>"),
        "pad_token": special.get("pad", "<pad>"),
        "add_bos_token": True,
        "add_eos_token": False,
        "model_max_length": int(args.max_position_embeddings or 2048),
    }
    (out / "tokenizer_config.json").write_text(json.dumps(tokenizer_config, indent=2))

    # BOS post-processor patch (reference convert-to-mlx-lm.py:91-140):
    # ensure single-sequence encodes get a BOS prefix.
    if (out / "tokenizer.json").exists():
        td = json.loads((out / "tokenizer.json").read_text())
        bos = tokenizer_config["bos_token"]
        vocab = td.get("model", {}).get("vocab", {})
        if bos in vocab and not td.get("post_processor"):
            td["post_processor"] = {
                "type": "TemplateProcessing",
                "single": [{"SpecialToken": {"id": bos, "type_id": 0}},
                           {"Sequence": {"id": "A", "type_id": 0}}],
                "pair": [{"SpecialToken": {"id": bos, "type_id": 0}},
                         {"Sequence": {"id": "A", "type_id": 0}},
                         {"Sequence": {"id": "B", "type_id": 0}}],
                "special_tokens": {bos: {"id": bos, "ids": [vocab[bos]],
                                         "tokens": [bos]}},
            }
            (out / "tokenizer.json").write_text(json.dumps(td))
    return out


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Export a run to mlx-lm/HF format")
    p.add_argument("--run", required=True, help="run directory (runs/<name>)")
    p.add_argument("--out-path", required=True)
    p.add_argument("--checkpoint", default="final")
    a = p.parse_args(argv)
    out = convert_run(a.run, a.out_path, a.checkpoint)
    print(f"exported to {out}")


if __name__ == "__main__":
    main()
