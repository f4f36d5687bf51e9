"""Tools: mlx-lm/HF export, tokenizer training CLI, model CLI."""
import json
import sys
from pathlib import Path

import pytest
import torch
import yaml

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO / "tools"))

from mlx_cuda_distributed_pretraining_amd.core.config import Config
from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer
from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs


def _tiny_run(tmp_path, name="tiny-export"):
    cfg = Config.from_yaml(REPO / "configs" / "model-config-sample.yaml")
    cfg.name = name
    cfg.overwrite = True
    cfg.data.synthetic = True
    cfg.training.hyperparameters["iters"] = 2
    cfg.training.hyperparameters["batch_size"] = 2
    cfg.data.preprocessing["max_context_size"] = 32
    cfg.logging.steps = {"logging_interval": 1, "checkpoint_interval": 0,
                         "validation_interval": 0}
    trainer = Trainer(cfg, runs_root=str(tmp_path / "runs"))
    trainer.train()
    return trainer, tmp_path / "runs" / name


def test_convert_to_mlx_lm_roundtrip(tmp_path):
    from convert_to_mlx_lm import convert_run, fused_to_hf_state

    trainer, run_dir = _tiny_run(tmp_path)
    out = convert_run(run_dir, tmp_path / "export")

    # layout
    assert (out / "model.safetensors").exists()
    assert (out / "config.json").exists()
    assert (out / "tokenizer_config.json").exists()

    cfg = json.loads((out / "config.json").read_text())
    assert cfg["model_type"] == "llama"
    assert cfg["hidden_size"] == trainer.model.args.hidden_size
    assert cfg["num_key_value_heads"] == trainer.model.args.num_kv_heads

    # weights: split tensors must reassemble exactly into the fused originals
    from safetensors.torch import load_file

    hf = load_file(out / "model.safetensors")
    state = {k: v.cpu() for k, v in trainer.model.state_dict().items()}
    args = trainer.model.args
    q = hf["model.layers.0.self_attn.q_proj.weight"]
    k = hf["model.layers.0.self_attn.k_proj.weight"]
    v = hf["model.layers.0.self_attn.v_proj.weight"]
    fused = torch.cat([q, k, v], dim=0)
    assert torch.equal(fused, state["layers.0.attention.wqkv.weight"])
    g = hf["model.layers.0.mlp.gate_proj.weight"]
    u = hf["model.layers.0.mlp.up_proj.weight"]
    assert torch.equal(torch.cat([g, u], 0), state["layers.0.mlp.w_gate_up.weight"])
    assert torch.equal(hf["model.embed_tokens.weight"], state["tok_embeddings.weight"])
    # tied embeddings -> no lm_head in export iff tied
    if args.tie_word_embeddings:
        assert "lm_head.weight" not in hf

    # no fused names may leak into the export
    assert not any("wqkv" in k or "w_gate_up" in k for k in hf)


def test_fused_to_hf_state_unit():
    from convert_to_mlx_lm import fused_to_hf_state

    args = ModelArgs(hidden_size=16, intermediate_size=32, num_layers=1,
                     num_heads=2, num_kv_heads=1, vocab_size=50,
                     tie_word_embeddings=False)
    model = Model(args)
    hf = fused_to_hf_state(model.state_dict(), args)
    assert "lm_head.weight" in hf
    assert hf["model.layers.0.self_attn.q_proj.weight"].shape == (16, 16)
    assert hf["model.layers.0.self_attn.k_proj.weight"].shape == (8, 16)
    assert hf["model.layers.0.mlp.gate_proj.weight"].shape == (32, 16)


def test_train_tokenizer_cli(tmp_path):
    import train_tokenizer

    data = tmp_path / "docs.jsonl"
    with open(data, "w") as f:
        for i in range(50):
            f.write(json.dumps({"text": f"the quick brown fox {i} jumps over the lazy dog"}) + "\n")
    cfg = {
        "data": {
            "input_file": str(data),
            "tokenizer": {"normal_vocab_size": 300,
                          "special_tokens": {"pad": "<pad>", "bos": "<bos>", "eos": "<eos>"}},
        }
    }
    cfg_path = tmp_path / "tok.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    train_tokenizer.main(["--config", str(cfg_path), "--out-dir", str(tmp_path / "tok")])
    tok_json = tmp_path / "tok" / "tokenizer.json"
    assert tok_json.exists()
    from tokenizers import Tokenizer

    t = Tokenizer.from_file(str(tok_json))
    ids = t.encode("the quick brown fox").ids
    assert ids and t.decode(ids).strip().startswith("the")


def test_model_cli_list_details_load(tmp_path, capsys):
    import model_cli

    _trainer, run_dir = _tiny_run(tmp_path, name="tiny-cli")
    runs_root = run_dir.parent

    runs = model_cli.list_runs(runs_root)
    assert any(r["name"] == "tiny-cli" for r in runs)

    info = model_cli.run_details(run_dir)
    assert info["name"] == "tiny-cli"
    assert info.get("steps_logged", 0) >= 1

    cli = model_cli.ModelCLI(str(runs_root))
    cli.cmd_list()
    out = capsys.readouterr().out
    assert "tiny-cli" in out
    cli.cmd_load("tiny-cli")
    assert cli.trainer is not None
    cli.cmd_generate("hello")
    out = capsys.readouterr().out
    assert "tok/s" in out


def test_export_loads_in_hf_transformers(tmp_path):
    """The exported directory must load through HF transformers' own Llama
    and produce logits matching our model — the real interop check behind
    the mlx-lm/HF export format."""
    transformers = pytest.importorskip("transformers")
    from convert_to_mlx_lm import convert_run

    trainer, run_dir = _tiny_run(tmp_path, name="tiny-hf")
    out = convert_run(run_dir, tmp_path / "hf_export")

    # config.json advertises bf16; load fp32 for an exact numerics compare
    hf_model = transformers.AutoModelForCausalLM.from_pretrained(
        str(out), torch_dtype=torch.float32)
    hf_model.eval()
    toks = torch.randint(3, 50, (1, 16))
    with torch.no_grad():
        ours = trainer.model(toks).float()
        theirs = hf_model(toks).logits.float()
    assert torch.allclose(ours, theirs, atol=2e-3, rtol=2e-3), \
        (ours - theirs).abs().max()
