from pathlib import Path

import yaml

from mlx_cuda_distributed_pretraining_amd.core.config import Config

REPO = Path(__file__).resolve().parents[1]


def test_sample_config_parses():
    cfg = Config.from_yaml(str(REPO / "configs" / "model-config-sample.yaml"))
    assert cfg.name == "llama-2m-sample"
    assert cfg.model.dimensions["hidden_size"] == 128
    assert cfg.training.hyperparameters["batch_size"] == 16
    assert cfg.system.seed == 42


def test_reference_style_config_parses(tmp_path):
    # the reference's schema fields (epochs at training top level, resume block)
    d = {
        "name": "t",
        "data": {"input_file": "x.jsonl", "preprocessing": {"max_context_size": 64}},
        "model": {"architecture": "llama", "dimensions": {"hidden_size": 32}},
        "training": {"epochs": 2, "hyperparameters": {"batch_size": 4}},
        "logging": {"log_dir": "logs", "checkpoint_dir": "ckpt"},
        "system": {"seed": 1, "device": "cpu"},
        "resume": {"checkpoint": "runs/t/checkpoints/step_10"},
    }
    p = tmp_path / "c.yaml"
    p.write_text(yaml.safe_dump(d))
    cfg = Config.from_yaml(str(p))
    assert cfg.training.epochs == 2
    assert cfg.resume.checkpoint.endswith("step_10")


def test_config_roundtrip(tmp_path):
    cfg = Config.from_yaml(str(REPO / "configs" / "model-config-1b.yaml"))
    out = tmp_path / "out.yaml"
    cfg.save_yaml(out)
    cfg2 = Config.from_yaml(str(out))
    assert cfg2.model.dimensions == cfg.model.dimensions
    assert cfg2.training.hyperparameters == cfg.training.hyperparameters


def test_all_baseline_configs_parse():
    for name in [
        "model-config-sample", "model-config-124m", "model-config-1b",
        "model-config-400m-muon", "model-config-256m-flex",
    ]:
        cfg = Config.from_yaml(str(REPO / "configs" / f"{name}.yaml"))
        assert cfg.name


def test_config_extends_deep_merge(tmp_path):
    """extends: base.yaml — child sections deep-merge over the base."""
    from mlx_cuda_distributed_pretraining_amd.core.config import Config

    (tmp_path / "base.yaml").write_text("""
name: base
data:
  synthetic: true
  preprocessing: {max_context_size: 512, chunk_overlap: 0}
model:
  architecture: llama
  dimensions: {hidden_size: 64, intermediate_size: 128, num_layers: 2}
training:
  hyperparameters: {batch_size: 4, learning_rate: 1.0e-3, iters: 10}
""")
    (tmp_path / "child.yaml").write_text("""
extends: base.yaml
name: child
training:
  hyperparameters: {learning_rate: 5.0e-4}
""")
    cfg = Config.from_yaml(str(tmp_path / "child.yaml"))
    assert cfg.name == "child"
    # overridden leaf
    assert cfg.training.hyperparameters["learning_rate"] == 5.0e-4
    # inherited siblings survive the deep merge
    assert cfg.training.hyperparameters["batch_size"] == 4
    assert cfg.model.dimensions["hidden_size"] == 64
    assert cfg.data.preprocessing["max_context_size"] == 512


def test_config_extends_cycle_guard(tmp_path):
    from mlx_cuda_distributed_pretraining_amd.core.config import Config

    (tmp_path / "a.yaml").write_text("extends: b.yaml\nname: a\n")
    (tmp_path / "b.yaml").write_text("extends: a.yaml\nname: b\n")
    import pytest as _pytest

    with _pytest.raises(ValueError, match="extends"):
        Config.from_yaml(str(tmp_path / "a.yaml"))
