"""End-to-end trainer on the 2M sample config (CPU)."""
import json
import re
from pathlib import Path

import torch

from mlx_cuda_distributed_pretraining_amd.core.config import Config
from mlx_cuda_distributed_pretraining_amd.core.trainer import EarlyStoppingMonitor, Trainer

REPO = Path(__file__).resolve().parents[1]


def small_cfg(tmp_runs, **overrides):
    cfg = Config.from_yaml(str(REPO / "configs" / "model-config-sample.yaml"))
    cfg.model.dimensions = {"hidden_size": 32, "intermediate_size": 64, "num_layers": 2}
    cfg.model.attention = {"num_heads": 2, "num_kv_heads": None, "head_dim": None,
                           "max_position_embeddings": None}
    cfg.data.preprocessing["max_context_size"] = 32
    cfg.training.hyperparameters["batch_size"] = 4
    cfg.training.hyperparameters["iters"] = 12
    cfg.logging.steps = {"logging_interval": 4, "checkpoint_interval": 0,
                         "validation_interval": 6}
    for k, v in overrides.items():
        setattr(cfg, k, v)
    return cfg


def test_end_to_end_training_run(tmp_runs):
    cfg = small_cfg(tmp_runs)
    trainer = Trainer(cfg, runs_root=tmp_runs)
    trainer.train()
    run_dir = Path(tmp_runs) / cfg.name
    assert (run_dir / "config.yaml").exists()
    log = (run_dir / "log.txt").read_text()
    # log-line contract (reference format: Step N: loss=... | ppl=... | ...)
    m = re.search(r"Step \d+: .*loss=\d\.\d+e[+-]\d+ \| ppl=[\d.]+ \| .*tok/s=[\d.]+K", log)
    assert m, f"log line format mismatch:\n{log}"
    assert (run_dir / "checkpoints" / "step_final_model.safetensors").exists()
    meta = json.loads((run_dir / "metadata.json").read_text())
    assert meta["checkpoints"]


def test_loss_decreases_over_run(tmp_runs):
    cfg = small_cfg(tmp_runs)
    cfg.training.hyperparameters["iters"] = 30
    cfg.training.hyperparameters["learning_rate"] = 1e-2
    cfg.training.scheduler = {"type": "constant"}
    trainer = Trainer(cfg, runs_root=tmp_runs)
    losses = []
    for _ in range(30):
        trainer.current_step = 0
        # same batch every time (step=0): the model must memorize it, so the
        # loss must fall well below the uniform-entropy floor.
        loss, _ = trainer.train_step(0)
        losses.append(float(loss))
    assert sum(losses[-5:]) / 5 < sum(losses[:5]) / 5 * 0.9


def test_checkpoint_resume_bitexact(tmp_runs):
    cfg = small_cfg(tmp_runs)
    cfg.training.hyperparameters["iters"] = 6
    cfg.logging.steps["checkpoint_interval"] = 3
    cfg.logging.steps["validation_interval"] = 0
    trainer = Trainer(cfg, runs_root=tmp_runs)
    trainer.train()
    final_sd = {k: v.clone() for k, v in trainer.model.state_dict().items()}

    # resume from step 3, retrain to 6, expect the same data path
    cfg2 = small_cfg(tmp_runs)
    cfg2.name = cfg.name
    cfg2.overwrite = True
    cfg2.training.hyperparameters["iters"] = 6
    cfg2.logging.steps["checkpoint_interval"] = 0
    cfg2.logging.steps["validation_interval"] = 0
    from mlx_cuda_distributed_pretraining_amd.core.config import ResumeConfig

    ckpt_base = str(Path(tmp_runs) / cfg.name / "checkpoints" / "step_3")
    cfg2.resume = ResumeConfig(checkpoint=ckpt_base)
    trainer2 = Trainer(cfg2, runs_root=tmp_runs)
    trainer2.train()
    for k, v in trainer2.model.state_dict().items():
        assert torch.allclose(v, final_sd[k], atol=1e-5), f"mismatch after resume: {k}"


def test_early_stopping_monitor():
    es = EarlyStoppingMonitor(patience=2, min_delta=0.01)
    assert not es.update(1.0)
    assert not es.update(0.5)   # improvement
    assert not es.update(0.5)   # no improvement (1)
    assert es.update(0.5)       # no improvement (2) -> stop


def test_lr_finder(tmp_runs):
    cfg = small_cfg(tmp_runs)
    cfg.training.lr_finder = {"enabled": True, "min_lr": 1e-6, "max_lr": 1e-1, "num_steps": 8}
    trainer = Trainer(cfg, runs_root=tmp_runs)
    trainer.train()
    csv_path = Path(tmp_runs) / cfg.name / "lr_finder.csv"
    assert csv_path.exists()
    lines = csv_path.read_text().strip().splitlines()
    assert len(lines) >= 3


def test_grad_accumulation_equivalence(tmp_runs):
    # accum=2 with bs=2 should equal accum=1 with the same data... we check it runs
    # and produces finite loss (exact equality needs identical batch composition).
    cfg = small_cfg(tmp_runs)
    cfg.name = "accum-test"
    cfg.training.hyperparameters["gradient_accumulation_steps"] = 2
    trainer = Trainer(cfg, runs_root=tmp_runs)
    loss, ntok = trainer.train_step(0)
    assert torch.isfinite(loss)
    assert ntok.item() > 0


def test_epochs_mode(tmp_runs):
    import json as _json

    data_file = Path(tmp_runs).parent / "train.jsonl"
    data_file.parent.mkdir(parents=True, exist_ok=True)
    data_file.write_text("\n".join(_json.dumps({"text": "hello world " * 5}) for _ in range(16)))
    cfg = small_cfg(tmp_runs)
    cfg.name = "epochs-test"
    cfg.data.synthetic = False
    cfg.data.input_file = str(data_file)
    cfg.training.epochs = 2
    trainer = Trainer(cfg, runs_root=tmp_runs)
    assert trainer.total_steps == trainer.steps_per_epoch * 2
    trainer.train()


def test_emergency_checkpoint_on_failure(tmp_runs):
    """Hot-loop failure -> emergency checkpoint saved, exception re-raised
    (SURVEY.md §5.3 recovery story)."""
    from pathlib import Path

    cfg = small_cfg(tmp_runs)
    cfg.name = "crash-run"
    cfg.overwrite = True
    cfg.data.synthetic = True
    cfg.training.hyperparameters["iters"] = 5
    trainer = Trainer(cfg, runs_root=tmp_runs)

    calls = {"n": 0}
    orig = trainer.train_step

    def exploding(step):
        calls["n"] += 1
        if calls["n"] >= 3:
            raise RuntimeError("simulated rank failure")
        return orig(step)

    trainer.train_step = exploding
    import pytest as _pytest

    with _pytest.raises(RuntimeError, match="simulated rank failure"):
        trainer.train()
    ckpts = list((Path(tmp_runs) / "crash-run" / "checkpoints").glob("step_emergency_*_model.safetensors"))
    assert ckpts, "no emergency checkpoint written"


def test_auto_resume_end_to_end(tmp_path):
    """--auto-resume closes the elastic-restart loop: run 4 iters with a
    checkpoint at 2, then re-invoke with --auto-resume and more iters — the
    second run must resume from step 2's checkpoint (not restart), and a
    fresh name with --auto-resume must start clean."""
    import yaml
    from mlx_cuda_distributed_pretraining_amd.core import training as T
    from mlx_cuda_distributed_pretraining_amd.core.checkpoint import latest_checkpoint

    cfg = {
        "name": "autoresume-test",
        "data": {"synthetic": True, "synthetic_vocab_size": 64,
                 "preprocessing": {"max_context_size": 32}},
        "model": {"dimensions": {"hidden_size": 32, "intermediate_size": 64, "num_layers": 1},
                  "attention": {"num_heads": 2, "num_kv_heads": 2,
                                "max_position_embeddings": 64}},
        "training": {"hyperparameters": {"iters": 4, "batch_size": 2, "learning_rate": 1e-3}},
        "logging": {"steps": {"logging_interval": 0, "checkpoint_interval": 2,
                              "validation_interval": 0}},
        "system": {"device": "cpu"},
    }
    cfg_path = tmp_path / "cfg.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    runs = tmp_path / "runs"

    # first start: no run dir -> auto-resume falls through to a clean start
    T.main(["--config", str(cfg_path), "--auto-resume", "--runs-root", str(runs)])
    latest = latest_checkpoint(runs / "autoresume-test")
    assert latest is not None and latest.endswith("step_4")

    # restart with more iters: must RESUME (run dir exists -> without resume
    # the unique-name check would raise)
    T.main(["--config", str(cfg_path), "--auto-resume", "--runs-root", str(runs),
            "--iters", "6"])
    assert latest_checkpoint(runs / "autoresume-test").endswith("step_6")
    text = (runs / "autoresume-test" / "log.txt").read_text()
    assert "Resumed from" in text


def test_latest_checkpoint_prefers_emergency_when_newer(tmp_path):
    from mlx_cuda_distributed_pretraining_amd.core.checkpoint import latest_checkpoint

    ck = tmp_path / "run" / "checkpoints"
    ck.mkdir(parents=True)
    for name in ("step_2_state.json", "step_emergency_5_state.json",
                 "step_final_state.json"):
        (ck / name).write_text("{}")
    assert latest_checkpoint(tmp_path / "run").endswith("step_emergency_5")
    # regular snapshot at the same (or higher) step wins
    (ck / "step_5_state.json").write_text("{}")
    assert latest_checkpoint(tmp_path / "run").endswith("step_5")
    (ck / "step_emergency_7_state.json").write_text("{}")
    assert latest_checkpoint(tmp_path / "run").endswith("step_emergency_7")
