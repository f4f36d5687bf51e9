"""Deterministic-seed loss-curve regression (SURVEY.md §4 implication):
the tiny-config curve must reproduce the stored reference exactly (same
seeds, same op order) — catches silent numerics changes in the op stack.
Regenerate tests/data/loss_curve_ref.json ONLY for intentional changes."""
import json
import tempfile
from pathlib import Path

import pytest

from mlx_cuda_distributed_pretraining_amd.core.config import Config
from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer

REPO = Path(__file__).resolve().parents[1]


def test_loss_curve_matches_reference():
    ref = json.load(open(REPO / "tests" / "data" / "loss_curve_ref.json"))
    cfg = Config.from_yaml(REPO / "configs" / "model-config-sample.yaml")
    cfg.name = "curve-check"
    cfg.overwrite = True
    cfg.data.synthetic = True
    cfg.model.dimensions = {"hidden_size": 64, "intermediate_size": 128, "num_layers": 2}
    cfg.model.attention = {"num_heads": 4, "num_kv_heads": 2, "head_dim": 16,
                           "max_position_embeddings": 64}
    cfg.data.preprocessing["max_context_size"] = 64
    cfg.training.hyperparameters.update({"batch_size": 4, "iters": 10,
                                         "learning_rate": 1e-3})
    cfg.logging.steps = {"logging_interval": 0, "checkpoint_interval": 0,
                         "validation_interval": 0}
    t = Trainer(cfg, runs_root=tempfile.mkdtemp())
    losses = []
    for i in range(10):
        loss, _ = t.train_step(i)
        losses.append(float(loss.detach()))
    for i, (got, want) in enumerate(zip(losses, ref)):
        assert got == pytest.approx(want, abs=1e-4), \
            f"step {i}: {got} != ref {want} — numerics changed"
