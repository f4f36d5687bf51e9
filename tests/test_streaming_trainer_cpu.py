"""End-to-end: Trainer consuming the streaming shard pipeline."""
import json
from pathlib import Path

from mlx_cuda_distributed_pretraining_amd.core.config import Config
from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer

REPO = Path(__file__).resolve().parent.parent


def test_trainer_with_streaming_source(tmp_path):
    shards = tmp_path / "shards"
    shards.mkdir()
    for s in range(2):
        with open(shards / f"part-{s}.jsonl", "w") as f:
            for i in range(40):
                f.write(json.dumps({"text": f"streaming doc {s}/{i} " + "word " * 30}) + "\n")

    cfg = Config.from_yaml(REPO / "configs" / "model-config-sample.yaml")
    cfg.name = "stream-run"
    cfg.overwrite = True
    cfg.data.synthetic = False
    cfg.data.streaming = {"source": str(shards), "cache_dir": str(tmp_path / "cache"),
                          "max_cache_gb": 1}
    cfg.data.input_file = None
    cfg.data.preprocessing["max_context_size"] = 32
    cfg.training.hyperparameters["iters"] = 3
    cfg.training.hyperparameters["batch_size"] = 2
    cfg.logging.steps = {"logging_interval": 1, "checkpoint_interval": 0,
                         "validation_interval": 0}
    trainer = Trainer(cfg, runs_root=str(tmp_path / "runs"))
    trainer.train()

    log = (tmp_path / "runs" / "stream-run" / "log.txt").read_text()
    assert "Step" in log
    # disk cache was populated by the stream
    assert any((tmp_path / "cache").iterdir())
