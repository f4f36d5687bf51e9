"""CLI: ``python -m mlx_cuda_distributed_pretraining_amd.core.training --config X``
(also reachable as ``python -m core.training`` via the repo-root shim).

Parity with the reference CLI (/root/reference/core/training.py:1907-2016):
--config plus overrides (run-id suffix, log interval, mixed precision,
gradient checkpointing, find-lr, tensorboard), merged into the config and
run. A ``train(config)`` helper accepts a dict or a path
(reference :2039-2083).
"""
from __future__ import annotations

import argparse
from typing import Any, Dict, Union

from .config import Config
from .trainer import Trainer


def build_arg_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description="MI355X-native Llama pretraining")
    p.add_argument("--config", type=str, required=True, help="YAML config path")
    p.add_argument("--run-id", type=str, default=None, help="suffix appended to the run name")
    p.add_argument("--log-interval", type=int, default=None)
    p.add_argument("--checkpoint-interval", type=int, default=None)
    p.add_argument("--iters", type=int, default=None)
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--mixed-precision", action="store_true")
    p.add_argument("--precision", type=str, default=None, choices=["float16", "bfloat16", "float32"])
    p.add_argument("--grad-checkpointing", action="store_true")
    p.add_argument("--find-lr", action="store_true")
    p.add_argument("--tensorboard", action="store_true")
    p.add_argument("--overwrite", action="store_true")
    p.add_argument("--resume", type=str, default=None, help="checkpoint base path to resume from")
    p.add_argument("--auto-resume", action="store_true",
                   help="resume from this run's newest step checkpoint if one exists "
                        "(closes the elastic-restart loop: torchrun --max-restarts N "
                        "-m ...training --config X --auto-resume)")
    p.add_argument("--runs-root", type=str, default="runs")
    return p


def apply_overrides(config: Config, args: argparse.Namespace) -> Config:
    if args.run_id:
        config.name = f"{config.name}-{args.run_id}"
    if args.log_interval is not None:
        config.logging.steps["logging_interval"] = args.log_interval
    if args.checkpoint_interval is not None:
        config.logging.steps["checkpoint_interval"] = args.checkpoint_interval
    if args.iters is not None:
        config.training.hyperparameters["iters"] = args.iters
        config.training.epochs = None
    if args.batch_size is not None:
        config.training.hyperparameters["batch_size"] = args.batch_size
    if args.mixed_precision:
        config.system.mixed_precision = True
    if args.precision:
        config.system.precision = args.precision
    if args.grad_checkpointing:
        config.system.gradient_checkpointing = True
    if args.find_lr:
        config.training.lr_finder["enabled"] = True
    if args.tensorboard:
        config.logging.tensorboard = True
    if args.overwrite:
        config.overwrite = True
    if args.resume:
        from .config import ResumeConfig

        config.resume = ResumeConfig(checkpoint=args.resume)
    elif getattr(args, "auto_resume", False):
        from pathlib import Path

        from .checkpoint import latest_checkpoint
        from .config import ResumeConfig

        latest = latest_checkpoint(Path(getattr(args, "runs_root", "runs")) / config.name)
        if latest:
            config.resume = ResumeConfig(checkpoint=latest)
        else:
            # crashed (or first start) before any checkpoint: start clean
            config.overwrite = True
    return config


def main(argv=None) -> None:
    args = build_arg_parser().parse_args(argv)
    config = Config.from_yaml(args.config)
    config = apply_overrides(config, args)
    trainer = Trainer(config, runs_root=args.runs_root)
    trainer.train()


def train(config: Union[str, Dict[str, Any], Config]) -> Trainer:
    """Programmatic entry: accepts a path, dict, or Config."""
    if isinstance(config, dict):
        config = Config.from_dict(config)
    elif isinstance(config, str):
        config = Config.from_yaml(config)
    trainer = Trainer(config)
    trainer.train()
    return trainer


if __name__ == "__main__":
    main()
