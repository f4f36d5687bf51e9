#!/usr/bin/env python3
"""Interactive REPL over trained runs: list / details / load / generate.

Parity surface: /root/reference/tools/model_cli.py:19-238.
"""
from __future__ import annotations

import argparse
import json
import sys
from pathlib import Path
from typing import Optional

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def list_runs(runs_root: Path) -> list:
    out = []
    if not runs_root.exists():
        return out
    for d in sorted(runs_root.iterdir()):
        if (d / "config.yaml").exists():
            meta = {}
            mp = d / "metadata.json"
            if mp.exists():
                try:
                    meta = json.loads(mp.read_text())
                except Exception:
                    pass
            ckpts = sorted((d / "checkpoints").glob("step_*_model.safetensors")) \
                if (d / "checkpoints").exists() else []
            out.append({"name": d.name, "path": str(d),
                        "n_checkpoints": len(ckpts), "metadata": meta})
    return out


def run_details(run_dir: Path) -> dict:
    info = {"name": run_dir.name}
    mp = run_dir / "metadata.json"
    if mp.exists():
        info["metadata"] = json.loads(mp.read_text())
    cfg = run_dir / "config.yaml"
    if cfg.exists():
        info["config"] = cfg.read_text()
    from mlx_cuda_distributed_pretraining_amd.utils.log_parse import parse_log_file

    records = parse_log_file(run_dir)
    if records:
        losses = [r.loss for r in records if r.loss is not None]
        info["steps_logged"] = len(records)
        info["final_loss"] = losses[-1] if losses else None
    return info


def load_model(run_dir: Path, checkpoint: str = "final"):
    from mlx_cuda_distributed_pretraining_amd.core.config import Config
    from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer

    cfg = Config.from_yaml(run_dir / "config.yaml")
    cfg.overwrite = True
    trainer = Trainer(cfg, for_training=False, runs_root=str(run_dir.parent))
    base = run_dir / "checkpoints" / f"step_{checkpoint}"
    trainer.load_checkpoint(str(base))
    return trainer


class ModelCLI:
    def __init__(self, runs_root: str = "runs"):
        self.runs_root = Path(runs_root)
        self.trainer = None
        self.current: Optional[str] = None

    def cmd_list(self, *_a) -> None:
        runs = list_runs(self.runs_root)
        if not runs:
            print(f"no runs under {self.runs_root}")
        for r in runs:
            print(f"  {r['name']:30s} checkpoints={r['n_checkpoints']}")

    def cmd_details(self, name: str, *_a) -> None:
        info = run_details(self.runs_root / name)
        print(json.dumps({k: v for k, v in info.items() if k != "config"}, indent=2))

    def cmd_load(self, name: str, checkpoint: str = "final", *_a) -> None:
        self.trainer = load_model(self.runs_root / name, checkpoint)
        self.current = name
        print(f"loaded {name} @ step_{checkpoint}")

    def cmd_generate(self, *words) -> None:
        if self.trainer is None:
            print("load a model first: load <run>")
            return
        prompt = " ".join(words)
        from mlx_cuda_distributed_pretraining_amd.inference.generate import generate

        text, stats = generate(
            self.trainer.model, self.trainer.tokenizer, prompt,
            max_tokens=128, temperature=0.7, top_p=0.9,
        )
        print(text)
        print(f"[{stats['generated_tokens']} toks @ {stats['tokens_per_second']:.1f} tok/s]")

    def repl(self) -> None:  # pragma: no cover - interactive
        print("commands: list | details <run> | load <run> [ckpt] | generate <prompt> | quit")
        while True:
            try:
                line = input(f"({self.current or 'no model'})> ").strip()
            except (EOFError, KeyboardInterrupt):
                break
            if not line:
                continue
            cmd, *rest = line.split()
            if cmd in ("quit", "exit", "q"):
                break
            fn = getattr(self, f"cmd_{cmd}", None)
            if fn is None:
                print(f"unknown command: {cmd}")
                continue
            try:
                fn(*rest)
            except Exception as e:
                print(f"error: {e}")


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Model management CLI")
    p.add_argument("--runs-root", default="runs")
    p.add_argument("command", nargs="*", help="one-shot command (default: REPL)")
    a = p.parse_args(argv)
    cli = ModelCLI(a.runs_root)
    if a.command:
        cmd, *rest = a.command
        getattr(cli, f"cmd_{cmd}")(*rest)
    else:  # pragma: no cover
        cli.repl()


if __name__ == "__main__":
    main()
