"""Training logger.

Keeps the reference's log-line contract so its plotting/monitoring regexes
keep working (/root/reference/core/training.py:197-321, :1396-1435):
``Step N: loss=1.234e+00 | ppl=3.43 | tok/s=123.45K | toks=8192 | lr=1.0e-03``
fanned out to console + runs/<name>/log.txt, optional TensorBoard.

Distributed addition: only rank 0 writes; per-rank metrics are reduced by the
trainer before logging.
"""
from __future__ import annotations

import math
import time
from pathlib import Path
from typing import Any, Dict, Optional


class Logger:
    def __init__(
        self,
        run_dir: Path,
        log_file: Path,
        config: Any = None,
        is_main: bool = True,
        use_tensorboard: bool = False,
    ):
        self.run_dir = Path(run_dir)
        self.log_file = Path(log_file)
        self.is_main = is_main
        self.tb = None
        if is_main and use_tensorboard:
            try:
                from torch.utils.tensorboard import SummaryWriter

                self.tb = SummaryWriter(log_dir=str(self.run_dir / "tensorboard"))
            except Exception:
                self.tb = None
        if self.is_main:
            self.log_file.parent.mkdir(parents=True, exist_ok=True)

    def log(self, message: str, console: bool = True) -> None:
        if not self.is_main:
            return
        if console:
            print(message, flush=True)
        with open(self.log_file, "a") as f:
            f.write(message + "\n")

    def log_metrics_line(self, step: int, metrics_str: str) -> None:
        self.log(f"Step {step}: {metrics_str}")

    def tb_scalars(self, step: int, scalars: Dict[str, float]) -> None:
        if self.tb is not None:
            for k, v in scalars.items():
                self.tb.add_scalar(k, v, step)

    def log_model_summary(self, model) -> None:
        if not self.is_main:
            return
        n_params = sum(p.numel() for p in model.parameters())
        n_trainable = sum(p.numel() for p in model.parameters() if p.requires_grad)
        self.log(f"Model parameters: {n_params/1e6:.2f}M total, {n_trainable/1e6:.2f}M trainable")

    def log_memory_usage(self) -> None:
        if not self.is_main:
            return
        try:
            import torch

            if torch.cuda.is_available():
                alloc = torch.cuda.memory_allocated() / 2**30
                peak = torch.cuda.max_memory_allocated() / 2**30
                self.log(f"Memory: allocated={alloc:.2f}GB peak={peak:.2f}GB")
                return
        except Exception:
            pass
        try:
            import psutil

            rss = psutil.Process().memory_info().rss / 2**30
            self.log(f"Memory: rss={rss:.2f}GB")
        except Exception:
            pass

    def close(self) -> None:
        if self.tb is not None:
            self.tb.close()


def format_metrics(
    step: int,
    loss: float,
    tokens: int,
    total_tokens: int,
    start_time: float,
    lr: float,
    val_loss: Optional[float] = None,
    metrics_flags: Optional[Dict[str, bool]] = None,
    epochs: Optional[int] = None,
    steps_per_epoch: Optional[int] = None,
    grad_accum_steps: int = 1,
    effective_batch_size: Optional[int] = None,
) -> str:
    """Build the per-step metric string in the reference's format."""
    flags = metrics_flags or {}
    parts = []
    if epochs is not None and steps_per_epoch:
        current_epoch = step // steps_per_epoch + 1
        epoch_step = step % steps_per_epoch + 1
        parts.append(f"epoch={current_epoch}/{epochs} ({epoch_step}/{steps_per_epoch})")
    if flags.get("log_loss", True):
        parts.append(f"loss={loss:.3e}")
        if val_loss is not None:
            parts.append(f"val_loss={val_loss:.3e}")
    if flags.get("log_perplexity", True):
        parts.append(f"ppl={math.exp(min(loss, 20.0)):.2f}")
        if val_loss is not None:
            parts.append(f"val_ppl={math.exp(min(val_loss, 20.0)):.2f}")
    if flags.get("log_tokens_per_second", True):
        elapsed = max(time.time() - start_time, 1e-9)
        parts.append(f"tok/s={total_tokens / (1000 * elapsed):.2f}K")
    if flags.get("log_tokens_processed", True):
        parts.append(f"toks={tokens}")
    if flags.get("log_learning_rate", True):
        parts.append(f"lr={lr:.3e}")
    if grad_accum_steps > 1:
        parts.append(f"accum={grad_accum_steps}")
        if effective_batch_size:
            parts.append(f"eff_bs={effective_batch_size}")
    return " | ".join(parts)
