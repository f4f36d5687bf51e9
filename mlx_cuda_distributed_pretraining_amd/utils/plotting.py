"""Loss-curve plotting from a run directory's log.txt.

Parity surface: /root/reference/utils/plotting.py (parses log.txt with
regexes, plots train/val loss + ppl + lr) and the root plot-logs*.py /
plot_curve.py scripts. CLI: ``python -m mlx_cuda_distributed_pretraining_amd.utils.plotting <run-or-log> [...]``.
"""
from __future__ import annotations

import argparse
import csv
import sys
from pathlib import Path
from typing import List, Optional

from .log_parse import LogRecord, parse_log_file


def records_to_csv(records: List[LogRecord], out_path: str | Path) -> None:
    fields = ["step", "loss", "val_loss", "ppl", "val_ppl", "lr", "toks", "tokens_per_sec", "grad_norm"]
    with open(out_path, "w", newline="") as f:
        w = csv.DictWriter(f, fieldnames=fields)
        w.writeheader()
        for r in records:
            w.writerow({k: getattr(r, k) for k in fields})


def plot_run(
    run_path: str | Path,
    out_path: Optional[str | Path] = None,
    smooth: int = 1,
    show: bool = False,
    title: Optional[str] = None,
):
    """Plot loss / val-loss / lr / tok-s curves for one run. Returns the
    figure (or None when matplotlib is unavailable)."""
    records = parse_log_file(run_path)
    if not records:
        raise ValueError(f"no parseable log lines under {run_path}")
    try:
        import matplotlib

        matplotlib.use("Agg" if not show else matplotlib.get_backend())
        import matplotlib.pyplot as plt
    except Exception:  # pragma: no cover - matplotlib always present in image
        print("matplotlib unavailable; use records_to_csv instead", file=sys.stderr)
        return None

    steps = [r.step for r in records if r.loss is not None]
    losses = [r.loss for r in records if r.loss is not None]
    if smooth > 1 and len(losses) > smooth:
        import numpy as np

        kern = np.ones(smooth) / smooth
        losses = list(np.convolve(losses, kern, mode="valid"))
        steps = steps[smooth - 1 :]

    fig, axes = plt.subplots(2, 2, figsize=(12, 8))
    axes[0][0].plot(steps, losses, label="train loss")
    vsteps = [r.step for r in records if r.val_loss is not None]
    vlosses = [r.val_loss for r in records if r.val_loss is not None]
    if vlosses:
        axes[0][0].plot(vsteps, vlosses, "o-", label="val loss")
    axes[0][0].set_xlabel("step"); axes[0][0].set_ylabel("loss"); axes[0][0].legend()
    axes[0][0].set_title(title or str(run_path))

    ppls = [(r.step, r.ppl) for r in records if r.ppl is not None]
    if ppls:
        axes[0][1].plot(*zip(*ppls)); axes[0][1].set_ylabel("perplexity"); axes[0][1].set_yscale("log")
    lrs = [(r.step, r.lr) for r in records if r.lr is not None]
    if lrs:
        axes[1][0].plot(*zip(*lrs)); axes[1][0].set_ylabel("learning rate")
    tps = [(r.step, r.tokens_per_sec) for r in records if r.tokens_per_sec is not None]
    if tps:
        axes[1][1].plot(*zip(*tps)); axes[1][1].set_ylabel("tokens/sec")
    for ax in axes.flat:
        ax.set_xlabel("step"); ax.grid(alpha=0.3)
    fig.tight_layout()

    if out_path is None and not show:
        p = Path(run_path)
        out_path = (p if p.is_dir() else p.parent) / "loss_curve.png"
    if out_path is not None:
        fig.savefig(out_path, dpi=120)
    if show:  # pragma: no cover
        plt.show()
    return fig


def compare_runs(run_paths: List[str], out_path: str = "comparison.png", metric: str = "loss"):
    """Overlay a metric across several runs (reference plot-logs.py behavior)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(10, 6))
    for rp in run_paths:
        records = parse_log_file(rp)
        pts = [(r.step, getattr(r, metric)) for r in records if getattr(r, metric) is not None]
        if pts:
            ax.plot(*zip(*pts), label=Path(rp).name)
    ax.set_xlabel("step"); ax.set_ylabel(metric); ax.legend(); ax.grid(alpha=0.3)
    fig.tight_layout(); fig.savefig(out_path, dpi=120)
    return fig


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Plot training curves from run log(s)")
    p.add_argument("runs", nargs="+", help="run directory(ies) or log.txt path(s)")
    p.add_argument("--out", default=None)
    p.add_argument("--smooth", type=int, default=1)
    p.add_argument("--csv", default=None, help="also dump parsed records to CSV")
    p.add_argument("--metric", default="loss", help="metric for multi-run comparison")
    args = p.parse_args(argv)
    if len(args.runs) == 1:
        plot_run(args.runs[0], out_path=args.out, smooth=args.smooth)
        if args.csv:
            records_to_csv(parse_log_file(args.runs[0]), args.csv)
    else:
        compare_runs(args.runs, out_path=args.out or "comparison.png", metric=args.metric)


if __name__ == "__main__":
    main()
