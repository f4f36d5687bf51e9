"""Live training monitoring: tail a run's log.txt, keep rolling statistics,
optionally live-update a matplotlib figure.

Parity surface: /root/reference/utils/monitoring.py (TrainingMonitor regex
parser + plot_real_time) and utils/realtime_plotting.py. The regex contract
lives in log_parse.py and is shared with plotting.py.
"""
from __future__ import annotations

import time
from pathlib import Path
from typing import Dict, List, Optional

from .log_parse import LogRecord, parse_log_line


class TrainingMonitor:
    """Incremental log follower. ``poll()`` reads any new lines; ``summary()``
    returns aggregate statistics of what has been seen so far."""

    def __init__(self, run_path: str | Path):
        p = Path(run_path)
        self.log_path = p / "log.txt" if p.is_dir() else p
        self.records: List[LogRecord] = []
        self._pos = 0

    def poll(self) -> List[LogRecord]:
        """Read newly appended lines; returns the new records."""
        if not self.log_path.exists():
            return []
        new: List[LogRecord] = []
        with open(self.log_path) as f:
            f.seek(self._pos)
            for line in f:
                rec = parse_log_line(line)
                if rec is not None:
                    new.append(rec)
            self._pos = f.tell()
        self.records.extend(new)
        return new

    def summary(self) -> Dict[str, Optional[float]]:
        losses = [r.loss for r in self.records if r.loss is not None]
        vlosses = [r.val_loss for r in self.records if r.val_loss is not None]
        tps = [r.tokens_per_sec for r in self.records if r.tokens_per_sec is not None]
        toks = [r.toks for r in self.records if r.toks is not None]
        return {
            "steps_seen": len(self.records),
            "last_step": self.records[-1].step if self.records else None,
            "last_loss": losses[-1] if losses else None,
            "min_loss": min(losses) if losses else None,
            "last_val_loss": vlosses[-1] if vlosses else None,
            "best_val_loss": min(vlosses) if vlosses else None,
            "mean_tokens_per_sec": sum(tps) / len(tps) if tps else None,
            "total_tokens": max(toks) if toks else None,
        }

    def follow(self, interval: float = 2.0, max_seconds: Optional[float] = None,
               callback=None) -> None:  # pragma: no cover - interactive loop
        """Poll until interrupted (or max_seconds), printing/forwarding new
        records. Reference behavior: utils/monitoring.py plot_real_time."""
        t0 = time.time()
        while max_seconds is None or time.time() - t0 < max_seconds:
            for rec in self.poll():
                if callback:
                    callback(rec)
                else:
                    print(f"step {rec.step}: loss={rec.loss} val={rec.val_loss} "
                          f"tok/s={rec.tokens_per_sec}")
            time.sleep(interval)


def plot_real_time(run_path: str, interval: float = 5.0,
                   max_seconds: Optional[float] = None) -> None:  # pragma: no cover
    """Live-updating loss plot (reference utils/realtime_plotting.py)."""
    import matplotlib.pyplot as plt

    mon = TrainingMonitor(run_path)
    plt.ion()
    fig, ax = plt.subplots()
    (line,) = ax.plot([], [], label="train loss")
    (vline,) = ax.plot([], [], "o-", label="val loss")
    ax.legend(); ax.grid(alpha=0.3); ax.set_xlabel("step"); ax.set_ylabel("loss")
    t0 = time.time()
    while max_seconds is None or time.time() - t0 < max_seconds:
        mon.poll()
        xs = [r.step for r in mon.records if r.loss is not None]
        ys = [r.loss for r in mon.records if r.loss is not None]
        line.set_data(xs, ys)
        vx = [r.step for r in mon.records if r.val_loss is not None]
        vy = [r.val_loss for r in mon.records if r.val_loss is not None]
        vline.set_data(vx, vy)
        ax.relim(); ax.autoscale_view()
        fig.canvas.draw_idle(); fig.canvas.flush_events()
        time.sleep(interval)


def main(argv=None) -> None:  # pragma: no cover
    import argparse

    p = argparse.ArgumentParser(description="Monitor a training run")
    p.add_argument("run")
    p.add_argument("--interval", type=float, default=2.0)
    p.add_argument("--plot", action="store_true")
    args = p.parse_args(argv)
    if args.plot:
        plot_real_time(args.run, interval=args.interval)
    else:
        TrainingMonitor(args.run).follow(interval=args.interval)


if __name__ == "__main__":  # pragma: no cover
    main()
