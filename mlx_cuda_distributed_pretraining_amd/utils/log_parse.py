"""Parser for the trainer's structured log line — the contract shared by
plotting/monitoring tools (reference parses its log.txt with regexes at
/root/reference/utils/monitoring.py:100-190 and utils/plotting.py:7-98; we
keep the same approach so post-hoc tools never need training internals).

Line format (Logger.format_metrics, core/logger.py):
  Step N: loss=1.234e+00 | ppl=3.43 [| val_loss=... | val_ppl=...]
      | tok/s=123.45K | toks=8192 | lr=1.0e-03 [| grad_norm=...]
"""
from __future__ import annotations

import re
from dataclasses import dataclass
from pathlib import Path
from typing import List, Optional

_STEP_RE = re.compile(r"Step\s+(\d+):")
_FIELD_RES = {
    "loss": re.compile(r"(?<!val_)loss=([\d.eE+-]+)"),
    "val_loss": re.compile(r"val_loss=([\d.eE+-]+)"),
    "ppl": re.compile(r"(?<!val_)ppl=([\d.eE+-]+)"),
    "val_ppl": re.compile(r"val_ppl=([\d.eE+-]+)"),
    "lr": re.compile(r"lr=([\d.eE+-]+)"),
    "toks": re.compile(r"toks=(\d+)"),
    "grad_norm": re.compile(r"grad_norm=([\d.eE+-]+)"),
}
_TPS_RE = re.compile(r"tok/s=([\d.]+)([KM]?)")


@dataclass
class LogRecord:
    step: int
    loss: Optional[float] = None
    val_loss: Optional[float] = None
    ppl: Optional[float] = None
    val_ppl: Optional[float] = None
    lr: Optional[float] = None
    toks: Optional[int] = None
    tokens_per_sec: Optional[float] = None
    grad_norm: Optional[float] = None


def parse_log_line(line: str) -> Optional[LogRecord]:
    m = _STEP_RE.search(line)
    if not m:
        return None
    rec = LogRecord(step=int(m.group(1)))
    for name, rx in _FIELD_RES.items():
        fm = rx.search(line)
        if fm:
            val = float(fm.group(1)) if name != "toks" else int(fm.group(1))
            setattr(rec, name, val)
    tm = _TPS_RE.search(line)
    if tm:
        mult = {"": 1.0, "K": 1e3, "M": 1e6}[tm.group(2)]
        rec.tokens_per_sec = float(tm.group(1)) * mult
    return rec


def parse_log_file(path: str | Path) -> List[LogRecord]:
    out: List[LogRecord] = []
    p = Path(path)
    if p.is_dir():
        p = p / "log.txt"
    if not p.exists():
        return out
    with open(p) as f:
        for line in f:
            rec = parse_log_line(line)
            if rec is not None:
                out.append(rec)
    return out
