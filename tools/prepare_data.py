#!/usr/bin/env python3
"""Data preparation utilities: validate JSONL, train/val split, token stats.

Parity surface: /root/reference/prepare_data_a100.py:13-141 (validate/split),
/root/reference/examine.py (token counting).
"""
from __future__ import annotations

import argparse
import json
import random
import sys
from pathlib import Path
from typing import Dict, Optional

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def validate_jsonl(path: str | Path, max_errors: int = 20) -> Dict:
    """Check every line parses and has a non-empty "text" field."""
    n_ok = n_bad = n_empty = 0
    errors = []
    with open(path) as f:
        for i, line in enumerate(f, 1):
            line = line.strip()
            if not line:
                continue
            try:
                doc = json.loads(line)
                text = doc.get("text", "")
                if not text:
                    n_empty += 1
                else:
                    n_ok += 1
            except json.JSONDecodeError as e:
                n_bad += 1
                if len(errors) < max_errors:
                    errors.append({"line": i, "error": str(e)})
    return {"ok": n_ok, "bad": n_bad, "empty": n_empty, "errors": errors,
            "valid": n_bad == 0}


def split_jsonl(path: str | Path, train_out: str | Path, val_out: str | Path,
                val_fraction: float = 0.01, seed: int = 42) -> Dict:
    """Shuffle-split a JSONL into train/val files."""
    lines = [ln for ln in open(path) if ln.strip()]
    rng = random.Random(seed)
    rng.shuffle(lines)
    n_val = max(1, int(len(lines) * val_fraction))
    with open(val_out, "w") as f:
        f.writelines(lines[:n_val])
    with open(train_out, "w") as f:
        f.writelines(lines[n_val:])
    return {"total": len(lines), "train": len(lines) - n_val, "val": n_val}


def token_stats(path: str | Path, tokenizer_json: Optional[str] = None,
                sample_docs: Optional[int] = None) -> Dict:
    """Count tokens (exact with a tokenizer.json, else whitespace estimate)."""
    tok = None
    if tokenizer_json:
        from tokenizers import Tokenizer

        tok = Tokenizer.from_file(str(tokenizer_json))
    n_docs = n_tokens = n_chars = 0
    lengths = []
    with open(path) as f:
        for line in f:
            line = line.strip()
            if not line:
                continue
            try:
                text = json.loads(line).get("text", "")
            except json.JSONDecodeError:
                continue
            n = len(tok.encode(text).ids) if tok else len(text.split())
            n_docs += 1
            n_tokens += n
            n_chars += len(text)
            lengths.append(n)
            if sample_docs and n_docs >= sample_docs:
                break
    lengths.sort()
    mid = lengths[len(lengths) // 2] if lengths else 0
    return {"docs": n_docs, "tokens": n_tokens, "chars": n_chars,
            "mean_tokens": n_tokens / max(n_docs, 1), "median_tokens": mid,
            "max_tokens": lengths[-1] if lengths else 0,
            "exact": tok is not None}


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Prepare/inspect JSONL training data")
    sub = p.add_subparsers(dest="cmd", required=True)
    v = sub.add_parser("validate")
    v.add_argument("path")
    s = sub.add_parser("split")
    s.add_argument("path")
    s.add_argument("--train-out", required=True)
    s.add_argument("--val-out", required=True)
    s.add_argument("--val-fraction", type=float, default=0.01)
    s.add_argument("--seed", type=int, default=42)
    e = sub.add_parser("stats")
    e.add_argument("path")
    e.add_argument("--tokenizer", default=None)
    e.add_argument("--sample-docs", type=int, default=None)
    a = p.parse_args(argv)
    if a.cmd == "validate":
        print(json.dumps(validate_jsonl(a.path), indent=2))
    elif a.cmd == "split":
        print(json.dumps(split_jsonl(a.path, a.train_out, a.val_out,
                                     a.val_fraction, a.seed), indent=2))
    else:
        print(json.dumps(token_stats(a.path, a.tokenizer, a.sample_docs), indent=2))


if __name__ == "__main__":
    main()
