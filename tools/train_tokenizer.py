#!/usr/bin/env python3
"""Train a byte-level BPE tokenizer from a YAML config.

Parity surface: /root/reference/tools/train-tokenizer.py:39-101 (HF
``tokenizers`` byte-level BPE, NFKC normalizer, special tokens, saved to
``tokenizer/tokenizer.json``). The heavy lifting lives in
mlx_cuda_distributed_pretraining_amd/data/tokenizer.py:train_bpe_tokenizer.
"""
from __future__ import annotations

import argparse
import sys
from pathlib import Path

import yaml

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

from mlx_cuda_distributed_pretraining_amd.data.tokenizer import (  # noqa: E402
    train_bpe_tokenizer,
)


def main(argv=None) -> None:
    p = argparse.ArgumentParser(description="Train a BPE tokenizer")
    p.add_argument("--config", required=True, help="training YAML (uses data.*)")
    p.add_argument("--out-dir", default=None,
                   help="output dir (default: tokenizer/ next to the data file)")
    a = p.parse_args(argv)

    cfg = yaml.safe_load(Path(a.config).read_text())
    data = cfg.get("data", {})
    input_files = data.get("input_file")
    if isinstance(input_files, str):
        input_files = [input_files]
    tok_cfg = data.get("tokenizer", {})
    vocab_size = int(tok_cfg.get("normal_vocab_size", 32000))
    special = tok_cfg.get("special_tokens", {"pad": "<pad>", "bos": "<bos>", "eos": "<eos>"})

    out_dir = Path(a.out_dir) if a.out_dir else Path(input_files[0]).parent / "tokenizer"
    path = train_bpe_tokenizer(
        input_files, vocab_size=vocab_size, out_dir=str(out_dir),
        special_tokens=list(special.values()),
    )
    print(f"tokenizer saved to {path}")


if __name__ == "__main__":
    main()
