"""Tokenizer management.

Parity with the reference TokenizerManager
(/root/reference/core/training.py:324-440): either an external HF
``tokenizers`` JSON (directory containing tokenizer.json) or a byte-level
fallback vocabulary (256 bytes + special tokens), with BOS/EOS framing and
optional truncation in ``tokenize_doc``.
"""
from __future__ import annotations

from pathlib import Path
from typing import Any, Dict, List, Optional


class TokenizerManager:
    def __init__(self, data_cfg: Any):
        tok_cfg: Dict[str, Any] = getattr(data_cfg, "tokenizer", None) or {}
        self.special_tokens: Dict[str, str] = dict(
            tok_cfg.get("special_tokens")
            or {"pad": "<pad>", "bos": "<bos>", "eos": "<eos>"}
        )
        self.normal_vocab_size = int(tok_cfg.get("normal_vocab_size", 256))
        self.external = None
        self.use_external_tokenizer = False

        tokenizer_path = getattr(data_cfg, "tokenizer_path", None)
        if tokenizer_path:
            tok_json = Path(tokenizer_path)
            if tok_json.is_dir():
                tok_json = tok_json / "tokenizer.json"
            if tok_json.exists():
                from tokenizers import Tokenizer

                self.external = Tokenizer.from_file(str(tok_json))
                self.use_external_tokenizer = True

        if self.use_external_tokenizer:
            self.vocab_size = self.external.get_vocab_size()
            vocab = self.external.get_vocab()
            self.PAD_TOKEN = vocab.get(self.special_tokens.get("pad", "<pad>"), 0)
            self.BOS_TOKEN = vocab.get(self.special_tokens.get("bos", "<bos>"), 1)
            self.EOS_TOKEN = vocab.get(self.special_tokens.get("eos", "<eos>"), 2)
        else:
            # Byte-level fallback: ids [0, normal_vocab_size) are raw bytes,
            # specials appended after.
            names = list(self.special_tokens.keys())
            self.special_ids = {
                name: self.normal_vocab_size + i for i, name in enumerate(names)
            }
            self.vocab_size = self.normal_vocab_size + len(names)
            self.PAD_TOKEN = self.special_ids.get("pad", self.normal_vocab_size)
            self.BOS_TOKEN = self.special_ids.get("bos", self.normal_vocab_size + 1)
            self.EOS_TOKEN = self.special_ids.get("eos", self.normal_vocab_size + 2)

    # -- encoding ---------------------------------------------------------
    def tokenize(self, text: str) -> List[int]:
        if self.use_external_tokenizer:
            return self.external.encode(text).ids
        return [b % self.normal_vocab_size for b in text.encode("utf-8")]

    def detokenize(self, tokens: List[int]) -> str:
        if self.use_external_tokenizer:
            return self.external.decode(list(tokens))
        data = bytes(t for t in tokens if 0 <= t < self.normal_vocab_size)
        return data.decode("utf-8", errors="replace")

    def tokenize_doc(self, doc: str, max_length: Optional[int] = None) -> List[int]:
        """BOS + tokens + EOS, truncated to max_length (BOS/EOS preserved)."""
        toks = self.tokenize(doc)
        if max_length is not None and len(toks) > max_length - 2:
            toks = toks[: max_length - 2]
        return [self.BOS_TOKEN] + toks + [self.EOS_TOKEN]

    def save(self, out_dir: Path) -> None:
        out_dir = Path(out_dir)
        out_dir.mkdir(parents=True, exist_ok=True)
        if self.use_external_tokenizer:
            self.external.save(str(out_dir / "tokenizer.json"))


def train_bpe_tokenizer(
    input_files: List[str],
    vocab_size: int,
    out_dir: str,
    special_tokens: Optional[List[str]] = None,
    min_frequency: int = 2,
) -> str:
    """Byte-level BPE training (parity: /root/reference/tools/train-tokenizer.py:39-101).

    Returns the path of the written tokenizer.json.
    """
    import json

    from tokenizers import Tokenizer, decoders, models, normalizers, pre_tokenizers, trainers

    tokenizer = Tokenizer(models.BPE())
    tokenizer.normalizer = normalizers.NFKC()
    tokenizer.pre_tokenizer = pre_tokenizers.ByteLevel(add_prefix_space=False)
    tokenizer.decoder = decoders.ByteLevel()
    specials = special_tokens or ["<pad>", "<bos>", "<eos>"]
    trainer = trainers.BpeTrainer(
        vocab_size=vocab_size,
        min_frequency=min_frequency,
        special_tokens=specials,
        show_progress=False,
    )

    def text_iterator():
        for path in input_files:
            with open(path) as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        doc = json.loads(line)
                        yield doc.get("text", "")
                    except json.JSONDecodeError:
                        yield line

    tokenizer.train_from_iterator(text_iterator(), trainer)
    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    out_path = out / "tokenizer.json"
    tokenizer.save(str(out_path))
    return str(out_path)
