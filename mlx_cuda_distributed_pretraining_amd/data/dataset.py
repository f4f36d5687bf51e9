"""Data pipeline.

Parity with the reference DataManager (/root/reference/core/training.py:442-543):
JSONL ``{"text": ...}`` docs tokenized with BOS/EOS, chunked to
``max_context_size`` with optional ``chunk_overlap``, batches built from
length-sorted-then-shuffled indices and padded to the longest row.

MI355X additions:
  - synthetic mode (random tokens of a fixed shape) for no-network benches,
  - rank sharding for data-parallel training (each rank sees a disjoint
    slice of batches via a rank-offset into the shuffled order).
"""
from __future__ import annotations

import json
from typing import Any, List, Optional

import numpy as np
import torch


class DataManager:
    def __init__(
        self,
        data_cfg: Any,
        tokenizer,
        batch_size: int,
        rank: int = 0,
        world_size: int = 1,
        seed: int = 42,
    ):
        self.cfg = data_cfg
        self.tokenizer = tokenizer
        self.batch_size = batch_size
        self.rank = rank
        self.world_size = world_size
        self.seed = seed

        prep = getattr(data_cfg, "preprocessing", None) or {}
        self.max_context_size = int(prep.get("max_context_size", 1024))
        self.chunk_overlap = int(prep.get("chunk_overlap", 0))

        self.synthetic = bool(getattr(data_cfg, "synthetic", False))
        self.docs: List[List[int]] = []
        self.val_docs: List[List[int]] = []
        self.batch_order: List[List[int]] = []
        self.val_ptr = 0

        if self.synthetic:
            self.synthetic_vocab = int(getattr(data_cfg, "synthetic_vocab_size", 32000))
            # synthetic ids feed the embedding directly: ids >= the model
            # vocab are an out-of-bounds gather = a GPU memory fault with no
            # python traceback (cost a day at 7B) — cap to the tokenizer
            # vocab and warn instead.
            tv = getattr(tokenizer, "vocab_size", None)
            if tv is not None and self.synthetic_vocab > tv:
                import warnings

                warnings.warn(
                    f"synthetic_vocab_size {self.synthetic_vocab} > tokenizer vocab {tv}; "
                    f"capping to {tv} (set data.tokenizer.normal_vocab_size to match)")
                self.synthetic_vocab = tv
            return

        self.stream = None
        stream_cfg = getattr(data_cfg, "streaming", None)
        if stream_cfg:
            from .streaming import DiskSpaceManager, StreamingTokenDataset

            mgr = None
            if stream_cfg.get("cache_dir"):
                mgr = DiskSpaceManager(
                    stream_cfg["cache_dir"],
                    max_bytes=int(float(stream_cfg.get("max_cache_gb", 10)) * 2**30),
                )
            ds = StreamingTokenDataset(
                stream_cfg["source"], tokenizer, seq_len=self.max_context_size,
                rank=rank, world_size=world_size,
                max_tokens=stream_cfg.get("max_tokens"), disk_manager=mgr,
            )
            self.stream = ds.iter_batches(batch_size)

        input_file = getattr(data_cfg, "input_file", None)
        if input_file:
            self.docs = self._load_jsonl(input_file)
        val_file = getattr(data_cfg, "validation_file", None)
        if val_file:
            self.val_docs = self._load_jsonl(val_file)
        self._build_batch_order()

    # -- loading ----------------------------------------------------------
    def _load_jsonl(self, path: str) -> List[List[int]]:
        chunks: List[List[int]] = []
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    text = json.loads(line).get("text", "")
                except json.JSONDecodeError:
                    text = line
                toks = self.tokenizer.tokenize_doc(text)
                chunks.extend(self._chunk(toks))
        return chunks

    def _chunk(self, toks: List[int]) -> List[List[int]]:
        """Split a token list into max_context_size chunks with overlap
        (reference :479-492)."""
        L = self.max_context_size
        if len(toks) <= L:
            return [toks]
        step = max(L - self.chunk_overlap, 1)
        out = []
        for start in range(0, len(toks), step):
            chunk = toks[start : start + L]
            if len(chunk) < 2:
                break
            out.append(chunk)
            if start + L >= len(toks):
                break
        return out

    def _build_batch_order(self) -> None:
        """Length-sorted then batch-shuffled indices (reference :458-464)."""
        if not self.docs:
            return
        order = sorted(range(len(self.docs)), key=lambda i: len(self.docs[i]))
        batches = [
            order[i : i + self.batch_size] for i in range(0, len(order), self.batch_size)
        ]
        rng = np.random.default_rng(self.seed)
        rng.shuffle(batches)
        self.batch_order = batches

    @property
    def num_batches(self) -> int:
        return len(self.batch_order)

    # -- batch generation -------------------------------------------------
    def _create_batch(self, indices: List[int], docs: List[List[int]]) -> torch.Tensor:
        rows = [docs[i] for i in indices]
        max_len = max(len(r) for r in rows)
        # Round up to a multiple of 8 to keep GEMMs MFMA-tile aligned.
        max_len = (max_len + 7) // 8 * 8
        pad = self.tokenizer.PAD_TOKEN
        batch = np.full((len(rows), max_len), pad, dtype=np.int64)
        for j, r in enumerate(rows):
            batch[j, : len(r)] = r
        return torch.from_numpy(batch)

    def fast_forward(self, num_batches: int) -> int:
        """Advance a streaming source past ``num_batches`` already-consumed
        batches (auto-resume: the stream otherwise replays from shard 0 and
        re-trains on seen data; tokens_emitted is re-advanced as a side
        effect, so the max_tokens budget resumes correctly too). No-op for
        the in-memory path, whose generate_batch is indexed by step.
        Returns the number of batches actually skipped."""
        if getattr(self, "stream", None) is None or num_batches <= 0:
            return 0
        skipped = 0
        for _ in range(num_batches):
            try:
                next(self.stream)
                skipped += 1
            except StopIteration:
                break
        return skipped

    def generate_batch(self, step: int) -> torch.Tensor:
        if self.synthetic:
            g = torch.Generator().manual_seed(self.seed * 1_000_003 + step * self.world_size + self.rank)
            return torch.randint(
                0,
                self.synthetic_vocab,
                (self.batch_size, self.max_context_size),
                generator=g,
                dtype=torch.long,
            )
        if getattr(self, "stream", None) is not None:
            try:
                return next(self.stream)
            except StopIteration:
                raise RuntimeError("streaming source exhausted (max_tokens reached)")
        if not self.batch_order:
            raise RuntimeError("No training data loaded")
        idx = (step * self.world_size + self.rank) % len(self.batch_order)
        return self._create_batch(self.batch_order[idx], self.docs)

    def generate_validation_batch(self, batch_idx: int) -> Optional[torch.Tensor]:
        if self.synthetic:
            g = torch.Generator().manual_seed(self.seed * 7_000_003 + batch_idx)
            return torch.randint(
                0,
                self.synthetic_vocab,
                (self.batch_size, self.max_context_size),
                generator=g,
                dtype=torch.long,
            )
        if not self.val_docs:
            return None
        start = batch_idx * self.batch_size
        if start >= len(self.val_docs):
            return None
        indices = list(range(start, min(start + self.batch_size, len(self.val_docs))))
        return self._create_batch(indices, self.val_docs)

    @property
    def num_validation_batches(self) -> int:
        if self.synthetic:
            return 4
        return (len(self.val_docs) + self.batch_size - 1) // self.batch_size
