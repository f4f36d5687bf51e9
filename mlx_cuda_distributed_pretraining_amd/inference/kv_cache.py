"""KV-cache variants for generation (K13).

Parity surface: /root/reference/core/generation_lite.py:75-94
(``maybe_quantize_kv_cache``, group size 64, ``kv_bits``) — the reference
quantizes MLX caches after a threshold offset; here the same policy applies
to our BSHD caches.

MI355X design notes: the growing torch.cat cache reallocates every decode
step; ChunkedKVCache preallocates in 256-row slabs so decode appends are a
slice write (no realloc, no copy), sized freely against 288 GB HBM3E.
QuantizedKVCache stores int8 with per-group scale/zero (group size along D)
— 2x memory saving vs bf16 with ~1e-2 dequant error, dequantized on read
(the flash-attention kernel consumes bf16; fused int8 dequant inside K1 is
a further optimization, not required for parity).
"""
from __future__ import annotations

from typing import List, Optional

import torch

from ..models.llama import KVCache, Model


class ChunkedKVCache(KVCache):
    """Preallocating cache: grows in ``chunk`` rows at a time."""

    def __init__(self, chunk: int = 256):
        super().__init__()
        self.chunk = chunk
        self._len = 0

    @property
    def offset(self) -> int:
        return self._len

    def trim(self, n: int) -> None:
        # Base-class trim slices capacity; here only the logical length moves
        # (later updates overwrite the trimmed rows in the preallocated slab).
        if n > 0:
            self._len = max(0, self._len - n)

    def update(self, k: torch.Tensor, v: torch.Tensor):
        B, S, H, D = k.shape
        need = self._len + S
        if self.k is None:
            cap = ((need + self.chunk - 1) // self.chunk) * self.chunk
            self.k = torch.empty(B, cap, H, D, dtype=k.dtype, device=k.device)
            self.v = torch.empty_like(self.k)
        elif need > self.k.shape[1]:
            cap = ((need + self.chunk - 1) // self.chunk) * self.chunk
            nk = torch.empty(B, cap, H, D, dtype=k.dtype, device=k.device)
            nv = torch.empty_like(nk)
            nk[:, : self._len] = self.k[:, : self._len]
            nv[:, : self._len] = self.v[:, : self._len]
            self.k, self.v = nk, nv
        self.k[:, self._len : need] = k
        self.v[:, self._len : need] = v
        self._len = need
        return self.k[:, :need], self.v[:, :need]


def _quantize_groups(t: torch.Tensor, bits: int, group: int):
    """[..., D] -> int8 codes [..., D], scales/zeros [..., D/group]."""
    *lead, D = t.shape
    assert D % group == 0, f"D={D} not divisible by group={group}"
    g = t.float().reshape(*lead, D // group, group)
    lo = g.min(dim=-1, keepdim=True).values
    hi = g.max(dim=-1, keepdim=True).values
    qmax = (1 << bits) - 1
    scale = (hi - lo).clamp_min(1e-8) / qmax
    codes = ((g - lo) / scale).round().clamp(0, qmax).to(torch.uint8)
    return codes.reshape(*lead, D), scale.squeeze(-1), lo.squeeze(-1)


def _dequantize_groups(codes: torch.Tensor, scale: torch.Tensor, zero: torch.Tensor,
                       group: int, dtype: torch.dtype) -> torch.Tensor:
    *lead, D = codes.shape
    g = codes.reshape(*lead, D // group, group).float()
    out = g * scale.unsqueeze(-1) + zero.unsqueeze(-1)
    return out.reshape(*lead, D).to(dtype)


class QuantizedKVCache(KVCache):
    """Group-quantized cache (default int8, group 64 along D)."""

    def __init__(self, bits: int = 8, group: int = 64):
        super().__init__()
        assert bits in (4, 8), "kv_bits must be 4 or 8"
        self.bits = bits
        self.group = group
        self._len = 0
        self._ck = self._cv = None  # codes
        self._sk = self._sv = None  # scales
        self._zk = self._zv = None  # zeros
        self.dtype = None

    @property
    def offset(self) -> int:
        return self._len

    def trim(self, n: int) -> None:
        # Base-class trim is a no-op here (self.k is None); slice the
        # quantized storage instead so speculative-decoding rewind works.
        if n <= 0 or self._len == 0:
            return
        keep = max(0, self._len - n)
        for name in ("_ck", "_cv", "_sk", "_sv", "_zk", "_zv"):
            t = getattr(self, name)
            if t is not None:
                setattr(self, name, t[:, :keep])
        self._len = keep

    def _append(self, codes, scale, zero, which: str):
        cn, sn, zn = f"_c{which}", f"_s{which}", f"_z{which}"
        if getattr(self, cn) is None:
            setattr(self, cn, codes)
            setattr(self, sn, scale)
            setattr(self, zn, zero)
        else:
            setattr(self, cn, torch.cat([getattr(self, cn), codes], dim=1))
            setattr(self, sn, torch.cat([getattr(self, sn), scale], dim=1))
            setattr(self, zn, torch.cat([getattr(self, zn), zero], dim=1))

    def update(self, k: torch.Tensor, v: torch.Tensor):
        self.dtype = k.dtype
        self.group = min(self.group, k.shape[-1])  # small head_dim models
        ck, sk, zk = _quantize_groups(k, self.bits, self.group)
        cv, sv, zv = _quantize_groups(v, self.bits, self.group)
        self._append(ck, sk, zk, "k")
        self._append(cv, sv, zv, "v")
        self._len += k.shape[1]
        kd = _dequantize_groups(self._ck, self._sk, self._zk, self.group, self.dtype)
        vd = _dequantize_groups(self._cv, self._sv, self._zv, self.group, self.dtype)
        return kd, vd

    @classmethod
    def from_cache(cls, cache: KVCache, bits: int = 8, group: int = 64) -> "QuantizedKVCache":
        q = cls(bits=bits, group=group)
        if cache.k is not None:
            q.group = min(group, cache.k.shape[-1])
            group = q.group
            k = cache.k[:, : cache.offset] if hasattr(cache, "_len") else cache.k
            v = cache.v[:, : cache.offset] if hasattr(cache, "_len") else cache.v
            q.dtype = k.dtype
            ck, sk, zk = _quantize_groups(k, bits, group)
            cv, sv, zv = _quantize_groups(v, bits, group)
            q._append(ck, sk, zk, "k")
            q._append(cv, sv, zv, "v")
            q._len = k.shape[1]
        return q


def make_cache(model: Model, kind: str = "chunked", **kw) -> List[KVCache]:
    ctor = {"simple": KVCache, "chunked": ChunkedKVCache,
            "quantized": QuantizedKVCache}[kind]
    return [ctor(**kw) for _ in range(len(model.layers))]


def maybe_quantize_kv_cache(
    cache: List[KVCache],
    quantized_kv_start: int = 0,
    kv_bits: Optional[int] = None,
    kv_group_size: int = 64,
) -> List[KVCache]:
    """Swap plain caches for quantized ones once the sequence passes
    ``quantized_kv_start`` (reference core/generation_lite.py:75-94)."""
    if kv_bits is None or not cache:
        return cache
    if isinstance(cache[0], QuantizedKVCache):
        return cache
    if cache[0].offset >= quantized_kv_start:
        return [QuantizedKVCache.from_cache(c, bits=kv_bits, group=kv_group_size)
                for c in cache]
    return cache
