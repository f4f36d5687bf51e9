"""Rotary position embedding (RoPE).

The reference builds a RoPE module but never applies it on its main path
(documented bug, /root/reference/models/llama.py:221-252 — SURVEY.md §2.2);
its standard-attention path applies it in Python
(/root/reference/models/llama_standard.py:51-133). Here RoPE is always
applied, via a CDNA4 HIP kernel (csrc/rope.hip) with a host-precomputed
cos/sin table (on-device trig turns the op VALU-bound — guide Appendix B).

Conventions (both supported, ``traditional`` flag as in the reference):
  traditional=True : interleaved pairs (x[2i], x[2i+1])
  traditional=False: split halves  (x[i], x[i+D/2])  — neox style
Backward = rotation by -theta (same kernel, conjugate flag).
"""
from __future__ import annotations

from typing import Optional

import torch

from ._ext import get_ext, use_hip


class RopeTable:
    """Precomputed cos/sin [S, D/2] fp32 table, grown on demand."""

    def __init__(self, head_dim: int, theta: float = 10000.0, scaling: Optional[float] = None):
        self.head_dim = head_dim
        self.theta = float(theta)
        self.scale = 1.0 / float(scaling) if scaling else 1.0
        self._cos: Optional[torch.Tensor] = None
        self._sin: Optional[torch.Tensor] = None

    def get(self, seq_len: int, device, offset: int = 0):
        need = seq_len + offset
        if self._cos is None or self._cos.shape[0] < need or self._cos.device != device:
            n = max(need, 2048 if self._cos is None else 2 * self._cos.shape[0])
            half = self.head_dim // 2
            inv_freq = self.theta ** (
                -torch.arange(0, half, dtype=torch.float32, device=device) / half
            )
            pos = torch.arange(n, dtype=torch.float32, device=device) * self.scale
            freqs = torch.outer(pos, inv_freq)
            self._cos = freqs.cos().contiguous()
            self._sin = freqs.sin().contiguous()
        return self._cos, self._sin


def rope_ref(
    x: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    traditional: bool = False,
    offset: int = 0,
    conj: bool = False,
) -> torch.Tensor:
    """Pure-torch reference. x: [B, S, H, D] (BSHD)."""
    S, D = x.shape[1], x.shape[-1]
    half = D // 2
    c = cos[offset : offset + S].to(torch.float32).view(S, 1, half)  # broadcast over H
    s = sin[offset : offset + S].to(torch.float32).view(S, 1, half)
    if conj:
        s = -s
    xf = x.float()
    if traditional:
        x0 = xf[..., 0::2]
        x1 = xf[..., 1::2]
        out = torch.empty_like(xf)
        out[..., 0::2] = x0 * c - x1 * s
        out[..., 1::2] = x0 * s + x1 * c
    else:
        x0 = xf[..., :half]
        x1 = xf[..., half:]
        out = torch.cat([x0 * c - x1 * s, x0 * s + x1 * c], dim=-1)
    return out.to(x.dtype)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, traditional: bool, offset: int):
        ctx.traditional = traditional
        ctx.offset = offset
        ctx.save_for_backward(cos, sin)
        if use_hip(x):
            ext = get_ext()
            return ext.rope_fwd(x, cos, sin, traditional, offset, False)
        return rope_ref(x, cos, sin, traditional, offset, conj=False)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        if use_hip(dy):
            ext = get_ext()
            dx = ext.rope_fwd(dy.contiguous(), cos, sin, ctx.traditional, ctx.offset, True)
        else:
            dx = rope_ref(dy, cos, sin, ctx.traditional, ctx.offset, conj=True)
        return dx, None, None, None, None


def apply_rope(
    x: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    traditional: bool = False,
    offset: int = 0,
) -> torch.Tensor:
    """x: [B, S, H, D] -> same shape with rotary embedding applied."""
    return _RopeFn.apply(x, cos, sin, traditional, offset)
