"""Fused SwiGLU activation.

Standard SwiGLU ``down(silu(gate(x)) * up(x))`` — a deliberate fix of the
reference's non-standard variant ``down(gate(x) * sigmoid(up(x)) * 2)``
(/root/reference/models/llama.py:149-151; SURVEY.md §2.2 documents the bug).

The gate and up projections are computed as ONE GEMM into [..., 2I]
(hipBLASLt via torch.matmul); this op fuses the elementwise
``silu(gate) * up`` (csrc/swiglu.hip) so the intermediate activations are
read/written exactly once — the op is HBM-bound, so fusion is the win.
"""
from __future__ import annotations

import torch

from ._ext import get_ext, use_hip


def swiglu_ref(gate_up: torch.Tensor) -> torch.Tensor:
    i = gate_up.shape[-1] // 2
    gate = gate_up[..., :i].float()
    up = gate_up[..., i:].float()
    return (torch.nn.functional.silu(gate) * up).to(gate_up.dtype)


class _SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate_up: torch.Tensor, emit_amax: bool = False):
        ctx.save_for_backward(gate_up)
        if use_hip(gate_up):
            if emit_amax:
                y, amax = get_ext().swiglu_fwd_amax(gate_up.contiguous(), True)
                y._mcdp_amax = amax  # fp8 quantizer skips its amax pass
                return y
            return get_ext().swiglu_fwd(gate_up.contiguous())
        return swiglu_ref(gate_up)

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (gate_up,) = ctx.saved_tensors
        if use_hip(gate_up):
            return get_ext().swiglu_bwd(gate_up.contiguous(), dy.contiguous()), None
        i = gate_up.shape[-1] // 2
        g = gate_up[..., :i].float()
        u = gate_up[..., i:].float()
        dyf = dy.float()
        sg = torch.sigmoid(g)
        silu_g = g * sg
        dgate = dyf * u * (sg * (1 + g * (1 - sg)))
        dup = dyf * silu_g
        return torch.cat([dgate, dup], dim=-1).to(gate_up.dtype), None


def swiglu(gate_up: torch.Tensor, emit_amax: bool = False) -> torch.Tensor:
    """gate_up: [..., 2I] -> [..., I] = silu(gate_up[...,:I]) * gate_up[...,I:].
    emit_amax: attach ``_mcdp_amax`` (max|out| float bits) for the fp8 path."""
    return _SwiGLUFn.apply(gate_up, emit_amax)
