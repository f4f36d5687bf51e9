"""Opt-in fp8 (OCP e4m3) GEMM path for the big projections.

MI355X fp8 MFMA peak is 2x bf16 (~5 PF/s dense, guide §3), and gfx950 takes
the OCP ``e4m3fn`` format (NOT the MI300X ``fnuz`` variant) — which is
exactly ``torch.float8_e4m3fn``/``torch._scaled_mm`` (hipBLASLt underneath).

Recipe (v1, dynamic per-tensor scaling):
  forward : y  = (x/sx)_e4m3 @ (W/sw)_e4m3^T  * sx*sw      (bf16 out)
  dgrad   : dx = (dy/sd)_e4m3 @ (W/sw)_e4m3   * sd*sw      (bf16 out)
  wgrad   : bf16 (dW = dy^T @ x)  — 1/3 of the GEMM flops stays bf16
Scales are amax/448 per tensor, computed on the fly (one extra read of each
operand — cheap next to the GEMM). Master weights stay fp32 in the fused
optimizer; the model's bf16 weights are quantized per use.

This is an OPT-IN precision (``model.misc.fp8: true`` or the llama-1b-fp8
config); the BASELINE headline remains bf16.
"""
from __future__ import annotations


import torch

E4M3_MAX = 448.0


def _amax_scale(t: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    return (t.abs().amax().float() / E4M3_MAX).clamp_min(eps)


def quantize_e4m3(t: torch.Tensor):
    """-> (codes e4m3, dequant scale fp32 scalar tensor)."""
    s = _amax_scale(t)
    q = (t.float() / s).clamp(-E4M3_MAX, E4M3_MAX).to(torch.float8_e4m3fn)
    return q, s


def dequantize(q: torch.Tensor, s: torch.Tensor, dtype=torch.bfloat16) -> torch.Tensor:
    return (q.float() * s).to(dtype)


def fp8_available() -> bool:
    return hasattr(torch, "_scaled_mm") and torch.cuda.is_available()


def _quant(t: torch.Tensor, transpose: bool = False, amax: torch.Tensor = None):
    # fused HIP quantize (csrc/fp8_quant.hip) on GPU; torch fallback on CPU.
    # amax: producer-accumulated |t|max float bits (rmsnorm/swiglu emit them)
    # -> the quantizer's own amax read of t is skipped.
    from ._ext import get_ext

    ext = get_ext()
    if ext is not None and t.is_cuda:
        if amax is not None:
            return ext.fp8_quantize_pre(t, amax, transpose)
        return ext.fp8_quantize(t, transpose)
    return quantize_e4m3(t.t().contiguous() if transpose else t)


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, x_amax: torch.Tensor = None):
        # x: [..., K] bf16; weight: [N, K] bf16
        shape = x.shape
        x2 = x.reshape(-1, shape[-1])
        x8, sx = _quant(x2, amax=x_amax)
        w8, sw = _quant(weight)
        y = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=sw,
                             out_dtype=x.dtype)
        ctx.save_for_backward(x2, weight)
        return y.reshape(*shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x2, weight = ctx.saved_tensors
        N = weight.shape[0]
        dy2 = dy.reshape(-1, N).contiguous()
        dy8, sd = _quant(dy2)
        # dx = dy @ W : b must be [N, K] column-major; the transposed-quantize
        # kernel emits W^T [K, N] row-major, whose .t() is that col-major view
        w8t, sw = _quant(weight, transpose=True)
        dx = torch._scaled_mm(dy8, w8t.t(), scale_a=sd, scale_b=sw,
                              out_dtype=dy.dtype)
        # wgrad in bf16 (outlier-sensitive)
        dw = dy2.t() @ x2
        return dx.reshape(*dy.shape[:-1], x2.shape[-1]), dw, None


def fp8_linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    # views drop python attributes, so the producer's amax rides in from the
    # ORIGINAL activation tensor here
    return _Fp8LinearFn.apply(x, weight, getattr(x, "_mcdp_amax", None))


def fp8_linear_ref(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    """Quant-dequant simulation (CPU test oracle for the numerics of the
    quantization itself — the GEMM is exact given the quantized operands)."""
    x8, sx = quantize_e4m3(x)
    w8, sw = quantize_e4m3(weight)
    return dequantize(x8, sx, torch.float32) @ dequantize(w8, sw, torch.float32).t()
