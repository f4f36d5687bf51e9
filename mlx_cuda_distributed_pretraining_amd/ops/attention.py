"""Attention ops: flash (tiled, online-softmax) + flex (block-sparse/score-mod).

Replaces the reference's three Python attention modules
(/root/reference/models/attention/{flash,flex,simple}_attention.py — all of
which materialize S×S scores; the "flash" one says so at flash_attention.py:100)
with a real tiled FlashAttention-2 fwd/bwd HIP kernel (csrc/attn_fwd.hip,
csrc/attn_bwd.hip): MFMA bf16 tiles, LDS-staged K/V with XOR swizzle, online
softmax, causal/sliding-window block skip, GQA without materializing repeated
K/V. FlexAttention's score/mask mods become kernel template variants
(MOD_* codes below) instead of per-element Python callbacks.

Tensor layout: BSHD — q [B, S, Hq, D], k/v [B, S, Hkv, D]; Hq % Hkv == 0.
(BSHD is what the QKV projection GEMM produces; keeping it end-to-end means
zero transpose copies on the hot path.)
"""
from __future__ import annotations

import math
from typing import Callable, Optional

import torch

from ._ext import get_ext, use_hip

# Mask/score-mod variant codes understood by the kernels.
MOD_NONE = 0  # full (bidirectional) attention
MOD_CAUSAL = 1
MOD_SLIDING_WINDOW = 2  # causal AND (q - kv) < window
MOD_PREFIX_LM = 3  # bidirectional over [0, prefix) then causal
MOD_ALIBI = 4  # causal + alibi slope * (kv - q)


def _mods_to_code(causal: bool, window: Optional[int], prefix_len: Optional[int], alibi: bool):
    if alibi:
        return MOD_ALIBI
    if window is not None:
        return MOD_SLIDING_WINDOW
    if prefix_len is not None:
        return MOD_PREFIX_LM
    return MOD_CAUSAL if causal else MOD_NONE


def attention_ref(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
    window: Optional[int] = None,
    prefix_len: Optional[int] = None,
    alibi_slopes: Optional[torch.Tensor] = None,
    score_mod: Optional[Callable] = None,
    mask_mod: Optional[Callable] = None,
    return_lse: bool = False,
):
    """fp32 reference composition (used on CPU and as the test oracle).

    q: [B, S, Hq, D]; k, v: [B, Skv, Hkv, D] (BSHD).
    score_mod(scores[B,H,S,S], b_idx, h_idx, q_idx, kv_idx) -> scores, applied
    to the scaled scores (FlexAttention semantics, vectorized over the full
    score tensor). mask_mod(b, h, q_idx, kv_idx) -> bool keep-mask. Index
    tensors arrive BROADCASTABLE (q [S,1] x kv [1,Skv]); write mods as plain
    expressions over them.
    """
    B, S, Hq, D = q.shape
    Hkv = k.shape[2]
    rep = Hq // Hkv
    scale = scale if scale is not None else 1.0 / math.sqrt(D)
    qf = q.float().transpose(1, 2)  # BHSD internally
    kf = k.float().transpose(1, 2)
    vf = v.float().transpose(1, 2)
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.einsum("bhsd,bhtd->bhst", qf, kf) * scale

    Skv = k.shape[1]
    # Absolute query positions: with a KV cache, the S query rows sit at the
    # END of the Skv keys (decode: S=1 at position Skv-1).
    q_idx = (torch.arange(S, device=q.device) + (Skv - S)).view(S, 1)
    kv_idx = torch.arange(Skv, device=q.device).view(1, Skv)
    keep = torch.ones(S, Skv, dtype=torch.bool, device=q.device)
    if causal or window is not None or prefix_len is not None:
        causal_keep = kv_idx <= q_idx
        if prefix_len is not None:
            keep = causal_keep | (kv_idx < prefix_len)
        elif window is not None:
            keep = causal_keep & (q_idx - kv_idx < window)
        elif causal:
            keep = causal_keep
    if alibi_slopes is not None:
        bias = alibi_slopes.float().view(1, Hq, 1, 1) * (kv_idx - q_idx).float().view(1, 1, S, Skv)
        scores = scores + bias
        keep = keep & (kv_idx <= q_idx)
    # mods receive broadcastable index tensors (q [S,1], kv [1,Skv]) so a
    # plain expression like (ki <= qi) evaluates as the outer grid — the same
    # convention CompiledBlockMask compiles with (vectorized FlexAttention
    # semantics).
    if score_mod is not None:
        b_idx = torch.arange(B, device=q.device).view(B, 1, 1, 1)
        h_idx = torch.arange(Hq, device=q.device).view(1, Hq, 1, 1)
        scores = score_mod(scores, b_idx, h_idx, q_idx.view(1, 1, S, 1),
                           kv_idx.view(1, 1, 1, Skv))
    if mask_mod is not None:
        b_idx = torch.arange(B, device=q.device)
        h_idx = torch.arange(Hq, device=q.device)
        keep = keep & mask_mod(b_idx, h_idx, q_idx, kv_idx)
    scores = scores.masked_fill(~keep.view(1, 1, S, Skv), float("-inf"))
    lse = torch.logsumexp(scores, dim=-1)
    p = torch.softmax(scores, dim=-1)
    # rows with no valid keys produce NaN from softmax over all -inf; zero them
    p = torch.nan_to_num(p, nan=0.0)
    o = torch.einsum("bhst,bhtd->bhsd", p, vf).transpose(1, 2)  # back to BSHD
    if return_lse:
        return o.to(q.dtype).contiguous(), lse
    return o.to(q.dtype).contiguous()


class _FlashAttnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal, scale, window, prefix_len, alibi_slopes):
        scale = scale if scale is not None else 1.0 / math.sqrt(q.shape[-1])
        if use_hip(q, k, v, dtypes=(torch.bfloat16,)):
            ext = get_ext()
            mod = _mods_to_code(causal, window, prefix_len, alibi_slopes is not None)
            modarg = int(
                window
                if window is not None
                else (prefix_len if prefix_len is not None else 0)
            )
            slopes = (
                alibi_slopes.float().contiguous()
                if alibi_slopes is not None
                else torch.empty(0, dtype=torch.float32, device=q.device)
            )
            o, lse = ext.attn_fwd(q, k, v, scale, mod, modarg, slopes)
            ctx.save_for_backward(q, k, v, o, lse, slopes)
            ctx.meta = (causal, scale, window, prefix_len, True)
            return o
        o, lse = attention_ref(
            q, k, v, causal=causal, scale=scale, window=window,
            prefix_len=prefix_len, alibi_slopes=alibi_slopes, return_lse=True,
        )
        slopes = (
            alibi_slopes.float().contiguous()
            if alibi_slopes is not None
            else torch.empty(0, dtype=torch.float32)
        )
        ctx.save_for_backward(q, k, v, o, lse, slopes)
        ctx.meta = (causal, scale, window, prefix_len, False)
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse, slopes = ctx.saved_tensors
        causal, scale, window, prefix_len, hip = ctx.meta
        if hip:
            ext = get_ext()
            mod = _mods_to_code(causal, window, prefix_len, slopes.numel() > 0)
            modarg = int(
                window if window is not None else (prefix_len if prefix_len is not None else 0)
            )
            dq, dk, dv = ext.attn_bwd(q, k, v, o, do, lse, scale, mod, modarg, slopes)
            return dq, dk, dv, None, None, None, None, None
        # CPU reference backward: recompute with autograd in fp32
        with torch.enable_grad():
            qf = q.detach().float().requires_grad_(True)
            kf = k.detach().float().requires_grad_(True)
            vf = v.detach().float().requires_grad_(True)
            alibi = slopes if slopes.numel() > 0 else None
            of = attention_ref(
                qf, kf, vf, causal=causal, scale=scale, window=window,
                prefix_len=prefix_len, alibi_slopes=alibi,
            )
            grads = torch.autograd.grad(of, [qf, kf, vf], do.float())
        return (
            grads[0].to(q.dtype), grads[1].to(k.dtype), grads[2].to(v.dtype),
            None, None, None, None, None,
        )



def _norm_alibi(alibi_slopes, num_heads: int, device) -> "Optional[torch.Tensor]":
    """Accept either a per-head slope tensor or an int head count (builds the
    standard 2^(-8(i+1)/H) slopes) — matches the reference's loose API."""
    if alibi_slopes is None or isinstance(alibi_slopes, torch.Tensor):
        return alibi_slopes
    h = int(num_heads)
    return torch.tensor([2 ** (-8.0 * (i + 1) / h) for i in range(h)],
                        dtype=torch.float32, device=device)


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    scale: Optional[float] = None,
    window: Optional[int] = None,
    prefix_len: Optional[int] = None,
    alibi_slopes: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Tiled attention, BSHD layout: [B,S,Hq,D] x [B,S,Hkv,D] -> [B,S,Hq,D]
    (q.shape[2] is the HEAD count — _norm_alibi relies on that)."""
    alibi_slopes = _norm_alibi(alibi_slopes, q.shape[2], q.device)
    return _FlashAttnFn.apply(q, k, v, causal, scale, window, prefix_len, alibi_slopes)


class _RopeAttnQKVFn(torch.autograd.Function):
    """Fused qkv-slice -> RoPE(q,k) -> flash attention, HIP path only.

    Takes the FUSED [B, S, (Hq+2*Hkv)*D] projection output directly so the
    backward can write dQ/dK/dV straight into ONE dqkv buffer (attn_bwd_out /
    rope_fwd_out strided-out variants) — eliminating autograd's
    split-backward torch.cat over the three grads (a full extra read+write
    of the qkv-grad per layer, the CatArrayBatchedCopy kernel in profiles)."""

    @staticmethod
    def forward(ctx, qkv, cos, sin, Hq, Hkv, D, traditional, causal, scale,
                window, prefix_len, alibi_slopes):
        B, S, _ = qkv.shape
        scale = scale if scale is not None else 1.0 / math.sqrt(D)
        q = qkv[..., : Hq * D].view(B, S, Hq, D)
        k = qkv[..., Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
        v = qkv[..., (Hq + Hkv) * D :].view(B, S, Hkv, D)
        ext = get_ext()
        qr = ext.rope_fwd(q, cos, sin, traditional, 0, False)
        kr = ext.rope_fwd(k, cos, sin, traditional, 0, False)
        mod = _mods_to_code(causal, window, prefix_len, alibi_slopes is not None)
        modarg = int(window if window is not None
                     else (prefix_len if prefix_len is not None else 0))
        slopes = (alibi_slopes.float().contiguous() if alibi_slopes is not None
                  else torch.empty(0, dtype=torch.float32, device=qkv.device))
        o, lse = ext.attn_fwd(qr, kr, v, scale, mod, modarg, slopes)
        ctx.save_for_backward(qkv, qr, kr, o, lse, cos, sin, slopes)
        ctx.meta = (Hq, Hkv, D, traditional, scale, mod, modarg)
        return o

    @staticmethod
    def backward(ctx, do):
        qkv, qr, kr, o, lse, cos, sin, slopes = ctx.saved_tensors
        Hq, Hkv, D, traditional, scale, mod, modarg = ctx.meta
        B, S, _ = qkv.shape
        ext = get_ext()
        v = qkv[..., (Hq + Hkv) * D :].view(B, S, Hkv, D)
        dqkv = torch.empty_like(qkv)
        dq_t = dqkv[..., : Hq * D].view(B, S, Hq, D)
        dk_t = dqkv[..., Hq * D : (Hq + Hkv) * D].view(B, S, Hkv, D)
        dv_t = dqkv[..., (Hq + Hkv) * D :].view(B, S, Hkv, D)
        none = torch.empty(0, dtype=qkv.dtype, device=qkv.device)
        dq_r, dk_r, _ = ext.attn_bwd_out(qr, kr, v, o, do, lse, scale, mod,
                                         modarg, slopes, none, none.clone(), dv_t)
        ext.rope_fwd_out(dq_r, cos, sin, traditional, 0, True, dq_t)
        ext.rope_fwd_out(dk_r, cos, sin, traditional, 0, True, dk_t)
        return (dqkv,) + (None,) * 11


def rope_flash_attention_qkv(
    qkv: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    n_heads: int,
    n_kv_heads: int,
    head_dim: int,
    traditional: bool = False,
    causal: bool = True,
    scale: Optional[float] = None,
    window: Optional[int] = None,
    prefix_len: Optional[int] = None,
    alibi_slopes: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """qkv: [B, S, (Hq+2*Hkv)*D] fused projection output -> o [B, S, Hq, D].

    HIP fast path (one fused autograd node, no grad cat); CPU falls back to
    the differentiable composition of the same ops."""
    if use_hip(qkv, dtypes=(torch.bfloat16,)):
        return _RopeAttnQKVFn.apply(qkv, cos, sin, n_heads, n_kv_heads, head_dim,
                                    traditional, causal, scale, window,
                                    prefix_len, alibi_slopes)
    from .rope import apply_rope

    B, S, _ = qkv.shape
    Hq, Hkv, D = n_heads, n_kv_heads, head_dim
    q, k, v = qkv.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    q = apply_rope(q.view(B, S, Hq, D), cos, sin, traditional, 0)
    k = apply_rope(k.view(B, S, Hkv, D), cos, sin, traditional, 0)
    return flash_attention(q, k, v.view(B, S, Hkv, D), causal=causal, scale=scale,
                           window=window, prefix_len=prefix_len,
                           alibi_slopes=alibi_slopes)


def flex_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    score_mod: Optional[Callable] = None,
    mask_mod: Optional[Callable] = None,
    block_mask=None,
    scale: Optional[float] = None,
) -> torch.Tensor:
    """Programmable attention (FlexAttention parity,
    /root/reference/models/attention/flex_attention.py:414-562).

    Named patterns route to the kernel (causal / sliding_window / prefix_lm /
    alibi via ``block_mask.pattern``); arbitrary Python callables run on the
    fp32 reference path (GPU included — they are inherently per-element
    Python and exist for parity/experimentation, not the hot path).
    """
    if isinstance(block_mask, CompiledBlockMask):
        return _flex_blockmask(q, k, v, block_mask, score_mod, scale)
    if block_mask is not None and score_mod is None and mask_mod is None:
        p = block_mask
        return flash_attention(
            q, k, v,
            causal=p.pattern in ("causal", "sliding_window", "alibi"),
            scale=scale,
            window=p.window if p.pattern == "sliding_window" else None,
            prefix_len=p.prefix_len if p.pattern == "prefix_lm" else None,
            alibi_slopes=getattr(p, "alibi_slopes", None),
        )
    if score_mod is None and mask_mod is None:
        return flash_attention(q, k, v, causal=True, scale=scale)
    if (mask_mod is not None and q.is_cuda and not torch.is_grad_enabled()
            and q.dtype == torch.bfloat16):
        bm = CompiledBlockMask(mask_mod, q.shape[0], q.shape[2], q.shape[1],
                               k.shape[1], device=q.device)
        return _flex_blockmask(q, k, v, bm, score_mod, scale)
    return attention_ref(q, k, v, causal=False, scale=scale, score_mod=score_mod, mask_mod=mask_mod)


def _flex_blockmask(q, k, v, bm: "CompiledBlockMask", score_mod, scale):
    """Kernel-path flex: compiled mask tensors + (optional) a precomputed
    additive bias tensor for score_mod (evaluated vectorized once; the
    kernel adds it pre-softmax). Inference/eval surface — with grads the
    caller stays on the autograd fp32 path."""
    from ._ext import require_ext

    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    bias = torch.empty(0, dtype=torch.bfloat16, device=q.device)
    if score_mod is not None:
        B, S, H, D = q.shape
        KV = k.shape[1]
        qi = torch.arange(S, device=q.device).view(1, 1, S, 1)
        ki = torch.arange(KV, device=q.device).view(1, 1, 1, KV)
        bi = torch.arange(B, device=q.device).view(B, 1, 1, 1)
        hi = torch.arange(H, device=q.device).view(1, H, 1, 1)
        # the kernel consumes score_mod as an ADDITIVE bias (f(s) = s + f(0));
        # probe additivity on a small slice and fall back to the fp32
        # composition for non-additive mods (e.g. multiplicative scaling)
        sp, kp = min(S, 8), min(KV, 8)
        probe0 = torch.zeros(B, H, sp, kp, device=q.device)
        f0 = score_mod(probe0, bi, hi, qi[:, :, :sp], ki[:, :, :, :kp]).float()
        f1 = score_mod(probe0 + 1.0, bi, hi, qi[:, :, :sp], ki[:, :, :, :kp]).float()
        if not torch.allclose(f1 - f0, torch.ones_like(f0), atol=1e-4):
            return attention_ref(q, k, v, causal=False, scale=scale,
                                 score_mod=score_mod,
                                 mask_mod=getattr(bm, "_mask_mod", None))
        zero = torch.zeros(B, H, S, KV, device=q.device)
        bias = score_mod(zero, bi, hi, qi, ki).to(torch.bfloat16)
    o, _ = require_ext().attn_fwd_blockmask(q, k, v, float(scale), bm.gran,
                                            bm.bits, bm.range, bias)
    return o


class BlockMask:
    """Named block-sparsity pattern (device-side block mask precompute lives in
    the kernel's block-skip logic; this object carries the pattern).

    Parity surface: create_block_mask
    (/root/reference/models/attention/flex_attention.py:356-411)."""

    def __init__(
        self,
        pattern: str = "causal",
        window: Optional[int] = None,
        prefix_len: Optional[int] = None,
        alibi_slopes: Optional[torch.Tensor] = None,
        block_size: int = 128,
    ):
        assert pattern in ("causal", "full", "sliding_window", "prefix_lm", "alibi")
        self.pattern = pattern
        self.window = window
        self.prefix_len = prefix_len
        self.alibi_slopes = alibi_slopes
        self.block_size = block_size


class CompiledBlockMask:
    """An arbitrary ``mask_mod`` compiled into device tensors that drive the
    K1 kernel's block-skip / per-element-bit machinery (VERDICT r1 #5;
    parity: /root/reference/models/attention/flex_attention.py:356-411 —
    the reference SAMPLED block midpoints and then materialized full S x S
    scores anyway; here the mask is exact and only live kv-tile ranges are
    visited).

    Tensors (uint8/int32, tiny vs S^2 fp32):
      gran  [B,H,ceil(Q/32),ceil(KV/64)]  0 = granule fully masked,
                                          1 = partial (per-element bits),
                                          2 = fully live (mask-free path)
      bits  [B,H,Q,ceil(KV/8)]            packed keep bits
      range [B,H,ceil(Q/256),2]           first / last+1 live kv tile per
                                          256-row kernel block
    """

    KVB = 64
    QPB = 256

    def __init__(self, mask_mod: Callable, B: int, H: int, Q_LEN: int, KV_LEN: int,
                 device="cuda"):
        self.B, self.H, self.Q_LEN, self.KV_LEN = B, H, Q_LEN, KV_LEN
        self._mask_mod = mask_mod  # kept for the non-additive score_mod fallback
        qi = torch.arange(Q_LEN, device=device)
        ki = torch.arange(KV_LEN, device=device)
        nq32 = (Q_LEN + 31) // 32
        nkv = (KV_LEN + self.KVB - 1) // self.KVB
        nqpb = (Q_LEN + self.QPB - 1) // self.QPB
        self.gran = torch.zeros(B, H, nq32, nkv, dtype=torch.uint8, device=device)
        self.bits = torch.zeros(B, H, Q_LEN, (KV_LEN + 7) // 8, dtype=torch.uint8,
                                device=device)
        self.range = torch.zeros(B, H, nqpb, 2, dtype=torch.int32, device=device)
        weights = (1 << torch.arange(8, device=device, dtype=torch.int32)).to(torch.uint8)
        for b in range(B):
            for h in range(H):
                keep = mask_mod(b, h, qi.unsqueeze(1), ki.unsqueeze(0))
                keep = keep.to(torch.bool)  # [Q, KV], exact
                kp = torch.zeros(Q_LEN, nkv * self.KVB // 8 * 8, dtype=torch.bool,
                                 device=device)
                kp[:, :KV_LEN] = keep
                packed = (kp.reshape(Q_LEN, -1, 8).to(torch.uint8) *
                          weights).sum(-1).to(torch.uint8)
                self.bits[b, h] = packed[:, : (KV_LEN + 7) // 8]
                # granule codes: pad to 32-row / 64-col granules
                gq = torch.zeros(nq32 * 32, nkv * self.KVB, dtype=torch.bool,
                                 device=device)
                gq[:Q_LEN, :KV_LEN] = keep
                g = gq.reshape(nq32, 32, nkv, self.KVB)
                any_live = g.any(dim=(1, 3))
                # "full" additionally requires every IN-RANGE element live
                gq2 = torch.ones(nq32 * 32, nkv * self.KVB, dtype=torch.bool,
                                 device=device)
                gq2[:Q_LEN, :KV_LEN] = keep
                all_live = gq2.reshape(nq32, 32, nkv, self.KVB).all(dim=(1, 3))
                self.gran[b, h] = any_live.to(torch.uint8) + (any_live & all_live).to(torch.uint8)
                # per 256-row block: first/last live kv tile. Slice the
                # 8 granule rows explicitly — a reshape(nqpb, -1) misassigns
                # rows whenever ceil(Q/32) is not a multiple of 8 (caught by
                # tests/test_attention_cpu.py::..._bruteforce at Q=300).
                for qb in range(nqpb):
                    rows = any_live[qb * 8:(qb + 1) * 8]
                    live = rows.any(dim=0).nonzero().flatten()
                    if live.numel():
                        self.range[b, h, qb, 0] = int(live.min())
                        self.range[b, h, qb, 1] = int(live.max()) + 1


def create_block_mask(
    mask_mod: Optional[Callable] = None,
    B: int = 1,
    H: int = 1,
    Q_LEN: int = 0,
    KV_LEN: int = 0,
    pattern: str = "causal",
    device=None,
    **kwargs,
) -> BlockMask:
    """Named patterns return a pattern BlockMask (kernel block-skip handles
    them natively); an arbitrary ``mask_mod`` callable with a CUDA device is
    compiled to a CompiledBlockMask driving the kernel (else it stays a
    python callable for the fp32 reference path)."""
    if mask_mod is not None and Q_LEN and KV_LEN and device is not None             and torch.device(device).type == "cuda":
        return CompiledBlockMask(mask_mod, B, H, Q_LEN, KV_LEN, device=device)
    return BlockMask(pattern=pattern, **kwargs)
