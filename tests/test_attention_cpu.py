"""Attention reference-path semantics (the same oracle the GPU kernels are
tested against in test_gpu_kernels.py)."""
import math

import torch

from mlx_cuda_distributed_pretraining_amd.ops import (
    BlockMask, attention_ref, flash_attention, flex_attention,
)

torch.manual_seed(0)


def naive_attention(q, k, v, causal=True, window=None, prefix_len=None):
    """Maximally naive BSHD attention for cross-checking attention_ref."""
    B, S, H, D = q.shape
    Hkv = k.shape[2]
    rep = H // Hkv
    out = torch.zeros_like(q, dtype=torch.float32)
    for b in range(B):
        for h in range(H):
            kk = k[b, :, h // rep].float()
            vv = v[b, :, h // rep].float()
            qq = q[b, :, h].float()
            scores = qq @ kk.t() / math.sqrt(D)
            for i in range(S):
                for j in range(S):
                    keep = True
                    if prefix_len is not None:
                        keep = j <= i or j < prefix_len
                    elif window is not None:
                        keep = j <= i and i - j < window
                    elif causal:
                        keep = j <= i
                    if not keep:
                        scores[i, j] = float("-inf")
            out[b, :, h] = torch.softmax(scores, -1) @ vv
    return out.to(q.dtype)


def test_attention_ref_causal_mha():
    B, S, H, D = 2, 16, 2, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    got = attention_ref(q, k, v, causal=True)
    want = naive_attention(q, k, v, causal=True)
    assert torch.allclose(got, want, atol=1e-5)


def test_attention_ref_gqa():
    B, S, Hq, Hkv, D = 2, 12, 4, 2, 8
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hkv, D)
    v = torch.randn(B, S, Hkv, D)
    got = attention_ref(q, k, v, causal=True)
    want = naive_attention(q, k, v, causal=True)
    assert torch.allclose(got, want, atol=1e-5)


def test_attention_ref_sliding_window():
    B, S, H, D = 1, 20, 2, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    got = attention_ref(q, k, v, causal=True, window=4)
    want = naive_attention(q, k, v, window=4)
    assert torch.allclose(got, want, atol=1e-5)


def test_attention_ref_prefix_lm():
    B, S, H, D = 1, 10, 2, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    got = attention_ref(q, k, v, causal=True, prefix_len=4)
    want = naive_attention(q, k, v, prefix_len=4)
    assert torch.allclose(got, want, atol=1e-5)


def test_attention_lse_matches():
    B, S, H, D = 1, 8, 1, 4
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    o, lse = attention_ref(q, k, v, causal=True, return_lse=True)
    scores = torch.einsum("bshd,bthd->bhst", q.float(), k.float()) / math.sqrt(D)
    mask = torch.tril(torch.ones(S, S, dtype=torch.bool))
    scores = scores.masked_fill(~mask, float("-inf"))
    want_lse = torch.logsumexp(scores, -1)
    assert torch.allclose(lse, want_lse, atol=1e-5)


def test_flash_attention_autograd_cpu():
    B, S, H, D = 2, 8, 2, 16
    q = torch.randn(B, S, H, D, requires_grad=True)
    k = torch.randn(B, S, H, D, requires_grad=True)
    v = torch.randn(B, S, H, D, requires_grad=True)
    o = flash_attention(q, k, v, causal=True)
    dy = torch.randn_like(o)
    o.backward(dy)

    q2, k2, v2 = (t.detach().clone().requires_grad_(True) for t in (q, k, v))
    o2 = attention_ref(q2, k2, v2, causal=True)
    o2.backward(dy)
    assert torch.allclose(o, o2, atol=1e-5)
    assert torch.allclose(q.grad, q2.grad, atol=1e-4)
    assert torch.allclose(k.grad, k2.grad, atol=1e-4)
    assert torch.allclose(v.grad, v2.grad, atol=1e-4)


def test_flex_attention_score_mod():
    B, S, H, D = 1, 8, 2, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))

    def score_mod(scores, b, h, qi, ki):
        return scores * 0.5

    got = flex_attention(q, k, v, score_mod=score_mod)
    want = attention_ref(q, k, v, causal=False, score_mod=score_mod)
    assert torch.allclose(got, want, atol=1e-6)


def test_flex_attention_named_patterns_route_to_flash():
    B, S, H, D = 1, 16, 2, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    bm = BlockMask(pattern="sliding_window", window=4)
    got = flex_attention(q, k, v, block_mask=bm)
    want = naive_attention(q, k, v, window=4)
    assert torch.allclose(got, want, atol=1e-5)


def test_alibi_attention():
    B, S, H, D = 1, 12, 4, 8
    q, k, v = (torch.randn(B, S, H, D) for _ in range(3))
    slopes = torch.tensor([2 ** (-(i + 1)) for i in range(H)])
    got = attention_ref(q, k, v, causal=True, alibi_slopes=slopes)
    # manual
    rep_out = torch.zeros_like(got, dtype=torch.float32)
    for h in range(H):
        qq, kk, vv = q[0, :, h].float(), k[0, :, h].float(), v[0, :, h].float()
        s = qq @ kk.t() / math.sqrt(D)
        for i in range(S):
            for j in range(S):
                if j > i:
                    s[i, j] = float("-inf")
                else:
                    s[i, j] += slopes[h] * (j - i)
        rep_out[0, :, h] = torch.softmax(s, -1) @ vv
    assert torch.allclose(got, rep_out.to(got.dtype), atol=1e-5)


def test_compiled_block_mask_tensors_match_bruteforce():
    """CompiledBlockMask's granule codes / packed bits / kv ranges vs a
    brute-force evaluation of the callable (CPU tensors; the GPU kernel
    consuming them is covered in test_gpu_kernels)."""
    import torch

    from mlx_cuda_distributed_pretraining_amd.ops.attention import CompiledBlockMask

    def mod(b, h, qi, ki):
        return (ki <= qi) & (qi - ki < 40) & ((ki % 97) != 3)

    B, H, S = 1, 2, 300
    bm = CompiledBlockMask(mod, B, H, S, S, device="cpu")
    qi = torch.arange(S).unsqueeze(1)
    ki = torch.arange(S).unsqueeze(0)
    keep = mod(0, 1, qi, ki)
    # bits
    for q in (0, 31, 64, 299):
        got = [(int(bm.bits[0, 1, q, k // 8]) >> (k % 8)) & 1 for k in range(S)]
        assert got == keep[q].int().tolist(), f"bits mismatch at q={q}"
    # granule codes: 0 = none live, 2 = all live (within bounds)
    nq, nkv = (S + 31) // 32, (S + 63) // 64
    for qg in range(nq):
        for kg in range(nkv):
            blk = keep[qg * 32:(qg + 1) * 32, kg * 64:(kg + 1) * 64]
            code = int(bm.gran[0, 1, qg, kg])
            assert (code > 0) == bool(blk.any()), (qg, kg)
            if code == 2:
                assert bool(blk.all()), (qg, kg)
    # kv ranges cover every live tile of each 256-row block
    nqpb = (S + 255) // 256
    for qb in range(nqpb):
        live = keep[qb * 256:(qb + 1) * 256]
        live_tiles = [kg for kg in range(nkv)
                      if live[:, kg * 64:(kg + 1) * 64].any()]
        lo, hi = int(bm.range[0, 1, qb, 0]), int(bm.range[0, 1, qb, 1])
        if live_tiles:
            assert lo <= live_tiles[0] and hi >= live_tiles[-1] + 1
