"""HIP kernel numerics on a real MI355X: each kernel vs the plain PyTorch
fp32 reference of the same op (tolerances per dtype)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


def dev():
    return torch.device("cuda:0")


@pytest.fixture(scope="module")
def ext():
    from mlx_cuda_distributed_pretraining_amd.ops import require_ext

    return require_ext()


# ---------------- MFMA layout self-checks (run these FIRST) ----------------
def test_mfma_tile_layout(ext):
    torch.manual_seed(0)
    # ASYMMETRIC inputs (guide: symmetric B passes transposed writes)
    A = (torch.arange(32 * 16, device=dev(), dtype=torch.float32).reshape(32, 16) % 7 - 3)
    B = (torch.arange(16 * 32, device=dev(), dtype=torch.float32).reshape(16, 32) % 5 - 2) * 0.5
    A += torch.randn(32, 16, device=dev())
    B += torch.randn(16, 32, device=dev())
    C = ext.mfma_tile_test(A.bfloat16(), B.bfloat16())
    want = A.bfloat16().float() @ B.bfloat16().float()
    assert torch.allclose(C, want, atol=1e-2, rtol=1e-2), (C - want).abs().max()


def test_afrag_transform(ext):
    torch.manual_seed(1)
    M = torch.randn(32, 32, device=dev(), dtype=torch.float32)
    A = ext.afrag_transform_test(M)
    want = M.t().bfloat16().float()
    assert torch.allclose(A, want, atol=1e-2), (A - want).abs().max()


# ---------------- pointwise / reduction kernels ----------------
@pytest.mark.parametrize("shape,H", [((4, 128), 2048), ((2, 64), 128), ((1, 7), 768)])
def test_rmsnorm_fwd_bwd(ext, shape, H):
    from mlx_cuda_distributed_pretraining_amd.ops import rms_norm

    torch.manual_seed(0)
    x = torch.randn(*shape, H, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    w = torch.randn(H, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    y = rms_norm(x, w, 1e-5)
    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-5) * wf
    assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy.float())
    assert torch.allclose(x.grad.float(), xf.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=5e-1, rtol=5e-2)


@pytest.mark.parametrize("traditional", [False, True])
@pytest.mark.parametrize("offset", [0, 5])
def test_rope_gpu_matches_ref(ext, traditional, offset):
    from mlx_cuda_distributed_pretraining_amd.ops import RopeTable, apply_rope
    from mlx_cuda_distributed_pretraining_amd.ops.rope import rope_ref

    torch.manual_seed(0)
    B, S, H, D = 2, 33, 4, 64
    table = RopeTable(D)
    cos, sin = table.get(S, dev(), offset)
    x = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    y = apply_rope(x, cos, sin, traditional, offset)
    ref = rope_ref(x.detach().float(), cos, sin, traditional, offset)
    assert torch.allclose(y.float(), ref, atol=2e-2, rtol=2e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    # backward is the inverse rotation
    dx_ref = rope_ref(dy.float(), cos, sin, traditional, offset, conj=True)
    assert torch.allclose(x.grad.float(), dx_ref, atol=2e-2, rtol=2e-2)


def test_swiglu_gpu(ext):
    from mlx_cuda_distributed_pretraining_amd.ops import swiglu

    torch.manual_seed(0)
    gu = torch.randn(8, 64, 2 * 256, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    y = swiglu(gu)
    guf = gu.detach().float().requires_grad_(True)
    ref = torch.nn.functional.silu(guf[..., :256]) * guf[..., 256:]
    assert torch.allclose(y.float(), ref, atol=3e-2, rtol=3e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    ref.backward(dy.float())
    assert torch.allclose(gu.grad.float(), guf.grad, atol=5e-2, rtol=5e-2)


def test_cross_entropy_gpu(ext):
    from mlx_cuda_distributed_pretraining_amd.ops import fused_cross_entropy

    torch.manual_seed(0)
    N, V = 512, 32000
    logits = torch.randn(N, V, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    targets = torch.randint(0, V, (N,), device=dev())
    targets[::5] = -100
    loss, ntok = fused_cross_entropy(logits, targets, ignore_index=-100)
    lf = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lf, targets, ignore_index=-100)
    assert ntok.item() == (targets != -100).sum().item()
    assert torch.allclose(loss, ref, atol=2e-3, rtol=2e-3)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad.float(), lf.grad, atol=2e-4, rtol=5e-2)


def test_sumsq_and_adamw_gpu(ext):
    from mlx_cuda_distributed_pretraining_amd.ops import fused_optim

    torch.manual_seed(0)
    N = 100_000
    g = torch.randn(N, device=dev(), dtype=torch.bfloat16)
    ss = fused_optim.grad_sumsq(g)
    assert torch.allclose(ss, g.float().pow(2).sum(), rtol=1e-3)

    master = torch.randn(N, device=dev(), dtype=torch.float32)
    param = master.bfloat16().clone()
    m = torch.zeros(N, device=dev())
    v = torch.zeros(N, device=dev())
    grad = torch.randn(N, device=dev(), dtype=torch.bfloat16)

    ref_p = torch.nn.Parameter(master.clone())
    opt = torch.optim.AdamW([ref_p], lr=1e-2, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.1)
    ref_p.grad = grad.float()
    opt.step()

    fused_optim.adamw_step(param, master, grad, m, v, 1, 1e-2, 0.9, 0.999, 1e-8, 0.1,
                           decay_boundary=N)
    torch.cuda.synchronize()
    assert torch.allclose(master, ref_p.detach(), atol=1e-5, rtol=1e-5)
    assert torch.allclose(param.float(), master, atol=1e-2, rtol=1e-2)


def test_lion_sgd_gpu(ext):
    from mlx_cuda_distributed_pretraining_amd.ops import fused_optim

    N = 4096
    for step_fn in ("lion", "sgd"):
        master = torch.randn(N, device=dev())
        param = master.bfloat16().clone()
        grad = torch.randn(N, device=dev(), dtype=torch.bfloat16)
        buf = torch.zeros(N, device=dev())
        if step_fn == "lion":
            fused_optim.lion_step(param, master, grad, buf, 1, 1e-3, 0.9, 0.99, 0.1, N)
        else:
            fused_optim.sgd_step(param, master, grad, buf, 1, 1e-2, 0.9, 0.1, N, True)
        torch.cuda.synchronize()
        assert torch.isfinite(master).all()
        assert torch.allclose(param.float(), master, atol=1e-2, rtol=1e-2)


def test_sample_token_gpu(ext):
    logits = torch.full((1000,), -10.0, device=dev())
    logits[123] = 10.0
    for seed in range(5):
        t = ext.sample_token(logits, 0.8, 1.0, 0.0, seed)
        assert t.item() == 123


# ---------------- attention ----------------
def attn_cases():
    # (B, Sq, Skv, Hq, Hkv, D, mod_kwargs)
    return [
        (1, 128, 128, 2, 2, 64, {}),
        (2, 256, 256, 4, 2, 128, {}),           # GQA
        (1, 200, 200, 2, 2, 64, {}),            # ragged seq
        (1, 256, 256, 2, 2, 64, {"window": 64}),
        (1, 256, 256, 2, 2, 64, {"prefix_len": 100}),
        (1, 128, 128, 4, 4, 64, {"alibi": True}),
        (1, 1, 96, 2, 2, 64, {}),               # decode: Sq=1 with cache
        (1, 128, 128, 2, 2, 64, {"causal": False}),
    ]


@pytest.mark.parametrize("case", attn_cases())
def test_attn_fwd_gpu(ext, case):
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention

    B, Sq, Skv, Hq, Hkv, D, kw = case
    torch.manual_seed(0)
    q = torch.randn(B, Sq, Hq, D, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(B, Skv, Hkv, D, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(B, Skv, Hkv, D, device=dev(), dtype=torch.bfloat16)
    causal = kw.get("causal", True)
    alibi = None
    if kw.get("alibi"):
        alibi = torch.tensor([2 ** (-(i + 1)) for i in range(Hq)], device=dev())
    o = flash_attention(q, k, v, causal=causal, window=kw.get("window"),
                        prefix_len=kw.get("prefix_len"), alibi_slopes=alibi)
    ref = attention_ref(q.float(), k.float(), v.float(), causal=causal,
                        scale=1.0 / math.sqrt(D), window=kw.get("window"),
                        prefix_len=kw.get("prefix_len"), alibi_slopes=alibi)
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"attn fwd max err {err} for case {case}"


def test_attn_fwd_rescale_branch_forced(ext):
    """Guide rule 26: the defer-max rescale branch is rare on random data —
    force it (spike K rows at a late tile so the running max jumps >> THR=8
    base-2 after earlier tiles accumulated) and check against the fp32
    oracle. Exercises the pipelined o/l rescale ordering (T13 hazard)."""
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention

    B, S, Hq, Hkv, D = 1, 512, 2, 2, 128
    torch.manual_seed(3)
    q = torch.randn(B, S, Hq, D, device=dev(), dtype=torch.bfloat16) * 0.3
    k = torch.randn(B, S, Hkv, D, device=dev(), dtype=torch.bfloat16) * 0.3
    v = torch.randn(B, S, Hkv, D, device=dev(), dtype=torch.bfloat16)
    # every q row gets a huge score against k rows 400..403 (tile 6 of 8):
    # k_spike = 40 * mean(q) makes q.k ~ 40*|q|^2/S >> previous maxima
    u = q.float().mean(dim=1, keepdim=True)  # [B,1,Hq,D]
    u = (40.0 * u / u.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
    for r in range(400, 404):
        k[:, r] = u[:, 0]
    o = flash_attention(q, k, v, causal=True)
    ref = attention_ref(q.float(), k.float(), v.float(), causal=True,
                        scale=1.0 / math.sqrt(D))
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"forced-rescale max err {err}"


def test_attn_fwd_lse_matches(ext):
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref
    from mlx_cuda_distributed_pretraining_amd.ops._ext import get_ext

    torch.manual_seed(0)
    B, S, H, D = 1, 128, 2, 64
    q = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    slopes = torch.empty(0, device=dev())
    o, lse = get_ext().attn_fwd(q, k, v, 1.0 / math.sqrt(D), 1, 0, slopes)
    _, ref_lse = attention_ref(q.float(), k.float(), v.float(), causal=True,
                               scale=1.0 / math.sqrt(D), return_lse=True)
    assert torch.allclose(lse, ref_lse, atol=3e-2, rtol=1e-2)


@pytest.mark.parametrize("case", attn_cases())
def test_attn_bwd_gpu(ext, case):
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref, flash_attention

    B, Sq, Skv, Hq, Hkv, D, kw = case
    if Sq != Skv:
        pytest.skip("bwd only used in training (Sq == Skv)")
    torch.manual_seed(0)
    mk = lambda *s: torch.randn(*s, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    q, k, v = mk(B, Sq, Hq, D), mk(B, Skv, Hkv, D), mk(B, Skv, Hkv, D)
    causal = kw.get("causal", True)
    alibi = None
    if kw.get("alibi"):
        alibi = torch.tensor([2 ** (-(i + 1)) for i in range(Hq)], device=dev())
    o = flash_attention(q, k, v, causal=causal, window=kw.get("window"),
                        prefix_len=kw.get("prefix_len"), alibi_slopes=alibi)
    dy = torch.randn_like(o)
    o.backward(dy)

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    ref = attention_ref(qf, kf, vf, causal=causal, scale=1.0 / math.sqrt(D),
                        window=kw.get("window"), prefix_len=kw.get("prefix_len"),
                        alibi_slopes=alibi)
    ref.backward(dy.float())
    for name, got, want in [("dq", q.grad, qf.grad), ("dk", k.grad, kf.grad),
                            ("dv", v.grad, vf.grad)]:
        err = (got.float() - want).abs().max().item()
        ref_mag = want.abs().max().item()
        assert err < 0.05 * max(ref_mag, 1.0), f"{name} max err {err} (ref mag {ref_mag}) case {case}"


def test_fused_qkv_rope_attention_matches_composition(ext):
    """Fused qkv->RoPE->attention autograd node vs the op composition
    (incl. dQKV written in place by attn_bwd_out/rope_fwd_out)."""
    from mlx_cuda_distributed_pretraining_amd.ops.attention import (
        flash_attention, rope_flash_attention_qkv,
    )
    from mlx_cuda_distributed_pretraining_amd.ops.rope import RopeTable, apply_rope

    torch.manual_seed(0)
    B, S, Hq, Hkv, D = 2, 256, 8, 4, 128
    table = RopeTable(D, 10000.0)
    cos, sin = table.get(S, dev(), 0)
    qkv = torch.randn(B, S, (Hq + 2 * Hkv) * D, device=dev(),
                      dtype=torch.bfloat16, requires_grad=True)

    o = rope_flash_attention_qkv(qkv, cos, sin, Hq, Hkv, D)
    do = torch.randn_like(o)
    o.backward(do)
    got_o, got_g = o.detach(), qkv.grad.clone()

    qkv2 = qkv.detach().clone().requires_grad_(True)
    q, k, v = qkv2.split([Hq * D, Hkv * D, Hkv * D], dim=-1)
    qr = apply_rope(q.view(B, S, Hq, D), cos, sin, False, 0)
    kr = apply_rope(k.view(B, S, Hkv, D), cos, sin, False, 0)
    o2 = flash_attention(qr, kr, v.view(B, S, Hkv, D), causal=True)
    o2.backward(do)

    assert torch.allclose(got_o.float(), o2.detach().float(), atol=3e-2, rtol=3e-2)
    assert torch.allclose(got_g.float(), qkv2.grad.float(), atol=5e-2, rtol=5e-2), \
        (got_g.float() - qkv2.grad.float()).abs().max()


def test_static_decode_gpu_matches_eager(ext):
    """GPU static-decode kernels (rope_decode/kv_append/attn_decode) +
    hipGraph capture vs the eager generate_step oracle."""
    from mlx_cuda_distributed_pretraining_amd.inference.generate import generate_step
    from mlx_cuda_distributed_pretraining_amd.inference.static_decode import GraphDecoder
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    torch.manual_seed(0)
    args = ModelArgs(hidden_size=256, intermediate_size=512, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=128, vocab_size=503)
    model = Model(args).to(dev(), torch.bfloat16).eval()
    prompt = [3, 17, 41, 5, 88, 23, 9, 101, 250, 77]
    n = 16

    oracle = list(generate_step(model, prompt, max_tokens=n))

    # eager static path (no graph)
    d1 = GraphDecoder(model, batch=1, max_len=512)
    d1.prefill(torch.tensor([prompt], device=dev()))
    got_eager = d1.decode(n)[0].tolist()
    assert got_eager == oracle, (got_eager, oracle)

    # graph-captured path
    d2 = GraphDecoder(model, batch=1, max_len=512)
    d2.prefill(torch.tensor([prompt], device=dev()))
    d2.capture()
    got_graph = d2.decode(n)[0].tolist()
    assert got_graph == oracle, (got_graph, oracle)


def test_attn_decode_kernel_numerics(ext):
    """Split-KV decode attention vs the fp32 reference at several lengths
    (chunk-boundary cases included)."""
    from mlx_cuda_distributed_pretraining_amd.ops.attention import attention_ref

    torch.manual_seed(0)
    B, Hq, Hkv, D, Lmax = 2, 4, 2, 128, 1024
    kc = torch.randn(B, Lmax, Hkv, D, device=dev(), dtype=torch.bfloat16)
    vc = torch.randn_like(kc)
    part = torch.empty(B, Hq, (Lmax + 255) // 256, D + 2, dtype=torch.float32, device=dev())
    for length in (1, 7, 255, 256, 257, 777, 1024):
        q = torch.randn(B, 1, Hq, D, device=dev(), dtype=torch.bfloat16)
        pos = torch.tensor([length - 1], dtype=torch.int32, device=dev())
        o = ext.attn_decode(q, kc, vc, pos, part, 1.0 / math.sqrt(D))
        want = attention_ref(q, kc[:, :length], vc[:, :length], causal=True)
        assert torch.allclose(o.float(), want.float(), atol=3e-2, rtol=3e-2), \
            (length, (o.float() - want.float()).abs().max())


def test_gemv_matches_matmul(ext):
    torch.manual_seed(0)
    for B, N, K in [(1, 2560, 2048), (1, 2048, 5632), (2, 512, 768), (1, 32003, 2048)]:
        x = torch.randn(B, 1, K, device=dev(), dtype=torch.bfloat16)
        W = torch.randn(N, K, device=dev(), dtype=torch.bfloat16)
        y = ext.gemv_bf16(x, W)
        want = torch.nn.functional.linear(x, W)
        assert y.shape == want.shape
        assert torch.allclose(y.float(), want.float(), atol=2e-1, rtol=2e-2), \
            (N, K, (y.float() - want.float()).abs().max())


def test_fp8_linear_gpu_matches_bf16(ext):
    """e4m3 _scaled_mm path vs bf16 linear (fwd + both grads) within fp8
    quantization tolerance."""
    from mlx_cuda_distributed_pretraining_amd.ops.fp8 import fp8_linear

    torch.manual_seed(0)
    M, K, N = 512, 2048, 2560
    x = torch.randn(M, K, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(N, K, device=dev(), dtype=torch.bfloat16) * 0.02).requires_grad_(True)
    y = fp8_linear(x, w)
    dy = torch.randn_like(y)
    y.backward(dy)

    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    y2 = torch.nn.functional.linear(x2, w2)
    y2.backward(dy)

    def relerr(a, b):
        return (a.float() - b.float()).norm() / b.float().norm()

    assert relerr(y, y2) < 0.05, relerr(y, y2)
    assert relerr(x.grad, x2.grad) < 0.05, relerr(x.grad, x2.grad)
    assert relerr(w.grad, w2.grad) < 0.05, relerr(w.grad, w2.grad)


def test_fp8_model_trains_on_gpu(ext):
    """fp8-flagged model: a few steps with finite non-diverging loss."""
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs
    from mlx_cuda_distributed_pretraining_amd.optim.flat_fused import FusedFlatAdamW
    from mlx_cuda_distributed_pretraining_amd.parallel.flat import FlatParamSpace

    torch.manual_seed(0)
    args = ModelArgs(hidden_size=256, intermediate_size=512, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=64, vocab_size=211,
                     fp8=True)
    model = Model(args).to(dev(), torch.bfloat16)
    space = FlatParamSpace(model, grad_mode="copy")
    opt = FusedFlatAdamW(space, lr=1e-3)
    losses = []
    for i in range(5):
        space.zero_grad()
        toks = torch.randint(0, 211, (4, 64), device=dev())
        logits = model(toks[:, :-1])
        loss = torch.nn.functional.cross_entropy(
            logits.float().reshape(-1, 211), toks[:, 1:].reshape(-1))
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert all(x == x and x < 20 for x in losses), losses
    assert losses[-1] <= losses[0] + 0.5, losses


def test_gemv_ex_modes(ext):
    """Fused GEMV staging modes vs the op compositions."""
    torch.manual_seed(0)
    K, N = 2048, 2560
    x = torch.randn(1, K, device=dev(), dtype=torch.bfloat16)
    W = torch.randn(N, K, device=dev(), dtype=torch.bfloat16) * 0.02
    nw = torch.randn(K, device=dev(), dtype=torch.bfloat16)
    res = torch.randn(1, N, device=dev(), dtype=torch.bfloat16)
    none = x.new_empty(0)

    def relerr(a, b):
        return (a.float() - b.float()).norm() / b.float().norm()

    # plain + residual epilogue
    y = ext.gemv_ex(x, W, 0, none, 0.0, res)
    want = torch.nn.functional.linear(x, W) + res
    assert relerr(y, want) < 2e-2, relerr(y, want)

    # rmsnorm staging
    y = ext.gemv_ex(x, W, 1, nw, 1e-5, none)
    xn = (x.float() * torch.rsqrt(x.float().pow(2).mean(-1, True) + 1e-5)
          * nw.float()).to(torch.bfloat16)
    want = torch.nn.functional.linear(xn, W)
    assert relerr(y, want) < 2e-2, relerr(y, want)

    # swiglu staging
    gu = torch.randn(1, 2 * K, device=dev(), dtype=torch.bfloat16)
    y = ext.gemv_ex(gu, W, 2, none, 0.0, none)
    g, u = gu.float().split(K, dim=-1)
    want = torch.nn.functional.linear(
        (torch.nn.functional.silu(g) * u).to(torch.bfloat16), W)
    assert relerr(y, want) < 2e-2, relerr(y, want)


def test_sampled_graph_decode(ext):
    """Temperature sampling inside the captured graph: tokens vary across
    positions (device-salt RNG) and stay in-vocab."""
    from mlx_cuda_distributed_pretraining_amd.inference.static_decode import GraphDecoder
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    torch.manual_seed(0)
    args = ModelArgs(hidden_size=256, intermediate_size=512, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=128, vocab_size=503)
    model = Model(args).to(dev(), torch.bfloat16).eval()
    d = GraphDecoder(model, batch=1, max_len=256, temperature=1.0, seed=7)
    d.prefill(torch.tensor([[3, 5, 7, 11]], device=dev()))
    d.capture()
    out = d.decode(32)[0]
    assert ((out >= 0) & (out < 503)).all()
    assert len(set(out.tolist())) > 4  # sampling, not a constant loop


def test_int8_kv_decode(ext):
    """Fused int8-KV decode: append+quantize kernel and dequant-fused
    attention vs the fp reference on the dequantized cache."""
    from mlx_cuda_distributed_pretraining_amd.inference.static_decode import GraphDecoder
    from mlx_cuda_distributed_pretraining_amd.models.llama import Model, ModelArgs

    torch.manual_seed(0)
    args = ModelArgs(hidden_size=256, intermediate_size=512, num_layers=2,
                     num_heads=4, num_kv_heads=2, head_dim=128, vocab_size=503)
    model = Model(args).to(dev(), torch.bfloat16).eval()
    prompt = torch.tensor([[3, 17, 41, 5, 88, 23, 9, 101]], device=dev())

    ref = GraphDecoder(model, batch=1, max_len=256)
    ref.prefill(prompt)
    ref_out = ref.decode(24)[0].tolist()

    q8 = GraphDecoder(model, batch=1, max_len=256, kv_bits=8)
    q8.prefill(prompt)
    q8.capture()
    q8_out = q8.decode(24)[0].tolist()
    # int8 KV perturbs logits slightly; greedy tokens should mostly agree
    agree = sum(a == b for a, b in zip(ref_out, q8_out)) / len(ref_out)
    assert agree >= 0.7, (agree, ref_out, q8_out)
    assert all(0 <= t < 503 for t in q8_out)


def test_add_rmsnorm_gpu(ext):
    from mlx_cuda_distributed_pretraining_amd.ops.rmsnorm import add_rms_norm

    torch.manual_seed(0)
    for rows, H in ((64, 2048), (33, 1024)):
        x = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16, requires_grad=True)
        r = torch.randn(rows, H, device=dev(), dtype=torch.bfloat16, requires_grad=True)
        w = torch.randn(H, device=dev(), dtype=torch.bfloat16, requires_grad=True)
        s, y = add_rms_norm(x, r, w, 1e-5)
        (s.float().sin().sum() + y.float().sum()).backward()

        x2 = x.detach().float().requires_grad_(True)
        r2 = r.detach().float().requires_grad_(True)
        w2 = w.detach().float().requires_grad_(True)
        s2 = x2 + r2
        y2 = s2 * torch.rsqrt(s2.pow(2).mean(-1, keepdim=True) + 1e-5) * w2
        (s2.sin().sum() + y2.sum()).backward()

        assert (s.float() - s2).abs().max() < 2e-2
        assert (y.float() - y2).abs().max() < 8e-2
        assert (x.grad.float() - x2.grad).abs().max() < 8e-2
        assert (r.grad.float() - r2.grad).abs().max() < 8e-2
        assert (w.grad.float() - w2.grad).abs().max() / w2.grad.abs().max() < 3e-2


def test_flex_custom_mask_mod_kernel_path(ext):
    """K2: arbitrary mask_mod compiled to a device block mask drives the
    MFMA kernel (MOD_BLOCKMASK) instead of the fp32 S^2 fallback (VERDICT
    r1 #5). Oracle: attention_ref with the same callable."""
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref
    from mlx_cuda_distributed_pretraining_amd.ops.attention import (
        CompiledBlockMask, flex_attention,
    )

    torch.manual_seed(7)
    B, S, H, D = 2, 512, 2, 64

    def doc_band_mask(b, h, qi, ki):
        # causal AND within a 128-token band, except documents of 200 tokens
        # don't attend across their boundary — an arbitrary non-named pattern
        return (ki <= qi) & (qi - ki < 128) & (qi // 200 == ki // 200)

    q = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    bm = CompiledBlockMask(doc_band_mask, B, H, S, S, device=dev())
    # live kv range honored: first 256-row block only touches tiles < 256/64
    assert int(bm.range[0, 0, 0, 1]) <= 4
    with torch.no_grad():
        o = flex_attention(q, k, v, block_mask=bm)
    ref = attention_ref(q.float(), k.float(), v.float(), causal=False,
                        scale=1.0 / math.sqrt(D), mask_mod=doc_band_mask)
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"flex blockmask kernel path err {err}"


def test_flex_score_mod_bias_kernel_path(ext):
    """score_mod -> precomputed additive bias consumed by the kernel."""
    from mlx_cuda_distributed_pretraining_amd.ops import attention_ref
    from mlx_cuda_distributed_pretraining_amd.ops.attention import flex_attention

    torch.manual_seed(8)
    B, S, H, D = 1, 256, 2, 64

    def rel_bias(score, b, h, qi, ki):
        return score - 0.05 * (qi - ki).abs().to(torch.float32)

    def causal_mask(b, h, qi, ki):
        return ki <= qi

    q = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    k = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    v = torch.randn(B, S, H, D, device=dev(), dtype=torch.bfloat16)
    with torch.no_grad():
        o = flex_attention(q, k, v, score_mod=rel_bias, mask_mod=causal_mask)
    ref = attention_ref(q.float(), k.float(), v.float(), causal=False,
                        scale=1.0 / math.sqrt(D),
                        score_mod=rel_bias, mask_mod=causal_mask)
    err = (o.float() - ref).abs().max().item()
    assert err < 3e-2, f"flex bias kernel path err {err}"


def test_vocab_parallel_ce_kernels(ext):
    """ce_vp_* kernels: world-1 full-vocab path == fused_cross_entropy;
    shard semantics (v0 offset, out-of-shard targets) checked by combining
    two shard calls by hand (ROADMAP r1 #15 / VERDICT #10)."""
    import torch.distributed as tdist

    from mlx_cuda_distributed_pretraining_amd.ops._ext import require_ext
    from mlx_cuda_distributed_pretraining_amd.ops.cross_entropy import fused_cross_entropy
    from mlx_cuda_distributed_pretraining_amd.parallel.tp import (
        vocab_parallel_cross_entropy,
    )

    e = require_ext()
    torch.manual_seed(9)
    N, V = 512, 1024
    logits = torch.randn(N, V, device=dev(), dtype=torch.bfloat16, requires_grad=True)
    tg = torch.randint(0, V, (N,), device=dev())
    tg[::7] = -100
    if not tdist.is_initialized():  # world-1: group reductions are no-ops
        loss, ntok = vocab_parallel_cross_entropy(logits, tg, 0, ignore_index=-100)
        l2 = logits.detach().clone().requires_grad_(True)
        ref, ntok_ref = fused_cross_entropy(l2, tg, -100)
        assert int(ntok) == int(ntok_ref)
        assert abs(float(loss) - float(ref)) < 2e-3
        loss.backward()
        ref.backward()
        gerr = (logits.grad.float() - l2.grad.float()).abs().max().item()
        assert gerr < 1e-4, f"vp-ce grad err {gerr}"

    # manual 2-shard combination equals full CE (shard algebra incl.
    # out-of-shard targets)
    lf = logits.detach()
    half = V // 2
    m0, t0 = e.ce_vp_stats(lf[:, :half].contiguous(), tg, 0, -100)
    m1, t1 = e.ce_vp_stats(lf[:, half:].contiguous(), tg, half, -100)
    m = torch.maximum(m0, m1)
    se = (e.ce_vp_sumexp(lf[:, :half].contiguous(), m) +
          e.ce_vp_sumexp(lf[:, half:].contiguous(), m))
    lse = m + torch.log(se)
    mask = tg != -100
    loss_sh = (torch.where(mask, lse - (t0 + t1), torch.zeros_like(lse)).sum()
               / mask.sum())
    ref2, _ = fused_cross_entropy(lf.clone().requires_grad_(True), tg, -100)
    assert abs(float(loss_sh) - float(ref2)) < 2e-3
