"""K8 Muon Newton-Schulz MFMA kernels vs plain-torch references.

Parity surface: /root/reference/optimizers/muon.py:54-83 (quintic NS-5).
Each kernel is checked against an fp32 torch composition; the whole NS chain
against the torch bf16 composition it replaces (same precision class).
"""
import pytest
import torch

pytestmark = pytest.mark.gpu

dev = "cuda:0"


def _ext():
    from mlx_cuda_distributed_pretraining_amd.ops._ext import require_ext
    return require_ext()


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (256, 384, 128), (512, 512, 5632)])
def test_muon_gemm_nt_numerics(M, N, K):
    ext = _ext()
    torch.manual_seed(0)
    X = (torch.randn(M, K, device=dev) / K**0.5).to(torch.bfloat16)
    Y = (torch.randn(N, K, device=dev) / K**0.5).to(torch.bfloat16)
    C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    none = torch.empty(0, dtype=torch.bfloat16, device=dev)
    ext.muon_gemm_nt(X, Y, C, 1.0, 0.0, none)
    ref = X.float() @ Y.float().t()
    err = (C.float() - ref).abs().max().item()
    assert err < 2e-2, f"NT gemm max err {err}"
    # fused beta*E epilogue (the quintic combine form)
    E = torch.randn(M, N, device=dev).to(torch.bfloat16)
    ext.muon_gemm_nt(X, Y, C, 2.0315, -4.775, E)
    ref2 = 2.0315 * ref - 4.775 * E.float()
    # C is stored bf16: tolerance is relative to the output magnitude
    tol = 1e-2 * ref2.abs().max().item() + 2e-2
    err2 = (C.float() - ref2).abs().max().item()
    assert err2 < tol, f"NT+E epilogue max err {err2} (tol {tol})"


@pytest.mark.parametrize("M,N", [(128, 128), (256, 640)])
def test_muon_gemm_nn_ax_numerics(M, N):
    ext = _ext()
    torch.manual_seed(1)
    B = (torch.randn(M, M, device=dev) / M**0.5).to(torch.bfloat16)
    X = torch.randn(M, N, device=dev).to(torch.bfloat16)
    C = torch.empty(M, N, dtype=torch.bfloat16, device=dev)
    ext.muon_gemm_nn_ax(B, X, C, 3.4445)
    ref = B.float() @ X.float() + 3.4445 * X.float()
    tol = 1e-2 * ref.abs().max().item() + 2e-2
    err = (C.float() - ref).abs().max().item()
    assert err < tol, f"NN+aX gemm max err {err} (tol {tol})"


@pytest.mark.parametrize("m,n", [(256, 512), (200, 300), (512, 384), (1024, 2816)])
def test_ns5_hip_matches_torch_chain(m, n):
    """Whole-chain check incl. zero-padding exactness (non-multiple shapes)
    and the transpose-if-tall path (512x384)."""
    from mlx_cuda_distributed_pretraining_amd.optim.muon import (
        NS_COEFFS, _ns5_hip, zeropower_via_newtonschulz5,
    )

    torch.manual_seed(2)
    G = torch.randn(m, n, device=dev)
    out = zeropower_via_newtonschulz5(G)  # dispatches to the HIP chain
    assert out.shape == G.shape

    # torch bf16 composition of the same chain (the reference semantics)
    a, b, c = NS_COEFFS
    X = G.t() if m > n else G
    X = (X / (X.norm() + 1e-7)).to(torch.bfloat16)
    for _ in range(5):
        A = X @ X.t()
        Bm = b * A + c * (A @ A)
        X = a * X + Bm @ X
    ref = X.float().t() if m > n else X.float()
    err = (out - ref).abs().max().item()
    assert err < 8e-2, f"NS5 chain max err {err} at {m}x{n}"
    # the orthogonalized factor has singular values near 1 (NS-5 band)
    sv = torch.linalg.svdvals(out)
    assert 0.3 < sv.min().item() and sv.max().item() < 1.6, (
        f"singular values out of NS-5 band: {sv.min()} .. {sv.max()}")


def test_muon_optimizer_uses_hip_chain():
    """A Muon step on GPU must run the kernel path (fail loudly otherwise)."""
    from mlx_cuda_distributed_pretraining_amd.optim.muon import Muon

    torch.manual_seed(3)
    p = torch.nn.Parameter(torch.randn(256, 384, device=dev))
    p.grad = torch.randn_like(p)
    opt = Muon([p], lr=0.02)
    before = p.detach().clone()
    opt.step()
    assert not torch.equal(before, p.detach())
    assert torch.isfinite(p).all()


def test_shampoo_stats_kernel_matches_torch():
    """K9 stats EMA on the MFMA kernel vs the fp32 torch composition."""
    from mlx_cuda_distributed_pretraining_amd.optim.shampoo import _stats_update_hip

    torch.manual_seed(4)
    for m, n in [(128, 256), (200, 100), (1024, 512)]:
        G = torch.randn(m, n, device=dev)
        S = torch.rand(m, m, device=dev)
        S = (S + S.t()) / 2
        S_ref = S.clone()
        _stats_update_hip(S, G, 0.95, left=True)
        ref = 0.95 * S_ref + 0.05 * (G @ G.t())
        rel = (S - ref).abs().max().item() / ref.abs().max().item()
        assert rel < 2e-2, f"left stats rel err {rel} at {m}x{n}"
        Sr = torch.rand(n, n, device=dev)
        Sr_ref = Sr.clone()
        _stats_update_hip(Sr, G, 0.95, left=False)
        refr = 0.95 * Sr_ref + 0.05 * (G.t() @ G)
        rel = (Sr - refr).abs().max().item() / refr.abs().max().item()
        assert rel < 2e-2, f"right stats rel err {rel} at {m}x{n}"


def test_shampoo_step_on_gpu_kernel_path():
    """A preconditioned Shampoo step runs the kernel-backed stats path and
    matches the CPU torch path closely."""
    from mlx_cuda_distributed_pretraining_amd.optim.shampoo import Shampoo, ShampooParams

    torch.manual_seed(5)
    w0 = torch.randn(128, 192)
    g0 = torch.randn(128, 192)
    outs = []
    for device in ("cpu", dev):
        p = torch.nn.Parameter(w0.clone().to(device))
        p.grad = g0.clone().to(device)
        opt = Shampoo([p], lr=1e-2,
                      hyperparams=ShampooParams(start_preconditioning_step=1, update_period=1))
        for _ in range(3):
            opt.step()
        outs.append(p.detach().cpu())
    # tolerance: the GPU stats are bf16-input-rounded (fp32 accumulate) and
    # the inverse-4th-root Newton amplifies stat deltas over repeated
    # preconditioned steps; the tight numerics check is the kernel-level
    # stats test above (2e-2 RELATIVE on the raw EMA).
    err = (outs[0] - outs[1]).abs().max().item()
    assert err < 3e-2, f"gpu kernel path diverged from cpu path: {err}"
