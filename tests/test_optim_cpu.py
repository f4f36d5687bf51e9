import math

import torch

from mlx_cuda_distributed_pretraining_amd.optim import (
    AdamWEnhanced, HybridOptimizer, LionEnhanced, Muon, OptimizationManager,
    SGDEnhanced, Shampoo, build_schedule, cosine_decay, join_schedules,
    linear_schedule, matrix_inverse_pth_root, zeropower_via_newtonschulz5,
)
from mlx_cuda_distributed_pretraining_amd.ops import fused_optim

torch.manual_seed(0)


# ---------------- schedules ----------------
def test_linear_schedule():
    s = linear_schedule(1.0, 0.0, 10)
    assert s(0) == 1.0
    assert abs(s(5) - 0.5) < 1e-9
    assert s(10) == 0.0
    assert s(20) == 0.0


def test_cosine_decay():
    s = cosine_decay(1.0, 100, 0.1)
    assert abs(s(0) - 1.0) < 1e-9
    assert abs(s(100) - 0.1) < 1e-9
    assert s(50) > s(75) > s(100)


def test_join_schedules_warmup_cosine():
    s = build_schedule({"type": "cosine_with_warmup", "warmup_steps": 10, "min_lr_ratio": 0.0},
                       1e-3, 110)
    assert s(0) < s(5) < s(10)
    assert abs(s(10) - 1e-3) < 1e-9
    assert s(50) < 1e-3


# ---------------- Muon ----------------
def test_newtonschulz_orthogonalizes():
    G = torch.randn(64, 32)
    X = zeropower_via_newtonschulz5(G, steps=5)
    # columns should be near-orthonormal: X^T X ~ I (for tall transpose-handled)
    gram = X.t() @ X
    I = torch.eye(32)
    assert (gram - I).abs().mean() < 0.35  # NS5 with these coeffs is approximate
    # singular values pushed toward 1
    sv = torch.linalg.svdvals(X)
    assert sv.min() > 0.3 and sv.max() < 1.6


def test_muon_step_decreases_loss():
    W = torch.nn.Parameter(torch.randn(16, 16))
    target = torch.randn(16, 16)
    opt = Muon([W], lr=0.05)
    losses = []
    for _ in range(20):
        loss = (W - target).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0]


# ---------------- Shampoo ----------------
def test_inverse_pth_root():
    A = torch.randn(16, 16)
    A = A @ A.t() + 0.1 * torch.eye(16)
    X = matrix_inverse_pth_root(A, p=4)
    # X should approximate A^(-1/4): X^4 @ A ~ I
    approx = torch.linalg.matrix_power(X, 4) @ A
    assert (approx - torch.eye(16)).abs().max() < 0.05


def test_shampoo_step_decreases_loss():
    W = torch.nn.Parameter(torch.randn(8, 8))
    target = torch.randn(8, 8)
    opt = Shampoo([W], lr=0.05)
    losses = []
    for _ in range(30):
        loss = (W - target).pow(2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] * 0.5


# ---------------- Enhanced ----------------
def test_adamw_enhanced_matches_torch_adamw():
    torch.manual_seed(1)
    p1 = torch.nn.Parameter(torch.randn(10, 10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = AdamWEnhanced([{"params": [p1], "weight_decay": 0.1}], lr=1e-2, weight_decay=0.1)
    o2 = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.1)
    for i in range(5):
        g = torch.randn(10, 10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6)


def test_lion_and_sgd_enhanced_run():
    for cls, kw in [(LionEnhanced, {}), (SGDEnhanced, {"momentum": 0.9})]:
        p = torch.nn.Parameter(torch.randn(5, 5))
        opt = cls([p], lr=1e-2, max_grad_norm=1.0, ema_decay=0.9, **kw)
        for _ in range(3):
            p.grad = torch.randn(5, 5)
            opt.step()
        assert torch.isfinite(p).all()
        assert opt.ema_state() is not None


def test_hybrid_routes_matrices():
    lin = torch.nn.Linear(8, 8)
    named = list(lin.named_parameters())
    opt = HybridOptimizer(named)
    assert opt.matrix_opt is not None  # weight (2D)
    assert opt.non_matrix_opt is not None  # bias (1D)
    lin.weight.grad = torch.randn_like(lin.weight)
    lin.bias.grad = torch.randn_like(lin.bias)
    opt.step()


# ---------------- manager ----------------
def test_optimization_manager_all_names():
    from mlx_cuda_distributed_pretraining_amd.core.config import TrainingConfig

    model = torch.nn.Sequential(torch.nn.Linear(4, 4), torch.nn.Linear(4, 4))
    for name in ["adamw", "adam", "sgd", "muon", "shampoo", "hybrid",
                 "adamw_enhanced", "sgd_enhanced", "lion"]:
        tc = TrainingConfig(optimization={"optimizer": name},
                            hyperparameters={"learning_rate": 1e-3})
        mgr = OptimizationManager(tc, total_steps=100)
        opt = mgr.create_optimizer(model)
        assert opt is not None
        sched = mgr.create_scheduler()
        assert sched(0) > 0


# ---------------- fused flat path (CPU reference semantics) ----------------
def test_fused_adamw_cpu_matches_torch():
    torch.manual_seed(2)
    N = 128
    master = torch.randn(N)
    param = master.clone()
    grad = torch.randn(N)
    m = torch.zeros(N)
    v = torch.zeros(N)

    p_ref = torch.nn.Parameter(master.clone())
    opt = torch.optim.AdamW([p_ref], lr=1e-2, betas=(0.9, 0.999), eps=1e-8, weight_decay=0.1)
    p_ref.grad = grad.clone()
    opt.step()

    fused_optim.adamw_step(param, master, grad, m, v, 1, 1e-2, 0.9, 0.999, 1e-8,
                           0.1, decay_boundary=N)
    assert torch.allclose(master, p_ref.detach(), atol=1e-6)


def test_fused_adamw_clip():
    N = 64
    master = torch.zeros(N)
    param = master.clone()
    grad = torch.ones(N) * 10.0
    m = torch.zeros(N)
    v = torch.zeros(N)
    sumsq = grad.pow(2).sum()
    fused_optim.adamw_step(param, master, grad, m, v, 1, 1e-2, 0.9, 0.999, 1e-8,
                           0.0, 0, sumsq=sumsq, max_grad_norm=1.0)
    # clipped grad has norm 1 -> each element 1/8 -> m = 0.1*1/8
    assert abs(m[0].item() - 0.1 / 8) < 1e-5
