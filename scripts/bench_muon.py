#!/usr/bin/env python3
"""Muon NS-5 chain: HIP MFMA kernels (csrc/muon.hip) vs torch/hipBLASLt."""
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from mlx_cuda_distributed_pretraining_amd.optim.muon import NS_COEFFS, _ns5_hip  # noqa: E402


def torch_chain(G, steps=5, eps=1e-7):
    a, b, c = NS_COEFFS
    X = (G / (G.norm() + eps)).to(torch.bfloat16)
    for _ in range(steps):
        A = X @ X.t()
        B = b * A + c * (A @ A)
        X = a * X + B @ X
    return X.float()


def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = "cuda:0"
    for m, n in [(1024, 2816), (2048, 5632), (2048, 2048)]:
        torch.manual_seed(0)
        G = torch.randn(m, n, device=dev)
        t_hip = bench(lambda: _ns5_hip(G, 5, 1e-7))
        t_torch = bench(lambda: torch_chain(G))
        # FLOPs of the full-grid chain actually launched (2*m^2*n for each of
        # XX^T and BX, 2*m^3 for A@A, x5 iters, x2 FLOP/MAC)
        fl = 5 * (2 * m * m * n * 2 + 2 * m**3)
        print(f"[{m}x{n}] hip: {t_hip*1e3:7.2f} ms ({fl/t_hip/1e12:6.1f} TF/s)   "
              f"torch/hipBLASLt: {t_torch*1e3:7.2f} ms ({fl/t_torch/1e12:6.1f} TF/s)   "
              f"speedup {t_torch/t_hip:.2f}x")


if __name__ == "__main__":
    main()
