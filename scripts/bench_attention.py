#!/usr/bin/env python3
"""Attention kernel micro-benchmark (within-process A/B, guide §5.4 rule 24).

Reports achieved TF/s for fwd and bwd at training shapes; used to drive the
attention optimization ladder with rocprof evidence.
"""
import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

from mlx_cuda_distributed_pretraining_amd.ops import flash_attention  # noqa: E402


def attn_flops(B, S, Hq, D, causal):
    # fwd: QK^T + PV = 4 * S^2 * D per head (x0.5 causal)
    f = 4.0 * S * S * D * Hq * B
    return f / 2 if causal else f


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=20)
    args = p.parse_args()
    dev = torch.device("cuda:0")
    shapes = [
        # (B, S, Hq, Hkv, D, label)
        (16, 2048, 16, 8, 128, "1B-shape"),
        (32, 1024, 12, 12, 64, "124M-shape"),
        (16, 2048, 16, 16, 128, "MHA-2k"),
    ]
    for B, S, Hq, Hkv, D, label in shapes:
        torch.manual_seed(0)
        q = torch.randn(B, S, Hq, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)
        v = torch.randn(B, S, Hkv, D, device=dev, dtype=torch.bfloat16, requires_grad=True)

        t_fwd = bench(lambda: flash_attention(q, k, v, causal=True), args.iters)
        fl = attn_flops(B, S, Hq, D, True)
        print(f"[{label}] fwd : {t_fwd*1e3:8.3f} ms  {fl/t_fwd/1e12:7.1f} TF/s")

        o = flash_attention(q, k, v, causal=True)
        dy = torch.randn_like(o)

        def fwdbwd():
            q.grad = k.grad = v.grad = None
            o = flash_attention(q, k, v, causal=True)
            o.backward(dy)

        t_fb = bench(fwdbwd, args.iters)
        t_bwd = t_fb - t_fwd
        # bwd ideal flops = 2.5x fwd (dQ, dK, dV, recomputed S, dP)
        print(f"[{label}] bwd : {t_bwd*1e3:8.3f} ms  {2.5*fl/t_bwd/1e12:7.1f} TF/s (ideal-flops basis)")


if __name__ == "__main__":
    main()
