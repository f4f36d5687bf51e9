#!/usr/bin/env python3
"""Flagship training benchmark: Llama-1B bf16 pretraining tokens/sec.

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
For N > 1 the driver launches via torch.distributed.run with one rank per
GPU over RCCL; ranks read RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* from env.
W untimed warmup steps, then EXACTLY K timed steps bracketed by
barrier + torch.cuda.synchronize on both sides; elapsed = MAX over ranks;
rank 0 prints ONE JSON line. Synthetic data (random tokens of the config's
shape), random-init weights, full optimizer step inside the timed region.
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time
from pathlib import Path

import torch

REPO = Path(__file__).resolve().parent
sys.path.insert(0, str(REPO))

from mlx_cuda_distributed_pretraining_amd.core.config import Config  # noqa: E402
from mlx_cuda_distributed_pretraining_amd.core.trainer import Trainer  # noqa: E402
from mlx_cuda_distributed_pretraining_amd.parallel.dist import (  # noqa: E402
    barrier, get_rank, get_world_size, is_distributed,
)

BASELINE_TOKS_PER_SEC = 25000.0  # reference: 1.3B @ ~25K tok/s on 2xA100-40GB


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--config", type=str, default=None)
    p.add_argument("--batch-size", type=int, default=None)
    p.add_argument("--seq-len", type=int, default=None)
    args = p.parse_args()

    on_gpu = torch.cuda.is_available()
    cfg_path = args.config or str(
        REPO / "configs" / ("model-config-1b.yaml" if on_gpu else "model-config-sample.yaml")
    )
    cfg = Config.from_yaml(cfg_path)
    cfg.overwrite = True
    cfg.name = f"bench-{cfg.name}-{os.environ.get('RANK', '0')}-{int(time.time())}"
    cfg.logging.steps = {"logging_interval": 0, "checkpoint_interval": 0, "validation_interval": 0}
    cfg.data.synthetic = True
    if args.batch_size:
        cfg.training.hyperparameters["batch_size"] = args.batch_size
    if args.seq_len:
        cfg.data.preprocessing["max_context_size"] = args.seq_len
        # long-context runs: don't let the trainer truncate to the config's
        # max_position_embeddings (the tiled attention is O(S) in memory)
        attn = cfg.model.attention or {}
        if attn.get("max_position_embeddings") and attn["max_position_embeddings"] < args.seq_len:
            attn["max_position_embeddings"] = args.seq_len
    cfg.training.hyperparameters["iters"] = args.steps + args.warmup + 1

    import tempfile

    trainer = Trainer(cfg, runs_root=tempfile.mkdtemp(prefix="bench_runs_"))
    world = get_world_size()
    rank = get_rank()
    if world != args.gpus and rank == 0:
        print(f"[bench] note: world_size={world} != --gpus {args.gpus}", file=sys.stderr)

    bsz = trainer.batch_size
    seq = cfg.data.preprocessing["max_context_size"]
    accum = trainer.grad_accum_steps
    tokens_per_step = bsz * seq * accum * world  # whole-job tokens per optimizer step

    for i in range(args.warmup):
        trainer.train_step(i)

    barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    last_loss = None
    for i in range(args.warmup, args.warmup + args.steps):
        loss, _ = trainer.train_step(i)
        last_loss = loss
    if on_gpu:
        torch.cuda.synchronize()
    barrier()
    t1 = time.perf_counter()

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if is_distributed():
        import torch.distributed as dist

        elapsed_dev = elapsed.to("cuda") if on_gpu else elapsed
        dist.all_reduce(elapsed_dev, op=dist.ReduceOp.MAX)
        elapsed = elapsed_dev.cpu()
    elapsed_s = float(elapsed.item())

    value = tokens_per_step * args.steps / elapsed_s
    ms_per_step = 1000.0 * elapsed_s / args.steps
    model_name = "Llama-1B" if "1b" in Path(cfg_path).stem else Path(cfg_path).stem

    if rank == 0:
        result = {
            "metric": "pretraining tokens/sec (whole node)",
            "value": value,
            "unit": "tokens/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / BASELINE_TOKS_PER_SEC,
            "dtype": (("bf16+fp8-gemm" if (cfg.model.misc or {}).get("fp8") else "bf16")
                      if on_gpu else "fp32"),
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": bsz * accum * world,
                "seq_len": seq,
                "parallelism": f"dp{world}",
                "final_loss": float(last_loss) if last_loss is not None else None,
            },
        }
        print(json.dumps(result))


if __name__ == "__main__":
    main()
