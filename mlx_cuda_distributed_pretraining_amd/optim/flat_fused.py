"""Fused flat-space AdamW — the GPU hot path, with optional ZeRO-1.

One HIP kernel call per step over the whole model (ops/fused_optim /
csrc/optim.hip): bf16 params + fp32 master + fp32 moments as flat buffers,
global-norm clip folded in (no host sync). With ``zero1=True`` the optimizer
state is sharded across ranks (collective C4):
reduce-scatter grads -> update own shard -> all-gather params, all over RCCL.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from ..ops import fused_optim
from ..parallel.dist import get_rank, get_world_size, is_distributed
from ..parallel.flat import FlatParamSpace


class FusedFlatAdamW:
    def __init__(
        self,
        space: FlatParamSpace,
        lr: float = 3e-4,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.01,
        max_grad_norm: float = 0.0,
        zero1: bool = False,
    ):
        self.space = space
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.max_grad_norm = max_grad_norm
        self.step_count = 0
        self.zero1 = zero1 and is_distributed()

        N = space.total
        dev = space.device
        if self.zero1:
            ws = get_world_size()
            assert N % 64 == 0
            self.shard_size = (N + ws - 1) // ws
            # pad so shard_size * ws == padded N
            self.padded = self.shard_size * ws
            if self.padded != N:
                # grow flat buffers? FlatParamSpace aligns to 64; pad shard math instead
                self.shard_size = self.padded // ws
            self.shard_start = get_rank() * self.shard_size
            self.shard_end = min(self.shard_start + self.shard_size, N)
            n_state = max(self.shard_end - self.shard_start, 0)
            self._grad_shard = torch.zeros(self.shard_size, dtype=space.grad_dtype, device=dev)
        else:
            n_state = N
            self.shard_start, self.shard_end = 0, N
        self.master = space.flat_param[self.shard_start : self.shard_end].float().clone()
        self.exp_avg = torch.zeros(n_state, dtype=torch.float32, device=dev)
        self.exp_avg_sq = torch.zeros(n_state, dtype=torch.float32, device=dev)

    def zero_grad(self) -> None:
        self.space.zero_grad()

    @torch.no_grad()
    def step(self, lr: Optional[float] = None) -> None:
        if lr is not None:
            self.lr = lr
        self.step_count += 1
        sp = self.space
        if self.zero1:
            ws = get_world_size()
            flat = sp.flat_grad
            if self.padded != sp.total:
                flat = torch.cat(
                    [flat, torch.zeros(self.padded - sp.total, dtype=flat.dtype, device=flat.device)]
                )
            dist.reduce_scatter_tensor(self._grad_shard, flat, op=dist.ReduceOp.SUM)
            self._grad_shard.div_(ws)
            grad = self._grad_shard[: self.shard_end - self.shard_start]
            sumsq = fused_optim.grad_sumsq(grad)
            if is_distributed():
                dist.all_reduce(sumsq, op=dist.ReduceOp.SUM)
            param_view = sp.flat_param[self.shard_start : self.shard_end]
            decay_boundary = min(max(sp.decay_numel - self.shard_start, 0),
                                 self.shard_end - self.shard_start)
        else:
            grad = sp.flat_grad
            sumsq = fused_optim.grad_sumsq(grad) if self.max_grad_norm > 0 else None
            param_view = sp.flat_param
            decay_boundary = sp.decay_numel

        fused_optim.adamw_step(
            param_view, self.master, grad, self.exp_avg, self.exp_avg_sq,
            self.step_count, self.lr, self.beta1, self.beta2, self.eps,
            self.weight_decay, decay_boundary,
            sumsq=sumsq, max_grad_norm=self.max_grad_norm,
        )
        if self.zero1:
            # all-gather updated param shards (bf16) into the flat param buffer
            shard_padded = torch.zeros(self.shard_size, dtype=sp.dtype, device=sp.device)
            shard_padded[: self.shard_end - self.shard_start] = param_view
            out = torch.empty(self.padded, dtype=sp.dtype, device=sp.device)
            dist.all_gather_into_tensor(out, shard_padded)
            sp.flat_param.copy_(out[: sp.total])

    # -- checkpoint compatibility ----------------------------------------
    def state_dict(self) -> dict:
        return {
            "step": self.step_count,
            "master": self.master,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
            "shard_start": self.shard_start,
            "shard_end": self.shard_end,
        }

    def load_state_dict(self, sd: dict) -> None:
        if sd["master"].numel() != self.master.numel():
            raise ValueError(
                f"optimizer state shard size mismatch: checkpoint has "
                f"{sd['master'].numel()} master elements, this run expects "
                f"{self.master.numel()} — ZeRO-1 checkpoints resume at the "
                f"same world size (re-sharding is a ROADMAP item)"
            )
        self.step_count = int(sd["step"])
        self.master.copy_(sd["master"].to(self.master.device))
        self.exp_avg.copy_(sd["exp_avg"].to(self.exp_avg.device))
        self.exp_avg_sq.copy_(sd["exp_avg_sq"].to(self.exp_avg_sq.device))
