"""Flat parameter space.

All trainable parameters become views into ONE contiguous buffer, ordered
[decay params | no-decay params]; gradients are views into one flat grad
buffer of the same layout (autograd accumulates in place into pre-set
``.grad`` views, so backward writes land directly in the flat buffer — no
copy pass). This enables:

  - single fused optimizer kernels over the whole model (ops/fused_optim),
  - contiguous gradient buckets for overlapped RCCL all-reduce (ddp.py),
  - ZeRO-1 by rank-slicing the same flat space (zero.py).

Sized for MI355X: 288 GB HBM3E means even a 1B-param model's flat fp32
master + moments (12 GB) is trivial; bigger shards, fewer and larger
collectives.
"""
from __future__ import annotations

from typing import Dict, List, Tuple

import torch

from ..optim.enhanced import _is_no_decay


def _align(n: int, a: int = 64) -> int:
    return (n + a - 1) // a * a


class FlatParamSpace:
    """grad_mode:
      - "views" (default): p.grad IS a view into flat_grad; autograd
        accumulates in place (an extra read-modify-write add per param per
        backward, plus a flat zero_ per step — but works with ANY optimizer
        that reads p.grad).
      - "copy": p.grad stays None; autograd ASSIGNS the computed grad (no
        add), and a post-accumulate hook copies it into the flat buffer then
        frees it. Saves ~2 memory passes per param per step. Only valid when
        the consumer reads flat_grad (FusedFlatAdamW) and every parameter
        receives a grad each step (untouched segments would go stale —
        zero_grad() does not fill in this mode).
    """

    def __init__(self, model: torch.nn.Module, grad_dtype: torch.dtype | None = None,
                 grad_mode: str = "views"):
        named = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
        # stable order: decay params first, then no-decay
        decay = [(n, p) for n, p in named if not _is_no_decay(n, p)]
        nodecay = [(n, p) for n, p in named if _is_no_decay(n, p)]
        ordered = decay + nodecay

        device = ordered[0][1].device
        dtype = ordered[0][1].dtype
        self.dtype = dtype
        self.grad_dtype = grad_dtype or dtype
        self.device = device

        self.segments: List[Tuple[str, int, int, torch.Size]] = []
        offset = 0
        for n, p in ordered:
            numel = p.numel()
            self.segments.append((n, offset, numel, p.shape))
            offset = _align(offset + numel)
        self.total = offset
        self.decay_numel = 0
        if decay:
            last_name, last_off, last_n, _ = self.segments[len(decay) - 1]
            self.decay_numel = _align(last_off + last_n)

        self.flat_param = torch.zeros(self.total, dtype=dtype, device=device)
        self.flat_grad = torch.zeros(self.total, dtype=self.grad_dtype, device=device)

        assert grad_mode in ("views", "copy")
        self.grad_mode = grad_mode
        self._seen: set = set()
        self.params: List[torch.nn.Parameter] = []
        self.name_to_param: Dict[str, torch.nn.Parameter] = {}
        self._offsets: Dict[int, Tuple[int, int]] = {}
        for (n, p), (_, off, numel, shape) in zip(ordered, self.segments):
            with torch.no_grad():
                self.flat_param[off : off + numel].copy_(p.detach().reshape(-1).to(dtype))
            p.data = self.flat_param[off : off + numel].view(shape)
            self._offsets[id(p)] = (off, numel)
            if grad_mode == "views":
                if self.grad_dtype == dtype:
                    p.grad = self.flat_grad[off : off + numel].view(shape)
            else:
                p.register_post_accumulate_grad_hook(self._copy_hook)
            self.params.append(p)
            self.name_to_param[n] = p

    def _copy_hook(self, p: torch.nn.Parameter) -> None:
        off, numel = self._offsets[id(p)]
        view = self.flat_grad[off : off + numel]
        g = p.grad.reshape(-1)
        if id(p) in self._seen:  # gradient accumulation micro-step > 0
            view.add_(g.to(self.grad_dtype))
        else:
            view.copy_(g.to(self.grad_dtype))
            self._seen.add(id(p))
        p.grad = None  # free; the flat buffer is the source of truth

    def zero_grad(self) -> None:
        if self.grad_mode == "copy":
            self._seen.clear()  # no GPU fill: every segment is overwritten
        else:
            self.flat_grad.zero_()

    def grad_view(self, start: int, end: int) -> torch.Tensor:
        return self.flat_grad[start:end]

    def param_segments_in(self, start: int, end: int):
        for n, off, numel, shape in self.segments:
            if off >= start and off + numel <= end:
                yield n, off, numel, shape
