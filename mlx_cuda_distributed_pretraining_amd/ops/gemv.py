"""Decode-path linear: custom bf16 GEMV for M<=8 rows (csrc/gemv.hip),
F.linear (hipBLASLt) otherwise. hipBLASLt's M=1 kernels leave ~7x HBM
bandwidth on the table for pure weight streaming; the GEMV kernel stages x
in LDS and streams W coalesced."""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from ._ext import get_ext, use_hip

GEMV_MAX_ROWS = 8


def linear_fast(x: torch.Tensor, weight: torch.Tensor,
                bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    if (
        bias is None
        and not torch.is_grad_enabled()
        and x.shape[:-1].numel() <= GEMV_MAX_ROWS
        and use_hip(x, weight, dtypes=(torch.bfloat16,))
    ):
        return get_ext().gemv_bf16(x, weight)
    return F.linear(x, weight, bias)


class FastLinear(torch.nn.Linear):
    """nn.Linear that routes tiny-M inference matmuls to the GEMV kernel and
    (when ``fp8`` is set by the model builder) big training matmuls to the
    e4m3 _scaled_mm path (ops/fp8.py)."""

    fp8: bool = False

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if (
            self.fp8
            and self.bias is None
            and torch.is_grad_enabled()
            and x.is_cuda
            and x.dtype == torch.bfloat16
            and x.shape[:-1].numel() >= 16
        ):
            from .fp8 import fp8_linear

            return fp8_linear(x, self.weight)
        return linear_fast(x, self.weight, self.bias)
