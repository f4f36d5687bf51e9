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
    """nn.Linear that routes tiny-M inference matmuls to the GEMV kernel."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return linear_fast(x, self.weight, self.bias)
