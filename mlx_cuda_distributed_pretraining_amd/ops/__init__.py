from .rmsnorm import RMSNorm, rms_norm, rms_norm_ref
from .rope import RopeTable, apply_rope, rope_ref
from .swiglu import swiglu, swiglu_ref
from .cross_entropy import fused_cross_entropy, cross_entropy_ref
from .attention import (
    BlockMask,
    attention_ref,
    create_block_mask,
    flash_attention,
    flex_attention,
)
from .sampling import make_sampler, make_logits_processors, sample_token
from .gemv import FastLinear, linear_fast
from .fp8 import fp8_linear, quantize_e4m3
from . import fused_optim
from ._ext import get_ext, require_ext

__all__ = [
    "RMSNorm", "rms_norm", "rms_norm_ref",
    "RopeTable", "apply_rope", "rope_ref",
    "swiglu", "swiglu_ref",
    "fused_cross_entropy", "cross_entropy_ref",
    "BlockMask", "attention_ref", "create_block_mask", "flash_attention", "flex_attention",
    "make_sampler", "make_logits_processors", "sample_token",
    "FastLinear", "linear_fast", "fp8_linear", "quantize_e4m3",
    "fused_optim", "get_ext", "require_ext",
]
