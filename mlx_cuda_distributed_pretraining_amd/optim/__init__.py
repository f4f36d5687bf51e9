from .schedules import build_schedule, cosine_decay, join_schedules, linear_schedule
from .enhanced import AdamWEnhanced, LionEnhanced, SGDEnhanced, clip_by_global_norm
from .muon import Muon, zeropower_via_newtonschulz5
from .shampoo import Shampoo, ShampooParams, matrix_inverse_pth_root
from .hybrid import HybridOptimizer
from .manager import OptimizationManager

__all__ = [
    "build_schedule", "cosine_decay", "join_schedules", "linear_schedule",
    "AdamWEnhanced", "LionEnhanced", "SGDEnhanced", "clip_by_global_norm",
    "Muon", "zeropower_via_newtonschulz5",
    "Shampoo", "ShampooParams", "matrix_inverse_pth_root",
    "HybridOptimizer", "OptimizationManager",
]
