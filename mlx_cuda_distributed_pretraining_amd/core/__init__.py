from .config import Config
from .trainer import Trainer, EarlyStoppingMonitor
from .checkpoint import CheckpointManager
from .logger import Logger, format_metrics

__all__ = ["Config", "Trainer", "EarlyStoppingMonitor", "CheckpointManager", "Logger", "format_metrics"]
