from .tokenizer import TokenizerManager
from .dataset import DataManager

__all__ = ["TokenizerManager", "DataManager"]
