from .tokenizer import TokenizerManager, train_bpe_tokenizer
from .dataset import DataManager
from .streaming import DiskSpaceManager, StreamingTokenDataset

__all__ = ["TokenizerManager", "train_bpe_tokenizer", "DataManager",
           "DiskSpaceManager", "StreamingTokenDataset"]
