from .generate import beam_search, generate, generate_step
from .kv_cache import (
    ChunkedKVCache, QuantizedKVCache, make_cache, maybe_quantize_kv_cache,
)
from .static_decode import GraphDecoder, StaticDecodeState
from .agent import AgentConfig, GenerationAgent

__all__ = [
    "generate", "generate_step", "beam_search",
    "ChunkedKVCache", "QuantizedKVCache", "make_cache", "maybe_quantize_kv_cache",
    "GraphDecoder", "StaticDecodeState",
    "AgentConfig", "GenerationAgent",
]
