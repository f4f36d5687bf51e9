from .generate import beam_search, generate, generate_step

__all__ = ["generate", "generate_step", "beam_search"]
