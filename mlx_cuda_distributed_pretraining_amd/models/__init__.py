from .llama import Model, ModelArgs

__all__ = ["Model", "ModelArgs"]
