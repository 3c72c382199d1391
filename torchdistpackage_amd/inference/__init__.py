from .generate import generate

__all__ = ["generate"]
