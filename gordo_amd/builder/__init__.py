from .build_model import ModelBuilder, build_model

__all__ = ["ModelBuilder", "build_model"]
