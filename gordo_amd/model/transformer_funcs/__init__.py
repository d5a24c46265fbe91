from .general import InfImputer, multiply

__all__ = ["multiply", "InfImputer"]
