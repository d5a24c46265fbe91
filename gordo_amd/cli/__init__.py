from .cli import gordo

__all__ = ["gordo"]
