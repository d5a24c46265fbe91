from . import disk_registry

__all__ = ["disk_registry"]
