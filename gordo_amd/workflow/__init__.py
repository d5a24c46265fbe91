from ..machine import Machine, NormalizedConfig
from .fleet import FleetBuilder
from .generator import generate_plan, generate_yaml, unique_tags

__all__ = [
    "Machine",
    "NormalizedConfig",
    "FleetBuilder",
    "generate_plan",
    "generate_yaml",
    "unique_tags",
]
