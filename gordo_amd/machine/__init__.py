from .sensor_tag import (
    SensorTag,
    SensorTagNormalizationError,
    normalize_sensor_tag,
    normalize_sensor_tags,
    to_list_of_strings,
)
from .machine import Machine, NormalizedConfig

__all__ = [
    "SensorTag",
    "SensorTagNormalizationError",
    "normalize_sensor_tag",
    "normalize_sensor_tags",
    "to_list_of_strings",
    "Machine",
    "NormalizedConfig",
]
