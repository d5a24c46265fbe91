from .base import GordoBaseDataProvider
from .providers import (
    CsvFileProvider,
    InfluxDataProvider,
    ParquetFileProvider,
    RandomDataProvider,
)

__all__ = [
    "GordoBaseDataProvider",
    "RandomDataProvider",
    "InfluxDataProvider",
    "ParquetFileProvider",
    "CsvFileProvider",
]
