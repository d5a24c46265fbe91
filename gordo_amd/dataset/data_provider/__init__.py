from .base import GordoBaseDataProvider
from .iroc_reader import IrocFileProvider
from .ncs_reader import NcsFileProvider
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
    "NcsFileProvider",
    "IrocFileProvider",
]
