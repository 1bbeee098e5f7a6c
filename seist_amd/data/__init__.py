from .registry import build_dataset, get_dataset_list, register_dataset
from .base import DatasetBase
from .preprocess import DataPreprocessor, SeismicDataset

# reader registrations (import side effect)
from . import diting  # noqa: F401
from . import pnw  # noqa: F401
from . import sos  # noqa: F401
from . import synthetic  # noqa: F401

__all__ = [
    "build_dataset", "get_dataset_list", "register_dataset",
    "DatasetBase", "DataPreprocessor", "SeismicDataset",
]
