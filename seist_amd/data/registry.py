"""Dataset registry (decorator pattern, parity with
/root/reference/datasets/_factory.py:12-50)."""

from typing import Callable, Dict, List

_DATASET_REGISTRY: Dict[str, Callable] = {}


def register_dataset(fn: Callable) -> Callable:
    name = fn.__name__
    if name in _DATASET_REGISTRY:
        raise Exception(f"Dataset '{name}' already exists.")
    _DATASET_REGISTRY[name] = fn
    return fn


def get_dataset_list() -> List[str]:
    return sorted(_DATASET_REGISTRY)


def build_dataset(dataset_name: str, **kwargs):
    if dataset_name not in _DATASET_REGISTRY:
        raise ValueError(
            f"Dataset '{dataset_name}' does not exist. "
            f"Registered: {get_dataset_list()}")
    return _DATASET_REGISTRY[dataset_name](**kwargs)
