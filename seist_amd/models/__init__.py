from ._registry import (
    create_model,
    get_model_list,
    load_checkpoint,
    register_model,
    save_checkpoint,
)
from .losses import (
    BCELoss,
    BinaryFocalLoss,
    CELoss,
    CombinationLoss,
    FocalLoss,
    HuberLoss,
    MousaviLoss,
    MSELoss,
)

# model registrations (import side effect)
from . import seist  # noqa: F401
from . import phasenet  # noqa: F401
from . import eqtransformer  # noqa: F401
from . import magnet  # noqa: F401
from . import ditingmotion  # noqa: F401
from . import baz_network  # noqa: F401
from . import distpt_network  # noqa: F401

__all__ = [
    "create_model", "get_model_list", "register_model",
    "save_checkpoint", "load_checkpoint",
    "CELoss", "BCELoss", "FocalLoss", "BinaryFocalLoss", "MSELoss",
    "CombinationLoss", "MousaviLoss", "HuberLoss",
]
