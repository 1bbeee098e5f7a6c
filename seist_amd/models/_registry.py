"""Model registry + ``.pth`` checkpoint I/O.

Parity surface with /root/reference/models/_factory.py: decorator-based name
registry (:17-56), checkpoint dict format {epoch, optimizer_dict, model_dict,
loss, use_compile, use_ddp} with wrapper unwrapping on save (:59-87) and
``module.`` / ``_orig_mod.`` key stripping on load (:90-126) so the
reference's 18 pretrained ``.pth`` files round-trip.
"""

import os
from typing import Callable, Dict, List, Optional

import torch
import torch.nn as nn

from ..utils.logger import logger

_MODEL_REGISTRY: Dict[str, Callable[..., nn.Module]] = {}


def register_model(fn: Callable[..., nn.Module]) -> Callable[..., nn.Module]:
    name = fn.__name__
    if name in _MODEL_REGISTRY:
        raise ValueError(f"Model '{name}' is already registered")
    _MODEL_REGISTRY[name] = fn
    return fn


def get_model_list() -> List[str]:
    return sorted(_MODEL_REGISTRY)


def create_model(model_name: str, **kwargs) -> nn.Module:
    if model_name not in _MODEL_REGISTRY:
        raise NotImplementedError(
            f"Unknown model: '{model_name}'. Registered: {get_model_list()}"
        )
    return _MODEL_REGISTRY[model_name](**kwargs)


def _unwrap(model: nn.Module) -> nn.Module:
    # DDP wrapper
    if hasattr(model, "module") and isinstance(model.module, nn.Module):
        model = model.module
    # torch.compile wrapper
    if hasattr(model, "_orig_mod"):
        model = model._orig_mod
    return model


def save_checkpoint(
    save_path: str,
    model: nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    epoch: int = 0,
    loss: float = 0.0,
    use_ddp: bool = False,
    use_compile: bool = False,
) -> None:
    """Checkpoints are always written with fp32 weights (the
    reference .pth format): under the bf16 training policy the fused
    optimizer's fp32 MASTER copies are saved in place of the bf16
    parameters (full-precision trajectory, not a bf16 round-trip), and
    any remaining non-fp32 floating tensors are upcast."""
    os.makedirs(os.path.dirname(save_path) or ".", exist_ok=True)
    mdl = _unwrap(model)
    sd = mdl.state_dict()
    masters = {}
    if optimizer is not None:
        for prm, st in getattr(optimizer, "state", {}).items():
            m = st.get("master") if isinstance(st, dict) else None
            if isinstance(m, torch.Tensor):
                masters[prm] = m
    needs_fix = masters or any(
        torch.is_tensor(v) and v.is_floating_point()
        and v.dtype != torch.float32 for v in sd.values())
    if needs_fix:
        pmap = dict(mdl.named_parameters())
        out = {}
        for k, v in sd.items():
            prm = pmap.get(k)
            if prm is not None and prm in masters:
                out[k] = masters[prm].detach().clone()
            elif (torch.is_tensor(v) and v.is_floating_point()
                  and v.dtype != torch.float32):
                out[k] = v.float()
            else:
                out[k] = v
        sd = out
    ckpt = {
        "epoch": epoch,
        "optimizer_dict": optimizer.state_dict() if optimizer is not None else None,
        "model_dict": sd,
        "loss": loss,
        "use_compile": use_compile,
        "use_ddp": use_ddp,
    }
    torch.save(ckpt, save_path)


def load_checkpoint(ckpt_path: str, device: torch.device = "cpu") -> dict:
    """Load a checkpoint; tolerates bare state-dict files and strips
    ``module.`` / ``_orig_mod.`` prefixes left by DDP / torch.compile."""
    ckpt = torch.load(ckpt_path, map_location=device, weights_only=False)
    if not isinstance(ckpt, dict) or "model_dict" not in ckpt:
        ckpt = {"model_dict": ckpt}

    cleaned = {}
    for k, v in ckpt["model_dict"].items():
        for prefix in ("module.", "_orig_mod."):
            while k.startswith(prefix):
                k = k[len(prefix):]
        cleaned[k] = v
    ckpt["model_dict"] = cleaned

    for flag in ("use_ddp", "use_compile"):
        if flag in ckpt and ckpt[flag]:
            logger.warning(
                f"Checkpoint '{os.path.basename(ckpt_path)}' was saved with {flag}=True"
            )
    return ckpt
