"""MI355X-native op library.

The compute path on GPU is the in-tree HIP extension ``seist_amd._C``
(hand-written CDNA4/gfx950 kernels: MFMA pointwise-conv GEMM, LDS-tiled
depthwise/grouped conv, fused BatchNorm+GELU, fused avg+max aggregation,
linear interpolation, fused pooled-KV attention, multi-tensor Adam). The CPU
path is a plain-PyTorch fp32 reference of the same semantics, used for tests
and CPU runs.

On a CUDA (ROCm) device the extension is REQUIRED: if an op's kernel is
missing the op raises instead of silently falling back to eager PyTorch
(set SEIST_AMD_ALLOW_FALLBACK=1 only for debugging).
"""

import os

import torch

_C = None
_EXT_ERR = None
try:
    from .. import _C as _C  # built in-tree by setup.py build_ext --inplace
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _EXT_ERR = e


def has_ext() -> bool:
    return _C is not None


def ext():
    """Return the native extension; raise loudly if it should exist."""
    if _C is None:
        if os.environ.get("SEIST_AMD_ALLOW_FALLBACK") == "1":
            return None
        raise RuntimeError(
            "seist_amd._C (HIP/gfx950 extension) is not built but a GPU op was "
            "requested. Build it with `python setup.py build_ext --inplace` "
            f"(import error: {_EXT_ERR})"
        )
    return _C


def use_native(x: torch.Tensor) -> bool:
    """True when the HIP path must run (tensor lives on a ROCm device)."""
    if not x.is_cuda:
        return False
    if _C is None and os.environ.get("SEIST_AMD_ALLOW_FALLBACK") == "1":
        return False
    return True


from .functional import (  # noqa: E402,F401
    auto_pad,
    auto_pad_lr,
    conv1d_stats,
    pointwise_conv_stats,
    avgmax_pool1d,
    droppath_add,
    droppath_dropout_add,
    bn_act,
    bn_act_cat,
    bn_act_pw,
    act_pw,
    conv1d,
    conv_transpose1d,
    gelu,
    interp_linear,
    max_pool1d,
    global_avg_pool1d,
    pointwise_conv,
    pointwise_conv_cat,
    pooled_attention,
    upsample2x,
    fused_prob_loss,
    nearest_resize,
    LOSS_BCE,
    LOSS_CE,
)
from .eqt import (  # noqa: E402,F401
    additive_attention_weights,
    layer_norm,
    lstm,
)
from .adam import FusedAdam  # noqa: E402,F401
