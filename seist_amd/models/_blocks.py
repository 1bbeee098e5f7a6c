"""Shared building helpers: run nn.Conv1d / nn.BatchNorm1d parameter
containers through the MI355X op layer."""

import os

import torch.nn as nn

from .. import ops

# Producer-side BN-stats fusion (FUSION_PLAN step 1) measured NET NEGATIVE
# on seist_m_dpk (35.7 -> 36.6 ms/step, same box A/B): the bn_sums pass it
# removes is L2-resident and cheap (~9 us/site) while the conv-epilogue
# reduction (+3.8 us/site) and the big-slab finalize (+7 us/site) cost
# more. Opt-in for experiments; see profiles/step_profile_r02.md.
_STATS_FUSION = os.environ.get("SEIST_AMD_STATS_FUSION") == "1"


def run_conv(conv: nn.Conv1d, x, padl: int = 0, padr: int = 0):
    """Apply an nn.Conv1d's parameters via the native conv path with
    explicit pre-padding (module padding is added on top)."""
    p = conv.padding[0] if isinstance(conv.padding, tuple) else int(conv.padding)
    return ops.conv1d(
        x, conv.weight, conv.bias,
        stride=conv.stride[0],
        padding=(padl + p, padr + p),
        groups=conv.groups,
        dilation=conv.dilation[0],
    )


def run_bn(bn, x, act: str = "none", part=None):
    """Fused BatchNorm1d (+act); falls through for Identity/other norms.

    A BN module tagged ``_sync_bn`` (see parallel.ddp.enable_native_syncbn)
    reduces its batch statistics across ranks inside the same fused path.
    ``part`` is an optional producer-collected (C, nsplit, 2) partial-sums
    slab (conv->BN fusion step 1).
    """
    if isinstance(bn, nn.BatchNorm1d):
        y = ops.bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                       bn.training, bn.momentum, bn.eps, act=act,
                       sync=getattr(bn, "_sync_bn", False), part=part)
        if bn.training and bn.track_running_stats \
                and not getattr(bn, "_managed_nbt", False):
            bn.num_batches_tracked += 1
        return y
    y = bn(x)
    if act == "relu":
        y = y.relu()
    elif act == "gelu":
        y = ops.gelu(y)
    return y


def manage_bn_counters(model: nn.Module):
    """Switch a model's BatchNorm ``num_batches_tracked`` updates from one
    tiny kernel per layer per forward (115 x ~4.5 us/step on seist_m) to a
    single ``torch._foreach_add_`` per step via the returned ``tick()``.
    The counter only matters for checkpoint interop (our bn_act uses a
    fixed momentum), so per-step batching is exact."""
    import torch
    nbts = []
    for m in model.modules():
        if isinstance(m, nn.BatchNorm1d) and m.track_running_stats:
            m._managed_nbt = True
            nbts.append(m.num_batches_tracked)

    def tick():
        if nbts:
            torch._foreach_add_(nbts, 1)
    return tick


def run_conv_bn(conv: nn.Conv1d, bn, x, act: str = "none", padl: int = 0,
                padr: int = 0, auto_pad: bool = False):
    """conv -> BatchNorm1d(+act) chain with the BN statistics accumulated
    in the conv kernel's epilogue (docs/FUSION_PLAN.md step 1): the
    standalone bn_sums pass over the conv output disappears."""
    k = conv.kernel_size[0]
    s = conv.stride[0]
    if auto_pad:
        pl, pr = ops.auto_pad_lr(x.size(-1), k, s)
    else:
        p = conv.padding[0] if isinstance(conv.padding, tuple) \
            else int(conv.padding)
        pl, pr = padl + p, padr + p
    collect = (isinstance(bn, nn.BatchNorm1d) and bn.training and x.is_cuda
               and _STATS_FUSION)
    if collect:
        y, part = ops.conv1d_stats(x, conv.weight, conv.bias, stride=s,
                                   padding=(pl, pr), groups=conv.groups,
                                   dilation=conv.dilation[0])
    else:
        y = ops.conv1d(x, conv.weight, conv.bias, stride=s,
                       padding=(pl, pr), groups=conv.groups,
                       dilation=conv.dilation[0])
        part = None
    return run_bn(bn, y, act=act, part=part)
