"""Functional op layer: one API, two implementations.

* GPU (ROCm/MI355X): hand-written HIP kernels from ``seist_amd._C``;
  dispatch is hard — a CUDA tensor hitting an op whose kernel exists must
  run the kernel (no silent eager fallback).
* CPU: plain PyTorch fp32 reference with identical semantics, used by the
  numerics tests as ground truth.

Layout convention is channels-first ``(N, C, L)`` throughout, matching the
reference design (models/seist.py:106-107 uses 1x1 Conv1d instead of Linear
to avoid transposes — here the 1x1 conv IS a GEMM on MFMA).
"""

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import has_ext, ext, use_native

_SQRT_2PI = math.sqrt(2.0 * math.pi)

_ACT_NONE = 0
_ACT_GELU = 1
_ACT_RELU = 2

def _trace_eager(x: torch.Tensor) -> bool:
    """True while torch.jit is tracing CPU tensors: run the plain eager
    composite instead of the autograd.Function wrapper (the tracer cannot
    trace custom Functions whose inputs are nn.Parameters). Keeps the
    models jit-traceable like the reference (its explicit-padding design
    exists for torch.jit/onnx export, reference models/seist.py:24)."""
    return x.device.type == "cpu" and torch.jit.is_tracing()


# ---------------------------------------------------------------------------
# padding semantics (reference models/seist.py:12-48 `_auto_pad_1d`)
# ---------------------------------------------------------------------------


def auto_pad_lr(length: int, kernel_size: int, stride: int = 1) -> Tuple[int, int]:
    """Left/right padding so conv output length is ceil(length / stride)."""
    assert kernel_size >= stride, (
        f"`kernel_size` must be >= `stride`, got {kernel_size}, {stride}"
    )
    pds = (stride - (length % stride)) % stride + kernel_size - stride
    return pds // 2, pds - pds // 2


def auto_pad(x: torch.Tensor, kernel_size: int, stride: int = 1,
             value: float = 0.0) -> torch.Tensor:
    """'same-for-strided' padding of the last dim (out = ceil(L/stride))."""
    pl, pr = auto_pad_lr(x.size(-1), kernel_size, stride)
    if pl == 0 and pr == 0:
        return x
    return F.pad(x, (pl, pr), "constant", value)


# ---------------------------------------------------------------------------
# gelu (erf form — PyTorch nn.GELU default)
# ---------------------------------------------------------------------------


def gelu(x: torch.Tensor) -> torch.Tensor:
    return F.gelu(x)


def _gelu_grad(x: torch.Tensor) -> torch.Tensor:
    cdf = 0.5 * (1.0 + torch.erf(x * (1.0 / math.sqrt(2.0))))
    pdf = torch.exp(-0.5 * x * x) / _SQRT_2PI
    return cdf + x * pdf


# ---------------------------------------------------------------------------
# pointwise (1x1) conv == batched GEMM  (MFMA kernel on GPU)
# ---------------------------------------------------------------------------


class _PointwiseConv(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        # x: (N, Ci, L), weight: (Co, Ci), bias: (Co,) or None
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        if use_native(x):
            return ext().pw_conv_fwd(x, weight, bias)
        y = torch.einsum("oc,ncl->nol", weight, x)
        if bias is not None:
            y = y + bias[:, None]
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        if use_native(x):
            dx, dw, db = ext().pw_conv_bwd(dy, x, weight, ctx.has_bias)
        else:
            dx = torch.einsum("oc,nol->ncl", weight, dy)
            dw = torch.einsum("nol,ncl->oc", dy, x)
            db = dy.sum(dim=(0, 2)) if ctx.has_bias else None
        return dx, dw, db


def pointwise_conv(x: torch.Tensor, weight: torch.Tensor,
                   bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """1x1 Conv1d: (N,Ci,L) x (Co,Ci[,1]) -> (N,Co,L)."""
    if weight.dim() == 3:
        weight = weight.squeeze(-1)
    if x.dtype != weight.dtype:
        x = x.to(weight.dtype)
    if _trace_eager(x):
        return F.conv1d(x, weight.unsqueeze(-1), bias)
    return _PointwiseConv.apply(x.contiguous(), weight.contiguous(), bias)


class _PointwiseConvStats(torch.autograd.Function):
    """Forward also returns the (Co, nsplit, 2) BN partial-sums slab,
    reduced in the conv epilogue (fusion step 1). The slab is
    non-differentiable auxiliary output."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        y, part = ext().pw_conv_fwd_stats(x, weight, bias)
        ctx.mark_non_differentiable(part)
        return y, part

    @staticmethod
    def backward(ctx, dy, dpart):
        x, weight = ctx.saved_tensors
        dx, dw, db = ext().pw_conv_bwd(dy.contiguous(), x, weight,
                                       ctx.has_bias)
        return dx, dw, db


def pointwise_conv_stats(x, weight, bias=None):
    """pointwise_conv + BN stats partials; (y, None) off-GPU."""
    if weight.dim() == 3:
        weight = weight.squeeze(-1)
    if x.dtype != weight.dtype:
        x = x.to(weight.dtype)
    if not use_native(x):
        return pointwise_conv(x, weight, bias), None
    return _PointwiseConvStats.apply(x.contiguous(), weight.contiguous(),
                                     bias)


class _Conv1dStats(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padl, padr, groups, dilation):
        ctx.save_for_backward(x, weight)
        ctx.conf = (stride, padl, padr, groups, dilation, bias is not None)
        y, part = ext().conv1d_fwd_stats(x, weight, bias, stride, padl,
                                         padr, groups, dilation)
        ctx.mark_non_differentiable(part)
        return y, part

    @staticmethod
    def backward(ctx, dy, dpart):
        x, weight = ctx.saved_tensors
        stride, padl, padr, groups, dilation, has_bias = ctx.conf
        dx, dw, db = ext().conv1d_bwd(dy.contiguous(), x, weight, stride,
                                      padl, padr, groups, dilation, has_bias)
        return dx, dw, db, None, None, None, None, None


def conv1d_stats(x, weight, bias=None, stride=1, padding=(0, 0), groups=1,
                 dilation=1):
    """conv1d + BN stats partials; (y, None) off-GPU."""
    padl, padr = padding
    if x.dtype != weight.dtype:
        x = x.to(weight.dtype)
    if not use_native(x):
        return conv1d(x, weight, bias, stride, padding, groups,
                      dilation), None
    if (weight.size(-1) == 1 and stride == 1 and padl == 0 and padr == 0
            and groups == 1):
        return pointwise_conv_stats(x, weight, bias)
    return _Conv1dStats.apply(x.contiguous(), weight.contiguous(), bias,
                              stride, padl, padr, groups, dilation)


# ---------------------------------------------------------------------------
# general direct conv1d (depthwise / grouped / dense, strided, pre-padded)
# ---------------------------------------------------------------------------


class _Conv1d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride, padl, padr, groups, dilation):
        ctx.save_for_backward(x, weight)
        ctx.conf = (stride, padl, padr, groups, dilation, bias is not None)
        if use_native(x):
            return ext().conv1d_fwd(x, weight, bias, stride, padl, padr,
                                    groups, dilation)
        xp = F.pad(x, (padl, padr)) if (padl or padr) else x
        return F.conv1d(xp, weight, bias, stride=stride, groups=groups,
                        dilation=dilation)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        stride, padl, padr, groups, dilation, has_bias = ctx.conf
        dy = dy.contiguous()
        if use_native(x):
            dx, dw, db = ext().conv1d_bwd(dy, x, weight, stride, padl, padr,
                                          groups, dilation, has_bias)
        else:
            xp = F.pad(x, (padl, padr)) if (padl or padr) else x
            xp = xp.detach().requires_grad_(True)
            w = weight.detach().requires_grad_(True)
            with torch.enable_grad():
                out = F.conv1d(xp, w, None, stride=stride, groups=groups,
                               dilation=dilation)
            dxp, dw = torch.autograd.grad(out, [xp, w], dy)
            Lp = dxp.size(-1)
            dx = dxp[..., padl: Lp - padr] if (padl or padr) else dxp
            db = dy.sum(dim=(0, 2)) if has_bias else None
        return dx, dw, db, None, None, None, None, None


def conv1d(x: torch.Tensor, weight: torch.Tensor,
           bias: Optional[torch.Tensor] = None, stride: int = 1,
           padding: Tuple[int, int] = (0, 0), groups: int = 1,
           dilation: int = 1) -> torch.Tensor:
    """Direct Conv1d with explicit (left, right) padding.

    weight: (Co, Ci/groups, K). Covers dense, grouped, depthwise
    (groups == Ci) and dilated-causal convolutions — the K2/K3/K4/K6 kernel
    family of SURVEY §2.4.
    """
    padl, padr = padding
    if x.dtype != weight.dtype:
        x = x.to(weight.dtype)
    if (weight.size(-1) == 1 and stride == 1 and padl == 0 and padr == 0
            and groups == 1):
        return pointwise_conv(x, weight, bias)
    if _trace_eager(x):
        xp = F.pad(x, (padl, padr)) if (padl or padr) else x
        return F.conv1d(xp, weight, bias, stride=stride, groups=groups,
                        dilation=dilation)
    return _Conv1d.apply(x.contiguous(), weight.contiguous(), bias, stride,
                         padl, padr, groups, dilation)


# ---------------------------------------------------------------------------
# fused BatchNorm1d (+ optional GELU) over (N, C, L)
# ---------------------------------------------------------------------------


def _sync_world(group):
    import torch.distributed as dist
    if not (dist.is_available() and dist.is_initialized()):
        return None, 1
    return group, dist.get_world_size(group)


class _BNAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, act, sync, group, part):
        import torch.distributed as dist
        world = 1
        if sync and training:
            group, world = _sync_world(group)
        ctx.sync = sync and training and world > 1
        ctx.group = group
        ctx.world = world
        if ctx.sync:
            # SyncBN: local (sum, sumsq) -> one small RCCL
            # all-reduce of (C, 2) floats -> finalize with the global count.
            # Keeps the native fused kernel in play under distributed
            # training (the reference's convert_sync_batchnorm drops it).
            n_local = x.size(0) * x.size(2)
            count = n_local * world
            if use_native(x):
                sums = (ext().bn_part_to_sums(part) if part is not None
                        else ext().bn_sums_only(x))
                dist.all_reduce(sums, group=group)
                y, mean, invstd = ext().bn_act_fwd_from_sums(
                    x, sums, float(count), gamma, beta, running_mean,
                    running_var, momentum, eps, act)
            else:
                x32 = x.float()
                sums = torch.stack([x32.sum(dim=(0, 2)),
                                    (x32 * x32).sum(dim=(0, 2))], dim=1)
                dist.all_reduce(sums, group=group)
                mean = sums[:, 0] / count
                var = (sums[:, 1] / count - mean * mean).clamp_min_(0)
                if running_mean is not None:
                    with torch.no_grad():
                        running_mean.mul_(1 - momentum).add_(momentum * mean)
                        unbiased = var * (count / max(count - 1, 1))
                        running_var.mul_(1 - momentum).add_(
                            momentum * unbiased)
                invstd = torch.rsqrt(var + eps)
                xhat = (x32 - mean[:, None]) * invstd[:, None]
                pre = xhat * gamma.float()[:, None] + beta.float()[:, None]
                if act == _ACT_GELU:
                    y = F.gelu(pre)
                elif act == _ACT_RELU:
                    y = F.relu(pre)
                else:
                    y = pre
                y = y.to(x.dtype)
            ctx.save_for_backward(x, gamma, beta, mean, invstd)
            ctx.training = training
            ctx.act = act
            return y
        if use_native(x):
            if part is not None and training:
                # producer-collected partials (fusion step 1): finalize
                # directly, no bn_sums pass over x
                y, mean, invstd = ext().bn_act_fwd_with_part(
                    x, part, gamma, beta, running_mean, running_var,
                    momentum, eps, act)
            else:
                y, mean, invstd = ext().bn_act_fwd(
                    x, gamma, beta, running_mean, running_var, training,
                    momentum, eps, act)
        else:
            x32 = x.float()
            if training:
                mean = x32.mean(dim=(0, 2))
                var = x32.var(dim=(0, 2), unbiased=False)
                n = x.size(0) * x.size(2)
                if running_mean is not None:
                    with torch.no_grad():
                        running_mean.mul_(1 - momentum).add_(momentum * mean)
                        unbiased = var * (n / max(n - 1, 1))
                        running_var.mul_(1 - momentum).add_(momentum * unbiased)
            else:
                mean = running_mean.float()
                var = running_var.float()
            invstd = torch.rsqrt(var + eps)
            xhat = (x32 - mean[:, None]) * invstd[:, None]
            pre = xhat * gamma.float()[:, None] + beta.float()[:, None]
            if act == _ACT_GELU:
                y = F.gelu(pre)
            elif act == _ACT_RELU:
                y = F.relu(pre)
            else:
                y = pre
            y = y.to(x.dtype)
        ctx.save_for_backward(x, gamma, beta, mean, invstd)
        ctx.training = training
        ctx.act = act
        return y

    @staticmethod
    def backward(ctx, dy):
        import torch.distributed as dist
        x, gamma, beta, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous()
        if ctx.sync:
            count = x.size(0) * x.size(2) * ctx.world
            if use_native(x):
                local = ext().bn_bwd_sums_only(dy, x, mean, invstd, gamma,
                                               beta, ctx.act)
                # dgamma/dbeta stay LOCAL sums (the later gradient
                # all-reduce averages them, torch SyncBatchNorm parity);
                # dx needs the GLOBAL correction sums.
                gsums = local.clone()
                dist.all_reduce(gsums, group=ctx.group)
                dx = ext().bn_bwd_dx_from_sums(dy, x, mean, invstd, gamma,
                                               beta, gsums, float(count),
                                               ctx.act)
                dbeta = local[:, 0].to(gamma.dtype)
                dgamma = local[:, 1].to(gamma.dtype)
            else:
                x32 = x.float()
                dy32 = dy.float()
                g = gamma.float()[:, None]
                b = beta.float()[:, None]
                xhat = (x32 - mean[:, None]) * invstd[:, None]
                if ctx.act != _ACT_NONE:
                    pre = xhat * g + b
                    if ctx.act == _ACT_GELU:
                        dy32 = dy32 * _gelu_grad(pre)
                    else:
                        dy32 = dy32 * (pre > 0).to(dy32.dtype)
                dbeta_l = dy32.sum(dim=(0, 2))
                dgamma_l = (dy32 * xhat).sum(dim=(0, 2))
                gsums = torch.stack([dbeta_l, dgamma_l], dim=1)
                dist.all_reduce(gsums, group=ctx.group)
                dx = (g * invstd[:, None] / count) * (
                    count * dy32 - gsums[:, 0][:, None]
                    - xhat * gsums[:, 1][:, None])
                dx = dx.to(x.dtype)
                dbeta = dbeta_l.to(gamma.dtype)
                dgamma = dgamma_l.to(gamma.dtype)
            return (dx, dgamma, dbeta, None, None, None, None, None, None,
                    None, None, None)
        if use_native(x):
            dx, dgamma, dbeta = ext().bn_act_bwd(
                dy, x, gamma, beta, mean, invstd, ctx.training, ctx.act)
        else:
            x32 = x.float()
            dy32 = dy.float()
            g = gamma.float()[:, None]
            b = beta.float()[:, None]
            xhat = (x32 - mean[:, None]) * invstd[:, None]
            if ctx.act != _ACT_NONE:
                pre = xhat * g + b
                if ctx.act == _ACT_GELU:
                    dy32 = dy32 * _gelu_grad(pre)
                else:
                    dy32 = dy32 * (pre > 0).to(dy32.dtype)
            dgamma = (dy32 * xhat).sum(dim=(0, 2))
            dbeta = dy32.sum(dim=(0, 2))
            if ctx.training:
                n = x.size(0) * x.size(2)
                dx = (g * invstd[:, None] / n) * (
                    n * dy32 - dbeta[:, None] - xhat * dgamma[:, None]
                )
            else:
                dx = dy32 * g * invstd[:, None]
            dx = dx.to(x.dtype)
            dgamma = dgamma.to(gamma.dtype)
            dbeta = dbeta.to(beta.dtype)
        return (dx, dgamma, dbeta, None, None, None, None, None, None, None,
                None, None)


def bn_act(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
           running_mean: Optional[torch.Tensor],
           running_var: Optional[torch.Tensor], training: bool,
           momentum: float = 0.1, eps: float = 1e-5,
           act: str = "none", sync: bool = False,
           process_group=None, part: Optional[torch.Tensor] = None
           ) -> torch.Tensor:
    """Fused BatchNorm1d (+GELU) — K7/K14 of SURVEY §2.4.

    BN statistics and parameters are fp32 regardless of activation dtype.
    With ``sync=True`` under an initialized process group, batch statistics
    are reduced across ranks (one (C,2)-float collective before finalize,
    one more in backward) — K7's SyncBN obligation.
    """
    if _trace_eager(x):
        y = F.batch_norm(x.float(), running_mean, running_var, gamma.float(),
                         beta.float(), training, momentum, eps)
        if act == "gelu":
            y = F.gelu(y)
        elif act == "relu":
            y = F.relu(y)
        return y.to(x.dtype)
    act_id = {"none": _ACT_NONE, "gelu": _ACT_GELU, "relu": _ACT_RELU}[act]
    return _BNAct.apply(x.contiguous(), gamma, beta, running_mean, running_var,
                        training, momentum, eps, act_id, sync, process_group,
                        part)


# ---------------------------------------------------------------------------
# fused avg+max pool (ceil mode) — the LocalAwareAggregation primitive
# ---------------------------------------------------------------------------


class _AvgMaxPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k):
        ctx.k = k
        ctx.in_len = x.size(-1)
        if use_native(x):
            y, idx = ext().avgmax_pool_fwd(x, k)
            ctx.save_for_backward(idx)
            return y
        ya = F.avg_pool1d(x, k, ceil_mode=True)
        ym, idx = F.max_pool1d(x.float(), k, ceil_mode=True, return_indices=True)
        ctx.save_for_backward(idx)
        return ya + ym.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dy = dy.contiguous()
        k, in_len = ctx.k, ctx.in_len
        if dy.is_cuda and has_ext():
            dx = ext().avgmax_pool_bwd(dy, idx, k, in_len)
            return dx, None
        N, C, Lo = dy.shape
        dx = dy.new_zeros(N, C, in_len)
        # max part
        dx.view(N, C, in_len).scatter_add_(2, idx, dy)
        # avg part: each output window [i*k, min((i+1)k, L)) gets dy/len
        for i in range(Lo):
            lo = i * k
            hi = min(lo + k, in_len)
            dx[:, :, lo:hi] += (dy[:, :, i] / (hi - lo))[:, :, None]
        return dx, None


def avgmax_pool1d(x: torch.Tensor, k: int) -> torch.Tensor:
    """avg_pool1d(x,k,ceil) + max_pool1d(x,k,ceil) in one kernel (K12)."""
    if k <= 1:
        # reference LocalAwareAggregationBlock builds no pools for k == 1
        # (models/seist.py:79-84) — identity, not avg+max.
        return x
    if _trace_eager(x):
        return (F.avg_pool1d(x, k, ceil_mode=True)
                + F.max_pool1d(x, k, ceil_mode=True))
    return _AvgMaxPool.apply(x.contiguous(), k)


class _MaxPool1d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k, ceil_mode):
        ctx.k = k
        ctx.in_len = x.size(-1)
        if use_native(x):
            y, idx = ext().max_pool1d_fwd(x, k, ceil_mode)
            ctx.save_for_backward(idx)
            return y
        y, idx = F.max_pool1d(x.float(), k, ceil_mode=ceil_mode,
                              return_indices=True)
        ctx.save_for_backward(idx)
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        dy = dy.contiguous()
        if dy.is_cuda and has_ext():
            return ext().max_pool1d_bwd(dy, idx, ctx.k, ctx.in_len), \
                None, None
        N, C, Lo = dy.shape
        dx = dy.new_zeros(N, C, ctx.in_len)
        dx.scatter_(2, idx, dy)
        return dx, None, None


def max_pool1d(x: torch.Tensor, k: int,
               ceil_mode: bool = False) -> torch.Tensor:
    """MaxPool1d with stride == kernel (K12): EQT/MagNet/DiTingMotion
    encoder pools."""
    if _trace_eager(x):
        return F.max_pool1d(x, k, ceil_mode=ceil_mode)
    return _MaxPool1d.apply(x.contiguous(), k, ceil_mode)


class _GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.in_len = x.size(-1)
        if use_native(x):
            return ext().gap_fwd(x)
        return x.mean(dim=-1, keepdim=True)

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        if dy.is_cuda and has_ext():
            return ext().gap_bwd(dy, ctx.in_len)
        return (dy / ctx.in_len).expand(dy.size(0), dy.size(1), ctx.in_len) \
            .contiguous()


def global_avg_pool1d(x: torch.Tensor) -> torch.Tensor:
    """AdaptiveAvgPool1d(1) (K12): classification/regression heads."""
    if _trace_eager(x):
        return x.mean(dim=-1, keepdim=True)
    return _GlobalAvgPool.apply(x.contiguous())


# ---------------------------------------------------------------------------
# linear interpolation resize (F.interpolate mode='linear', align_corners=False)
# ---------------------------------------------------------------------------


class _InterpLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, out_len):
        ctx.in_len = x.size(-1)
        if use_native(x):
            return ext().interp_linear_fwd(x, out_len)
        return F.interpolate(x, size=out_len, mode="linear", align_corners=False)

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        if dy.is_cuda and has_ext():
            return ext().interp_linear_bwd(dy, ctx.in_len), None
        # CPU: use autograd of F.interpolate
        x = dy.new_zeros(dy.size(0), dy.size(1), ctx.in_len).requires_grad_(True)
        with torch.enable_grad():
            y = F.interpolate(x, size=dy.size(-1), mode="linear",
                              align_corners=False)
        (dx,) = torch.autograd.grad(y, [x], dy)
        return dx, None


def interp_linear(x: torch.Tensor, out_len: int) -> torch.Tensor:
    """K13: 1d linear resize to arbitrary size."""
    if out_len == x.size(-1):
        return x
    if _trace_eager(x):
        return F.interpolate(x, size=out_len, mode="linear",
                             align_corners=False)
    return _InterpLinear.apply(x.contiguous(), out_len)


# ---------------------------------------------------------------------------
# pooled-KV attention (K9): softmax((q/sqrt(E))^T k) @ v^T
# ---------------------------------------------------------------------------


class _PooledAttnTrain(torch.autograd.Function):
    """Fused training attention: forward saves the per-query softmax stats
    and a bit-packed dropout mask (32x smaller than the (Lq x Lk)
    probability tensor the composite materializes); backward is a
    flash-style recompute split into a per-query kernel (dQ + correction
    term) and a per-key kernel (dK, dV)."""

    @staticmethod
    def forward(ctx, q, k, v, p):
        out, stats, mask = ext().pooled_attn_train_fwd(q, k, v, p)
        ctx.save_for_backward(q, k, v, out, stats, mask)
        ctx.p = p
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, stats, mask = ctx.saved_tensors
        dq, dk, dv = ext().pooled_attn_bwd(q, k, v, out, dout.contiguous(),
                                           stats, mask, ctx.p)
        return dq, dk, dv, None


def pooled_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                     attn_dropout: float = 0.0,
                     training: bool = False) -> torch.Tensor:
    """SeisT pooled-KV attention core (reference models/seist.py:368-393).

    q: (N, H, E, Lq) from the full-length sequence; k/v: (N, H, E, Lk) from
    the aggregated sequence (Lk == Lq / aggr_ratio; 128 at every stage for
    the published configs). Returns (N, H, E, Lq).

    A fully fused single-kernel HIP path serves inference; training runs
    the fused stats+mask forward with a flash-style backward. The batched
    GEMM + softmax composite remains the CPU/fallback path.
    """
    E = q.size(2)
    fusable = (use_native(q) and E in (8, 16, 32) and k.size(3) <= 256
               and E * k.size(3) <= 4096)
    if fusable and not training and not torch.is_grad_enabled():
        return ext().pooled_attn_fwd(q.contiguous(), k.contiguous(),
                                     v.contiguous())
    if fusable and training and hasattr(ext(), "pooled_attn_train_fwd"):
        return _PooledAttnTrain.apply(q.contiguous(), k.contiguous(),
                                      v.contiguous(), float(attn_dropout))
    attn = torch.matmul(q.transpose(-1, -2), k) * (1.0 / math.sqrt(E))
    attn = attn.softmax(dim=-1)
    if attn_dropout > 0.0 and training:
        attn = F.dropout(attn, p=attn_dropout, training=True)
    out = torch.matmul(attn, v.transpose(-1, -2)).transpose(-1, -2)
    return out


# ---------------------------------------------------------------------------
# fused residual + DropPath (stochastic depth): z = x + mask[n]/keep * y
# ---------------------------------------------------------------------------


class _DropPathAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, y, mask, scale):
        ctx.scale = scale
        ctx.save_for_backward(mask if mask is not None else torch.empty(0))
        if use_native(x) and hasattr(ext(), "row_scale_add"):
            return ext().row_scale_add(x, y, mask, scale)
        m = scale if mask is None else mask.view(
            -1, *([1] * (x.dim() - 1))) * scale
        return x + m * y

    @staticmethod
    def backward(ctx, dz):
        (mask,) = ctx.saved_tensors
        mask = mask if mask.numel() else None
        dz = dz.contiguous()
        if dz.is_cuda and has_ext() and hasattr(ext(), "row_scale"):
            dy = ext().row_scale(dz, mask, ctx.scale)
        else:
            m = ctx.scale if mask is None else mask.view(
                -1, *([1] * (dz.dim() - 1))) * ctx.scale
            dy = dz * m
        return dz, dy, None, None


def droppath_add(x: torch.Tensor, y: torch.Tensor, drop_prob: float,
                 training: bool) -> torch.Tensor:
    """Fused residual + stochastic depth: replaces ``x + DropPath(p)(y)``
    (one elementwise pass; the reference chain is bernoulli/div/mul/add)."""
    if drop_prob == 0.0 or not training:
        if _trace_eager(x):
            return x + y
        return _DropPathAdd.apply(x.contiguous(), y.contiguous(), None, 1.0)
    keep = 1.0 - drop_prob
    mask = torch.bernoulli(
        torch.full((x.size(0),), keep, device=x.device, dtype=torch.float32))
    return _DropPathAdd.apply(x.contiguous(), y.contiguous(), mask,
                              1.0 / keep)


# ---------------------------------------------------------------------------
# ConvTranspose1d (K5) — PhaseNet decoder
# ---------------------------------------------------------------------------


class _ConvTranspose1d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride):
        ctx.save_for_backward(x, weight)
        ctx.stride = stride
        ctx.has_bias = bias is not None
        if use_native(x) and hasattr(ext(), "conv_transpose1d_fwd"):
            return ext().conv_transpose1d_fwd(x, weight, bias, stride)
        return F.conv_transpose1d(x, weight, bias, stride=stride)

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        if use_native(x) and hasattr(ext(), "conv_transpose1d_bwd"):
            dx, dw, db = ext().conv_transpose1d_bwd(dy, x, weight,
                                                    ctx.stride, ctx.has_bias)
        else:
            xd = x.detach().requires_grad_(True)
            wd = weight.detach().requires_grad_(True)
            with torch.enable_grad():
                out = F.conv_transpose1d(xd, wd, None, stride=ctx.stride)
            dx, dw = torch.autograd.grad(out, [xd, wd], dy)
            db = dy.sum(dim=(0, 2)) if ctx.has_bias else None
        return dx, dw, db, None


def conv_transpose1d(x: torch.Tensor, weight: torch.Tensor,
                     bias: Optional[torch.Tensor] = None,
                     stride: int = 1) -> torch.Tensor:
    """ConvTranspose1d, weight (Ci, Co, K), no padding (PhaseNet's usage)."""
    if _trace_eager(x):
        return F.conv_transpose1d(x, weight, bias, stride=stride)
    return _ConvTranspose1d.apply(x.contiguous(), weight.contiguous(), bias,
                                  stride)


# ---------------------------------------------------------------------------
# nearest 2x upsample (EQTransformer decoders)
# ---------------------------------------------------------------------------


class _Upsample2x(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if use_native(x) and hasattr(ext(), "upsample2x_fwd"):
            return ext().upsample2x_fwd(x)
        return F.interpolate(x, scale_factor=2, mode="nearest")

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        if dy.is_cuda and has_ext() and hasattr(ext(), "upsample2x_bwd"):
            return ext().upsample2x_bwd(dy)
        return dy.view(dy.size(0), dy.size(1), -1, 2).sum(-1)


def upsample2x(x: torch.Tensor) -> torch.Tensor:
    """Nearest-neighbour x2 upsample along the last dim."""
    if _trace_eager(x):
        return F.interpolate(x, scale_factor=2, mode="nearest")
    return _Upsample2x.apply(x.contiguous())


# ---------------------------------------------------------------------------
# concat-fused pointwise conv: y = W @ cat(xs, dim=1) without the cat
# ---------------------------------------------------------------------------


class _PointwiseConvMulti(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight, bias, *xs):
        ctx.save_for_backward(weight, *xs)
        ctx.has_bias = bias is not None
        return ext().pw_conv_multi_fwd(list(xs), weight, bias)

    @staticmethod
    def backward(ctx, dy):
        weight, *xs = ctx.saved_tensors
        dy = dy.contiguous()
        dxs = ext().pw_conv_multi_dx(dy, weight, [x.size(1) for x in xs])
        # dw slices per input (three small hipblaslt GEMMs on the virtual
        # concat's pieces), concatenated on the tiny (Co, Ci) weight grad
        dw = torch.cat(
            [ext().sum_batch(torch.bmm(dy, x.transpose(1, 2)))
             for x in xs], dim=1).to(weight.dtype)
        db = ext().channel_sum(dy).to(weight.dtype) if ctx.has_bias else None
        return (dw, db) + tuple(dxs)


def pointwise_conv_cat(xs, weight, bias=None):
    """1x1 conv over the channel-concat of ``xs`` with the concat fused
    into the kernel's LDS staging (forward) and the input gradient written
    straight into per-input contiguous tensors (backward) — the stem's
    3-path concat never exists (reference models/seist.py:187-195)."""
    if weight.dim() == 3:
        weight = weight.squeeze(-1)
    if (use_native(xs[0]) and 2 <= len(xs) <= 3
            and xs[0].dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16
            and sum(x.size(1) for x in xs) >= 16):
        return _PointwiseConvMulti.apply(
            weight.contiguous(), bias, *[x.contiguous() for x in xs])
    return pointwise_conv(torch.cat(xs, dim=1), weight, bias)


# ---------------------------------------------------------------------------
# fused BN(+act) -> pointwise conv (FUSION_PLAN step 2): the normalized/
# activated tensor never exists in HBM — it is produced inside the
# consumer conv's LDS staging (forward) and inside the split-K weight-
# gradient kernel's staging (backward).
# ---------------------------------------------------------------------------


class _BNActPw(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, mean, invstd, weight, bias, act,
                bn_training, act_only=False):
        scale = (gamma.float() * invstd).contiguous()
        shift = (beta.float() - mean * scale).contiguous()
        y = ext().pw_conv_pre_fwd(x, weight, bias, scale, shift, act)
        ctx.save_for_backward(x, gamma, beta, mean, invstd, weight, scale,
                              shift)
        ctx.act = act
        ctx.has_bias = bias is not None
        ctx.bn_training = bn_training
        ctx.act_only = act_only
        return y

    @staticmethod
    def backward(ctx, dy):
        (x, gamma, beta, mean, invstd, weight, scale,
         shift) = ctx.saved_tensors
        dy = dy.contiguous()
        dz = ext().pw_conv_dx(dy, weight)
        dw = ext().pw_dw_pre(dy, x, scale, shift, ctx.act,
                             weight.dtype)
        db = ext().channel_sum(dy).to(weight.dtype) if ctx.has_bias else None
        if ctx.act_only:
            # identity-BN: dx = act_grad(x) * dz, no dgamma/dbeta passes
            dx = ext().bn_bwd_dx_eval(dz, x, mean, invstd, gamma, beta,
                                      ctx.act)
            dgamma = dbeta = None
        else:
            dx, dgamma, dbeta = ext().bn_act_bwd(dz, x, gamma, beta, mean,
                                                 invstd, ctx.bn_training,
                                                 ctx.act)
        return (dx, dgamma, dbeta, None, None, dw, db, None, None, None)


def bn_act_pw(x, bn, act, weight, bias):
    """BatchNorm(+act) fused into the following 1x1 conv. ``bn`` is the
    nn.BatchNorm1d module (stats/running updates handled here); the module
    tree and checkpoint format are untouched."""
    if weight.dim() == 3:
        weight = weight.squeeze(-1)
    act_id = {"none": _ACT_NONE, "gelu": _ACT_GELU, "relu": _ACT_RELU}[act]
    import os
    fusable = (use_native(x) and x.dtype == torch.bfloat16
               and weight.dtype == torch.bfloat16 and x.size(1) >= 16
               and weight.size(0) >= 16  # dx GEMM reduces over Co
               and not getattr(bn, "_sync_bn", False)
               # net win in inference (no backward); A/B-negative in
               # training (profiles/step_profile_r02.md) -> opt-in there
               and (not torch.is_grad_enabled()
                    or os.environ.get("SEIST_AMD_PW_FUSION") == "1"))
    if not fusable:
        y = bn_act(x, bn.weight, bn.bias, bn.running_mean, bn.running_var,
                   bn.training, bn.momentum, bn.eps, act=act,
                   sync=getattr(bn, "_sync_bn", False))
        if bn.training and bn.track_running_stats \
                and not getattr(bn, "_managed_nbt", False):
            bn.num_batches_tracked += 1
        return pointwise_conv(y, weight, bias)
    x = x.contiguous()
    if bn.training:
        sums = ext().bn_sums_only(x)
        count = x.size(0) * x.size(2)
        mean, invstd = ext().bn_finalize_only(
            sums, float(count), bn.running_mean, bn.running_var,
            bn.momentum, bn.eps)
        if bn.track_running_stats \
                and not getattr(bn, "_managed_nbt", False):
            bn.num_batches_tracked += 1
    else:
        mean = bn.running_mean.float()
        invstd = torch.rsqrt(bn.running_var.float() + bn.eps)
    return _BNActPw.apply(x, bn.weight, bn.bias, mean, invstd,
                          weight.contiguous(), bias, act_id, bn.training)


def act_pw(x, act, weight, bias, module=None):
    """Elementwise activation fused into the following 1x1 conv (the MLP's
    GELU between lin0 and lin1 — reference models/seist.py:99-121)."""
    if weight.dim() == 3:
        weight = weight.squeeze(-1)
    import os
    if not (use_native(x) and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and x.size(1) >= 16
            and weight.size(0) >= 16
            and (not torch.is_grad_enabled()
                 or os.environ.get("SEIST_AMD_PW_FUSION") == "1")):
        y = gelu(x) if act == "gelu" else (x.relu() if act == "relu" else x)
        return pointwise_conv(y, weight, bias)
    act_id = {"none": _ACT_NONE, "gelu": _ACT_GELU, "relu": _ACT_RELU}[act]
    x = x.contiguous()
    C = x.size(1)
    cache = getattr(module, "_actpw_cache", None) if module is not None \
        else None
    if cache is None or cache[0].numel() != C \
            or cache[0].device != x.device:
        ones = torch.ones(C, dtype=torch.float32, device=x.device)
        zeros = torch.zeros(C, dtype=torch.float32, device=x.device)
        cache = (ones, zeros)
        if module is not None:
            module._actpw_cache = cache
    ones, zeros = cache
    return _BNActPw.apply(x, ones, zeros, zeros, ones,
                          weight.contiguous(), bias, act_id, False, True)


# ---------------------------------------------------------------------------
# BN(+act) over a virtual channel-concat (MSMC/MPT concat->norm sites)
# ---------------------------------------------------------------------------


class _BNActCat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gamma, beta, running_mean, running_var, training,
                momentum, eps, act, *xs):
        y, mean, invstd = ext().bn_act_cat_fwd(
            list(xs), gamma, beta, running_mean, running_var, training,
            momentum, eps, act)
        ctx.save_for_backward(gamma, beta, mean, invstd, *xs)
        ctx.training = training
        ctx.act = act
        return y

    @staticmethod
    def backward(ctx, dy):
        gamma, beta, mean, invstd, *xs = ctx.saved_tensors
        outs = ext().bn_act_cat_bwd(dy.contiguous(), list(xs), gamma, beta,
                                    mean, invstd, ctx.training, ctx.act)
        *dxs, dgamma, dbeta = outs
        return (dgamma, dbeta, None, None, None, None, None, None) \
            + tuple(dxs)


def bn_act_cat(xs, bn, act: str = "none"):
    """BatchNorm(+act) over the channel-concat of ``xs`` with the concat
    virtual: the kernels route channel reads by range, and backward writes
    per-input contiguous gradients (no cat, no narrow+copy)."""
    act_id = {"none": _ACT_NONE, "gelu": _ACT_GELU, "relu": _ACT_RELU}[act]
    if (use_native(xs[0]) and 2 <= len(xs) <= 3
            and not getattr(bn, "_sync_bn", False)):
        y = _BNActCat.apply(bn.weight, bn.bias, bn.running_mean,
                            bn.running_var, bn.training, bn.momentum,
                            bn.eps, act_id, *[x.contiguous() for x in xs])
        if bn.training and bn.track_running_stats \
                and not getattr(bn, "_managed_nbt", False):
            bn.num_batches_tracked += 1
        return y
    y = bn_act(torch.cat(xs, dim=1), bn.weight, bn.bias, bn.running_mean,
               bn.running_var, bn.training, bn.momentum, bn.eps, act=act,
               sync=getattr(bn, "_sync_bn", False))
    if bn.training and bn.track_running_stats \
            and not getattr(bn, "_managed_nbt", False):
        bn.num_batches_tracked += 1
    return y


# ---------------------------------------------------------------------------
# fused residual + DropPath + elementwise Dropout (K14)
# ---------------------------------------------------------------------------

_DPD_BASE = [0]


class _DropPathDropoutAdd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, y, path_p, drop_p, base):
        z, slot = ext().droppath_dropout_add(x, y, path_p, drop_p, base)
        ctx.save_for_backward(slot)
        ctx.conf = (path_p, drop_p, base)
        return z

    @staticmethod
    def backward(ctx, dz):
        (slot,) = ctx.saved_tensors
        path_p, drop_p, base = ctx.conf
        dz = dz.contiguous()
        dy = ext().droppath_dropout_scale(dz, slot, path_p, drop_p, base)
        return dz, dy, None, None, None


def droppath_dropout_add(x, y, path_p: float, drop_p: float,
                         training: bool):
    """z = x + DropPath(path_p)(Dropout(drop_p)(y)) in ONE pass: both
    masks are regenerated from a seed the kernel snapshots per call, so
    no mask tensors exist and hipGraph replays draw fresh masks."""
    if not training or (path_p == 0.0 and drop_p == 0.0):
        return droppath_add(x, y, 0.0, False)
    if not use_native(x):
        y = F.dropout(y, p=drop_p, training=True) if drop_p > 0.0 else y
        return droppath_add(x, y, path_p, True)
    # distinct RNG stream per call site; the captured value is frozen per
    # graph node while the device seed varies per replay
    _DPD_BASE[0] = (_DPD_BASE[0] + 1) & 0x3FFFFFFF
    base = _DPD_BASE[0] * (1 << 28)
    return _DropPathDropoutAdd.apply(x.contiguous(), y.contiguous(),
                                     float(path_p), float(drop_p), base)


# ---------------------------------------------------------------------------
# fused probability-input losses (K15): BCE / CE forward + backward
# ---------------------------------------------------------------------------

LOSS_BCE = 0
LOSS_CE = 1


class _FusedProbLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, p, t, w, kind, inv_div):
        ctx.kind = kind
        ctx.inv_div = inv_div
        ctx.save_for_backward(p, t, w)
        return ext().loss_sum_fwd(p, t, w, kind, inv_div)

    @staticmethod
    def backward(ctx, gout):
        p, t, w = ctx.saved_tensors
        dp = ext().loss_sum_bwd(p, t, w, gout.contiguous(), ctx.kind,
                                ctx.inv_div)
        return dp, None, None, None, None


def fused_prob_loss(preds: torch.Tensor, targets: torch.Tensor,
                    weight: torch.Tensor, kind: int):
    """K15: single-pass BCE/CE over probability inputs (reference
    models/loss.py:8-61 semantics, eps=1e-6, scalar weight). Returns the
    0-dim fp32 loss, or ``None`` when the fused path does not apply (CPU,
    non-fp32, per-channel weight, weight still on host) — callers fall
    back to the eager composite, which stays the numerics ground truth.

    * BCE: mean over ALL elements -> inv_div = 1/numel
    * CE : ``loss.sum(1).mean()`` -> inv_div = size(1)/numel
    ``weight`` and grad_output are read via device pointers so the loss is
    hipGraph-capturable inside the benchmark's captured compute graph.
    """
    if not (preds.is_cuda
            and preds.dtype == torch.float32
            and targets.dtype == torch.float32
            and targets.is_cuda
            and preds.shape == targets.shape
            and weight.dim() == 0
            and weight.is_cuda
            and not weight.requires_grad
            and not targets.requires_grad):
        return None
    if kind == LOSS_CE and preds.dim() < 2:
        return None
    if not use_native(preds):  # ALLOW_FALLBACK debug escape hatch
        return None
    p = preds.contiguous()
    t = targets.contiguous()
    div = p.numel() if kind == LOSS_BCE else p.numel() // p.size(1)
    if div == 0:
        return None
    return _FusedProbLoss.apply(p, t, weight, kind, 1.0 / div)


def nearest_resize(x: torch.Tensor, out_len: int) -> torch.Tensor:
    """``F.interpolate(x, out_len)`` (nearest, the reference's default) with
    fast paths: an integer-ratio downsample IS a strided slice
    (src = floor(i * Lin/Lout) = i * step), which replaces the
    launch-bound at::native nearest kernels (measured 6 calls x ~130 us
    per ditingmotion step at (500, 2, 1024)->16); 2x upsample routes to
    the native kernel."""
    L = x.size(-1)
    if out_len == L:
        return x
    if out_len < L and L % out_len == 0:
        return x[..., :: L // out_len]
    if out_len == 2 * L:
        return upsample2x(x)
    return F.interpolate(x, out_len)
