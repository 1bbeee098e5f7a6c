"""EQTransformer-family ops: LayerNorm (K8), additive/banded attention
(K10) and LSTM (K11) of SURVEY §2.4.

GPU: fused HIP kernels from ``seist_amd._C`` (eqt.hip). CPU: plain fp32
PyTorch composites with identical semantics (reference
models/eqtransformer.py:135-198, 245-262; models/magnet.py:95-101),
used by the numerics tests as ground truth.
"""

from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import ext, use_native

_EPS = 1e-6


# ---------------------------------------------------------------------------
# K8: LayerNorm over the last dim (channel-last rows)
# ---------------------------------------------------------------------------


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        if use_native(x):
            y, mean, rstd = ext().ln_fwd(x, gamma, beta, eps)
        else:
            x32 = x.float()
            mean = x32.mean(-1)
            var = x32.var(-1, unbiased=False)
            rstd = torch.rsqrt(var + eps)
            y = ((x32 - mean[..., None]) * rstd[..., None]
                 * gamma.float() + beta.float()).to(x.dtype)
            mean = mean.reshape(-1)
            rstd = rstd.reshape(-1)
        ctx.save_for_backward(x, gamma, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        if use_native(x):
            dx, dgamma, dbeta = ext().ln_bwd(dy, x, gamma, mean, rstd)
        else:
            C = x.size(-1)
            shape = x.shape
            x32 = x.float().reshape(-1, C)
            dy32 = dy.float().reshape(-1, C)
            xhat = (x32 - mean[:, None]) * rstd[:, None]
            g = dy32 * gamma.float()
            dbeta = dy32.sum(0).to(gamma.dtype)
            dgamma = (dy32 * xhat).sum(0).to(gamma.dtype)
            dx = ((g - g.mean(-1, keepdim=True)
                   - xhat * (g * xhat).mean(-1, keepdim=True))
                  * rstd[:, None]).reshape(shape).to(x.dtype)
        return dx, dgamma, dbeta, None


def layer_norm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
               eps: float = 1e-5) -> torch.Tensor:
    """LayerNorm over the last dimension (EQT transformer stage, K8)."""
    if x.device.type == "cpu" and torch.jit.is_tracing():
        return F.layer_norm(x, x.shape[-1:], gamma, beta, eps)
    return _LayerNorm.apply(x.contiguous(), gamma, beta, eps)


# ---------------------------------------------------------------------------
# K10: additive (Bahdanau-style) attention with optional band mask
# ---------------------------------------------------------------------------


def _band_limits(L: int, attn_width: Optional[int]) -> Tuple[int, int]:
    """tril/triu diagonals of the reference band mask
    (eqtransformer.py:180-186): keep tril(w//2 - 1) ∩ triu(-w//2)."""
    if attn_width is None:
        return L, -L
    return attn_width // 2 - 1, -attn_width // 2


class _AdditiveAttn(torch.autograd.Function):
    """Fused scores+softmax: the (N,L,L,d) tanh tensor never exists; only
    the (N,L,L) attention matrix is materialized (the reference composite
    keeps the full 4-d intermediate alive through autograd)."""

    @staticmethod
    def forward(ctx, q, k, bh, wa, ba, tril_k, triu_k):
        # ba stays a device scalar (a host float() here would D2H-sync and
        # break hipGraph capture)
        attn, ssum, amax = ext().addattn_fwd(q, k, bh, wa, ba.reshape(1),
                                             tril_k, triu_k)
        ctx.save_for_backward(q, k, bh, wa, attn, amax)
        return attn

    @staticmethod
    def backward(ctx, dattn):
        q, k, bh, wa, attn, amax = ctx.saved_tensors
        dq, dk, dwa, dba, descore = ext().addattn_bwd(q, k, bh, wa, attn,
                                                      dattn, amax)
        # dbh_d = sum over (n,i) of dq — identical reduction
        dbh = dq.sum(dim=(0, 1))
        return dq, dk, dbh, dwa, dba, None, None


def additive_attention_weights(q: torch.Tensor, k: torch.Tensor,
                               bh: torch.Tensor, wa: torch.Tensor,
                               ba: torch.Tensor,
                               attn_width: Optional[int] = None
                               ) -> torch.Tensor:
    """a_ij = normalize(band(exp(Wa·tanh(q_i + k_j + bh) + ba − rowmax)))
    with the reference's sum+1e-6 normalisation. q/k: (N, L, d) fp32."""
    L = q.size(1)
    tril_k, triu_k = _band_limits(L, attn_width)
    if use_native(q) and L % 4 == 0 and L <= 256 and q.size(2) <= 64:
        return _AdditiveAttn.apply(q.contiguous(), k.contiguous(), bh,
                                   wa.reshape(-1), ba.reshape(()), tril_k,
                                   triu_k)
    # composite reference path
    h = torch.tanh(q.unsqueeze(2) + k.unsqueeze(1) + bh)     # (N,L,L,d)
    e = (h * wa.reshape(1, 1, 1, -1)).sum(-1) + ba
    e = torch.exp(e - torch.max(e, dim=-1, keepdim=True).values)
    if attn_width is not None:
        mask = (torch.ones(e.shape[-2:], dtype=torch.bool, device=e.device)
                .tril(attn_width // 2 - 1)
                .triu(-attn_width // 2))
        e = e.where(mask, torch.zeros((), dtype=e.dtype, device=e.device))
    s = torch.sum(e, dim=-1, keepdim=True)
    return e / (s + _EPS)


# ---------------------------------------------------------------------------
# K11: LSTM (input projections = one GEMM; recurrence = persistent kernel)
# ---------------------------------------------------------------------------


class _LSTMDir(torch.autograd.Function):
    """One direction of the recurrence. ``pre`` = x @ W_ih^T + b_ih + b_hh
    (N, L, 4H); writes its H-slice of the shared output y in place."""

    @staticmethod
    def forward(ctx, pre, whh, y, dir_, training):
        # grad mode is disabled inside Function.forward, so the caller
        # decides whether the backward stashes are needed
        out = ext().lstm_fwd(pre, whh, y, dir_, training)
        if training:
            cstash, gstash = out
        else:
            cstash = gstash = None
        H = whh.size(1)
        ctx.dir = dir_
        ctx.dirs = y.size(2) // H
        # save a CLONE of this direction's h-slice (not y itself: the other
        # direction writes y in place after us, which would trip the saved-
        # tensor version counter)
        ys = y[:, :, dir_ * H:(dir_ + 1) * H].clone() if training \
            else torch.empty(0)
        ctx.save_for_backward(pre, whh, ys,
                              cstash if cstash is not None else
                              torch.empty(0),
                              gstash if gstash is not None else
                              torch.empty(0))
        ctx.mark_dirty(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        pre, whh, ys, cstash, gstash = ctx.saved_tensors
        H = whh.size(1)
        dirs, dir_ = ctx.dirs, ctx.dir
        dy = dy.contiguous()
        dgates = ext().lstm_bwd(dy, ys, cstash, gstash, whh, dirs, dir_)
        # h_{t-1} per output position from the saved slice
        N, L = ys.size(0), ys.size(1)
        zero = ys.new_zeros(N, 1, H)
        if dir_ == 0:
            hp = torch.cat([zero, ys[:, :-1]], dim=1)
        else:
            hp = torch.cat([ys[:, 1:], zero], dim=1)
        dwhh = torch.einsum("ntg,nth->gh", dgates, hp)
        # the input y's slice for this direction was overwritten: its
        # incoming values contribute nothing
        grad_y = dy.clone()
        grad_y[:, :, dir_ * H:(dir_ + 1) * H] = 0
        return dgates, dwhh, grad_y, None, None


def lstm(x: torch.Tensor, module: torch.nn.LSTM) -> torch.Tensor:
    """nn.LSTM-equivalent forward (batch_first, 1 layer, h0=c0=0): the
    input/bias projections run as one GEMM per direction, the sequential
    recurrence as a persistent HIP kernel (K11). x: (N, L, in) fp32;
    returns (N, L, D*H)."""
    assert module.num_layers == 1
    H = module.hidden_size
    dirs = 2 if module.bidirectional else 1
    if not use_native(x):
        out, _ = module(x.float() if x.dtype != torch.float32 else x)
        return out
    # pre-projections stay fp32 even for bf16 activations: a bf16 GEMM
    # pair here measured SLOWER same-box (eqt 20.49 vs 19.79 ms/step —
    # hipblaslt's tall-skinny fp32 dW tunings beat the bf16 ones at
    # K=16 shapes), so the upcast is the measured-faster path.
    x = x.contiguous()
    if x.dtype != torch.float32:
        x = x.float()
    N, L, _ = x.shape
    y = x.new_empty(N, L, dirs * H, dtype=torch.float32)
    training = torch.is_grad_enabled() and (
        x.requires_grad
        or any(p.requires_grad for p in module.parameters()))
    for dir_ in range(dirs):
        sfx = "_reverse" if dir_ == 1 else ""
        w_ih = getattr(module, f"weight_ih_l0{sfx}")
        w_hh = getattr(module, f"weight_hh_l0{sfx}")
        pre = x.matmul(w_ih.t())
        if module.bias:
            b_ih = getattr(module, f"bias_ih_l0{sfx}")
            b_hh = getattr(module, f"bias_hh_l0{sfx}")
            pre = pre + (b_ih + b_hh)
        y = _LSTMDir.apply(pre.contiguous(), w_hh.contiguous(), y, dir_,
                           training)
    return y
