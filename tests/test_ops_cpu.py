"""CPU semantics of the functional op layer vs plain PyTorch. These same
semantics are the ground truth the GPU kernels are tested against in
test_gpu_ops.py."""

import math

import pytest
import torch
import torch.nn.functional as F

from seist_amd import ops
from seist_amd.ops.functional import auto_pad, auto_pad_lr


def test_auto_pad_matches_reference_semantics():
    # out length must be ceil(L / stride) (reference seist.py:12-48)
    for L in (100, 101, 8192, 17):
        for k, s in ((7, 2), (11, 2), (5, 1), (3, 3)):
            x = torch.randn(1, 2, L)
            xp = auto_pad(x, k, s)
            y = F.conv1d(xp, torch.randn(2, 2, k), stride=s)
            assert y.shape[-1] == math.ceil(L / s), (L, k, s)


def test_pointwise_conv_matches_conv1d():
    x = torch.randn(3, 8, 50, requires_grad=True)
    w = torch.randn(16, 8, 1, requires_grad=True)
    b = torch.randn(16, requires_grad=True)
    y = ops.pointwise_conv(x, w, b)
    y_ref = F.conv1d(x, w, b)
    assert torch.allclose(y, y_ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    gx, gw, gb = x.grad.clone(), w.grad.clone(), b.grad.clone()
    x.grad = w.grad = b.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)
    assert torch.allclose(gw.squeeze(-1), w.grad.squeeze(-1), atol=1e-4)
    assert torch.allclose(gb, b.grad, atol=1e-4)


@pytest.mark.parametrize("groups,k,stride,dil", [
    (1, 7, 1, 1), (1, 7, 4, 1), (8, 3, 1, 1), (8, 5, 2, 1),
    (2, 3, 1, 1), (1, 2, 1, 8),
])
def test_conv1d_matches_torch(groups, k, stride, dil):
    Ci, Co = 8, 8
    x = torch.randn(2, Ci, 64, requires_grad=True)
    w = torch.randn(Co, Ci // groups, k, requires_grad=True)
    padl, padr = (k - 1) * dil, 0
    y = ops.conv1d(x, w, None, stride=stride, padding=(padl, padr),
                   groups=groups, dilation=dil)
    xp = F.pad(x, (padl, padr))
    y_ref = F.conv1d(xp, w, None, stride=stride, groups=groups, dilation=dil)
    assert torch.allclose(y, y_ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    gx, gw = x.grad.clone(), w.grad.clone()
    x.grad = w.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)
    assert torch.allclose(gw, w.grad, atol=1e-4)


@pytest.mark.parametrize("training", [True, False])
@pytest.mark.parametrize("act", ["none", "gelu", "relu"])
def test_bn_act_matches_batchnorm(training, act):
    torch.manual_seed(0)
    N, C, L = 4, 6, 32
    x = torch.randn(N, C, L, requires_grad=True)
    gamma = torch.randn(C).abs().add(0.5).requires_grad_(True)
    beta = torch.randn(C, requires_grad=True)
    rm = torch.zeros(C)
    rv = torch.ones(C)
    rm2, rv2 = rm.clone(), rv.clone()

    y = ops.bn_act(x, gamma, beta, rm, rv, training, 0.1, 1e-5, act)
    y_ref = F.batch_norm(x, rm2, rv2, gamma, beta, training, 0.1, 1e-5)
    if act == "gelu":
        y_ref = F.gelu(y_ref)
    elif act == "relu":
        y_ref = F.relu(y_ref)
    assert torch.allclose(y, y_ref, atol=1e-5), (y - y_ref).abs().max()
    assert torch.allclose(rm, rm2, atol=1e-6)
    assert torch.allclose(rv, rv2, atol=1e-5)

    g = torch.randn_like(y)
    y.backward(g)
    gx, gg, gb = x.grad.clone(), gamma.grad.clone(), beta.grad.clone()
    x.grad = gamma.grad = beta.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-4), (gx - x.grad).abs().max()
    assert torch.allclose(gg, gamma.grad, atol=1e-4)
    assert torch.allclose(gb, beta.grad, atol=1e-4)


@pytest.mark.parametrize("L,k", [(32, 2), (33, 2), (100, 8), (31, 4)])
def test_avgmax_pool(L, k):
    x = torch.randn(2, 3, L, requires_grad=True)
    y = ops.avgmax_pool1d(x, k)
    y_ref = F.avg_pool1d(x, k, ceil_mode=True) + \
        F.max_pool1d(x, k, ceil_mode=True)
    assert torch.allclose(y, y_ref, atol=1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    gx = x.grad.clone()
    x.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)


@pytest.mark.parametrize("Li,Lo", [(128, 256), (100, 273), (256, 100),
                                   (64, 8192)])
def test_interp_linear(Li, Lo):
    x = torch.randn(2, 3, Li, requires_grad=True)
    y = ops.interp_linear(x, Lo)
    y_ref = F.interpolate(x, size=Lo, mode="linear", align_corners=False)
    assert torch.allclose(y, y_ref, atol=1e-6)
    g = torch.randn_like(y)
    y.backward(g)
    gx = x.grad.clone()
    x.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)


def test_pooled_attention_matches_manual():
    torch.manual_seed(0)
    N, H, E, Lq, Lk = 2, 4, 8, 64, 16
    q = torch.randn(N, H, E, Lq)
    k = torch.randn(N, H, E, Lk)
    v = torch.randn(N, H, E, Lk)
    out = ops.pooled_attention(q, k, v)
    attn = (q / math.sqrt(E)).transpose(-1, -2) @ k
    ref = (attn.softmax(-1) @ v.transpose(-1, -2)).transpose(-1, -2)
    assert torch.allclose(out, ref, atol=1e-6)


@pytest.mark.parametrize("Ci,Co,K,stride", [(8, 4, 7, 4), (16, 8, 7, 4),
                                            (4, 4, 3, 2)])
def test_conv_transpose1d_matches_torch(Ci, Co, K, stride):
    x = torch.randn(2, Ci, 64, requires_grad=True)
    w = (torch.randn(Ci, Co, K) * 0.2).requires_grad_(True)
    b = torch.randn(Co, requires_grad=True)
    y = ops.conv_transpose1d(x, w, b, stride=stride)
    y_ref = F.conv_transpose1d(x, w, b, stride=stride)
    assert torch.allclose(y, y_ref, atol=1e-5)
    g = torch.randn_like(y_ref)
    y.backward(g)
    gx, gw, gb = x.grad.clone(), w.grad.clone(), b.grad.clone()
    x.grad = w.grad = b.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad, atol=1e-5)
    assert torch.allclose(gw, w.grad, atol=1e-4)
    assert torch.allclose(gb, b.grad, atol=1e-4)


def test_droppath_add_eval_is_plain_add():
    x = torch.randn(4, 3, 16)
    y = torch.randn(4, 3, 16)
    out = ops.droppath_add(x, y, 0.5, training=False)
    assert torch.allclose(out, x + y)


def test_droppath_add_train_masks_rows():
    torch.manual_seed(0)
    x = torch.zeros(64, 2, 8, requires_grad=True)
    y = torch.ones(64, 2, 8)
    out = ops.droppath_add(x, y, 0.5, training=True)
    rows = out.flatten(1).sum(1)
    keep_scale = 1.0 / 0.5
    # each row is either fully dropped or fully kept (scaled)
    assert set(rows.round(decimals=4).unique().tolist()) <= {0.0, 16.0 * keep_scale}
    # grad of x passes through unchanged
    out.sum().backward()
    assert torch.allclose(x.grad, torch.ones_like(x))


def test_upsample2x_matches_interpolate():
    x = torch.randn(2, 3, 37, requires_grad=True)
    y = ops.upsample2x(x)
    y_ref = F.interpolate(x, scale_factor=2, mode="nearest")
    assert torch.equal(y, y_ref)
    g = torch.randn_like(y)
    y.backward(g)
    gx = x.grad.clone()
    x.grad = None
    y_ref.backward(g)
    assert torch.allclose(gx, x.grad)


def test_droppath_dropout_add_cpu_fallback():
    """CPU path: dropout + droppath composite semantics; eval identity."""
    import torch
    from seist_amd import ops
    torch.manual_seed(0)
    x = torch.zeros(32, 4, 256)
    y = torch.ones(32, 4, 256)
    z = ops.droppath_dropout_add(x, y, 0.0, 0.0, training=True)
    assert torch.allclose(z, x + y)
    z = ops.droppath_dropout_add(x, y, 0.5, 0.5, training=False)
    assert torch.allclose(z, x + y)
    z = ops.droppath_dropout_add(x, y, 0.25, 0.2, training=True)
    row = z.reshape(32, -1)
    alive = row.abs().sum(1) > 0
    assert 0.4 < alive.float().mean().item() < 1.0
    live_mean = row[alive].mean().item() * 0.75
    assert abs(live_mean - 1.0) < 0.15


def test_bn_act_cat_cpu_fallback():
    import torch
    import torch.nn as nn
    from seist_amd import ops
    torch.manual_seed(1)
    xs = [torch.randn(4, 6, 64, requires_grad=True),
          torch.randn(4, 10, 64, requires_grad=True)]
    bn = nn.BatchNorm1d(16).train()
    y = ops.bn_act_cat(xs, bn, act="relu")
    y.sum().backward()
    bn2 = nn.BatchNorm1d(16).train()
    xs2 = [x.detach().clone().requires_grad_(True) for x in xs]
    ref = torch.relu(bn2(torch.cat(xs2, dim=1)))
    assert torch.allclose(y, ref, atol=1e-5)
    assert all(x.grad is not None for x in xs)


def test_nearest_resize_matches_interpolate():
    import torch.nn.functional as F
    from seist_amd import ops
    torch.manual_seed(0)
    for L, out in [(1024, 16), (512, 8), (256, 8), (100, 7), (64, 128),
                   (128, 128)]:
        x = torch.randn(3, 2, L, requires_grad=True)
        y = ops.nearest_resize(x, out)
        ref = F.interpolate(x, out) if out != L else x
        assert torch.equal(y.reshape(-1), ref.reshape(-1)), (L, out)
        if out != L:
            g = torch.randn_like(ref)
            (dx,) = torch.autograd.grad(y.sum() * 0 + (y * g).sum(), [x])
            xr = x.detach().clone().requires_grad_(True)
            (dxr,) = torch.autograd.grad(
                (F.interpolate(xr, out) * g).sum(), [xr])
            assert torch.allclose(dx, dxr), (L, out)
