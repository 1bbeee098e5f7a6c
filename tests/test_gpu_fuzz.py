"""Randomized shape fuzz over the conv/bn/pool dispatch paths vs the CPU
fp32 reference. Seeded, so failures are reproducible; shapes are drawn to
cross every dispatcher boundary (MFMA tap-gather vs VALU conv-GEMM vs
depthwise, dw_mfma diagonal/pair tiles vs direct, strided/dilated,
fp32/bf16)."""

import random

import pytest
import torch

pytestmark = pytest.mark.gpu

from seist_amd import ops  # noqa: E402


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    from seist_amd.ops import has_ext
    assert has_ext()
    return torch.device("cuda:0")


def _cmp(a, b, atol, rtol, msg):
    diff = (a.detach().float().cpu() - b.detach().float().cpu()).abs().max()
    denom = b.detach().float().abs().max().item() + 1e-8
    assert diff.item() <= atol + rtol * denom, \
        f"{msg}: max diff {diff.item():.3g} (ref scale {denom:.3g})"


@pytest.mark.parametrize("trial", range(24))
def test_conv1d_fuzz(dev, trial):
    rng = random.Random(1234 + trial)
    dtype = rng.choice([torch.float32, torch.bfloat16])
    N = rng.choice([1, 2, 3])
    L = rng.choice([63, 128, 257, 500, 1024, 2049])
    k = rng.choice([1, 2, 3, 5, 7, 11, 16, 19])
    stride = rng.choice([1, 1, 1, 2, 4])
    dil = 1 if stride > 1 else rng.choice([1, 1, 1, 2, 8])
    if (k - 1) * dil >= L:
        dil = 1
    # channel structure: dense, grouped (Cog==Cig), or depthwise
    kind = rng.choice(["dense", "grouped", "depthwise"])
    if kind == "dense":
        Ci = rng.choice([1, 3, 6, 16, 40])
        Co = rng.choice([1, 4, 8, 24, 48])
        groups = 1
    elif kind == "grouped":
        Cog = rng.choice([2, 4, 8, 16, 32])
        groups = rng.choice([2, 4])
        Ci = Co = Cog * groups
    else:
        Ci = Co = groups = rng.choice([2, 3, 8, 16, 24])
    padl = rng.randint(0, (k - 1) * dil)
    padr = rng.randint(0, (k - 1) * dil)
    Lo = (L + padl + padr - (k - 1) * dil - 1) // stride + 1
    if Lo <= 0:
        padr += (k - 1) * dil
        Lo = (L + padl + padr - (k - 1) * dil - 1) // stride + 1
    has_bias = rng.random() < 0.5

    torch.manual_seed(trial)
    x32 = torch.randn(N, Ci, L).to(dtype).float()
    w32 = (torch.randn(Co, Ci // groups, k) * 0.2).to(dtype).float()
    b32 = (torch.randn(Co) * 0.1).to(dtype).float() if has_bias else None

    xg = x32.to(dev, dtype).requires_grad_(True)
    wg = w32.to(dev, dtype).requires_grad_(True)
    bg = b32.to(dev, dtype).requires_grad_(True) if has_bias else None
    y = ops.conv1d(xg, wg, bg, stride=stride, padding=(padl, padr),
                   groups=groups, dilation=dil)

    xc = x32.clone().requires_grad_(True)
    wc = w32.clone().requires_grad_(True)
    bc = b32.clone().requires_grad_(True) if has_bias else None
    y_ref = ops.conv1d(xc, wc, bc, stride=stride, padding=(padl, padr),
                       groups=groups, dilation=dil)

    msg = (f"{kind} N{N} Ci{Ci} Co{Co} L{L} k{k} s{stride} d{dil} "
           f"g{groups} pad({padl},{padr}) {dtype}")
    tol = (1e-4, 1e-4) if dtype == torch.float32 else (5e-2, 2e-2)
    _cmp(y, y_ref, *tol, "fwd " + msg)

    g32 = torch.randn_like(y_ref).to(dtype).float()
    y.backward(g32.to(dev, dtype))
    y_ref.backward(g32)
    dtol = (2e-4, 1e-4) if dtype == torch.float32 else (1e-1, 2e-2)
    wtol = (1e-3, 1e-3) if dtype == torch.float32 else (0.0, 4e-2)
    _cmp(xg.grad, xc.grad, *dtol, "dx " + msg)
    _cmp(wg.grad, wc.grad, *wtol, "dw " + msg)
    if has_bias:
        _cmp(bg.grad, bc.grad, *wtol, "db " + msg)


@pytest.mark.parametrize("trial", range(8))
def test_bn_pool_interp_fuzz(dev, trial):
    rng = random.Random(77 + trial)
    dtype = rng.choice([torch.float32, torch.bfloat16])
    N = rng.choice([1, 2, 5])
    C = rng.choice([1, 3, 16, 33])
    L = rng.choice([17, 64, 255, 1000])
    torch.manual_seed(trial)
    x32 = torch.randn(N, C, L).to(dtype).float()
    tol = (1e-4, 1e-4) if dtype == torch.float32 else (4e-2, 2e-2)
    msg = f"N{N} C{C} L{L} {dtype}"

    # fused avg+max pool
    k = rng.choice([2, 3, 4, 7])
    xg = x32.to(dev, dtype).requires_grad_(True)
    xc = x32.clone().requires_grad_(True)
    y = ops.avgmax_pool1d(xg, k)
    y_ref = ops.avgmax_pool1d(xc, k)
    _cmp(y, y_ref, *tol, "pool fwd " + msg)
    g = torch.randn_like(y_ref)
    y.backward(g.to(dev, dtype))
    y_ref.backward(g)
    _cmp(xg.grad, xc.grad, *tol, "pool bwd " + msg)

    # linear interpolation (up and down)
    out_len = rng.choice([max(2, L // 3), L * 2, L * 4 + 1])
    xg = x32.to(dev, dtype).requires_grad_(True)
    xc = x32.clone().requires_grad_(True)
    y = ops.interp_linear(xg, out_len)
    y_ref = ops.interp_linear(xc, out_len)
    _cmp(y, y_ref, *tol, f"interp fwd {out_len} " + msg)
    g = torch.randn_like(y_ref)
    y.backward(g.to(dev, dtype))
    y_ref.backward(g)
    _cmp(xg.grad, xc.grad, tol[0] * 4, tol[1] * 2,
         f"interp bwd {out_len} " + msg)


@pytest.mark.parametrize("seed", range(6))
def test_fuzz_new_fused_ops(seed):
    """Randomized shapes through the round-2 fused ops: cat-BN, cat-pw,
    dropout-residual, pools."""
    import torch.nn as nn
    from seist_amd import ops

    rng = torch.Generator().manual_seed(1000 + seed)

    def ri(lo, hi):
        return int(torch.randint(lo, hi + 1, (1,), generator=rng))

    dev = "cuda:0"
    N = ri(2, 6)
    L = ri(33, 700)
    dtype = torch.bfloat16 if seed % 2 == 0 else torch.float32

    # --- cat-BN ---
    npieces = ri(2, 3)
    widths = [ri(4, 40) for _ in range(npieces)]
    C = sum(widths)
    xs = [torch.randn(N, c, L, device=dev, dtype=dtype, requires_grad=True)
          for c in widths]
    bn = nn.BatchNorm1d(C).to(dev).train()
    act = ["none", "gelu", "relu"][seed % 3]
    y = ops.bn_act_cat(xs, bn, act=act)
    bn2 = nn.BatchNorm1d(C).to(dev).train()
    xs2 = [x.detach().clone().requires_grad_(True) for x in xs]
    ref = ops.bn_act(torch.cat(xs2, 1), bn2.weight, bn2.bias,
                     bn2.running_mean, bn2.running_var, True, bn2.momentum,
                     bn2.eps, act=act)
    assert torch.allclose(y.float(), ref.float(), atol=5e-2), \
        (y - ref).abs().max().item()
    dy = torch.randn_like(y)
    g1 = torch.autograd.grad(y, xs + [bn.weight, bn.bias], dy)
    g2 = torch.autograd.grad(ref, xs2 + [bn2.weight, bn2.bias], dy)
    for a, b in zip(g1, g2):
        s = b.float().abs().max().item() or 1.0
        assert (a.float() - b.float()).abs().max().item() / s < 5e-2

    # --- cat-pw (bf16 only path; fallback otherwise — both must agree) ---
    Co = ri(8, 64)
    w = (torch.randn(Co, C, device=dev, dtype=dtype, generator=None) * 0.1
         ).requires_grad_(True)
    xs3 = [x.detach().clone().requires_grad_(True) for x in xs]
    z1 = ops.pointwise_conv_cat(xs3, w, None)
    xs4 = [x.detach().clone().requires_grad_(True) for x in xs]
    w2 = w.detach().clone().requires_grad_(True)
    z2 = ops.pointwise_conv(torch.cat(xs4, 1), w2, None)
    s = z2.float().abs().max().item() or 1.0
    assert (z1.float() - z2.float()).abs().max().item() / s < 2e-2
    dz = torch.randn_like(z1)
    h1 = torch.autograd.grad(z1, xs3 + [w], dz)
    h2 = torch.autograd.grad(z2, xs4 + [w2], dz)
    for a, b in zip(h1, h2):
        s = b.float().abs().max().item() or 1.0
        assert (a.float() - b.float()).abs().max().item() / s < 2e-2

    # --- dropout residual: backward mask equals forward ---
    x0 = torch.zeros(N, widths[0], L, device=dev, dtype=dtype)
    y0 = torch.ones(N, widths[0], L, device=dev, dtype=dtype,
                    requires_grad=True)
    pp = [0.0, 0.3][seed % 2]
    dp = [0.2, 0.0][seed % 2]
    z = ops.droppath_dropout_add(x0, y0, pp, dp, training=True)
    g = torch.autograd.grad(z, y0, torch.ones_like(z))[0]
    assert torch.equal(g.float(), z.float())

    # --- pools at odd lengths ---
    k = ri(2, 5)
    xp = torch.randn(N, widths[0], L, device=dev, dtype=torch.float32,
                     requires_grad=True)
    yp = ops.max_pool1d(xp, k, ceil_mode=True)
    ref = torch.nn.functional.max_pool1d(xp.detach(), k, ceil_mode=True)
    assert torch.allclose(yp, ref, atol=1e-6)
