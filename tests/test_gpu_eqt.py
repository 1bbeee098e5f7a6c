"""GPU numerics for the EQT-family kernels (K8 LayerNorm, K10 additive
attention, K11 LSTM) against the plain fp32 PyTorch composites."""

import pytest
import torch

from seist_amd import ops

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _assert_close(a, b, atol, what):
    d = (a.float() - b.float()).abs().max().item()
    assert d <= atol, f"{what}: max diff {d}"


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(500, 64, 16), (7, 33, 48)])
def test_layer_norm_fwd_bwd(dtype, shape):
    torch.manual_seed(0)
    x = torch.randn(shape, device=DEV, dtype=dtype)
    g = torch.rand(shape[-1], device=DEV) + 0.5
    b = torch.randn(shape[-1], device=DEV)

    x1 = x.clone().requires_grad_(True)
    g1 = g.clone().requires_grad_(True)
    b1 = b.clone().requires_grad_(True)
    y1 = ops.layer_norm(x1, g1, b1, 1e-5)
    dy = torch.randn_like(y1)
    y1.backward(dy)

    x2 = x.float().cpu().requires_grad_(True)
    g2 = g.cpu().requires_grad_(True)
    b2 = b.cpu().requires_grad_(True)
    y2 = torch.nn.functional.layer_norm(x2, (shape[-1],), g2, b2, 1e-5)
    y2.backward(dy.float().cpu())

    tol = 5e-2 if dtype == torch.bfloat16 else 2e-5
    _assert_close(y1.cpu(), y2, tol, "ln fwd")
    _assert_close(x1.grad.cpu(), x2.grad, tol, "ln dx")
    _assert_close(g1.grad.cpu(), g2.grad, 2e-2 if dtype == torch.bfloat16
                  else 2e-3, "ln dgamma")
    _assert_close(b1.grad.cpu(), b2.grad, 2e-2 if dtype == torch.bfloat16
                  else 2e-3, "ln dbeta")


def _attn_composite(q, k, bh, wa, ba, width):
    h = torch.tanh(q.unsqueeze(2) + k.unsqueeze(1) + bh)
    e = (h * wa.reshape(1, 1, 1, -1)).sum(-1) + ba
    e = torch.exp(e - torch.max(e, dim=-1, keepdim=True).values)
    if width is not None:
        mask = (torch.ones(e.shape[-2:], dtype=torch.bool, device=e.device)
                .tril(width // 2 - 1).triu(-width // 2))
        e = e.where(mask, torch.zeros((), dtype=e.dtype, device=e.device))
    s = torch.sum(e, dim=-1, keepdim=True)
    return e / (s + 1e-6)


@pytest.mark.parametrize("width", [None, 3])
@pytest.mark.parametrize("nld", [(16, 64, 32), (5, 48, 24)])
def test_additive_attention_fused_vs_composite(width, nld):
    torch.manual_seed(1)
    N, L, d = nld
    mk = lambda *s: torch.randn(*s, device=DEV, dtype=torch.float32)
    q0, k0, bh0, wa0 = mk(N, L, d), mk(N, L, d), mk(d), mk(d)
    ba0 = mk(1)[0]

    args1 = [t.clone().requires_grad_(True) for t in (q0, k0, bh0, wa0, ba0)]
    a1 = ops.additive_attention_weights(*args1, attn_width=width)
    da = torch.randn_like(a1)
    a1.backward(da)

    args2 = [t.clone().requires_grad_(True) for t in (q0, k0, bh0, wa0, ba0)]
    a2 = _attn_composite(*args2, width)
    a2.backward(da)

    _assert_close(a1, a2, 1e-5, "attn weights")
    for t1, t2, name in zip(args1, args2, ["dq", "dk", "dbh", "dwa", "dba"]):
        _assert_close(t1.grad, t2.grad, 1e-3, f"attn {name}")


@pytest.mark.parametrize("bidir", [False, True])
@pytest.mark.parametrize("geo", [(8, 64, 64, 16), (3, 512, 32, 100)])
def test_lstm_vs_nn(bidir, geo):
    torch.manual_seed(2)
    N, L, Cin, H = geo
    m = torch.nn.LSTM(Cin, H, batch_first=True, bidirectional=bidir).to(DEV)
    m_cpu = torch.nn.LSTM(Cin, H, batch_first=True, bidirectional=bidir)
    m_cpu.load_state_dict({k: v.cpu() for k, v in m.state_dict().items()})

    x = torch.randn(N, L, Cin, device=DEV)
    x1 = x.clone().requires_grad_(True)
    y1 = ops.lstm(x1, m)
    assert y1.shape == (N, L, (2 if bidir else 1) * H)
    dy = torch.randn_like(y1)
    y1.backward(dy)

    x2 = x.cpu().requires_grad_(True)
    y2, _ = m_cpu(x2)
    y2.backward(dy.cpu())

    _assert_close(y1.cpu(), y2, 1e-4, "lstm fwd")
    _assert_close(x1.grad.cpu(), x2.grad, 1e-4, "lstm dx")
    for (n1, p1), (n2, p2) in zip(m.named_parameters(),
                                  m_cpu.named_parameters()):
        assert n1 == n2
        _assert_close(p1.grad.cpu(), p2.grad, 5e-3, f"lstm {n1}")


def test_eqtransformer_gpu_matches_cpu():
    from seist_amd.models import create_model
    torch.manual_seed(3)
    m = create_model("eqtransformer").eval()
    x = torch.randn(2, 3, 8192)
    with torch.no_grad():
        y_cpu = m(x)
        y_gpu = m.to(DEV)(x.to(DEV))
    _assert_close(y_gpu.cpu(), y_cpu, 1e-3, "eqt forward")


def test_eqtransformer_gpu_train_step():
    from seist_amd.models import create_model
    torch.manual_seed(4)
    m = create_model("eqtransformer").to(DEV).train()
    x = torch.randn(4, 3, 8192, device=DEV)
    y = m(x)
    loss = (y - 0.5).pow(2).mean()
    loss.backward()
    for n, p in m.named_parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all(), n


def test_magnet_gpu_matches_cpu():
    from seist_amd.models import create_model
    torch.manual_seed(5)
    m = create_model("magnet", in_channels=3).eval()
    x = torch.randn(2, 3, 8192)
    with torch.no_grad():
        y_cpu = m(x)
        y_gpu = m.to(DEV)(x.to(DEV))
    _assert_close(y_gpu.cpu(), y_cpu, 1e-3, "magnet forward")
