"""Model zoo: registry, parameter counts, output shapes, and golden parity
with the reference implementation (state-dict keys + eval-mode forward)."""

import pytest
import torch

from seist_amd.models import create_model, get_model_list
from seist_amd.utils.misc import count_parameters

from _refload import load_ref_models, reference_available

# verified against the reference by instantiation
PARAM_COUNTS = {
    "seist_s_dpk": 125717, "seist_m_dpk": 380805, "seist_l_dpk": 662173,
    "seist_s_emg": 98283, "seist_m_emg": 312043, "seist_l_emg": 529291,
    "phasenet": 268443, "eqtransformer": 335623,
}


def _kwargs(name):
    if name in ("magnet", "distpt_network"):
        return dict(in_channels=3)
    if name == "ditingmotion":
        return dict(in_channels=2)
    if name == "baz_network":
        return dict(in_channels=3, in_samples=8192)
    return {}


def test_registry_has_all_21_models():
    assert len(get_model_list()) == 21


@pytest.mark.parametrize("name,expected", sorted(PARAM_COUNTS.items()))
def test_param_counts(name, expected):
    assert count_parameters(create_model(name)) == expected


@pytest.mark.parametrize("name,out_shape", [
    ("seist_s_dpk", (2, 3, 8192)),
    ("phasenet", (2, 3, 8192)),
    ("eqtransformer", (2, 3, 8192)),
    ("seist_s_emg", (2, 1)),
    ("seist_s_pmp", (2, 2)),
])
def test_forward_shapes(name, out_shape):
    m = create_model(name).eval()
    with torch.no_grad():
        y = m(torch.randn(2, 3, 8192))
    assert tuple(y.shape) == out_shape


def test_multi_output_shapes():
    m = create_model("magnet", in_channels=3).eval()
    assert tuple(m(torch.randn(2, 3, 8192)).shape) == (2, 2)
    m = create_model("ditingmotion", in_channels=2).eval()
    clr, pmp = m(torch.randn(2, 2, 8192))
    assert tuple(clr.shape) == (2, 2) and tuple(pmp.shape) == (2, 2)
    m = create_model("baz_network", in_channels=3, in_samples=8192).eval()
    c, s = m(torch.randn(2, 3, 8192))
    assert tuple(c.shape) == (2, 1) and tuple(s.shape) == (2, 1)
    m = create_model("distpt_network", in_channels=3).eval()
    d, p = m(torch.randn(2, 3, 2048))
    assert tuple(d.shape) == (2, 2) and tuple(p.shape) == (2, 2)


@pytest.mark.skipif(not reference_available(), reason="reference absent")
@pytest.mark.parametrize("name", ["seist_m_dpk", "seist_s_emg", "phasenet",
                                  "eqtransformer", "magnet", "ditingmotion",
                                  "baz_network", "distpt_network"])
def test_reference_state_dict_and_forward_parity(name):
    ref_models = load_ref_models()
    kw = _kwargs(name)
    torch.manual_seed(0)
    ref = ref_models.create_model(name, **kw).eval()
    ours = create_model(name, **kw).eval()

    rk, ok = set(ref.state_dict()), set(ours.state_dict())
    assert rk == ok, f"missing:{sorted(rk - ok)[:5]} extra:{sorted(ok - rk)[:5]}"
    ours.load_state_dict(ref.state_dict())

    C = 2 if name == "ditingmotion" else 3
    L = 2048 if name == "distpt_network" else 8192
    x = torch.randn(2, C, L)
    if name == "baz_network":
        # Our build uses the symmetric eigensolver (closed-form K16 path)
        # instead of torch.linalg.eig: eigenvector order/sign are arbitrary,
        # so exact forward parity is not defined. Check the covariance
        # path and eigenvalue sets instead (no pretrained baz checkpoint
        # exists, so .pth interop is unaffected).
        cov_r = ref._cov(x)
        cov_o = ours._cov(x)
        assert torch.allclose(cov_r, cov_o, atol=1e-4)
        ev_r, _ = ref._eig(cov_r)
        ev_o, _ = torch.linalg.eigh(cov_o.float())
        assert torch.allclose(ev_r.squeeze(-1).sort(-1).values,
                              ev_o.sort(-1).values, atol=1e-3)
        with torch.no_grad():
            yo = ours(x)
        assert all(torch.isfinite(t).all() for t in yo)
        return
    with torch.no_grad():
        yr, yo = ref(x), ours(x)
    if isinstance(yr, tuple):
        diff = max((a - b).abs().max().item() for a, b in zip(yr, yo))
    else:
        diff = (yr - yo).abs().max().item()
    assert diff < 1e-5, f"{name} forward diff {diff}"


def test_checkpoint_roundtrip(tmp_path):
    from seist_amd.models import load_checkpoint, save_checkpoint
    m = create_model("seist_s_dpk")
    path = str(tmp_path / "ck.pth")
    save_checkpoint(path, model=m, epoch=3, loss=0.5)
    ck = load_checkpoint(path)
    assert ck["epoch"] == 3 and ck["loss"] == 0.5
    m2 = create_model("seist_s_dpk")
    m2.load_state_dict(ck["model_dict"])
    for a, b in zip(m.state_dict().values(), m2.state_dict().values()):
        assert torch.equal(a, b)


def test_checkpoint_strips_wrapper_prefixes(tmp_path):
    import torch as t
    m = create_model("phasenet")
    sd = {f"module.{k}": v for k, v in m.state_dict().items()}
    path = str(tmp_path / "pref.pth")
    t.save({"model_dict": sd, "epoch": 0, "loss": 0.0, "use_ddp": False,
            "use_compile": False, "optimizer_dict": None}, path)
    from seist_amd.models import load_checkpoint
    ck = load_checkpoint(path)
    m.load_state_dict(ck["model_dict"])


def test_sym3_eig_analytic():
    """K16: closed-form symmetric 3x3 eig vs LAPACK eigh."""
    from seist_amd.models.baz_network import _sym3_eig
    torch.manual_seed(0)
    x = torch.randn(32, 3, 500)
    d = x - x.mean(-1, keepdim=True)
    cov = torch.matmul(d, d.transpose(1, 2)) / 499
    vals, vecs = _sym3_eig(cov)
    ref = torch.linalg.eigh(cov)[0].flip(-1)
    assert torch.allclose(vals.squeeze(-1), ref, atol=1e-4, rtol=1e-4)
    # eigen equation A v = lambda v
    err = (torch.matmul(cov, vecs) - vals.transpose(1, 2) * vecs).abs().max()
    assert err < 1e-4


def test_seist_activation_checkpointing_matches():
    """use_checkpoint=True (reference seist.py:841-847 parity) must not
    change forward or gradients."""
    import torch
    from seist_amd.models import create_model
    torch.manual_seed(0)
    m1 = create_model("seist_s_dpk", in_channels=3, in_samples=512)
    m2 = create_model("seist_s_dpk", in_channels=3, in_samples=512,
                      use_checkpoint=True)
    m2.load_state_dict(m1.state_dict())
    m1.train(); m2.train()
    # droppath/dropout randomness: evaluate in eval mode for exactness
    m1.eval(); m2.eval()
    x = torch.randn(2, 3, 512)
    y1 = m1(x)
    y2 = m2(x)
    assert torch.allclose(y1, y2, atol=1e-6)

    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    m1(x1).sum().backward()
    m2(x2).sum().backward()
    assert torch.allclose(x1.grad, x2.grad, atol=1e-6)


@pytest.mark.parametrize("name", ["seist_l_dpk", "seist_l_pmp", "seist_l_emg",
                                  "seist_l_baz", "seist_l_dis"])
def test_seist_l_five_heads_train_step(name):
    """configs[4] milestone: all five seist_l task heads run a full
    CPU train step (forward, loss, backward, optimizer)."""
    from seist_amd.config import Config
    from seist_amd.ops import FusedAdam

    torch.manual_seed(0)
    m = create_model(name, in_channels=3, in_samples=2048).train()
    opt = FusedAdam(m.parameters(), lr=1e-4)
    x = torch.randn(3, 3, 2048)
    loss_fn = Config.get_loss(name)
    labels, tgt_trans = Config.get_model_config_(
        name, "labels", "targets_transform_for_loss")
    if labels == [["det", "ppk", "spk"]]:
        t = torch.rand(3, 3, 2048)
    elif labels == ["pmp"]:
        t = torch.eye(2)[torch.randint(0, 2, (3,))]
    else:
        t = torch.rand(3, 1) * 5.0
    if tgt_trans is not None:
        t = tgt_trans(t)
    out = m(x)
    out = [o.float() for o in out] if isinstance(out, (list, tuple)) \
        else out.float()
    loss = loss_fn(out, t)
    opt.zero_grad()
    loss.backward()
    opt.step()
    assert torch.isfinite(loss).item()
    grads = [p.grad for p in m.parameters() if p.requires_grad]
    assert all(g is not None and torch.isfinite(g).all() for g in grads)


def test_torch_compile_smoke():
    """--use-torch-compile parity capability: dynamo traces through the
    custom autograd Functions (fwd + bwd). backend="eager" exercises the
    dynamo capture (the part our op layer could break) without paying
    ~2 min of CPU inductor codegen, which is torch's own machinery."""
    import warnings
    import torch
    from seist_amd.models import create_model
    with warnings.catch_warnings():
        warnings.simplefilter("ignore")
        m = create_model("phasenet", in_channels=3, in_samples=1024)
        mc = torch.compile(m, backend="eager")
        x = torch.randn(2, 3, 1024)
        y = mc(x)
        assert y.shape == (2, 3, 1024)
        y.float().pow(2).mean().backward()
        assert all(p.grad is not None for p in m.parameters())
