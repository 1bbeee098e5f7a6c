"""GPU integration: model forward/backward with the native op path, a full
bench-style training step, and verification that the in-tree extension is
the code that actually ran."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_native_extension_loaded(dev):
    import seist_amd._C as C
    assert "seist_amd" in C.__file__, C.__file__
    # the op layer must dispatch to it for CUDA tensors
    from seist_amd.ops import has_ext, use_native
    assert has_ext()
    assert use_native(torch.zeros(1, device=dev))


@pytest.mark.parametrize("name", ["seist_m_dpk", "seist_s_dpk", "phasenet"])
def test_model_gpu_forward_matches_cpu(dev, name):
    from seist_amd.models import create_model
    torch.manual_seed(0)
    m = create_model(name).eval()
    x = torch.randn(2, 3, 8192)
    with torch.no_grad():
        y_cpu = m(x)
        y_gpu = m.to(dev)(x.to(dev))
    diff = (y_gpu.float().cpu() - y_cpu).abs().max().item()
    assert diff < 1e-3, f"{name}: GPU/CPU forward diff {diff}"


def test_seist_bf16_train_step(dev):
    from seist_amd.config import Config
    from seist_amd.engine.precision import convert_to_bf16
    from seist_amd.models import create_model
    from seist_amd.ops import FusedAdam
    from seist_amd.parallel.ddp import FlatReplica

    torch.manual_seed(0)
    model = convert_to_bf16(
        create_model("seist_m_dpk", in_channels=3, in_samples=8192))
    model = model.to(dev).train()
    rep = FlatReplica(model)
    opt = FusedAdam(model.parameters(), lr=1e-4)
    loss_fn = Config.get_loss("seist_m_dpk").to(dev)

    x = torch.randn(8, 3, 8192, device=dev, dtype=torch.bfloat16)
    t = torch.rand(8, 3, 8192, device=dev)
    losses = []
    for _ in range(5):
        rep.zero_grad()
        loss = loss_fn(model(x).float(), t)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    # optimizing on a fixed batch must reduce the loss
    assert losses[-1] < losses[0]


def test_eqtransformer_gpu_step(dev):
    from seist_amd.config import Config
    from seist_amd.models import create_model
    from seist_amd.ops import FusedAdam

    torch.manual_seed(0)
    model = create_model("eqtransformer").to(dev).train()
    opt = FusedAdam(model.parameters(), lr=1e-4)
    loss_fn = Config.get_loss("eqtransformer").to(dev)
    x = torch.randn(4, 3, 8192, device=dev)
    t = torch.rand(4, 3, 8192, device=dev)
    loss = loss_fn(model(x), t)
    opt.zero_grad()
    loss.backward()
    opt.step()
    assert torch.isfinite(loss).item()


def test_train_engine_one_epoch_gpu(dev, tmp_path):
    from seist_amd.cli import get_args, main_worker
    args = get_args([
        "--mode", "train_test", "--model-name", "seist_s_dpk",
        "--dataset-name", "synthetic", "--dataset-size", "32",
        "--dataset-samples", "9000", "--batch-size", "8", "--epochs", "1",
        "--workers", "0", "--device", "cuda:0", "--use-tensorboard",
        "false", "--log-base", str(tmp_path), "--warmup-steps", "2",
        "--down-steps", "3", "--log-step", "100", "--augmentation",
        "false", "--precision", "bf16",
    ])
    args.distributed = False
    main_worker(args, dev)
    import glob
    assert glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth"))


@pytest.mark.parametrize("name,C,L", [
    ("magnet", 3, 8192), ("ditingmotion", 2, 8192),
    ("baz_network", 3, 8192), ("distpt_network", 3, 2048),
    ("seist_s_emg", 3, 8192), ("seist_s_pmp", 3, 8192),
])
def test_other_model_families_gpu_step(dev, name, C, L):
    from seist_amd.config import Config
    from seist_amd.models import create_model
    from seist_amd.ops import FusedAdam

    torch.manual_seed(0)
    kw = {"in_channels": C, "in_samples": L}
    model = create_model(name, **kw).to(dev).train()
    opt = FusedAdam(model.parameters(), lr=1e-4)
    x = torch.randn(4, C, L, device=dev)
    if name == "distpt_network":
        # config-disabled in the reference (no travel-time data); exercise
        # the dilated-causal kernels with a direct MSE
        do, po = model(x)
        loss = ((do - 1.0) ** 2).mean() + ((po - 1.0) ** 2).mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
        assert torch.isfinite(loss).item()
        return
    loss_fn = Config.get_loss(name).to(dev)
    labels, tgt_trans = Config.get_model_config_(
        name, "labels", "targets_transform_for_loss")
    if name == "ditingmotion":
        t = (torch.eye(2, device=dev)[torch.randint(0, 2, (4,))],
             torch.eye(2, device=dev)[torch.randint(0, 2, (4,))])
    elif name == "seist_s_pmp":
        t = torch.eye(2, device=dev)[torch.randint(0, 2, (4,))]
    else:
        t = torch.rand(4, 1, device=dev) * 4
    if tgt_trans is not None:
        t = tgt_trans(t)
    out = model(x)
    loss = loss_fn(out, t)
    opt.zero_grad()
    loss.backward()
    opt.step()
    assert torch.isfinite(loss).item()


def test_postprocess_gpu_matches_cpu(dev):
    from seist_amd.engine.postprocess import _detect_event, _pick_phase
    torch.manual_seed(2)
    out = torch.rand(64, 2048)
    a = _pick_phase(out.to(dev), 0.6, 50, 3, -7).cpu()
    b = _pick_phase(out, 0.6, 50, 3, -7)
    assert torch.equal(a, b)
    a = _detect_event(out.to(dev), 0.8, 2).cpu()
    b = _detect_event(out, 0.8, 2)
    assert torch.equal(a, b)


def test_flat_replica_lazy_grads_match(dev):
    """Lazy FlatReplica (steal + one _foreach_copy_ pack) must produce the
    same flat-buffer gradients as accumulate-mode, eager AND under
    hipGraph capture/replay."""
    import copy
    from seist_amd.parallel.ddp import FlatReplica
    from seist_amd.models import create_model
    from seist_amd.engine.precision import convert_to_bf16

    torch.manual_seed(0)
    # zero drop rates: the fused attention draws dropout masks from a
    # device-resident seed that bumps every forward, so stochastic runs
    # are not comparable
    kw = dict(in_samples=2048, path_drop_rate=0.0, attn_drop_rate=0.0,
              key_drop_rate=0.0, mlp_drop_rate=0.0, other_drop_rate=0.0)
    m1 = convert_to_bf16(create_model("seist_s_dpk", **kw)).to(dev)
    m2 = copy.deepcopy(m1)
    m1.train()
    m2.train()
    r1 = FlatReplica(m1)
    r2 = FlatReplica(m2, lazy=True)
    x = torch.randn(4, 3, 2048, device=dev, dtype=torch.bfloat16)

    def run(rep, model):
        rep.zero_grad()
        y = model(x)
        y.float().pow(2).mean().backward()
        rep.allreduce()

    run(r1, m1)
    run(r2, m2)
    for dt in r1.buffers:
        # accumulate-mode and steal-mode round bf16 adds in a different
        # order; compare relative to the gradient scale
        scale = r1.buffers[dt].float().abs().max().item() or 1.0
        d = (r1.buffers[dt].float()
             - r2.buffers[dt].float()).abs().max().item() / scale
        tol = 2e-2 if dt == torch.bfloat16 else 1e-5
        assert d < tol, f"eager lazy mismatch {dt}: {d}"

    # graph capture of the lazy step: replay must rewrite the same grads
    ref = {dt: b.clone() for dt, b in r2.buffers.items()}
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        run(r2, m2)
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        run(r2, m2)
    for dt in r2.buffers:
        r2.buffers[dt].zero_()
    g.replay()
    torch.cuda.synchronize()
    for dt in r2.buffers:
        d = (r2.buffers[dt].float() - ref[dt].float()).abs().max()
        assert d.item() < 1e-5, f"replay mismatch {dt}: {d.item()}"
    # second replay must be stable too
    g.replay()
    torch.cuda.synchronize()
    for dt in r2.buffers:
        d = (r2.buffers[dt].float() - ref[dt].float()).abs().max()
        assert d.item() < 1e-5, f"replay2 mismatch {dt}: {d.item()}"


def test_flat_replica_lazy_split_graph(dev):
    """The bench's distributed pattern: compute (zero+backward+pack) is
    captured, all-reduce stays eager. The captured pack must refresh the
    flat buffers on EVERY replay."""
    import copy
    from seist_amd.parallel.ddp import FlatReplica
    from seist_amd.models import create_model
    from seist_amd.engine.precision import convert_to_bf16

    torch.manual_seed(1)
    kw = dict(in_samples=2048, path_drop_rate=0.0, attn_drop_rate=0.0,
              key_drop_rate=0.0, mlp_drop_rate=0.0, other_drop_rate=0.0)
    m = convert_to_bf16(create_model("seist_s_dpk", **kw)).to(dev).train()
    rep = FlatReplica(m, lazy=True)
    x = torch.randn(4, 3, 2048, device=dev, dtype=torch.bfloat16)

    def compute():
        rep.zero_grad()
        y = m(x)
        y.float().pow(2).mean().backward()
        rep.pack()

    for _ in range(2):
        compute()
    torch.cuda.synchronize()
    ref = {dt: b.clone() for dt, b in rep.buffers.items()}

    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        compute()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        compute()
    for _ in range(3):
        for b in rep.buffers.values():
            b.zero_()
        g.replay()
        torch.cuda.synchronize()
        for dt in rep.buffers:
            d = (rep.buffers[dt].float() - ref[dt].float()).abs().max()
            assert d.item() < 1e-6, f"split-graph pack stale {dt}: {d.item()}"
