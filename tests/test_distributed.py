"""Multi-process distributed correctness on CPU (gloo, world_size=2):
collectives, flat-bucket gradient replica, metric sync, and a 2-rank
mini training run."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

WORLD = 2


def _run(rank, fn, port, *args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    dist.init_process_group("gloo", rank=rank, world_size=WORLD)
    try:
        fn(rank, *args)
    finally:
        dist.destroy_process_group()


def _spawn(fn, port, *args):
    mp.start_processes(_run, args=(fn, port, *args), nprocs=WORLD,
                       start_method="spawn", join=True)


# --- worker fns (module-level for pickling) ---------------------------------

def _w_collectives(rank):
    from seist_amd.parallel import dist as pdist
    t = torch.tensor([float(rank + 1)])
    s = pdist.reduce_tensor(t, "SUM")
    assert s.item() == 3.0
    a = pdist.reduce_tensor(t, "AVG")
    assert a.item() == pytest.approx(1.5)
    lst = pdist.gather_tensors_to_list(t)
    assert sorted(x.item() for x in lst) == [1.0, 2.0]
    obj = pdist.broadcast_object({"v": rank} if rank == 0 else None, src=0)
    assert obj == {"v": 0}


def _w_flat_replica(rank):
    from seist_amd.parallel.ddp import FlatReplica
    torch.manual_seed(100 + rank)  # different init per rank
    net = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    rep = FlatReplica(net)
    # params must now be identical (broadcast from rank 0)
    for p in net.parameters():
        lst = [torch.zeros_like(p) for _ in range(WORLD)]
        dist.all_gather(lst, p.data)
        assert torch.equal(lst[0], lst[1])
    # per-rank data -> averaged grads must match manual average
    torch.manual_seed(rank)
    x = torch.randn(16, 4)
    rep.zero_grad()
    loss = net(x).sum()
    loss.backward()
    local = [p.grad.clone() for p in net.parameters()]
    rep.allreduce()
    for p, lg in zip(net.parameters(), local):
        lst = [torch.zeros_like(lg) for _ in range(WORLD)]
        dist.all_gather(lst, lg)
        manual = (lst[0] + lst[1]) / 2
        assert torch.allclose(p.grad, manual, atol=1e-6)


def _w_metrics_sync(rank):
    from seist_amd.engine.metrics import Metrics
    m = Metrics(task="emg", metric_names=["mean", "rmse", "mae"],
                sampling_rate=50, time_threshold=0.1, num_samples=8192,
                device=torch.device("cpu"))
    torch.manual_seed(rank)
    t = torch.rand(8, 1) * 5
    p = t + 1.0
    m.compute(t, p, reduce=True)
    # both ranks see the merged value; mae == 1 exactly by construction
    assert m.get_metric("mae") == pytest.approx(1.0, abs=1e-5)
    d = m._data["data_size"].item()
    assert d == 16


def _w_train(rank, tmpdir, mode="train", extra=()):
    from seist_amd.cli import get_args, main_worker
    args = get_args([
        "--mode", mode, "--model-name", "phasenet",
        "--dataset-name", "synthetic", "--dataset-size", "24",
        "--dataset-samples", "9000", "--batch-size", "2", "--epochs", "1",
        "--workers", "0", "--device", "cpu", "--use-tensorboard", "false",
        "--log-base", tmpdir, "--warmup-steps", "2", "--down-steps", "3",
        "--log-step", "100", "--augmentation", "false", "--sync-bn", "false",
    ] + list(extra))
    args.distributed = True
    main_worker(args, torch.device("cpu"))


# --- tests ------------------------------------------------------------------

def _w_syncbn(rank):
    """Native SyncBN math vs single-process BN over the concatenated global
    batch (the definition torch SyncBatchNorm implements)."""
    from seist_amd import ops

    torch.manual_seed(10 + rank)
    N, C, L = 3, 5, 16
    x = torch.randn(N, C, L, dtype=torch.float32)
    dy = torch.randn(N, C, L, dtype=torch.float32)

    # global reference on the concatenated batch
    xs = [torch.zeros_like(x) for _ in range(WORLD)]
    dys = [torch.zeros_like(dy) for _ in range(WORLD)]
    dist.all_gather(xs, x)
    dist.all_gather(dys, dy)
    xg = torch.cat(xs, 0).requires_grad_(True)
    bn_ref = torch.nn.BatchNorm1d(C)
    torch.manual_seed(7)
    with torch.no_grad():
        bn_ref.weight.copy_(torch.rand(C) + 0.5)
        bn_ref.bias.copy_(torch.randn(C))
    yg = bn_ref(xg)
    yg.backward(torch.cat(dys, 0))

    for act in ("none", "gelu"):
        gamma = bn_ref.weight.detach().clone().requires_grad_(True)
        beta = bn_ref.bias.detach().clone().requires_grad_(True)
        rm = torch.zeros(C)
        rv = torch.ones(C)
        xl = x.clone().requires_grad_(True)
        y = ops.bn_act(xl, gamma, beta, rm, rv, training=True,
                       momentum=0.1, eps=1e-5, act=act, sync=True)
        if act == "gelu":
            # recompute the reference with gelu applied
            xg2 = torch.cat(xs, 0).requires_grad_(True)
            bn2 = torch.nn.BatchNorm1d(C)
            with torch.no_grad():
                bn2.weight.copy_(gamma)
                bn2.bias.copy_(beta)
            yg2 = torch.nn.functional.gelu(bn2(xg2))
            yg2.backward(torch.cat(dys, 0))
            y_ref = yg2.detach()
            dx_ref = xg2.grad
            dg_ref, db_ref = bn2.weight.grad, bn2.bias.grad
            rm_ref, rv_ref = bn2.running_mean, bn2.running_var
        else:
            y_ref = yg.detach()
            dx_ref = xg.grad
            dg_ref, db_ref = bn_ref.weight.grad, bn_ref.bias.grad
            rm_ref, rv_ref = bn_ref.running_mean, bn_ref.running_var

        lo, hi = rank * N, (rank + 1) * N
        assert torch.allclose(y, y_ref[lo:hi], atol=1e-5), act
        y.backward(dy)
        assert torch.allclose(xl.grad, dx_ref[lo:hi], atol=1e-5), act
        # local dgamma/dbeta summed over ranks == global grads
        dg = gamma.grad.clone()
        db = beta.grad.clone()
        dist.all_reduce(dg)
        dist.all_reduce(db)
        assert torch.allclose(dg, dg_ref, atol=1e-4), act
        assert torch.allclose(db, db_ref, atol=1e-4), act
        # running stats updated with global-batch statistics on every rank
        assert torch.allclose(rm, rm_ref, atol=1e-5), act
        assert torch.allclose(rv, rv_ref, atol=1e-5), act


def _w_syncbn_model(rank):
    """enable_native_syncbn end-to-end on a seist model: each rank's
    output must equal a plain single-process model forwarded on the
    concatenated global batch (the definition of SyncBN)."""
    from seist_amd.models import create_model
    from seist_amd.parallel.ddp import enable_native_syncbn

    torch.manual_seed(0)  # identical weights on both ranks
    m = create_model("seist_s_dpk", in_samples=1024)
    m_ref = create_model("seist_s_dpk", in_samples=1024)
    m_ref.load_state_dict(m.state_dict())
    enable_native_syncbn(m)
    m.train()
    m_ref.train()

    torch.manual_seed(50 + rank)
    x = torch.randn(2, 3, 1024)
    xs = [torch.zeros_like(x) for _ in range(WORLD)]
    dist.all_gather(xs, x)
    with torch.no_grad():
        y = m(x)
        y_ref = m_ref(torch.cat(xs, 0))
    lo, hi = rank * 2, (rank + 1) * 2
    # one-pass (sum, sumsq) variance — the same formulation torch
    # SyncBatchNorm uses — drifts from the plain two-pass reference by
    # fp32 rounding that the ~40-layer BN cascade amplifies; the exact
    # per-layer math is pinned by test_native_syncbn_math
    assert torch.allclose(y, y_ref[lo:hi], atol=2e-2), \
        (y - y_ref[lo:hi]).abs().max().item()


def _w_flat_replica_mixed(rank):
    """eqtransformer's shape of replica: bf16 conv params + fp32 LSTM/LN
    params -> TWO flat buckets, one all-reduce each (lazy mode, the
    bench default). Grads of both dtypes must land averaged."""
    from seist_amd.parallel.ddp import FlatReplica
    torch.manual_seed(200 + rank)
    net = torch.nn.Sequential(torch.nn.Linear(4, 8), torch.nn.Linear(8, 2))
    net[0].to(torch.bfloat16)  # mixed precision module tree
    rep = FlatReplica(net, lazy=True)
    assert set(rep.buffers) == {torch.bfloat16, torch.float32}
    torch.manual_seed(rank)
    x = torch.randn(16, 4, dtype=torch.bfloat16)
    rep.zero_grad()
    loss = net[1](net[0](x).float()).sum()
    loss.backward()
    local = [p.grad.clone() for p in net.parameters()]
    rep.allreduce()
    for p, lg in zip(net.parameters(), local):
        lst = [torch.zeros_like(lg) for _ in range(WORLD)]
        dist.all_gather(lst, lg)
        manual = ((lst[0].float() + lst[1].float()) / 2).to(lg.dtype)
        tol = 2e-2 if lg.dtype == torch.bfloat16 else 1e-6
        assert torch.allclose(p.grad.float(), manual.float(), atol=tol), \
            p.grad.dtype


def test_collectives():
    _spawn(_w_collectives, 29511)


def test_flat_replica():
    _spawn(_w_flat_replica, 29512)


def test_flat_replica_mixed_dtype():
    _spawn(_w_flat_replica_mixed, 29517)


def test_metrics_sync():
    _spawn(_w_metrics_sync, 29513)


def test_native_syncbn_math():
    _spawn(_w_syncbn, 29516)


def test_native_syncbn_model_parity():
    _spawn(_w_syncbn_model, 29517)


def test_two_rank_training(tmp_path):
    _spawn(_w_train, 29514, str(tmp_path))
    import glob
    assert glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth"))


def test_two_rank_train_test(tmp_path):
    """train_test mode under 2 ranks: exercises the validate/test reduce
    paths, the best-checkpoint broadcast into test_worker, and the rank-0
    ResultSaver gating (exactly one CSV)."""
    _spawn(_w_train, 29515, str(tmp_path), "train_test")
    import glob
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1


def test_two_rank_training_torch_ddp(tmp_path):
    """--use-torch-ddp: the reference-parity torch DDP wrapper path (vs
    the native FlatReplica default) trains under gloo world_size 2.
    (torch SyncBatchNorm needs GPU process groups, so sync-bn stays off
    here; its GPU path is covered by the engine default tests.)"""
    _spawn(_w_train, 29518, str(tmp_path), "train",
           ("--use-torch-ddp", "true"))
    import glob
    ckpts = glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth"))
    assert ckpts
