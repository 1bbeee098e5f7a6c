"""Metrics engine vs the reference implementation on random data."""

import numpy as np
import pytest
import torch

from seist_amd.engine.metrics import Metrics

from _refload import load_ref_utils, reference_available

DEV = torch.device("cpu")


def _ours(task, names):
    return Metrics(task=task, metric_names=names, sampling_rate=50,
                   time_threshold=0.1, num_samples=8192, device=DEV)


def test_picking_metrics_hand_case():
    m = _ours("ppk", ["precision", "recall", "f1", "mae"])
    targets = torch.tensor([[1000], [2000], [-10000000]])
    preds = torch.tensor([[1002], [2100], [-10000000]])
    # thr = 0.1*50 = 5 samples: first within, second out, third invalid
    m.compute(targets, preds)
    assert m.get_metric("precision") == pytest.approx(1 / 2, abs=1e-4)
    assert m.get_metric("recall") == pytest.approx(1 / 2, abs=1e-4)


def test_det_metrics_hand_case():
    m = _ours("det", ["precision", "recall", "f1"])
    targets = torch.tensor([[100, 200], [1, 0]])   # second row: empty pad
    preds = torch.tensor([[150, 250], [1, 0]])
    m.compute(targets, preds)
    # overlap exists in sample space -> tp >= 1
    assert m.get_metric("recall") > 0


def test_onehot_metrics():
    m = _ours("pmp", ["precision", "recall", "f1"])
    targets = torch.tensor([[1.0, 0.0], [0.0, 1.0], [1.0, 0.0]])
    preds = torch.tensor([[0.9, 0.1], [0.2, 0.8], [0.3, 0.7]])
    m.compute(targets, preds)
    # class0: tp=1, pred=1, poss=2 ; class1: tp=1, pred=2, poss=1
    assert m.get_metric("precision") == pytest.approx((1 / 1 + 1 / 2) / 2,
                                                      abs=1e-4)


def test_baz_wraparound():
    m = _ours("baz", ["mean", "rmse", "mae"])
    targets = torch.tensor([[359.0]])
    preds = torch.tensor([[1.0]])
    m.compute(targets, preds)
    assert m.get_metric("mae") == pytest.approx(2.0, abs=1e-4)


def test_add_and_merge():
    a = _ours("emg", ["mean", "rmse", "mae", "r2"])
    b = _ours("emg", ["mean", "rmse", "mae", "r2"])
    t1, p1 = torch.rand(8, 1) * 5, torch.rand(8, 1) * 5
    t2, p2 = torch.rand(8, 1) * 5, torch.rand(8, 1) * 5
    a.compute(t1, p1)
    b.compute(t2, p2)
    merged = a + b
    whole = _ours("emg", ["mean", "rmse", "mae", "r2"])
    whole.compute(torch.cat([t1, t2]), torch.cat([p1, p2]))
    # accumulators over disjoint batches must equal one big batch
    for k in ("mean", "rmse", "mae"):
        assert merged.get_metric(k) == pytest.approx(whole.get_metric(k),
                                                     abs=1e-5)


@pytest.mark.skipif(not reference_available(), reason="reference absent")
@pytest.mark.parametrize("task,shape", [
    ("ppk", (16, 3)), ("spk", (16, 1)), ("emg", (16, 1)), ("baz", (16, 1)),
    ("pmp", (16, 2)),
])
def test_matches_reference(task, shape):
    refm = load_ref_utils().metrics
    names = {"ppk": ["precision", "recall", "f1", "mean", "rmse", "mae",
                     "mape"],
             "spk": ["precision", "recall", "f1", "mae"],
             "emg": ["mean", "rmse", "mae", "r2"],
             "baz": ["mean", "rmse", "mae", "r2"],
             "pmp": ["precision", "recall", "f1"]}[task]
    torch.manual_seed(0)
    if task in ("ppk", "spk"):
        targets = torch.randint(-5, 8192, shape)
        preds = targets + torch.randint(-10, 10, shape)
    elif task == "pmp":
        targets = torch.eye(2)[torch.randint(0, 2, (shape[0],))]
        preds = torch.rand(shape)
    else:
        targets = torch.rand(shape) * 300
        preds = targets + torch.randn(shape) * 20

    ours = _ours(task, names)
    ours.compute(targets.clone(), preds.clone())
    theirs = refm.Metrics(task=task, metric_names=names, sampling_rate=50,
                          time_threshold=0.1, num_samples=8192, device=DEV)
    theirs.compute(targets.clone(), preds.clone())
    for n in names:
        assert ours.get_metric(n) == pytest.approx(
            theirs.get_metric(n), rel=1e-5, abs=1e-6), (task, n)


def test_det_matches_reference():
    if not reference_available():
        pytest.skip("reference absent")
    refm = load_ref_utils().metrics
    torch.manual_seed(1)
    targets = torch.tensor([[100, 400, 1, 0], [5000, 6000, 1, 0]])
    preds = torch.tensor([[120, 380, 1, 0], [1, 0, 1, 0]])
    names = ["precision", "recall", "f1"]
    ours = _ours("det", names)
    ours.compute(targets.clone(), preds.clone())
    theirs = refm.Metrics(task="det", metric_names=names, sampling_rate=50,
                          time_threshold=0.1, num_samples=8192, device=DEV)
    theirs.compute(targets.clone(), preds.clone())
    for n in names:
        assert ours.get_metric(n) == pytest.approx(theirs.get_metric(n),
                                                   abs=1e-6)


@pytest.mark.skipif(not reference_available(), reason="reference absent")
@pytest.mark.parametrize("seed", [1, 2, 3, 4])
def test_phase_metrics_reference_sweep(seed):
    """Batched greedy phase ordering (_order_phases, K argmin rounds) vs
    the reference loop across random draws plus tie/padding adversaries:
    all-padded rows, duplicate predicted times, equidistant pairs."""
    refm = load_ref_utils().metrics
    names = ["precision", "recall", "f1", "mean", "rmse", "mae", "mape"]
    torch.manual_seed(seed)
    K = 3
    targets = torch.randint(-5, 8192, (12, K))
    preds = targets + torch.randint(-12, 12, (12, K))
    preds[0] = -10000000                  # all padding
    preds[1, 1] = preds[1, 0]             # duplicate predicted time
    targets[2, 1] = targets[2, 0]         # duplicate target time
    if K >= 2:
        # two preds exactly equidistant from one target (argmin tie)
        targets[3, 0] = 1000
        preds[3, 0] = 996
        preds[3, 1] = 1004
    ours = _ours("ppk", names)
    ours.compute(targets.clone(), preds.clone())
    theirs = refm.Metrics(task="ppk", metric_names=names, sampling_rate=50,
                          time_threshold=0.1, num_samples=8192, device=DEV)
    theirs.compute(targets.clone(), preds.clone())
    for n in names:
        assert ours.get_metric(n) == pytest.approx(
            theirs.get_metric(n), rel=1e-5, abs=1e-6), (seed, n)
