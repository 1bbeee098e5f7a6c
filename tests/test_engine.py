"""End-to-end engine tests on CPU with the synthetic dataset."""

import glob
import os

import pytest
import torch

from seist_amd.cli import get_args, main_worker


def _args(tmp_path, extra):
    argv = [
        "--mode", "train_test", "--dataset-name", "synthetic",
        "--dataset-size", "24", "--dataset-samples", "9000",
        "--batch-size", "4", "--epochs", "1", "--workers", "0",
        "--device", "cpu", "--use-tensorboard", "false",
        "--log-base", str(tmp_path), "--warmup-steps", "2",
        "--down-steps", "3", "--log-step", "100", "--augmentation", "false",
    ] + extra
    return get_args(argv)


def test_phasenet_train_test_cpu(tmp_path):
    args = _args(tmp_path, ["--model-name", "phasenet"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    # checkpoint written, results CSV written
    ckpts = glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth"))
    assert len(ckpts) >= 1
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1
    losses = glob.glob(str(tmp_path / "*" / "loss" / "*.npy"))
    assert len(losses) == 3


def test_seist_regression_task_cpu(tmp_path):
    args = _args(tmp_path, ["--model-name", "seist_s_emg"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1


def test_regression_task_with_noise_augmentation_cpu(tmp_path):
    """Value-type labels + augmentation: generate_noise_rate must be forced
    off (a noise-replaced window has no magnitude/baz/dis target — the
    collate would see [] vs [1])."""
    args = _args(tmp_path, ["--model-name", "seist_s_dis",
                            "--augmentation", "true",
                            "--generate-noise-rate", "0.9"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1


def test_resume_from_checkpoint(tmp_path):
    args = _args(tmp_path, ["--model-name", "phasenet", "--mode", "train"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    ckpts = sorted(glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth")))
    assert ckpts
    # resume training from the saved checkpoint
    args2 = _args(tmp_path, ["--model-name", "phasenet", "--mode", "train",
                             "--start-epoch", "1", "--epochs", "2"])
    args2.checkpoint = ckpts[-1]
    args2.distributed = False
    main_worker(args2, torch.device("cpu"))


def test_early_stopping(tmp_path):
    args = _args(tmp_path, ["--model-name", "phasenet", "--mode", "train",
                            "--epochs", "4", "--patience", "0"])
    args.distributed = False
    # with patience 0 the run must stop after the first non-improving epoch
    main_worker(args, torch.device("cpu"))


def test_ditingmotion_train_test_cpu(tmp_path):
    # 2-channel [z, dz] input, dual clarity+polarity heads, onehot labels
    args = _args(tmp_path, ["--model-name", "ditingmotion",
                            "--in-samples", "256"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1


def test_magnet_train_test_cpu(tmp_path):
    # conv+BiLSTM regression with MousaviLoss (mag, log-var) head
    args = _args(tmp_path, ["--model-name", "magnet"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) == 1


def test_scaled_activation_heads_cpu(tmp_path):
    # baz head uses cos/sin transforms end to end
    args = _args(tmp_path, ["--model-name", "baz_network", "--mode",
                            "train"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))


def test_step_time_tracing(tmp_path, capsys):
    args = _args(tmp_path, ["--model-name", "phasenet", "--mode", "train",
                            "--trace-step-time", "true"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    # breakdown line must be in the train log
    logs = glob.glob(str(tmp_path / "*" / "train.log"))
    text = open(logs[0]).read() if logs else ""
    assert "step breakdown" in text


def test_torchrun_bench_cpu(tmp_path):
    """The driver's exact multi-rank launch contract must work (gloo/CPU)."""
    import subprocess
    import sys
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", "bench.py", "--gpus", "2",
         "--steps", "1", "--warmup", "0"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert res.returncode == 0, res.stderr[-2000:]
    import json as _json
    line = [l for l in res.stdout.splitlines() if l.startswith("{")][-1]
    out = _json.loads(line)
    assert out["n_gpus"] == 2 and out["value"] > 0


def test_demo_predict_end_to_end(tmp_path):
    """Train one tiny epoch, then run demo_predict.py against the saved
    checkpoint (the README deployment recipe)."""
    import subprocess
    import sys
    args = _args(tmp_path, ["--model-name", "seist_s_dpk", "--mode", "train"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    ckpts = sorted(glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth")))
    assert ckpts
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "demo_predict.py", "--checkpoint", ckpts[-1],
         "--model-name", "seist_s_dpk", "--save-dir", str(tmp_path)],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert res.returncode == 0, res.stderr[-1500:]
    assert glob.glob(str(tmp_path / "*demo_prediction*.png"))


def test_demo_predict_with_real_pretrained(tmp_path):
    """demo_predict.py against one of the reference's shipped pretrained
    checkpoints — the deployment recipe on real weights."""
    import subprocess
    import sys
    ckpt = "/root/reference/pretrained/seist_m_dpk_diting.pth"
    if not os.path.exists(ckpt):
        pytest.skip("reference pretrained dir absent")
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        [sys.executable, "demo_predict.py", "--checkpoint", ckpt,
         "--model-name", "seist_m_dpk", "--save-dir", str(tmp_path)],
        capture_output=True, text=True, timeout=600, cwd=root)
    assert res.returncode == 0, res.stderr[-1500:]
    assert glob.glob(str(tmp_path / "*demo_prediction*.png"))


def test_bf16_precision_train_cpu(tmp_path):
    """--precision bf16: the engine converts the model (fp32 BN/LSTM kept),
    casts inputs, and still produces finite losses and a checkpoint that
    loads back into an fp32 model (fp32 master save)."""
    args = _args(tmp_path, ["--model-name", "seist_s_dpk", "--mode", "train",
                            "--precision", "bf16"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    ckpts = glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth"))
    assert len(ckpts) >= 1
    from seist_amd.models import create_model, load_checkpoint
    m = create_model("seist_s_dpk", in_channels=3, in_samples=9000)
    ckpt = load_checkpoint(ckpts[-1])
    # the save must be reference-format fp32 (the fused optimizer's fp32
    # masters substitute for bf16 params), never raw bf16 tensors
    for k, v in ckpt["model_dict"].items():
        if torch.is_tensor(v) and v.is_floating_point():
            assert v.dtype == torch.float32, k
    m.load_state_dict(ckpt["model_dict"])
    assert all(p.dtype == torch.float32 for p in m.parameters())
    losses = glob.glob(str(tmp_path / "*" / "loss" / "*.npy"))
    import numpy as np
    for f in losses:
        assert np.isfinite(np.load(f)).all()


def test_scheduler_matches_reference_conversion():
    """build_scheduler's fractional warmup/down conversion and CyclicLR
    parameters reproduce the reference's LR sequence exactly
    (reference training/train.py:328-354)."""
    import pytest
    from types import SimpleNamespace
    from seist_amd.engine.train import build_scheduler
    for (w, d, steps) in ((2.0, 3.0, 100), (0.2, 0.3, 50), (0.0, 0.0, 40),
                          (5.0, 0.5, 64)):
        args = SimpleNamespace(use_lr_scheduler=True, warmup_steps=w,
                               down_steps=d, steps=steps, base_lr=8e-5,
                               max_lr=1e-3, lr_scheduler_mode="exp_range",
                               start_epoch=0)
        opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=8e-5)
        sch = build_scheduler(args, opt, steps_per_epoch=10)
        ours = []
        for _ in range(steps):
            opt.step()
            sch.step()
            ours.append(sch.get_last_lr()[0])
        rw = w if w >= 1 else (int(steps * w) if w > 0 else 1)
        rd = d if d >= 1 else (int(steps * d) if d > 0 else steps - rw)
        opt2 = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=8e-5)
        ref = torch.optim.lr_scheduler.CyclicLR(
            opt2, base_lr=8e-5, max_lr=1e-3, step_size_up=int(rw),
            step_size_down=int(rd), mode="exp_range",
            gamma=8e-5 ** ((steps * 2) ** -1), cycle_momentum=False,
            last_epoch=-1)
        theirs = []
        for _ in range(steps):
            opt2.step()
            ref.step()
            theirs.append(ref.get_last_lr()[0])
        assert ours == pytest.approx(theirs, rel=1e-12), (w, d)


def test_standalone_test_mode(tmp_path):
    """--mode test --checkpoint ... (the reference README's test recipe,
    README.md:186-224): builds the test split, loads the checkpoint, and
    writes the results CSV."""
    args = _args(tmp_path, ["--model-name", "phasenet", "--mode", "train"])
    args.distributed = False
    main_worker(args, torch.device("cpu"))
    ckpts = sorted(glob.glob(str(tmp_path / "*" / "checkpoints" / "*.pth")))
    assert ckpts
    args2 = _args(tmp_path, ["--model-name", "phasenet", "--mode", "test"])
    args2.checkpoint = ckpts[-1]
    args2.distributed = False
    main_worker(args2, torch.device("cpu"))
    csvs = glob.glob(str(tmp_path / "*" / "test_results_*.csv"))
    assert len(csvs) >= 1
