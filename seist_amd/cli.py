"""CLI: argument parser + process entry.

Flag-for-flag parity with /root/reference/main.py:8-179 (same names and
defaults for every reference flag), plus MI355X-specific additions:
``--precision`` (bf16 policy), ``--sync-metrics-per-step`` /
``--sync-bn`` (distributed sync discipline), ``--dataset-size`` /
``--dataset-samples`` (synthetic dataset sizing). ``--use-torch-compile``
is accepted but is a no-op: kernel fusion here comes from the HIP op
library, not a tracing compiler.
"""

import argparse
import os

import torch

from .config import Config
from .engine import test_worker, train_worker
from .parallel import dist as pdist
from .utils.logger import logger
from .utils.misc import get_time_str, setup_seed, strfargs


def bool_(x):
    return False if str(x).strip().lower() in ("0", "false", "f", "no", "n") \
        else bool(x)


def get_args(argv=None):
    parser = argparse.ArgumentParser(
        description="seist_amd training/testing arguments")

    # Mode
    parser.add_argument("--mode", type=str, default="train_test",
                        help="train/test/train_test")
    # Model
    parser.add_argument("--model-name", default="seist_m_dpk", type=str)
    parser.add_argument("--checkpoint", default="", type=str)
    parser.add_argument("--use-torch-compile", type=bool_, default=False,
                        help="accepted for CLI parity; fusion is native")
    # Seed
    parser.add_argument("--seed", default=0, type=int)
    # Logs
    parser.add_argument("--log-base", default="./logs", type=str)
    parser.add_argument("--log-step", default=4, type=int)
    parser.add_argument("--use-tensorboard", default=True, type=bool_)
    # Results
    parser.add_argument("--save-test-results", default=True, type=bool_)
    # Distributed
    parser.add_argument("--find-unused-parameters", type=bool_, default=False)
    parser.add_argument("--sync-bn", type=bool_, default=True)
    parser.add_argument("--use-torch-ddp", type=bool_, default=False,
                        help="use torch DDP + SyncBatchNorm instead of the "
                             "native flat-bucket replica + fused SyncBN")
    parser.add_argument("--trace-step-time", type=bool_, default=False,
                        help="log a data/h2d/fwd/loss/bwd/opt/comm/"
                             "postprocess/metrics step-time breakdown")
    parser.add_argument("--train-metrics-interval", type=int, default=1,
                        help="compute train metrics/postprocess every Nth "
                             "step (1 = reference parity)")
    parser.add_argument("--sync-metrics-per-step", type=bool_, default=False,
                        help="reference-parity per-step metric collectives "
                             "(default: per-epoch)")
    # Device / precision
    parser.add_argument("--device", type=str, default="cuda:0")
    parser.add_argument("--precision", type=str, default="fp32",
                        choices=["fp32", "bf16"])
    # Dataset
    parser.add_argument("--data", default="./data", type=str)
    parser.add_argument("--dataset-name", default="diting_light", type=str)
    parser.add_argument("--data-split", type=bool_, default=True)
    parser.add_argument("--train-size", type=float, default=0.8)
    parser.add_argument("--val-size", type=float, default=0.1)
    parser.add_argument("--dataset-size", type=int, default=256,
                        help="synthetic dataset only: number of events")
    parser.add_argument("--dataset-samples", type=int, default=12288,
                        help="synthetic dataset only: raw trace length")
    # Loader
    parser.add_argument("--shuffle", type=bool_, default=True)
    parser.add_argument("--workers", default=8, type=int)
    parser.add_argument("--pin-memory", default=True, type=bool_)
    # Preprocess
    parser.add_argument("--in-samples", default=8192, type=int)
    parser.add_argument("--label-width", type=float, default=0.5)
    parser.add_argument("--label-shape", type=str, default="gaussian")
    parser.add_argument("--coda-ratio", default=2.0, type=float)
    parser.add_argument("--norm-mode", default="std", type=str)
    parser.add_argument("--min-snr", type=float, default=-float("inf"))
    parser.add_argument("--p-position-ratio", type=float, default=-1)
    # Augmentation
    parser.add_argument("--augmentation", type=bool_, default=True)
    parser.add_argument("--add-event-rate", default=0.0, type=float)
    parser.add_argument("--max-event-num", default=1, type=int)
    parser.add_argument("--shift-event-rate", default=0.2, type=float)
    parser.add_argument("--add-noise-rate", default=0.4, type=float)
    parser.add_argument("--add-gap-rate", default=0.4, type=float)
    parser.add_argument("--min-event-gap", default=0.5, type=float)
    parser.add_argument("--drop-channel-rate", default=0.4, type=float)
    parser.add_argument("--scale-amplitude-rate", default=0.4, type=float)
    parser.add_argument("--pre-emphasis-rate", default=0.4, type=float)
    parser.add_argument("--pre-emphasis-ratio", default=0.97, type=float)
    parser.add_argument("--generate-noise-rate", default=0.05, type=float)
    parser.add_argument("--mask-percent", default=0, type=int)
    parser.add_argument("--noise-percent", default=0, type=int)
    # Train
    parser.add_argument("--epochs", default=200, type=int)
    parser.add_argument("--patience", default=30, type=int)
    parser.add_argument("--steps", default=0, type=int)
    parser.add_argument("--start-epoch", default=0, type=int)
    parser.add_argument("--batch-size", default=500, type=int,
                        help="batch size of each worker (process)")
    parser.add_argument("--optim", default="Adam", type=str)
    parser.add_argument("--momentum", default=0.9, type=float)
    parser.add_argument("--weight_decay", default=0.0, type=float)
    parser.add_argument("--use-lr-scheduler", default=True, type=bool_)
    parser.add_argument("--lr-scheduler-mode", default="exp_range", type=str)
    parser.add_argument("--base-lr", default=8e-5, type=float)
    parser.add_argument("--max-lr", default=1e-3, type=float)
    parser.add_argument("--warmup-steps", default=2000, type=float)
    parser.add_argument("--down-steps", default=3000, type=float)
    # Val/Test
    parser.add_argument("--time-threshold", default=0.1, type=float)
    parser.add_argument("--min-peak-dist", default=1.0, type=float)
    parser.add_argument("--ppk-threshold", default=0.3, type=float)
    parser.add_argument("--spk-threshold", default=0.3, type=float)
    parser.add_argument("--det-threshold", default=0.5, type=float)
    parser.add_argument("--max-detect-event-num", default=1, type=int)

    args = parser.parse_args(argv)

    if not 0 <= args.p_position_ratio <= 1:
        args.p_position_ratio = -1

    args.log_base = os.path.abspath(args.log_base)
    args.data = os.path.abspath(args.data)
    if args.checkpoint:
        args.checkpoint = os.path.abspath(args.checkpoint)
    args.dataset_kwargs = (
        {"size": args.dataset_size, "num_samples": args.dataset_samples}
        if args.dataset_name == "synthetic" else {})
    return args


def main_worker(args, device):
    log_dir = (os.path.join(
        args.log_base,
        f"{get_time_str()}_{args.model_name}_{args.dataset_name}")
        if not args.checkpoint
        else args.checkpoint.split("checkpoints")[0])
    if pdist.is_main_process():
        logger.set_logdir(log_dir)
    logger.set_logger("global")

    if pdist.is_main_process():
        logger.info(f"device: {device}")
        logger.info(f"pid: {os.getpid()}")
        logger.info(f"\n{strfargs(args, Config)}")

    mode = args.mode.split("_")
    if "train" in mode:
        setup_seed(args.seed)
        ckpt_path = train_worker(args, device)
        args.checkpoint = ckpt_path
    if "test" in mode:
        setup_seed(args.seed)
        test_worker(args, device)
    if not ({"train", "test"} & set(mode)):
        raise ValueError(
            f"`mode` must be 'train','test' or 'train_test', "
            f"got '{args.mode}'")


def main(argv=None):
    args = get_args(argv)
    args.distributed = pdist.init_distributed_mode()
    if args.distributed and torch.cuda.is_available():
        args.device = f"cuda:{pdist.get_local_rank()}"
    elif not torch.cuda.is_available():
        args.device = "cpu"
    device = torch.device(args.device)
    main_worker(args, device)


if __name__ == "__main__":
    main()
