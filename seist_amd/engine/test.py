"""Test entry (flow parity with /root/reference/training/test.py:10-88)."""

import torch

from ..config import Config
from ..data import SeismicDataset
from ..models import create_model, load_checkpoint
from ..parallel import dist as pdist
from ..parallel.ddp import wrap_distributed
from ..utils.logger import logger
from ..utils.misc import count_parameters
from .precision import convert_to_bf16
from .validate import validate


def test_worker(args, device) -> float:
    logger.set_logger("test")

    model_inputs, model_labels, model_tasks = Config.get_model_config_(
        args.model_name, "inputs", "labels", "eval")
    in_channels = Config.get_num_inchannels(model_name=args.model_name)
    test_dataset = SeismicDataset(args=args, input_names=model_inputs,
                                  label_names=model_labels,
                                  task_names=model_tasks, mode="test")
    test_sampler = (torch.utils.data.DistributedSampler(test_dataset)
                    if pdist.is_dist() else None)
    test_loader = torch.utils.data.DataLoader(
        test_dataset, batch_size=args.batch_size,
        shuffle=(test_sampler is None and args.shuffle),
        pin_memory=args.pin_memory, num_workers=args.workers,
        sampler=test_sampler)

    if not args.checkpoint:
        raise ValueError("checkpoint is None.")
    checkpoint = load_checkpoint(args.checkpoint, device=device)
    logger.info(f"Model loaded: {args.checkpoint}")

    loss_fn = Config.get_loss(model_name=args.model_name).to(device)

    model = create_model(model_name=args.model_name, in_channels=in_channels,
                         in_samples=args.in_samples)
    if checkpoint is not None and "model_dict" in checkpoint:
        model.load_state_dict(checkpoint["model_dict"])
        logger.info("model.load_state_dict")
    if pdist.is_main_process():
        logger.info(f"Model parameters: {count_parameters(model)}")

    if args.precision == "bf16":
        model = convert_to_bf16(model)
    model = model.to(device)
    model = wrap_distributed(model, args)

    test_loss, test_metrics_dict = validate(
        args, model_tasks, model, loss_fn, test_loader, 0, device,
        testing=True)

    if pdist.is_main_process():
        s = "* "
        for task in model_tasks:
            s += f"[{task.upper()}]{test_metrics_dict[task]} "
        logger.info(s)
    return test_loss
