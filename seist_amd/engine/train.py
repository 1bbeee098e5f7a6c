"""Training loop + worker.

Flow parity with /root/reference/training/train.py:20-484 (same CLI
contract, CyclicLR schedule, early stopping, best-checkpoint policy,
loss .npy dumps, ckpt-path broadcast), redesigned for MI355X:

* bf16 compute policy via ``--precision bf16`` (fp32 BN stats + fp32 Adam
  master weights in the fused optimizer); the reference's TF32 has no
  gfx950 equivalent.
* distributed sync discipline: the reference all-reduces loss + batch size
  + every task's metric counters EVERY STEP, each bracketed by barriers
  (train.py:124-131, metrics.py:87-96). Here per-step collectives are one
  loss/batch-size all-reduce with no barrier, and metric counters merge
  locally, synchronizing once per epoch (restore per-step sync with
  --sync-metrics-per-step).
* fused multi-tensor Adam (K17) instead of eager per-tensor Adam.
"""

import datetime
import math
import os
from typing import Union

import numpy as np
import torch

from ..config import Config
from ..data import SeismicDataset
from ..models import create_model, load_checkpoint, save_checkpoint
from ..models._blocks import manage_bn_counters
from ..ops import FusedAdam
from ..parallel import dist as pdist
from ..parallel.ddp import FlatReplica, enable_native_syncbn, wrap_distributed
from ..utils.logger import logger
from ..utils.meters import AverageMeter, ProgressMeter
from ..utils.misc import count_parameters, get_safe_path, strftimedelta
from .metrics import Metrics
from .postprocess import process_outputs
from .precision import cast_inputs, convert_to_bf16
from .scalars import ScalarWriter
from .tracing import StepTimer
from .validate import validate


def _to_device(x, device, dtype=None):
    if isinstance(x, (list, tuple)):
        return [xi.to(device, non_blocking=True) for xi in x]
    return x.to(device, non_blocking=True)


def train(args, tasks, model, optimizer, scheduler, loss_fn, train_loader,
          epoch, device, scalar_writer, replica=None) -> Union[list, dict]:
    model.train()
    # one _foreach_add_ per step for all BN counters instead of one tiny
    # kernel per layer per forward
    bn_tick = manage_bn_counters(model)

    train_loss_per_step = []
    average_meters = {}
    metrics_merged = {}
    sampling_rate = train_loader.dataset.sampling_rate()
    compute_dtype = (torch.bfloat16 if args.precision == "bf16"
                     else torch.float32)

    def new_metrics(task):
        return Metrics(task=task, metric_names=Config.get_metrics(task),
                       sampling_rate=sampling_rate,
                       time_threshold=args.time_threshold,
                       num_samples=args.in_samples, device=device)

    for task in tasks:
        metrics_merged[task] = new_metrics(task)
        for metric in metrics_merged[task].metric_names():
            average_meters[f"{task}_{metric}"] = AverageMeter(
                f"[{task.upper()}]{metric}", ":6.4f")
    average_meters["loss"] = AverageMeter("Loss", ":6.4f")
    progress = ProgressMeter(len(train_loader),
                             list(average_meters.values()),
                             prefix=f"Train: [{epoch}/{args.epochs}]")

    (label_names, tgts_trans_for_loss, outs_trans_for_loss,
     outs_trans_for_res) = Config.get_model_config_(
        args.model_name, "labels", "targets_transform_for_loss",
        "outputs_transform_for_loss", "outputs_transform_for_results")

    sync_per_step = pdist.is_dist() and args.sync_metrics_per_step
    timer = StepTimer(getattr(args, "trace_step_time", False),
                      torch.device(device))

    loader_it = iter(train_loader)
    step = -1
    while True:
        with timer.phase("data"):
            batch = next(loader_it, None)
        if batch is None:
            break
        step += 1
        x, loss_targets, metrics_targets, _ = batch
        with timer.phase("h2d"):
            x = cast_inputs(_to_device(x, device), compute_dtype)
            loss_targets = _to_device(loss_targets, device)

        with timer.phase("forward"):
            outputs = model(x)

        with timer.phase("loss"):
            outputs_for_loss = (outs_trans_for_loss(outputs)
                                if outs_trans_for_loss is not None
                                else outputs)
            loss_targets = (tgts_trans_for_loss(loss_targets)
                            if tgts_trans_for_loss is not None
                            else loss_targets)
            if isinstance(outputs_for_loss, (list, tuple)):
                outputs_for_loss = [o.float() for o in outputs_for_loss]
            else:
                outputs_for_loss = outputs_for_loss.float()
            loss = loss_fn(outputs_for_loss, loss_targets)

        with timer.phase("backward"):
            if replica is not None:
                replica.zero_grad()
            else:
                optimizer.zero_grad(set_to_none=True)
            loss.backward()
        if replica is not None:
            with timer.phase("comm"):
                replica.allreduce()
        with timer.phase("optimizer"):
            optimizer.step()
            bn_tick()
            if scheduler is not None:
                scheduler.step()
                lr = scheduler.get_last_lr()[0]
            else:
                lr = optimizer.param_groups[0]["lr"]

        step_batch_size = (x[0] if isinstance(x, (list, tuple))
                           else x).size(0)
        with timer.phase("comm"):
            if pdist.is_dist():
                loss = pdist.reduce_tensor(loss, "AVG")
                sbs = torch.tensor(step_batch_size, device=device,
                                   dtype=torch.int32)
                step_batch_size = pdist.reduce_tensor(sbs).item()

        average_meters["loss"].update(loss.item(), step_batch_size)
        train_loss_per_step.append(loss.item())

        interval = max(1, getattr(args, "train_metrics_interval", 1))
        if step % interval != 0:
            timer.step()
            if step % args.log_step == 0 and pdist.is_main_process():
                logger.info(progress.get_str(
                    batch_idx=step, name=f"{args.model_name}_train"))
            continue

        with timer.phase("postprocess"):
            outputs_for_metrics = (outs_trans_for_res(outputs)
                                   if outs_trans_for_res is not None
                                   else outputs)
            if isinstance(outputs_for_metrics, (list, tuple)):
                outputs_for_metrics = [o.float()
                                       for o in outputs_for_metrics]
            else:
                outputs_for_metrics = outputs_for_metrics.float()
            results = process_outputs(args, outputs_for_metrics, label_names,
                                      sampling_rate)

        with timer.phase("metrics"):
            tasks_metrics = {}
            for task in tasks:
                metrics = new_metrics(task)
                tasks_metrics[task] = metrics
                metrics.compute(targets=metrics_targets[task],
                                preds=results[task], reduce=sync_per_step)
                for metric in metrics.metric_names():
                    average_meters[f"{task}_{metric}"].update(
                        metrics.get_metric(metric), step_batch_size)
                metrics_merged[task].add(metrics)
        timer.step()

        if scalar_writer is not None and pdist.is_main_process():
            gstep = epoch * len(train_loader) + step
            scalar_writer.add_scalar("learning-rate/step", lr, gstep)
            scalar_writer.add_scalar("train-loss/step", loss.item(), gstep)
            for task in tasks:
                scalar_writer.add_scalars(
                    f"train.{task}.metrics/step",
                    tasks_metrics[task].get_all_metrics(), gstep)

        if step % args.log_step == 0 and pdist.is_main_process():
            logger.info(progress.get_str(batch_idx=step,
                                         name=f"{args.model_name}_train"))

    # one metric sync per epoch instead of per step
    if pdist.is_dist() and not sync_per_step:
        for task in tasks:
            metrics_merged[task].synchronize_between_processes()

    if timer.enabled and pdist.is_main_process():
        msg = timer.format()
        if msg:
            logger.info(f"* {msg}")

    return train_loss_per_step, metrics_merged


def build_optimizer(args, model):
    optim_lower = args.optim.lower()
    params = [{"params": model.parameters(), "initial_lr": args.base_lr}]
    if optim_lower in ("adam", "adamw"):
        return FusedAdam(params, lr=args.base_lr,
                         weight_decay=args.weight_decay,
                         adamw=(optim_lower == "adamw"))
    if optim_lower == "sgd":
        return torch.optim.SGD(params, lr=args.base_lr,
                               momentum=args.momentum,
                               weight_decay=args.weight_decay)
    raise ValueError(f"Unsupported optimizer:'{args.optim}'")


def build_scheduler(args, optimizer, steps_per_epoch):
    if not args.use_lr_scheduler:
        return None
    if args.warmup_steps < 1:
        args.warmup_steps = (int(args.steps * args.warmup_steps)
                             if args.warmup_steps > 0 else 1)
        logger.info(f"`args.warmup_steps` -> {args.warmup_steps}")
    if args.down_steps < 1:
        args.down_steps = (int(args.steps * args.down_steps)
                           if args.down_steps > 0
                           else args.steps - args.warmup_steps)
        logger.info(f"`args.down_steps` -> {args.down_steps}")
    return torch.optim.lr_scheduler.CyclicLR(
        optimizer=optimizer, base_lr=args.base_lr, max_lr=args.max_lr,
        step_size_up=int(args.warmup_steps),
        step_size_down=int(args.down_steps), mode=args.lr_scheduler_mode,
        gamma=args.base_lr ** ((args.steps * 2) ** -1),
        cycle_momentum=False,
        last_epoch=args.start_epoch * steps_per_epoch - 1)


def train_worker(args, device) -> str:
    logger.set_logger("train")
    log_dir = logger.logdir() or "."
    checkpoint_save_dir = os.path.join(log_dir, "checkpoints")
    scalar_writer = (ScalarWriter(os.path.join(log_dir, "tensorboard"))
                     if args.use_tensorboard else None)
    if pdist.is_main_process():
        os.makedirs(checkpoint_save_dir, exist_ok=True)
        if args.use_tensorboard:
            # convenience launcher next to the run dir (reference
            # train.py:193-194 writes the same helper)
            helper = os.path.join(log_dir, "run_tensorboard.sh")
            with open(helper, "w") as f:
                f.write("#!/bin/bash\ntensorboard --logdir "
                        f"{os.path.join(log_dir, 'tensorboard')} "
                        "--port ${1:-6006}\n")
            os.chmod(helper, 0o755)

    model_inputs, model_labels, model_tasks = Config.get_model_config_(
        args.model_name, "inputs", "labels", "eval")
    in_channels = Config.get_num_inchannels(model_name=args.model_name)

    train_dataset = SeismicDataset(args=args, input_names=model_inputs,
                                   label_names=model_labels,
                                   task_names=model_tasks, mode="train")
    val_dataset = SeismicDataset(args=args, input_names=model_inputs,
                                 label_names=model_labels,
                                 task_names=model_tasks, mode="val")
    logger.info(f"train size: {len(train_dataset)}, "
                f"val size:{len(val_dataset)}")

    train_sampler = (torch.utils.data.DistributedSampler(train_dataset)
                     if pdist.is_dist() else None)
    val_sampler = (torch.utils.data.DistributedSampler(val_dataset)
                   if pdist.is_dist() else None)
    train_loader = torch.utils.data.DataLoader(
        train_dataset, batch_size=args.batch_size,
        shuffle=(train_sampler is None and args.shuffle),
        pin_memory=args.pin_memory, num_workers=args.workers,
        sampler=train_sampler, persistent_workers=args.workers > 0,
        drop_last=False)
    val_loader = torch.utils.data.DataLoader(
        val_dataset, batch_size=args.batch_size,
        shuffle=False, pin_memory=args.pin_memory,
        num_workers=args.workers, sampler=val_sampler,
        persistent_workers=args.workers > 0)

    if args.steps > 0:
        args.epochs = math.ceil(args.steps / len(train_loader))
    args.steps = args.epochs * len(train_loader)
    logger.info(f"`args.epochs` -> {args.epochs}, `args.steps` -> {args.steps}")

    checkpoint = None
    if args.checkpoint:
        checkpoint = load_checkpoint(args.checkpoint, device=device)
        logger.info(f"Model loaded: {args.checkpoint}")

    loss_fn = Config.get_loss(model_name=args.model_name).to(device)
    best_loss = (checkpoint["loss"] if checkpoint and "loss" in checkpoint
                 and checkpoint["loss"] is not None else float("inf"))

    model = create_model(model_name=args.model_name, in_channels=in_channels,
                         in_samples=args.in_samples)
    if checkpoint is not None and "model_dict" in checkpoint:
        model.load_state_dict(checkpoint["model_dict"])
        logger.info("model.load_state_dict")

    if pdist.is_main_process():
        # back up the model source next to the run artifacts
        # (reference train.py:288-291)
        try:
            import inspect
            import shutil
            src = inspect.getfile(model.__class__)
            shutil.copy2(src, get_safe_path(
                os.path.join(log_dir, "model_backup.py")))
        except Exception as e:
            logger.warning(f"model source backup skipped: {e}")
        logger.info(f"Model parameters: {count_parameters(model)}")

    if getattr(args, "use_torch_compile", False):
        # the reference wraps the model in torch.compile (train.py:369);
        # this build's compute path is a fixed fused HIP kernel library, so
        # a tracing compiler adds nothing — accept the flag but say so
        logger.warning("--use-torch-compile is a no-op: the MI355X build "
                       "runs a fixed fused-kernel path (see docs/DESIGN.md)")
    if args.precision == "bf16":
        model = convert_to_bf16(model)
    model = model.to(device)

    optimizer = build_optimizer(args, model)
    if checkpoint is not None and checkpoint.get("optimizer_dict"):
        try:
            optimizer.load_state_dict(checkpoint["optimizer_dict"])
            logger.info("optimizer.load_state_dict")
        except Exception as e:
            logger.warning(f"optimizer state not restored: {e}")

    scheduler = build_scheduler(args, optimizer, len(train_loader))

    losses_dict = {n: [] for n in ("train_loss_per_step",
                                   "train_loss_per_epoch",
                                   "val_loss_per_epoch")}

    # Distributed wiring. Default: FlatReplica (pre-aliased flat gradient
    # buckets -> ONE RCCL all-reduce per dtype per step) + native fused
    # SyncBN (a (C,2)-float collective per BN layer, statistics math
    # matching torch SyncBatchNorm) — the same step the flagship bench
    # measures. torch DDP + SyncBatchNorm remains behind --use-torch-ddp.
    replica = None
    if pdist.is_dist():
        if (getattr(args, "use_torch_ddp", False)
                or getattr(args, "find_unused_parameters", False)):
            model = wrap_distributed(model, args)
        else:
            if getattr(args, "sync_bn", True):
                enable_native_syncbn(model)
            # lazy: backward steals grads (no per-param accumulate adds);
            # allreduce() packs them with one _foreach_copy_
            replica = FlatReplica(model, lazy=torch.cuda.is_available())
    elif torch.cuda.is_available():
        # single GPU: the lazy replica gives the fused optimizer stable
        # flat-view grad pointers (plain set_to_none steals fresh tensors
        # every step, forcing a repack of the Adam chunk metadata per step)
        replica = FlatReplica(model, lazy=True)

    ckpt_path = None
    num_saved = 0
    epochs_since_improvement = 0
    cost_time = datetime.timedelta()
    for i, epoch in enumerate(range(args.start_epoch, args.epochs)):
        epoch_start_time = datetime.datetime.now()
        if train_sampler is not None:
            train_sampler.set_epoch(epoch)

        train_losses, train_metrics_dict = train(
            args, model_tasks, model, optimizer, scheduler, loss_fn,
            train_loader, epoch, device, scalar_writer, replica=replica)
        train_loss = float(np.mean(train_losses))
        losses_dict["train_loss_per_step"].extend(train_losses)
        losses_dict["train_loss_per_epoch"].append(train_loss)

        val_loss, val_metrics_dict = validate(
            args, model_tasks, model, loss_fn, val_loader, epoch, device)
        losses_dict["val_loss_per_epoch"].append(val_loss)

        if pdist.is_main_process():
            if val_loss < best_loss:
                best_loss = val_loss
                ckpt_path = os.path.join(checkpoint_save_dir,
                                         f"model-{epoch}.pth")
                save_checkpoint(ckpt_path, model=model, optimizer=optimizer,
                                epoch=epoch, loss=best_loss,
                                use_ddp=pdist.is_dist())
                logger.info(f"Model saved: {ckpt_path}")
                num_saved += 1
                epochs_since_improvement = 0
            else:
                epochs_since_improvement += 1
                logger.info(
                    f"Epochs since last improvement:{epochs_since_improvement}")

            if scalar_writer is not None:
                scalar_writer.add_scalars(
                    "train-val.loss/epoch",
                    {"train": train_loss, "val": val_loss}, epoch)
                for task in model_tasks:
                    scalar_writer.add_scalars(
                        f"train.{task}.metrics/epoch",
                        train_metrics_dict[task].get_all_metrics(), epoch)
                    scalar_writer.add_scalars(
                        f"val.{task}.metrics/epoch",
                        val_metrics_dict[task].get_all_metrics(), epoch)
                    scalar_writer.add_scalars(
                        f"val.{task}.allvalues/epoch",
                        val_metrics_dict[task].to_dict(), epoch)

            tm = "* [Train Metrics]"
            vm = "* [Val Metrics]"
            for task in model_tasks:
                tm += f"[{task.upper()}]{train_metrics_dict[task]} "
                vm += f"[{task.upper()}]{val_metrics_dict[task]} "
            logger.info(tm)
            logger.info(vm)

        # keep early-stop decision consistent across ranks
        stop = epochs_since_improvement > args.patience
        if pdist.is_dist():
            stop = pdist.broadcast_object(stop, src=0)
        if stop:
            logger.info("* Stop training (early stopping).")
            break

        epoch_end_time = datetime.datetime.now()
        epoch_cost_time = epoch_end_time - epoch_start_time
        cost_time += epoch_cost_time
        if pdist.is_main_process():
            est_end = ((cost_time / (i + 1)) * 0.1
                       + epoch_cost_time * 0.9) \
                * (args.epochs - (i + 1)) + epoch_end_time
            logger.info(f"* Epoch cost time: {strftimedelta(epoch_cost_time)}")
            logger.info(f"* Estimated end time: "
                        f"{est_end.strftime('%Y-%m-%d %H:%M:%S')}")

    if pdist.is_main_process():
        loss_save_dir = os.path.join(log_dir, "loss")
        os.makedirs(loss_save_dir, exist_ok=True)
        for name, t in losses_dict.items():
            np.save(os.path.join(loss_save_dir,
                                 f"{args.model_name}_{name}.npy"), np.array(t))
    if scalar_writer is not None:
        scalar_writer.close()

    if pdist.is_dist():
        ckpt_path = pdist.broadcast_object(ckpt_path, src=0)
    return ckpt_path
