"""Validation / test loop (shared via ``testing`` flag).

Flow parity with /root/reference/training/validate.py:10-134; per-step
metric collectives replaced by one per-epoch sync (see train.py note).
"""

import json
import os
from typing import Union

import torch

from ..config import Config
from ..parallel import dist as pdist
from ..utils.logger import logger
from ..utils.meters import AverageMeter, ProgressMeter
from ..utils.misc import get_safe_path
from .metrics import Metrics
from .postprocess import ResultSaver, process_outputs
from .precision import cast_inputs


def validate(args, tasks, model, loss_fn, val_loader, epoch, device,
             testing=False) -> Union[float, dict]:
    model.eval()

    (model_labels, tgts_trans_for_loss, outs_trans_for_loss,
     outs_trans_for_res) = Config.get_model_config_(
        args.model_name, "labels", "targets_transform_for_loss",
        "outputs_transform_for_loss", "outputs_transform_for_results")

    compute_dtype = (torch.bfloat16 if args.precision == "bf16"
                     else torch.float32)
    average_meters = {}
    metrics_merged = {}
    sampling_rate = val_loader.dataset.sampling_rate()

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
    progress = ProgressMeter(
        len(val_loader), list(average_meters.values()),
        prefix=f"{'Test' if testing else 'Val'}: [{epoch}/{args.epochs}]")

    # Every rank accumulates its shard's rows; they are gathered to rank 0
    # before the CSV write, so the saved file covers the whole test set
    # (the reference saves only rank 0's 1/world_size shard).
    results_saver = (ResultSaver(item_names=tasks)
                     if testing and args.save_test_results else None)
    sync_per_step = pdist.is_dist() and args.sync_metrics_per_step

    with torch.no_grad():
        for step, (x, loss_targets, metrics_targets,
                   meta_data_jsons) in enumerate(val_loader):
            if isinstance(x, (list, tuple)):
                x = [xi.to(device) for xi in x]
            else:
                x = x.to(device)
            x = cast_inputs(x, compute_dtype)
            if isinstance(loss_targets, (list, tuple)):
                loss_targets = [yi.to(device) for yi in loss_targets]
            else:
                loss_targets = loss_targets.to(device)

            outputs = model(x)

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

            step_batch_size = (x[0] if isinstance(x, (list, tuple))
                               else x).size(0)
            if pdist.is_dist():
                loss = pdist.reduce_tensor(loss, "AVG")
                sbs = torch.tensor(step_batch_size, device=device,
                                   dtype=torch.int32)
                step_batch_size = pdist.reduce_tensor(sbs).item()
            average_meters["loss"].update(loss.item(), step_batch_size)

            outputs_for_metrics = (outs_trans_for_res(outputs)
                                   if outs_trans_for_res is not None
                                   else outputs)
            if isinstance(outputs_for_metrics, (list, tuple)):
                outputs_for_metrics = [o.float() for o in outputs_for_metrics]
            else:
                outputs_for_metrics = outputs_for_metrics.float()
            results = process_outputs(args, outputs_for_metrics, model_labels,
                                      sampling_rate)

            if results_saver is not None:
                meta_data_dict = {k: [] for k in
                                  json.loads(meta_data_jsons[0]).keys()}
                for j in meta_data_jsons:
                    for k, v in json.loads(j).items():
                        meta_data_dict[k].append(v)
                results_saver.append(meta_data_dict, metrics_targets, results)

            for task in tasks:
                metrics = new_metrics(task)
                metrics.compute(targets=metrics_targets[task],
                                preds=results[task], reduce=sync_per_step)
                for metric in metrics.metric_names():
                    average_meters[f"{task}_{metric}"].update(
                        metrics.get_metric(metric), step_batch_size)
                metrics_merged[task].add(metrics)

            if pdist.is_main_process() and step % args.log_step == 0:
                logger.info(progress.get_str(
                    batch_idx=step,
                    name=f"{args.model_name}_{'test' if testing else 'val'}"))

    if pdist.is_dist() and not sync_per_step:
        for task in tasks:
            metrics_merged[task].synchronize_between_processes()

    if results_saver is not None:
        if pdist.is_dist():
            results_saver.gather_to_main()
        if pdist.is_main_process():
            path = get_safe_path(os.path.join(
                logger.logdir() or ".",
                f"test_results_{val_loader.dataset.name()}.csv"))
            results_saver.save_as_csv(path)

    return average_meters["loss"].avg, metrics_merged
