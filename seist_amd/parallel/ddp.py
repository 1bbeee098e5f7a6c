"""Data parallelism over RCCL/xGMI.

Two implementations:

* ``wrap_distributed`` — torch DDP (+SyncBatchNorm, parity with reference
  train.py:367-374) used by the generic training engine.
* ``FlatReplica`` — the MI355X-native path used by the benchmark/serving
  step: gradients of all parameters live pre-aliased inside per-dtype flat
  buffers, so a whole backward produces ready-packed buckets and the
  gradient exchange is ONE RCCL all-reduce per dtype per step with zero
  pack/unpack kernels. For SeisT-class models (<= 2.6 MB of gradients,
  SURVEY §2.5 C2) the exchange is latency-bound, so minimizing calls —
  not overlapping many buckets — is the right xGMI design.
"""

from typing import List

import torch
import torch.distributed as dist
import torch.nn as nn

from . import dist as pdist


def wrap_distributed(model: nn.Module, args) -> nn.Module:
    """Reference-parity distributed wrapper: DDP + SyncBN."""
    if not pdist.is_dist():
        return model
    local_rank = pdist.get_local_rank()
    device_ids = [local_rank] if torch.cuda.is_available() else None
    model = torch.nn.parallel.DistributedDataParallel(
        model, device_ids=device_ids,
        find_unused_parameters=getattr(args, "find_unused_parameters", False))
    if getattr(args, "sync_bn", True):
        model = torch.nn.SyncBatchNorm.convert_sync_batchnorm(model)
    return model


def enable_native_syncbn(model: nn.Module, flag: bool = True) -> nn.Module:
    """Tag every BatchNorm1d so the fused bn_act path reduces batch
    statistics across ranks (one (C,2)-float collective around the
    finalize) instead of converting the module tree to
    torch SyncBatchNorm — which would bypass the native kernel entirely.
    The module tree and checkpoint format are unchanged."""
    for m in model.modules():
        if isinstance(m, nn.BatchNorm1d):
            m._sync_bn = flag
    return model


class FlatReplica:
    """Flat-bucket gradient replica for one-process-per-GPU training.

    Usage::

        rep = FlatReplica(model)          # broadcasts params from rank 0
        rep.zero_grad()
        loss.backward()                   # grads land in the flat buffers
        rep.allreduce()                   # one RCCL all-reduce per dtype
        optimizer.step()
    """

    def __init__(self, model: nn.Module, process_group=None,
                 lazy: bool = False):
        self.model = model
        self.group = process_group
        self.world_size = pdist.get_world_size()
        self.lazy = lazy
        self.params: List[torch.nn.Parameter] = [
            p for p in model.parameters() if p.requires_grad]

        # broadcast initial parameters from rank 0
        if pdist.is_dist():
            for p in self.params:
                dist.broadcast(p.data, src=0, group=self.group)

        # per-dtype flat gradient buffers; p.grad aliases a view
        buckets = {}
        for p in self.params:
            buckets.setdefault(p.dtype, []).append(p)
        self.buffers = {}
        self._bucket_params = buckets
        self._views = {}
        for dtype, ps in buckets.items():
            total = sum(p.numel() for p in ps)
            buf = torch.zeros(total, dtype=dtype, device=ps[0].device)
            off = 0
            views = []
            for p in ps:
                v = buf[off:off + p.numel()].view_as(p)
                if not lazy:
                    p.grad = v
                views.append(v)
                off += p.numel()
            self.buffers[dtype] = buf
            self._views[dtype] = views

    def zero_grad(self):
        if self.lazy:
            # backward STEALS each gradient (no per-param accumulate-add
            # kernel); pack() copies them into the flat buffers afterwards
            for p in self.params:
                p.grad = None
            return
        for buf in self.buffers.values():
            buf.zero_()

    def pack(self):
        """Lazy mode: one _foreach_copy_ of all stolen grads into the flat
        buffers, then re-alias p.grad to the flat views so the fused
        optimizer's packed pointers stay stable across steps. Params that
        received no gradient this step zero their slice (stale values
        from the previous step must not survive the all-reduce)."""
        if not self.lazy:
            return
        for dtype, ps in self._bucket_params.items():
            views = self._views[dtype]
            dst, src, missing = [], [], []
            for p, v in zip(ps, views):
                if p.grad is None:
                    missing.append(v)
                elif p.grad.data_ptr() != v.data_ptr():
                    dst.append(v)
                    src.append(p.grad)
                # else: already packed (double pack without a new backward)
            if dst:
                torch._foreach_copy_(dst, src)
            if missing:
                torch._foreach_zero_(missing)
            for p, v in zip(ps, views):
                p.grad = v

    def allreduce(self):
        self.pack()
        if not pdist.is_dist() or self.world_size == 1:
            return
        for buf in self.buffers.values():
            dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=self.group)
            buf.div_(self.world_size)

    def __call__(self, *a, **k):
        return self.model(*a, **k)
