#!/usr/bin/env python3
"""Flagship training-step benchmark (driver contract).

Measures the BASELINE.json headline metric: waveforms/sec (whole node) for
seist_m_dpk training at batch 500 per GPU on synthetic 3x8192 bf16
waveforms with random-init weights, at N GPUs of one node (weak scaling,
one rank per GPU over RCCL/xGMI).

Single GPU:   python bench.py --gpus 1 --steps 20 --warmup 5
Multi GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N ...

The timed region covers the full training step: H2D-free synthetic batch
(resident in HBM), forward, loss, backward, flat-bucket RCCL all-reduce,
fused-Adam update. Timing is bracketed by barrier+synchronize on both
sides; the max over ranks is reported by rank 0 as one JSON line.
"""

import argparse
import json
import os
import time

import torch

from seist_amd.config import Config
from seist_amd.engine.precision import convert_to_bf16
from seist_amd.models import create_model
from seist_amd.models._blocks import manage_bn_counters
from seist_amd.ops import FusedAdam
from seist_amd.parallel import dist as pdist
from seist_amd.parallel.ddp import FlatReplica


def make_batch(model_name, batch, in_samples, device, dtype, seed):
    """Synthetic waveforms + soft labels of the flagship task's shape."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(batch, 3, in_samples, generator=g)
    labels = Config.get_model_config_(model_name, "labels")
    if labels == [["det", "ppk", "spk"]] or labels == [["non", "ppk", "spk"]]:
        t = torch.rand(batch, 3, in_samples, generator=g)
    elif labels == ["emg"] or labels == ["baz"] or labels == ["dis"]:
        t = torch.rand(batch, 1, generator=g) * 5.0
    elif labels == ["pmp"]:  # one-hot polarity
        t = torch.eye(2)[torch.randint(0, 2, (batch,), generator=g)]
    elif labels == ["clr", "pmp"]:
        # ditingmotion: (clarity, polarity) one-hot pair
        t = tuple(torch.eye(2)[torch.randint(0, 2, (batch,), generator=g)]
                  for _ in labels)
    else:
        t = torch.rand(batch, 3, in_samples, generator=g)
    if isinstance(t, tuple):
        return (x.to(device=device, dtype=dtype),
                tuple(ti.to(device=device, dtype=torch.float32)
                      for ti in t))
    return (x.to(device=device, dtype=dtype),
            t.to(device=device, dtype=torch.float32))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=500)
    p.add_argument("--in-samples", type=int, default=8192)
    p.add_argument("--model", type=str, default="seist_m_dpk")
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32"])
    p.add_argument("--graph", type=int, default=1,
                   help="capture the step in a hipGraph (1=try, 0=off)")
    p.add_argument("--lazy-grads", type=int, default=1,
                   help="FlatReplica lazy mode: steal grads + one foreach "
                        "pack instead of per-param accumulate adds")
    p.add_argument("--mode", type=str, default="train",
                   choices=["train", "infer"],
                   help="train = full step; infer = forward-only (serving)")
    args = p.parse_args()

    distributed = pdist.init_distributed_mode()
    world = pdist.get_world_size()
    rank = pdist.get_rank()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        device = torch.device(f"cuda:{pdist.get_local_rank()}")
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
        if args.batch_size > 16:
            args.batch_size = 8  # CPU sanity mode only

    dtype = torch.bfloat16 if (args.dtype == "bf16" and use_cuda) \
        else torch.float32

    torch.manual_seed(1234 + rank)
    model = create_model(args.model, in_channels=3,
                         in_samples=args.in_samples)
    if dtype == torch.bfloat16:
        model = convert_to_bf16(model)
    model = model.to(device).train()

    replica = FlatReplica(model, lazy=bool(args.lazy_grads)
                          and use_cuda)
    bn_tick = manage_bn_counters(model)
    optimizer = FusedAdam(model.parameters(), lr=8e-5)
    loss_fn = Config.get_loss(args.model).to(device)
    tgt_trans = Config.get_model_config_(args.model,
                                         "targets_transform_for_loss")

    x, t = make_batch(args.model, args.batch_size, args.in_samples, device,
                      dtype, seed=1234 + rank)
    if tgt_trans is not None:
        t = tgt_trans(t)

    if args.mode == "infer":
        model = model.eval()

        @torch.no_grad()
        def compute():
            return model(x)

        def exchange():
            pass
    else:
        def compute():
            replica.zero_grad()
            out = model(x)
            if isinstance(out, (list, tuple)):
                out = [o.float() for o in out]
            else:
                out = out.float()
            loss = loss_fn(out, t)
            loss.backward()
            # pack INSIDE the captured region: the replayed backward
            # writes the graph-pool grad tensors, and only a captured
            # _foreach_copy_ moves them into the flat buffers every
            # replay (an eager pack would see p.grad already re-aliased
            # and skip — silently desynchronizing ranks)
            replica.pack()
            return loss

        def exchange():
            replica.allreduce()
            optimizer.step()
            bn_tick()

    def step():
        compute()
        exchange()

    # warmup (also materializes optimizer state + adam packing)
    for _ in range(args.warmup):
        step()
    if use_cuda:
        torch.cuda.synchronize()

    # hipGraph capture. In distributed mode the RCCL all-reduce stays
    # EAGER between two captured graphs (compute: fwd+loss+bwd; update:
    # fused Adam), so the 8-rank step replays ~2 graphs + 1 collective
    # instead of ~4k eager dispatches.
    graphed = False
    if args.graph and use_cuda:
        try:
            stream = torch.cuda.Stream()
            with torch.cuda.stream(stream):
                step()
            torch.cuda.current_stream().wait_stream(stream)
            g_compute = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g_compute):
                compute()
            if distributed and args.mode != "infer":
                g_update = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g_update):
                    optimizer.step()
                    bn_tick()

                def step():  # noqa: F811
                    g_compute.replay()
                    replica.allreduce()
                    g_update.replay()
            else:
                g_exchange = None
                if args.mode != "infer":
                    g_exchange = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(g_exchange):
                        exchange()

                def step():  # noqa: F811
                    g_compute.replay()
                    if g_exchange is not None:
                        g_exchange.replay()
            graphed = True
        except Exception as e:
            print(f"# hipGraph capture failed ({e}); eager steps", flush=True)
            graphed = False

    pdist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    pdist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if distributed:
        e = torch.tensor([elapsed], device=device if use_cuda else "cpu")
        e = pdist.reduce_tensor(e, "MAX")
        elapsed = e.item()

    if rank == 0:
        global_batch = args.batch_size * world
        value = global_batch * args.steps / elapsed
        ms_per_step = elapsed / args.steps * 1e3
        print(json.dumps({
            "metric": "waveforms/sec (whole node) seist_m_dpk bs=500 "
                      "3x8192 at 1/2/4/8 MI355X"
                      if args.model == "seist_m_dpk"
                      else f"waveforms/sec (whole node) {args.model}",
            "value": round(value, 2),
            "unit": "waveforms/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype if use_cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.in_samples,
                "parallelism": f"dp{world}",
                "mode": args.mode,
                "graph": graphed,
            },
        }), flush=True)


if __name__ == "__main__":
    main()
    import torch.distributed as _dist
    if _dist.is_available() and _dist.is_initialized():
        _dist.destroy_process_group()
