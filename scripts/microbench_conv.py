#!/usr/bin/env python3
"""Small-C conv kernel microbenchmark (fwd + dx) at the hot model shapes.

Run twice for an A/B: once plain (small-C kernel dispatched) and once with
SEIST_AMD_NO_SMALLC=1 (tap-gather kernel). The dispatch env var is read
once per process. Also cross-checks the dispatched kernel against the
im2col/tap result computed in the same process via F.conv1d eager fp32.
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.nn.functional as F

from seist_amd import ops

# (Co, Ci, K, L) at stride 1, dilation 1, groups 1 — from the model survey
SHAPES = [
    (8, 3, 7, 8192), (8, 8, 7, 8192), (8, 16, 7, 8192),   # phasenet
    (16, 8, 7, 2048), (16, 32, 7, 2048),
    (1, 8, 11, 8192), (8, 3, 11, 8192), (8, 16, 11, 8192),  # eqt
    (16, 8, 9, 4096), (16, 16, 9, 4096), (16, 32, 7, 2048),
    (8, 3, 3, 8192), (8, 8, 5, 8192), (2, 27, 3, 1024),     # ditingmotion
    (8, 27, 5, 1024), (8, 8, 3, 4096),
]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=500)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--check", action="store_true")
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    tag = "tap" if os.environ.get("SEIST_AMD_NO_SMALLC") else "smallc"
    torch.manual_seed(0)
    print(f"# path={tag} batch={args.batch}")
    for (co, ci, k, L) in SHAPES:
        x = torch.randn(args.batch, ci, L, device=dev,
                        dtype=torch.bfloat16)
        w = torch.randn(co, ci, k, device=dev, dtype=torch.bfloat16) \
            * (ci * k) ** -0.5
        pl, pr = ops.auto_pad_lr(L, k, 1)
        dy = torch.randn(args.batch, co, L, device=dev,
                         dtype=torch.bfloat16)

        if args.check:
            y = ops.conv1d(x, w, None, stride=1, padding=(pl, pr))
            ref = F.conv1d(F.pad(x.float(), (pl, pr)), w.float())
            d = (y.float() - ref).abs().max().item()
            scale = ref.abs().max().item()
            assert d <= 1e-2 * scale + 1e-2, \
                f"fwd mismatch {co},{ci},{k},{L}: {d} vs {scale}"
            xg = x.float().requires_grad_(True)
            yr = F.conv1d(F.pad(xg, (pl, pr)), w.float())
            (dxr,) = torch.autograd.grad(yr, [xg], dy.float())
            xq = x.clone().requires_grad_(True)
            yq = ops.conv1d(xq, w, None, stride=1, padding=(pl, pr))
            (dxq,) = torch.autograd.grad(yq, [xq], dy)
            d = (dxq.float() - dxr).abs().max().item()
            scale = dxr.abs().max().item()
            assert d <= 1e-2 * scale + 1e-2, \
                f"dx mismatch {co},{ci},{k},{L}: {d} vs {scale}"

        def timeit(fn):
            for _ in range(3):
                fn()
            torch.cuda.synchronize()
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            for _ in range(args.iters):
                fn()
            e.record()
            torch.cuda.synchronize()
            return s.elapsed_time(e) / args.iters * 1e3  # us

        t_f = timeit(lambda: ops.conv1d(x, w, None, stride=1,
                                        padding=(pl, pr)))
        xr = x.requires_grad_(True)

        def bwd():
            xr.grad = None
            y = ops.conv1d(xr, w, None, stride=1, padding=(pl, pr))
            y.backward(dy)

        t_b = timeit(bwd)
        roof = (args.batch * (ci + 2 * co) * L * 2) / 6.3e12 * 1e6
        print(f"{tag} Co={co:3d} Ci={ci:3d} K={k:2d} L={L:5d} "
              f"fwd {t_f:8.1f} us  fwd+bwd {t_b:8.1f} us  "
              f"(fwd roof ~{roof:5.1f})")


if __name__ == "__main__":
    main()
