"""Shape sweep: VALU pw_gemm vs MFMA pw kernel on the seist_m_dpk pointwise
shapes (within-process interleaved timing)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import seist_amd._C as C

assert torch.cuda.is_available()
dev = torch.device("cuda:0")

# (Ci, Co, L) pointwise shapes in seist_m_dpk at batch 500
SHAPES = [
    (3, 3, 8192), (16, 16, 4096), (8, 8, 4096), (16, 16, 2048),
    (48, 16, 2048), (16, 24, 1024), (24, 24, 1024), (24, 48, 1024),
    (32, 32, 512), (64, 64, 256), (64, 128, 256), (96, 96, 128),
    (96, 192, 128), (192, 96, 128),
]
N = 500


def timeit(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1e3  # us


print(f"{'Ci':>4} {'Co':>4} {'L':>5} | {'fwd us':>8} {'ref us':>8} | ok")
for Ci, Co, L in SHAPES:
    x = torch.randn(N, Ci, L, device=dev, dtype=torch.bfloat16)
    w = torch.randn(Co, Ci, device=dev, dtype=torch.bfloat16) * 0.1
    y = C.pw_conv_fwd(x, w, None)
    ref = torch.einsum("oc,ncl->nol", w.float(), x.float())
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    us = timeit(lambda: C.pw_conv_fwd(x, w, None))
    # roofline: bytes through HBM
    byts = (N * Ci * L + N * Co * L) * 2
    roof = byts / 6.3e12 * 1e6
    print(f"{Ci:4d} {Co:4d} {L:5d} | {us:8.1f} roof={roof:6.1f} | "
          f"relerr={err / scale:.2e}")
