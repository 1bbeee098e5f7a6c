import torch, time
from seist_amd import ops
def timeit(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6
N = 500
shapes = [  # (Ci, Co, K, L) eqt encoder + decoder upsampling convs
    (3, 8, 11, 8192), (8, 16, 9, 4096), (16, 16, 7, 2048), (16, 32, 7, 1024),
    (32, 32, 5, 512), (32, 64, 5, 256), (64, 64, 3, 128),
    (16, 16, 9, 4096), (16, 8, 11, 8192), (8, 3, 11, 8192),
]
for Ci, Co, K, L in shapes:
    x = torch.randn(N, Ci, L, device="cuda:0", dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(Co, Ci, K, device="cuda:0", dtype=torch.bfloat16) * 0.1).requires_grad_(True)
    b = torch.randn(Co, device="cuda:0", dtype=torch.bfloat16)
    pl, pr = (K-1)//2, K-1-(K-1)//2
    y = ops.conv1d(x, w, b, stride=1, padding=(pl, pr))
    t_f = timeit(lambda: ops.conv1d(x.detach(), w, b, stride=1, padding=(pl, pr)))
    dy = torch.randn_like(y)
    def bwd():
        yy = ops.conv1d(x, w, b, stride=1, padding=(pl, pr))
        torch.autograd.grad(yy, [x, w], dy)
    t_fb = timeit(bwd)
    bytes_f = (N*Ci*L + N*Co*L) * 2
    roof = bytes_f / 6.3e3 / 1e3  # us
    print(f"Ci={Ci:3d} Co={Co:3d} K={K:2d} L={L:5d}: fwd {t_f:7.1f}us (roof {roof:6.1f}) fwd+bwd {t_fb:7.1f}us")
