import torch, time
from seist_amd import ops
def timeit(fn, iters=50):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/iters*1e6
N = 500
for Ci, Co, K, L in [(3,8,11,8192),(8,16,9,4096),(16,8,11,8192),(16,16,7,2048),(8,3,11,8192)]:
    x = torch.randn(N, Ci, L, device="cuda:0", dtype=torch.bfloat16)
    w = torch.randn(Co, Ci, K, device="cuda:0", dtype=torch.bfloat16) * 0.1
    pl, pr = (K-1)//2, K-1-(K-1)//2
    t = timeit(lambda: ops.conv1d(x, w, None, stride=1, padding=(pl, pr)))
    print(f"Ci={Ci:3d} Co={Co:3d} K={K:2d} L={L:5d}: {t:7.1f}us")
