"""A/B fp32 conv2-dgrad variants (within-probe, guide rule 24)."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from dist_mnist_amd.ops import native

ext = native.ext()
dev = "cuda:0"
torch.manual_seed(0)
N = 8192
dy = torch.randn(N, 20, 8, 8, device=dev)
w = torch.randn(20, 10, 5, 5, device=dev) * 0.1
x_fp = torch.randn(N, 10, 12, 12, device=dev)
wf = torch.randn(50, 320, device=dev) * 0.1
xf = torch.randn(N, 320, device=dev)
bf = torch.randn(50, device=dev)

def timeit(fn, iters=60, warmup=15):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for _ in range(2):  # interleave rounds
    for tag, env in [("gather", None), ("smallimg_mw4", "1"), ("smallimg_mw8", "8")]:
        if env is None:
            os.environ.pop("DMA_F32_SMALLIMG_DGRAD", None)
        else:
            os.environ["DMA_F32_SMALLIMG_DGRAD"] = env
        t = timeit(lambda: ext.conv2d_bwd_input(dy, w, 12, 12, 1, 0))
        print(f"dgrad {tag:14} {t:8.1f} us")
    print(f"linear_fwd fc1    {timeit(lambda: ext.linear_fwd(xf, wf, bf, True)):8.1f} us")
os.environ.pop("DMA_F32_SMALLIMG_DGRAD", None)
# numerics spot check for mw8
os.environ["DMA_F32_SMALLIMG_DGRAD"] = "8"
got = ext.conv2d_bwd_input(dy, w, 12, 12, 1, 0)
os.environ.pop("DMA_F32_SMALLIMG_DGRAD", None)
ref = torch.nn.grad.conv2d_input((N, 10, 12, 12), w, dy)
print("mw8 max|d| vs torch:", (got - ref).abs().max().item())
