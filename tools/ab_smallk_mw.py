"""A/B smallk MW (within-probe)."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from dist_mnist_amd.ops import native
ext = native.ext()
dev = "cuda:0"
torch.manual_seed(0)
x = torch.randn(8192, 12, 12, 16, device=dev).to(torch.bfloat16)
w = torch.randn(20, 10, 5, 5, device=dev) * 0.1
b = torch.randn(20, device=dev)
dy = torch.randn(8192, 8, 8, 24, device=dev).to(torch.bfloat16)

def timeit(fn, iters=60, warmup=15):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

for _ in range(2):
    for mw in ["2", "4", "8"]:
        os.environ["DMA_SMALLK_MW"] = mw
        f = timeit(lambda: ext.conv2d_fwd_bf16(x, w, b, 24, 1, 0, False))
        d = timeit(lambda: ext.conv2d_dgrad_bf16(dy, w, 12, 12, 16, 1, 0))
        print(f"MW={mw}: fwd {f:7.1f}  dgrad {d:7.1f} us")
os.environ["DMA_SMALLK_MW"] = "8"
got = ext.conv2d_dgrad_bf16(dy, w, 12, 12, 16, 1, 0)
gf = ext.conv2d_fwd_bf16(x, w, b, 24, 1, 0, False)
del os.environ["DMA_SMALLK_MW"]
ref = ext.conv2d_dgrad_bf16(dy, w, 12, 12, 16, 1, 0)
rf = ext.conv2d_fwd_bf16(x, w, b, 24, 1, 0, False)
print("mw8 dgrad bitwise ok:", bool((got == ref).all()), " fwd:", bool((gf == rf).all()))
