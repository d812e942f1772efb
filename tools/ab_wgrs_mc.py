"""A/B region-wgrad chunk sizes after the round-2 split-budget change."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from dist_mnist_amd.ops import native
ext = native.ext()
dev = "cuda:0"
torch.manual_seed(0)
shapes = [("l1", 256, 56, 56, 64, 64), ("l2", 256, 28, 28, 128, 128),
          ("l3", 256, 14, 14, 256, 256), ("l4", 256, 7, 7, 512, 512)]
def timeit(fn, iters=40, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6
for rnd in range(2):
    for mc in ["128", "256"]:
        os.environ["DMA_WGRS_MC"] = mc
        row = []
        for name, N, H, W, CI, CO in shapes:
            x = torch.randn(N, H, W, CI, device=dev).to(torch.bfloat16)
            dy = torch.randn(N, H, W, CO, device=dev).to(torch.bfloat16)
            t = timeit(lambda: ext.conv2d_wgrad_bf16(dy, x, CI, CO, 3, 3, 1, 1, True, torch.Tensor()))
            row.append(f"{name}={t:7.1f}")
        print(f"MC={mc}: " + " ".join(row))
os.environ.pop("DMA_WGRS_MC", None)
