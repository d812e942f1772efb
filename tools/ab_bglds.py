"""A/B the glds-pipelined B panel (DMA_TILED_BGLDS) on the tiled conv,
with a correctness check first (wrap in timeout-able small sizes)."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from dist_mnist_amd.ops import native
ext = native.ext()
dev = "cuda:0"
torch.manual_seed(0)

# correctness: small l1-like shape vs the default engine
x = torch.randn(8, 56, 56, 64, device=dev).to(torch.bfloat16)
w = torch.randn(64, 64, 3, 3, device=dev) * 0.1
ref = ext.conv_tiled_s1_bf16(x, w, torch.Tensor(), 64, 1, False, False)
os.environ["DMA_TILED_BGLDS"] = "1"
got = ext.conv_tiled_s1_bf16(x, w, torch.Tensor(), 64, 1, False, False)
torch.cuda.synchronize()
ok = bool((got == ref).all())
print("BGLDS correctness (bitwise vs default):", ok)
assert ok

def timeit(fn, iters=40, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

shapes = [("l1", 256, 56, 56, 64, 64), ("l2", 256, 28, 28, 128, 128)]
for rnd in range(2):
    for tag, env in [("default", None), ("bglds", "1")]:
        if env is None: os.environ.pop("DMA_TILED_BGLDS", None)
        else: os.environ["DMA_TILED_BGLDS"] = env
        row = []
        for name, N, H, W, CI, CO in shapes:
            xs = torch.randn(N, H, W, CI, device=dev).to(torch.bfloat16)
            ws = torch.randn(CO, CI, 3, 3, device=dev) * 0.1
            t = timeit(lambda: ext.conv_tiled_s1_bf16(xs, ws, torch.Tensor(), CO, 1, False, False))
            row.append(f"{name}={t:7.1f}")
        print(f"{tag:8}: " + " ".join(row))
os.environ.pop("DMA_TILED_BGLDS", None)
