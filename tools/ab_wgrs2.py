"""A/B the S=2 region-wgrad chunk size (DMA_WGRS_MC2) on the three ResNet
3x3/s2 shapes. Usage: DMA_WGRS_MC2=96 python tools/ab_wgrs2.py"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402

from dist_mnist_amd.ops import native  # noqa: E402

ext = native.ext()


def timeit(fn, iters=30, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1000 / iters


SHAPES = [  # name, N, H, W, CIp, CI, CO, COp  (3x3 s2 pad1)
    ("l2.b0.conv1", 256, 56, 56, 64, 64, 128, 128),
    ("l3.b0.conv1", 256, 28, 28, 128, 128, 256, 256),
    ("l4.b0.conv1", 256, 14, 14, 256, 256, 512, 512),
]

print(f"DMA_WGRS_MC2={os.environ.get('DMA_WGRS_MC2', '(default 64)')}")
for name, N, H, W, CIp, CI, CO, COp in SHAPES:
    OH, OW = H // 2, W // 2
    x = torch.randn(N, H, W, CIp, device="cuda", dtype=torch.bfloat16)
    dy = torch.randn(N, OH, OW, COp, device="cuda", dtype=torch.bfloat16)
    us = timeit(lambda: ext.conv2d_wgrad_bf16(dy, x, CI, CO, 3, 3, 2, 1,
                                              False, torch.Tensor()))
    print(f"{name:14} wgrad={us:9.1f} us")
