#!/usr/bin/env python3
"""Per-op A/B microbenchmark on the GPU box (guide §5.4: within-probe timing).

Times the bf16 conv engines on the model shapes so routing decisions are
measured, not guessed. Columns call the HOST dispatchers, so "fwd/dgrad/
wgrad" always measure the currently-routed engine; the explicit
conv_tiled_s1 columns force the tiled engine for comparison.

    gpurun -- 'python tools/microbench.py > gpurun_out/micro.txt'
"""

import sys
import time

import torch

sys.path.insert(0, ".")
from dist_mnist_amd.ops import native  # noqa: E402


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    ext = native.ext()
    dev = "cuda:0"
    torch.manual_seed(0)

    # (name, N, H, W, CIp, CIcan, CO, COp, KH, stride, pad)
    shapes = [
        ("lenet.conv2", 8192, 12, 12, 16, 10, 20, 24, 5, 1, 0),
        ("rn.l1.3x3", 256, 56, 56, 64, 64, 64, 64, 3, 1, 1),
        ("rn.l2.3x3", 256, 28, 28, 128, 128, 128, 128, 3, 1, 1),
        ("rn.l3.3x3", 256, 14, 14, 256, 256, 256, 256, 3, 1, 1),
        ("rn.l4.3x3", 256, 7, 7, 512, 512, 512, 512, 3, 1, 1),
    ]
    print(f"{'shape':14} {'fwd_gather':>11} {'fwd_tiled':>10} {'dg_gather':>10} "
          f"{'dg_tiled':>9} {'wgrad':>8}   (us/call)")
    for name, N, H, W, CIp, CIcan, CO, COp, KH, stride, pad in shapes:
        x = torch.randn(N, H, W, CIp, device=dev).to(torch.bfloat16)
        w = torch.randn(CO, CIcan, KH, KH, device=dev) * 0.1
        b = torch.randn(CO, device=dev)
        OH = (H + 2 * pad - KH) // stride + 1
        dy = torch.randn(N, OH, OH, COp, device=dev).to(torch.bfloat16)

        fg = timeit(lambda: ext.conv2d_fwd_bf16(x, w, b, COp, stride, pad, False))
        ft = timeit(lambda: ext.conv_tiled_s1_bf16(x, w, b, COp, pad, False, False)) \
            if stride == 1 else float("nan")
        dg = timeit(lambda: ext.conv2d_dgrad_bf16(dy, w, H, W, CIp, stride, pad))
        dt = timeit(lambda: ext.conv_tiled_s1_bf16(dy, w, torch.Tensor(), CIp,
                                                   KH - 1 - pad, False, True)) \
            if stride == 1 else float("nan")
        wg = timeit(lambda: ext.conv2d_wgrad_bf16(dy, x, CIcan, CO, KH, KH,
                                                  stride, pad, True,
                                                  torch.Tensor()))
        print(f"{name:14} {fg:11.1f} {ft:10.1f} {dg:10.1f} {dt:9.1f} {wg:8.1f}")

    # stride-2 shapes: the host dispatchers route these to the stem /
    # S=2 tiled / parity-class engines automatically, so the numbers below
    # are the ROUTED engine. A/B the region-wgrad chunk sizes with
    # DMA_WGRS_MC=32|64|128|256 and DMA_WGST_MC=32|64|96 in the env.
    s2 = [
        ("rn.stem7x7s2", 64, 224, 224, 8, 3, 64, 64, 7, 2, 3),
        ("rn.l2.ds3x3s2", 256, 56, 56, 64, 64, 128, 128, 3, 2, 1),
        ("rn.l2.1x1s2", 256, 56, 56, 64, 64, 128, 128, 1, 2, 0),
    ]
    print()
    for name, N, H, W, CIp, CIcan, CO, COp, KH, stride, pad in s2:
        x = torch.randn(N, H, W, CIp, device=dev).to(torch.bfloat16)
        w = torch.randn(CO, CIcan, KH, KH, device=dev) * 0.1
        b = torch.randn(CO, device=dev)
        OH = (H + 2 * pad - KH) // stride + 1
        dy = torch.randn(N, OH, OH, COp, device=dev).to(torch.bfloat16)
        fg = timeit(lambda: ext.conv2d_fwd_bf16(x, w, b, COp, stride, pad, False))
        dg = timeit(lambda: ext.conv2d_dgrad_bf16(dy, w, H, W, CIp, stride, pad))
        wg = timeit(lambda: ext.conv2d_wgrad_bf16(dy, x, CIcan, CO, KH, KH,
                                                  stride, pad, True,
                                                  torch.Tensor()))
        print(f"{name:14} fwd={fg:9.1f} dgrad={dg:9.1f} wgrad={wg:9.1f}")

    # LeNet fused stem (conv1+pool+relu) fwd + its fused wgrad
    N = 8192
    xs = torch.randn(N, 28, 28, 1, device=dev).to(torch.bfloat16)
    ws = torch.randn(10, 1, 5, 5, device=dev) * 0.1
    bs = torch.randn(10, device=dev)
    p, idx = ext.conv_pool_stem_bf16(xs, ws, bs, 16)
    dpool = torch.randn_like(p)
    sf = timeit(lambda: ext.conv_pool_stem_bf16(xs, ws, bs, 16))
    sw = timeit(lambda: ext.conv2d_wgrad_pool_bf16(dpool, p, idx, xs, 1, 10, 5, 5))
    print(f"\n{'lenet.stem':14} fwd={sf:9.1f} wgrad_pool={sw:9.1f}")


if __name__ == "__main__":
    main()
