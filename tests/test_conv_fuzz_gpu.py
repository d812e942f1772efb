"""Seeded shape-fuzz for the bf16 conv engines (GPU): random (N, C, K, stride,
pad) combinations hit every dispatch path (gather MW2/MW4, tiled (MW,NIMG),
parity s2 dgrad, NW tiers, tile remainders) against the torch fp32 oracle on
bf16-quantized inputs."""

import random

import pytest
import torch
import torch.nn.functional as F

from dist_mnist_amd.ops import functional_bf16 as fb

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _case(rng):
    kh = rng.choice([1, 3, 5, 7])
    stride = rng.choice([1, 1, 1, 2])
    pad = rng.randint(0, kh // 2) if kh > 1 else 0
    ci = rng.choice([1, 3, 8, 10, 24, 40, 64])
    co = rng.choice([4, 10, 16, 24, 48, 64, 96])
    n = rng.choice([1, 2, 3, 5])
    # pick H so the output is valid and >= 1
    h = rng.randint(kh + stride, 36)
    w = rng.randint(kh + stride, 36)
    return n, ci, h, w, co, kh, stride, pad


@pytest.mark.timeout(600)
def test_conv_bf16_shape_fuzz():
    import os

    rng = random.Random(int(os.environ.get("FUZZ_SEED", 20260913)))
    torch.manual_seed(0)
    for trial in range(24):
        n, ci, h, w, co, kh, stride, pad = _case(rng)
        oh = (h + 2 * pad - kh) // stride + 1
        ow = (w + 2 * pad - kh) // stride + 1
        if oh < 1 or ow < 1:
            continue
        x = torch.randn(n, ci, h, w, device=DEV)
        wt = (torch.randn(co, ci, kh, kh, device=DEV) * 0.2).requires_grad_(True)
        b = torch.randn(co, device=DEV).requires_grad_(True)
        dy = torch.randn(n, co, oh, ow, device=DEV)

        xb = fb.nchw_to_nhwc_padded(x).requires_grad_(True)
        y = fb.conv2d_bf16(xb, wt, b, stride=stride, pad=pad)
        y.backward(fb.nchw_to_nhwc_padded(dy, cp=y.shape[3]))

        q = lambda t: t.to(torch.bfloat16).float()
        xr = q(x).requires_grad_(True)
        wr = q(wt.detach()).requires_grad_(True)
        br = b.detach().clone().requires_grad_(True)
        ref = F.conv2d(xr, wr, br, stride=stride, padding=pad)
        ref.backward(q(dy))

        msg = f"trial {trial}: n{n} ci{ci} h{h} w{w} co{co} k{kh} s{stride} p{pad}"
        torch.testing.assert_close(fb.nhwc_to_nchw(y, co), ref, rtol=3e-2,
                                   atol=5e-2, msg=msg + " (fwd)")
        torch.testing.assert_close(fb.nhwc_to_nchw(xb.grad, ci), xr.grad,
                                   rtol=4e-2, atol=6e-2, msg=msg + " (dx)")
        wtol = 3e-2 * max(1.0, wr.grad.abs().max().item())
        torch.testing.assert_close(wt.grad, wr.grad, rtol=3e-2, atol=wtol,
                                   msg=msg + " (dw)")
        btol = 3e-2 * max(1.0, br.grad.abs().max().item())
        torch.testing.assert_close(b.grad, br.grad, rtol=3e-2, atol=btol,
                                   msg=msg + " (db)")
