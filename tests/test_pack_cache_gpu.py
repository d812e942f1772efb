"""Packed-weight cache semantics (native/ext_pack.inc): the tiled conv
engines read weights through persistent per-weight pack buffers, refreshed
by one batched kernel after each optimizer step. These tests guard the
staleness rules — an out-of-band in-place weight mutation (version-counter
bump) must re-pack before the next use, and repack_refresh_all must bring
every cached pack back in sync after a raw-pointer mutation that does NOT
bump the version counter (exactly what the fused SGD kernel does)."""

import pytest
import torch

from dist_mnist_amd.ops import functional_bf16 as fb
from dist_mnist_amd.ops import native

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _conv(x, w):
    # 3x3 s1 p1, the tiled-engine shape that uses the pack cache
    return fb.conv2d_bf16(x, w, stride=1, pad=1, cop=64)


def _ref(x, w):
    import torch.nn.functional as F
    xr = x.float().permute(0, 3, 1, 2)
    y = F.conv2d(xr, w.reshape(64, 64, 3, 3), stride=1, padding=1)
    return y.permute(0, 2, 3, 1).to(torch.bfloat16)


def test_copy_mutation_repacks():
    # w.copy_ bumps the tensor version counter -> pack_get must re-pack
    g = torch.Generator().manual_seed(0)
    x = torch.randn(2, 28, 28, 64, generator=g).to(DEV).to(torch.bfloat16)
    w = (torch.randn(64, 64, 3, 3, generator=g) * 0.1).to(DEV)
    w2 = (torch.randn(64, 64, 3, 3, generator=g) * 0.1).to(DEV)

    y1 = _conv(x, w)
    assert torch.allclose(y1.float(), _ref(x, w).float(), atol=0.25, rtol=0.1)
    w.copy_(w2)
    y2 = _conv(x, w)  # stale pack would reproduce y1 here
    assert torch.allclose(y2.float(), _ref(x, w2).float(), atol=0.25, rtol=0.1)
    assert not torch.allclose(y1.float(), y2.float())


def test_raw_pointer_mutation_plus_refresh():
    # mutate the weight storage WITHOUT a version bump (as fused_sgd_step
    # does), then repack_refresh_all must resync the cached pack
    g = torch.Generator().manual_seed(1)
    x = torch.randn(2, 28, 28, 64, generator=g).to(DEV).to(torch.bfloat16)
    w = (torch.randn(64, 64, 3, 3, generator=g) * 0.1).to(DEV)
    _conv(x, w)  # populate the cache

    with torch.no_grad():
        delta = (torch.randn(64, 64, 3, 3, generator=g) * 0.05).to(DEV)
        # .data bypasses the version counter like the raw-pointer kernel
        w.data.add_(delta)
    native.ext().repack_refresh_all()
    y = _conv(x, w)
    assert torch.allclose(y.float(), _ref(x, w).float(), atol=0.25, rtol=0.1)


def test_cache_entries_and_clear():
    ext = native.ext()
    g = torch.Generator().manual_seed(2)
    x = torch.randn(2, 28, 28, 64, generator=g).to(DEV).to(torch.bfloat16)
    w = (torch.randn(64, 64, 3, 3, generator=g) * 0.1).to(DEV)
    before = ext.repack_cache_size()
    _conv(x, w)
    assert ext.repack_cache_size() >= before + 1
    ext.repack_cache_clear()
    assert ext.repack_cache_size() == 0
    # cache repopulates transparently
    y = _conv(x, w)
    assert torch.allclose(y.float(), _ref(x, w).float(), atol=0.25, rtol=0.1)
