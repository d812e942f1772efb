"""bf16 NHWC autograd ops — the MI355X-first hot path (BASELINE.json config 4).

Layout contract (matches native/bf16.hip):
- activations NHWC bf16, channels padded to a multiple of 8;
- master weights fp32 canonical NCHW / [CO, CI] (exact reference state_dict);
- weight gradients come back fp32 canonical (feed DDP buckets + SGD directly);
- fully-connected layers take an int32 perm table mapping the padded
  (possibly NHWC-flattened) feature index to the canonical column, so the
  checkpoint ordering never changes.

GPU-only: these Functions raise on CPU tensors (the fp32 torch path is the
CPU reference; numerics tests compare this path against it on the GPU box).
"""

from __future__ import annotations

import torch

from . import native


def pad8(c: int) -> int:
    return (c + 7) // 8 * 8


def nchw_to_nhwc_padded(x: torch.Tensor, cp: int | None = None) -> torch.Tensor:
    """fp32/bf16 NCHW -> bf16 NHWC with channel padding (host or device)."""
    n, c, h, w = x.shape
    cp = cp or pad8(c)
    out = torch.zeros(n, h, w, cp, dtype=torch.bfloat16, device=x.device)
    out[..., :c] = x.permute(0, 2, 3, 1).to(torch.bfloat16)
    return out


def nhwc_to_nchw(x: torch.Tensor, c: int) -> torch.Tensor:
    """bf16 NHWC padded -> fp32 NCHW canonical (for tests/inspection)."""
    return x[..., :c].permute(0, 3, 1, 2).float()


def build_flatten_perm(h: int, w: int, cp: int, c_can: int,
                       device) -> torch.Tensor:
    """perm for an FC layer consuming a flattened NHWC [h][w][cp] block whose
    canonical (reference NCHW-flatten) feature order is c*h*w + y*w + x."""
    perm = torch.full((h * w * cp,), -1, dtype=torch.int32)
    for y in range(h):
        for x in range(w):
            for c in range(cp):
                if c < c_can:
                    perm[(y * w + x) * cp + c] = c * h * w + y * w + x
    return perm.to(device)


def build_identity_perm(cik: int, c_can: int, device) -> torch.Tensor:
    perm = torch.arange(cik, dtype=torch.int32)
    perm[c_can:] = -1
    return perm.to(device)


def _use_tiled(x_cip, w, stride, dgrad=False):
    """hot-path router (tools/microbench.py on MI355X, MW=2 kernel): the
    LDS-tiled input-stationary kernel wins every s1 dgrad (264-530us vs
    544-686 gather at 64-512ch; 154 vs 172 for LeNet's 5x5) and 3x3 s1
    forwards from 64 channels (331-530 vs 415-620). Thin-channel forwards
    (LeNet conv2 5x5: 107 gather vs 122 tiled) stay on the gather kernel."""
    if not (stride == 1 and w.shape[2] == w.shape[3] and w.shape[2] == 3):
        return False  # 5x5 (LeNet conv2): gather wins both directions now
    return True if dgrad else x_cip >= 64


class _Conv2dBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride, pad, cop, relu, aff):
        xc, wc = x.contiguous(), w.contiguous()
        bc = b.contiguous() if b is not None else torch.Tensor()
        if aff is not None:
            # x is the RAW previous conv output; the BN+ReLU apply runs
            # inside the tiled staging (fused-BN route: 3x3 s1 only)
            y = native.ext().conv_tiled_s1_aff_bf16(xc, wc, aff, cop, pad)
        elif _use_tiled(x.shape[3], w, stride, dgrad=False):
            y = native.ext().conv_tiled_s1_bf16(xc, wc, bc, cop, pad, relu, False)
        else:
            y = native.ext().conv2d_fwd_bf16(xc, wc, bc, cop, stride, pad, relu)
        ctx.save_for_backward(x, w, aff if aff is not None else torch.Tensor())
        ctx.meta = (stride, pad, b is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, aff = ctx.saved_tensors
        stride, pad, has_bias = ctx.meta
        dy = dy.contiguous()
        m = native.ext()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            # dgrad reads dy (channels = COp of the conv): route on that.
            # With fused BN, dx here is d(y_bn) — autograd routes it into
            # _BnVirtualFn.backward, which turns it into d(x_raw).
            if _use_tiled(dy.shape[3], w, stride, dgrad=True):
                # s1 dgrad of a KxK conv is a KxK s1 conv with pad' = K-1-pad
                dx = m.conv_tiled_s1_bf16(dy, w.contiguous(), torch.Tensor(),
                                          x.shape[3], w.shape[2] - 1 - pad,
                                          False, True)
            else:
                dx = m.conv2d_dgrad_bf16(dy, w, x.shape[1], x.shape[2],
                                         x.shape[3], stride, pad)
        if ctx.needs_input_grad[1] or (has_bias and ctx.needs_input_grad[2]):
            # with fused BN the wgrad's region staging re-applies the affine,
            # reproducing the never-materialized y_bn
            dw, db = m.conv2d_wgrad_bf16(dy, x.contiguous(), w.shape[1],
                                         w.shape[0], w.shape[2], w.shape[3],
                                         stride, pad, has_bias, aff)
        if not has_bias:
            db = None
        return dx, dw, db, None, None, None, None, None


def conv2d_bf16(x, w, b=None, stride=1, pad=0, cop=None, relu=False, aff=None):
    cop = cop or pad8(w.shape[0])
    return _Conv2dBF16Fn.apply(x, w, b, stride, pad, cop, relu, aff)


class _MaxPoolReluBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = native.ext().maxpool2x2_relu_bf16_fwd(x.contiguous())
        ctx.save_for_backward(y, idx)
        ctx.hw = (x.shape[1], x.shape[2])
        return y

    @staticmethod
    def backward(ctx, dy):
        y, idx = ctx.saved_tensors
        return native.ext().maxpool2x2_relu_bf16_bwd(dy.contiguous(), y, idx,
                                                     *ctx.hw)


def maxpool2x2_relu_bf16(x):
    return _MaxPoolReluBF16Fn.apply(x)


_ctr_dev: dict = {}


def _dropout_ctr(device):
    """Per-device RNG stream counter living in DEVICE memory so hipGraph
    replays of a captured step still draw fresh dropout masks (the bump
    kernel runs inside the capture)."""
    key = str(device)
    if key not in _ctr_dev:
        _ctr_dev[key] = torch.zeros(1, dtype=torch.int64, device=device)
    return _ctr_dev[key]


class _DropoutBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, channelwise, seed):
        y, mask = native.ext().dropout_bf16_fwd(x.contiguous(), p,
                                                bool(channelwise), int(seed),
                                                _dropout_ctr(x.device))
        ctx.save_for_backward(mask)
        ctx.meta = (p, channelwise, x.shape)
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        p, channelwise, _ = ctx.meta
        return native.ext().dropout_bf16_bwd(dy.contiguous(), mask, p,
                                             bool(channelwise)), None, None, None


def dropout_bf16(x, p=0.5, training=True, channelwise=False, seed=0):
    if not training or p == 0.0:
        return x
    return _DropoutBF16Fn.apply(x, p, channelwise, seed)


class _LinearBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, perm, cop, relu):
        y = native.ext().linear_fwd_bf16(
            x.contiguous(), w.contiguous(),
            b.contiguous() if b is not None else torch.Tensor(), perm, cop,
            relu)
        ctx.save_for_backward(x, w, y, perm)
        ctx.meta = (relu, b is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y, perm = ctx.saved_tensors
        relu, has_bias = ctx.meta
        dy = dy.contiguous()
        m = native.ext()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = m.linear_dx_bf16(dy, w, y, perm, relu)
        if ctx.needs_input_grad[1] or (has_bias and ctx.needs_input_grad[2]):
            dw, db = m.linear_dw_bf16(dy, x, y, perm, w.shape[0], w.shape[1],
                                      relu, has_bias)
        if not has_bias:
            db = None
        return dx, dw, db, None, None, None


def linear_bf16(x, w, b=None, perm=None, cop=None, relu=False):
    cop = cop or pad8(w.shape[0])
    if perm is None:
        perm = build_identity_perm(x.shape[1], w.shape[1], x.device)
    return _LinearBF16Fn.apply(x, w, b, perm, cop, relu)


class _LogSoftmaxNllBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, target, c):
        logp, loss = native.ext().logsoftmax_nll_bf16_fwd(x.contiguous(),
                                                          target.contiguous(), c)
        ctx.save_for_backward(logp, target)
        ctx.cop = x.shape[1]
        ctx.mark_non_differentiable(logp)
        return logp, loss

    @staticmethod
    def backward(ctx, dlogp_unused, dloss):
        logp, target = ctx.saved_tensors
        dx = native.ext().logsoftmax_nll_bf16_bwd(
            logp, target, dloss.reshape(1).contiguous().float(), ctx.cop)
        return dx, None, None


def logsoftmax_nll_bf16(x, target, c=10):
    return _LogSoftmaxNllBF16Fn.apply(x, target, c)


class _ConvPoolStemBF16Fn(torch.autograd.Function):
    """Fused Conv2d(CI<=2, s1 p0) + MaxPool2x2 + ReLU (the LeNet stem).
    Backward produces only (dw, db) — the stem is the input layer, so no
    dgrad is ever needed (needs_input_grad[0] is False for loader tensors)."""

    @staticmethod
    def forward(ctx, x, w, b, cop):
        p, idx = native.ext().conv_pool_stem_bf16(x.contiguous(), w.contiguous(),
                                                  b.contiguous(), cop)
        ctx.save_for_backward(x, p, idx, w)
        return p

    @staticmethod
    def backward(ctx, dpool):
        x, p, idx, w = ctx.saved_tensors
        if ctx.needs_input_grad[0]:
            raise RuntimeError("conv_pool_stem has no dgrad (input layer only)")
        dw, db = native.ext().conv2d_wgrad_pool_bf16(
            dpool.contiguous(), p, idx, x, w.shape[1], w.shape[0], w.shape[2],
            w.shape[3])
        return None, dw, db, None


def conv_pool_stem_bf16(x, w, b, cop=None):
    cop = cop or pad8(w.shape[0])
    return _ConvPoolStemBF16Fn.apply(x, w, b, cop)
