"""Autograd-integrated ops.

Each op dispatches to the hand-written HIP/CDNA4 kernels (``native/``) for CUDA
(= ROCm/HIP) tensors and to a plain PyTorch fp32 reference on CPU. The CPU path
exists for the no-GPU test tier and the reference's gloo plumbing config
(BASELINE.json config 1); on a GPU box the native kernels
are the only path — a missing extension raises (ops/native.py).

Reference op surface being reproduced:
  K1/K2 Conv2d fwd/bwd (im2col-GEMM), K3 ReLU, K4 MaxPool2d 2x2, K5 Linear,
  K6 Dropout/Dropout2d, K7 LogSoftmax+NLLLoss, K8 SGD step.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from . import native


def _nc(t):
    """contiguous or None-passthrough"""
    return t.contiguous() if t is not None else None


# ---------------------------------------------------------------------------
# Conv2d (im2col-GEMM on device)
# ---------------------------------------------------------------------------


class _Conv2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride, padding):
        ctx.save_for_backward(x, w)
        ctx.stride = stride
        ctx.padding = padding
        ctx.has_bias = b is not None
        if x.is_cuda:
            return native.ext().conv2d_fwd(
                _nc(x), _nc(w), _nc(b) if b is not None else torch.Tensor(), stride, padding
            )
        return F.conv2d(x, w, b, stride=stride, padding=padding)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        stride, padding = ctx.stride, ctx.padding
        dy = _nc(dy)
        dx = dw = db = None
        if dy.is_cuda:
            m = native.ext()
            if ctx.needs_input_grad[0]:
                dx = m.conv2d_bwd_input(dy, _nc(w), x.shape[2], x.shape[3], stride, padding)
            if ctx.needs_input_grad[1] or ctx.needs_input_grad[2]:
                dw, db = m.conv2d_bwd_weight(
                    dy, _nc(x), w.shape[2], w.shape[3], stride, padding, ctx.has_bias
                )
        else:
            if ctx.needs_input_grad[0]:
                dx = torch.nn.grad.conv2d_input(x.shape, w, dy, stride=stride, padding=padding)
            if ctx.needs_input_grad[1]:
                dw = torch.nn.grad.conv2d_weight(x, w.shape, dy, stride=stride, padding=padding)
            if ctx.has_bias and ctx.needs_input_grad[2]:
                db = dy.sum(dim=(0, 2, 3))
        if not ctx.has_bias:
            db = None
        return dx, dw, db, None, None


def conv2d(x, w, b=None, stride=1, padding=0):
    return _Conv2dFn.apply(x, w, b, stride, padding)


class _ConvPoolStemF32Fn(torch.autograd.Function):
    """Fused Conv2d(CI==1, s1 p0) + MaxPool2x2 + ReLU, fp32 NCHW (the LeNet
    stem, GPU only — the fp32 twin of ops.functional_bf16.conv_pool_stem).
    Backward produces only (dw, db): the stem is the input layer."""

    @staticmethod
    def forward(ctx, x, w, b):
        p, idx = native.ext().conv_pool_stem_f32(_nc(x), _nc(w), _nc(b))
        ctx.save_for_backward(x, p, idx, w)
        return p

    @staticmethod
    def backward(ctx, dpool):
        x, p, idx, w = ctx.saved_tensors
        if ctx.needs_input_grad[0]:
            raise RuntimeError("conv_pool_stem has no dgrad (input layer only)")
        dw, db = native.ext().conv2d_wgrad_pool_f32(
            _nc(dpool), p, idx, _nc(x), w.shape[2], w.shape[3])
        return None, dw, db


def conv_pool_stem(x, w, b):
    """GPU fp32 fused stem; callers fall back to conv2d+maxpool on CPU."""
    return _ConvPoolStemF32Fn.apply(x, w, b)


# ---------------------------------------------------------------------------
# Fused MaxPool2d(2) + ReLU  (reference computes relu(max_pool2d(conv(x), 2));
# relu is monotone so relu(max4(x)) == the same graph, one kernel fwd + bwd)
# ---------------------------------------------------------------------------


class _MaxPool2ReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            y, idx = native.ext().maxpool2x2_relu_fwd(_nc(x))
            ctx.save_for_backward(y, idx)
            ctx.in_shape = x.shape
            return y
        p = F.max_pool2d(x, 2)
        y = F.relu(p)
        ctx.save_for_backward(x, y)
        ctx.in_shape = x.shape
        return y

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        if dy.is_cuda:
            y, idx = a, b
            return native.ext().maxpool2x2_relu_bwd(_nc(dy), y, idx, ctx.in_shape[2], ctx.in_shape[3])
        x, y = a, b
        with torch.enable_grad():
            x2 = x.detach().requires_grad_(True)
            out = F.relu(F.max_pool2d(x2, 2))
            (dx,) = torch.autograd.grad(out, x2, dy)
        return dx


def maxpool2x2_relu(x):
    return _MaxPool2ReluFn.apply(x)


# ---------------------------------------------------------------------------
# ReLU (standalone, for fc1; fused into linear epilogue on the native path)
# ---------------------------------------------------------------------------


class _ReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            y = native.ext().relu_fwd(_nc(x))
        else:
            y = F.relu(x)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        if dy.is_cuda:
            return native.ext().relu_bwd(_nc(dy), y)
        return dy * (y > 0).to(dy.dtype)


def relu(x):
    return _ReluFn.apply(x)


# ---------------------------------------------------------------------------
# Linear (optionally with fused ReLU epilogue)
# ---------------------------------------------------------------------------


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, act_relu):
        ctx.act_relu = act_relu
        ctx.has_bias = b is not None
        if x.is_cuda:
            y = native.ext().linear_fwd(
                _nc(x), _nc(w), _nc(b) if b is not None else torch.Tensor(), act_relu
            )
        else:
            y = F.linear(x, w, b)
            if act_relu:
                y = F.relu(y)
        ctx.save_for_backward(x, w, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        dy = _nc(dy)
        dx = dw = db = None
        if dy.is_cuda:
            dx, dw, db = native.ext().linear_bwd(
                dy, _nc(x), _nc(w), y, ctx.act_relu, ctx.has_bias
            )
        else:
            if ctx.act_relu:
                dy = dy * (y > 0).to(dy.dtype)
            dx = dy @ w
            dw = dy.t() @ x
            db = dy.sum(0) if ctx.has_bias else None
        if not ctx.has_bias:
            db = None
        if not ctx.needs_input_grad[0]:
            dx = None
        return dx, dw, db, None


def linear(x, w, b=None, act_relu=False):
    return _LinearFn.apply(x, w, b, act_relu)


# ---------------------------------------------------------------------------
# LogSoftmax + NLLLoss (fused fwd+bwd on device; returns (logp, loss))
# The reference computes output = log_softmax(model_out) then
# F.nll_loss(output, target). logp is returned for the
# caller's logging/accuracy use and is marked non-differentiable; the loss
# carries the gradient (softmax - onehot)/N.
# ---------------------------------------------------------------------------


class _LogSoftmaxNllFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, target):
        if x.is_cuda:
            logp, loss = native.ext().logsoftmax_nll_fwd(_nc(x), _nc(target))
        else:
            logp = F.log_softmax(x, dim=1)
            loss = F.nll_loss(logp, target)
        ctx.save_for_backward(logp, target)
        ctx.mark_non_differentiable(logp)
        return logp, loss

    @staticmethod
    def backward(ctx, dlogp_unused, dloss):
        logp, target = ctx.saved_tensors
        if logp.is_cuda:
            dx = native.ext().logsoftmax_nll_bwd(logp, target, dloss.reshape(1).contiguous())
        else:
            n = logp.shape[0]
            dx = logp.exp()
            dx[torch.arange(n), target] -= 1.0
            dx = dx * (dloss / n)
        return dx, None


def logsoftmax_nll(x, target):
    """Fused log_softmax(dim=1) + mean NLL loss. Returns (log_probs, loss)."""
    return _LogSoftmaxNllFn.apply(x, target)


class _LogSoftmaxFn(torch.autograd.Function):
    """Standalone log_softmax (model output when no target is supplied,
    e.g. eval/serving). Generic backward."""

    @staticmethod
    def forward(ctx, x):
        if x.is_cuda:
            logp = native.ext().logsoftmax_fwd(_nc(x))
        else:
            logp = F.log_softmax(x, dim=1)
        ctx.save_for_backward(logp)
        return logp

    @staticmethod
    def backward(ctx, dy):
        (logp,) = ctx.saved_tensors
        if dy.is_cuda:
            return native.ext().logsoftmax_bwd(_nc(dy), logp)
        return dy - logp.exp() * dy.sum(dim=1, keepdim=True)


def log_softmax(x):
    return _LogSoftmaxFn.apply(x)


# ---------------------------------------------------------------------------
# Dropout / Dropout2d — counter-based (Philox-style) RNG on device, mask saved
# as u8. Channelwise=True zeroes whole (n, c) feature maps (Dropout2d).
# ---------------------------------------------------------------------------

_dropout_ctr_dev: dict = {}


def _dropout_ctr(device):
    """Device-side RNG stream counter (shared scheme with the bf16 path):
    the bump kernel runs on-stream, so hipGraph replays of a captured step
    still draw fresh masks."""
    key = str(device)
    if key not in _dropout_ctr_dev:
        _dropout_ctr_dev[key] = torch.zeros(1, dtype=torch.int64, device=device)
    return _dropout_ctr_dev[key]


class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, channelwise, seed):
        if p == 0.0:
            ctx.p = 0.0
            return x
        if x.is_cuda:
            y, mask = native.ext().dropout_fwd(_nc(x), p, bool(channelwise), int(seed),
                                               _dropout_ctr(x.device))
        else:
            if channelwise:
                keep = (torch.rand(x.shape[0], x.shape[1], 1, 1, device=x.device) >= p).to(x.dtype)
            else:
                keep = (torch.rand_like(x.float()) >= p).to(x.dtype)
            mask = keep
            y = x * keep / (1.0 - p)
        ctx.save_for_backward(mask)
        ctx.p = p
        ctx.channelwise = channelwise
        return y

    @staticmethod
    def backward(ctx, dy):
        if ctx.p == 0.0:
            return dy, None, None, None
        (mask,) = ctx.saved_tensors
        if dy.is_cuda:
            dx = native.ext().dropout_bwd(_nc(dy), mask, ctx.p, bool(ctx.channelwise))
        else:
            dx = dy * mask / (1.0 - ctx.p)
        return dx, None, None, None


def dropout(x, p=0.5, training=True, channelwise=False, seed=0):
    if not training or p == 0.0:
        return x
    return _DropoutFn.apply(x, p, channelwise, seed)
