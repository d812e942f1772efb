"""ResNet-family bf16 NHWC autograd ops (BASELINE.json config 5): BatchNorm2d
with fp32 statistics, MaxPool 3x3/s2/p1 (deterministic gather backward),
global average pool, fused residual add+ReLU, standalone ReLU."""

from __future__ import annotations

import torch

from . import native


class _BatchNormBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, relu):
        y, mean, invstd = native.ext().bn_fwd_bf16(
            x.contiguous(), gamma.contiguous(), beta.contiguous(),
            running_mean, running_var, bool(training), momentum, eps, relu)
        ctx.save_for_backward(x, gamma, beta, mean, invstd)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, beta, mean, invstd = ctx.saved_tensors
        dy = dy.contiguous()
        # relu gate is RECOMPUTED from (x, affine) — bitwise the forward's
        # yact>0 (bn_bwd_bf16 docstring) — so y is never saved or re-read
        dx, dgamma, dbeta = native.ext().bn_bwd_bf16(
            dy, x, mean, invstd, gamma, beta, ctx.relu)
        return dx, dgamma, dbeta, None, None, None, None, None, None


def batchnorm_bf16(x, gamma, beta, running_mean, running_var, training=True,
                   momentum=0.1, eps=1e-5, relu=False):
    return _BatchNormBF16Fn.apply(x, gamma, beta, running_mean, running_var,
                                  training, momentum, eps, relu)


class _BnVirtualFn(torch.autograd.Function):
    """BatchNorm whose APPLY is fused into the consumer (conv staging or the
    residual add): forward computes only the batch statistics and the
    per-channel affine aff=[2][Cp] (scale, shift); the normalized tensor is
    never materialized. Backward gates (relu=True) by RECOMPUTING
    bf16(scale*x+shift) > 0 — bitwise the fused forward's value — instead of
    reading a saved activation."""

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, training,
                momentum, eps, relu):
        mean, invstd, aff = native.ext().bn_stats_bf16(
            x.contiguous(), gamma.contiguous(), beta.contiguous(),
            running_mean, running_var, bool(training), momentum, eps)
        ctx.save_for_backward(x, gamma, beta, mean, invstd)
        ctx.relu = relu
        ctx.mark_non_differentiable(aff)
        # no zero-tensor materialization for d_aff (it is ignored below) —
        # saves a fill launch per BN backward inside the captured step
        ctx.set_materialize_grads(False)
        return x.view_as(x), aff

    @staticmethod
    def backward(ctx, d_out, d_aff_unused):
        x, gamma, beta, mean, invstd = ctx.saved_tensors
        dx, dgamma, dbeta = native.ext().bn_bwd_fused_bf16(
            d_out.contiguous(), x, mean, invstd, gamma, beta, ctx.relu)
        return dx, dgamma, dbeta, None, None, None, None, None, None


def bn_virtual_bf16(x, gamma, beta, running_mean, running_var, training=True,
                    momentum=0.1, eps=1e-5, relu=False):
    """Returns (x_token, aff): pass x_token + aff to the fused consumer."""
    return _BnVirtualFn.apply(x, gamma, beta, running_mean, running_var,
                              training, momentum, eps, relu)


class _AddAffReluBF16Fn(torch.autograd.Function):
    """Residual add+ReLU with the BN-apply of one/both inputs fused in; the
    gradients returned for a and b are d(bn_out) for each side, which
    autograd routes into the matching _BnVirtualFn backward."""

    @staticmethod
    def forward(ctx, a, b, affa, affb):
        out = native.ext().add_aff_relu_bf16(
            a.contiguous(), b.contiguous(), affa,
            affb if affb is not None else torch.Tensor())
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, dy):
        (out,) = ctx.saved_tensors
        g = native.ext().relu_gate_bf16(dy.contiguous(), out)
        return g, g, None, None


def add_aff_relu_bf16(a, b, affa, affb=None):
    return _AddAffReluBF16Fn.apply(a, b, affa, affb)


class _MaxPool3x3s2BF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = native.ext().maxpool3x3s2_fwd_bf16(x.contiguous())
        ctx.save_for_backward(idx)
        ctx.hw = (x.shape[1], x.shape[2])
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        return native.ext().maxpool3x3s2_bwd_bf16(dy.contiguous(), idx, *ctx.hw)


def maxpool3x3s2_bf16(x):
    return _MaxPool3x3s2BF16Fn.apply(x)


class _GapBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.hw = (x.shape[1], x.shape[2])
        return native.ext().gap_fwd_bf16(x.contiguous())

    @staticmethod
    def backward(ctx, dy):
        return native.ext().gap_bwd_bf16(dy.contiguous(), *ctx.hw)


def global_avgpool_bf16(x):
    return _GapBF16Fn.apply(x)


class _AddReluBF16Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        out = native.ext().add_bf16(a.contiguous(), b.contiguous(), True)
        ctx.save_for_backward(out)
        return out

    @staticmethod
    def backward(ctx, dy):
        (out,) = ctx.saved_tensors
        g = native.ext().relu_gate_bf16(dy.contiguous(), out)
        return g, g


def add_relu_bf16(a, b):
    return _AddReluBF16Fn.apply(a, b)
