"""The reference's 2-conv LeNet-style Net.

Topology (pytorch/examples MNIST lineage, recalled — the reference tree at v0
was empty):

    Conv2d(1, 10, 5) -> maxpool2 -> relu
    Conv2d(10, 20, 5) -> Dropout2d -> maxpool2 -> relu
    flatten(320) -> Linear(320, 50) -> relu -> Dropout -> Linear(50, 10)
    -> log_softmax

state_dict compatibility surface (byte-compatible via
torch.save): conv1.weight [10,1,5,5], conv1.bias [10], conv2.weight
[20,10,5,5], conv2.bias [20], fc1.weight [50,320], fc1.bias [50],
fc2.weight [10,50], fc2.bias [10].

Forward dispatches through dist_mnist_amd.ops: hand-written HIP/CDNA4 kernels
on GPU, plain PyTorch fp32 reference on CPU.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from .. import ops


class Net(nn.Module):
    def __init__(self):
        super().__init__()
        # nn.Conv2d/nn.Linear used purely as parameter containers so that
        # parameter names, shapes, dtypes and default init match the
        # reference's state_dict exactly; forward goes through our ops.
        self.conv1 = nn.Conv2d(1, 10, kernel_size=5)
        self.conv2 = nn.Conv2d(10, 20, kernel_size=5)
        self.conv2_drop = nn.Dropout2d()
        self.fc1 = nn.Linear(320, 50)
        self.fc2 = nn.Linear(50, 10)
        self._bf16_native = False
        self._fc1_perm = None
        self._fc2_perm = None

    def bf16_native_(self, enable: bool = True):
        """Switch forward to the bf16 NHWC kernel path (GPU only). Inputs
        must then be NHWC bf16 [N, 28, 28, C] (C=1 unpadded for the fused
        stem, or padded-to-8 for the generic conv path).
        Master weights stay fp32 canonical — checkpoints are unchanged."""
        self._bf16_native = enable
        return self

    def forward(self, x, target=None):
        if self._bf16_native and x.is_cuda:
            return self._forward_bf16(x, target)
        if x.is_cuda:
            # fused stem (conv1 + pool + relu): the fp32 twin of the bf16
            # path's fused stem — no conv-y round trip (2x189 MB at b8192)
            x = ops.conv_pool_stem(x, self.conv1.weight, self.conv1.bias)
        else:
            x = ops.conv2d(x, self.conv1.weight, self.conv1.bias)
            x = ops.maxpool2x2_relu(x)
        x = ops.conv2d(x, self.conv2.weight, self.conv2.bias)
        x = ops.dropout(x, p=self.conv2_drop.p, training=self.training, channelwise=True)
        x = ops.maxpool2x2_relu(x)
        x = x.reshape(-1, 320)
        x = ops.linear(x, self.fc1.weight, self.fc1.bias, act_relu=True)
        x = ops.dropout(x, p=0.5, training=self.training)
        x = ops.linear(x, self.fc2.weight, self.fc2.bias)
        if target is not None:
            # fused LogSoftmax+NLLLoss: returns (log_probs, loss)
            return ops.logsoftmax_nll(x, target)
        return ops.log_softmax(x)

    def _forward_bf16(self, x, target=None):
        from ..ops import functional_bf16 as fb

        if self._fc1_perm is None or self._fc1_perm.device != x.device:
            # pool2 output is NHWC [4, 4, 24]; reference flatten order is
            # canonical NCHW (c, h, w) -> perm maps padded k -> canonical ci
            self._fc1_perm = fb.build_flatten_perm(4, 4, 24, 20, x.device)
            self._fc2_perm = fb.build_identity_perm(56, 50, x.device)
        # fused stem: conv1 + pool + relu in one kernel (no y1 round-trip)
        x = fb.conv_pool_stem_bf16(x, self.conv1.weight, self.conv1.bias, cop=16)
        x = fb.conv2d_bf16(x, self.conv2.weight, self.conv2.bias, cop=24)
        x = fb.dropout_bf16(x, p=self.conv2_drop.p, training=self.training,
                            channelwise=True)
        x = fb.maxpool2x2_relu_bf16(x)
        x = x.reshape(-1, 4 * 4 * 24)
        x = fb.linear_bf16(x, self.fc1.weight, self.fc1.bias,
                           perm=self._fc1_perm, cop=56, relu=True)
        x = fb.dropout_bf16(x, p=0.5, training=self.training)
        x = fb.linear_bf16(x, self.fc2.weight, self.fc2.bias,
                           perm=self._fc2_perm, cop=16)
        if target is not None:
            from ..ops.functional_bf16 import logsoftmax_nll_bf16

            return logsoftmax_nll_bf16(x, target, c=10)
        # standalone log-probs (eval): reuse the fused kernel's fwd only
        logp, _ = fb.logsoftmax_nll_bf16(x, torch.zeros(x.shape[0], dtype=torch.long,
                                                        device=x.device), c=10)
        return logp
