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

    def forward(self, x, target=None):
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
