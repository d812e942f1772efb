"""ResNet-18 on the bf16 NHWC engine (BASELINE.json config 5: "ResNet-18 on
synthetic 3x224x224, DDP 8xMI355X — stress MFMA conv + RCCL").

Parameter containers are standard nn modules with torchvision-resnet18
naming (conv1, bn1, layer{1..4}.{0,1}.conv{1,2} / bn{1,2} / downsample.{0,1},
fc) so state_dicts interchange with the stock model. Forward runs entirely on
the hand-written HIP kernels: implicit-GEMM bf16 convs, BatchNorm with fp32
stats, deterministic 3x3/s2 maxpool, fused residual add+ReLU, global avgpool,
MFMA linear. Input: NHWC bf16 [N, 224, 224, 8] (RGB in channels 0-2).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional_bf16 as fb
from ..ops import functional_resnet as fr


def _pad8(c):
    return (c + 7) // 8 * 8


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        # BN-apply fusion (docs/ROADMAP.md round 2): bn1's apply+ReLU runs
        # inside conv2's tiled staging, bn2's (and the downsample bn's)
        # inside the residual add — no normalized activation is ever
        # written/re-read. bn_virtual computes stats + the per-channel
        # affine only; backward recomputes the relu gate from (x, affine).
        t = self.training
        out = fb.conv2d_bf16(x, self.conv1.weight, stride=self.stride, pad=1,
                             cop=_pad8(self.conv1.out_channels))
        out, aff1 = fr.bn_virtual_bf16(out, self.bn1.weight, self.bn1.bias,
                                       self.bn1.running_mean,
                                       self.bn1.running_var, training=t,
                                       momentum=self.bn1.momentum,
                                       eps=self.bn1.eps, relu=True)
        out = fb.conv2d_bf16(out, self.conv2.weight, pad=1,
                             cop=_pad8(self.conv2.out_channels), aff=aff1)
        out, aff2 = fr.bn_virtual_bf16(out, self.bn2.weight, self.bn2.bias,
                                       self.bn2.running_mean,
                                       self.bn2.running_var, training=t,
                                       momentum=self.bn2.momentum,
                                       eps=self.bn2.eps)
        if self.downsample is not None:
            conv, bn = self.downsample[0], self.downsample[1]
            idn = fb.conv2d_bf16(x, conv.weight, stride=conv.stride[0],
                                 cop=_pad8(conv.out_channels))
            idn, affd = fr.bn_virtual_bf16(idn, bn.weight, bn.bias,
                                           bn.running_mean, bn.running_var,
                                           training=t, momentum=bn.momentum,
                                           eps=bn.eps)
        else:
            idn, affd = x, None
        return fr.add_aff_relu_bf16(out, idn, aff2, affd)


class ResNet18(nn.Module):
    def __init__(self, num_classes: int = 1000):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.layer1 = self._make_layer(64, 64, 2, stride=1)
        self.layer2 = self._make_layer(64, 128, 2, stride=2)
        self.layer3 = self._make_layer(128, 256, 2, stride=2)
        self.layer4 = self._make_layer(256, 512, 2, stride=2)
        self.fc = nn.Linear(512, num_classes)
        self._fc_perm = None

    @staticmethod
    def _make_layer(inplanes, planes, blocks, stride):
        downsample = None
        if stride != 1 or inplanes != planes:
            downsample = nn.Sequential(
                nn.Conv2d(inplanes, planes, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes),
            )
        layers = [BasicBlock(inplanes, planes, stride, downsample)]
        for _ in range(1, blocks):
            layers.append(BasicBlock(planes, planes))
        return nn.Sequential(*layers)

    def forward(self, x, target=None):
        t = self.training
        x = fb.conv2d_bf16(x, self.conv1.weight, stride=2, pad=3, cop=64)
        x = fr.batchnorm_bf16(x, self.bn1.weight, self.bn1.bias,
                              self.bn1.running_mean, self.bn1.running_var,
                              training=t, momentum=self.bn1.momentum,
                              eps=self.bn1.eps, relu=True)
        x = fr.maxpool3x3s2_bf16(x)
        for layer in (self.layer1, self.layer2, self.layer3, self.layer4):
            for block in layer:
                x = block(x)
        x = fr.global_avgpool_bf16(x)  # [N, 512]
        if self._fc_perm is None or self._fc_perm.device != x.device:
            self._fc_perm = fb.build_identity_perm(512, 512, x.device)
        x = fb.linear_bf16(x, self.fc.weight, self.fc.bias, perm=self._fc_perm,
                           cop=_pad8(self.fc.out_features))
        if target is not None:
            return fb.logsoftmax_nll_bf16(x, target, c=self.fc.out_features)
        logp, _ = fb.logsoftmax_nll_bf16(
            x, torch.zeros(x.shape[0], dtype=torch.long, device=x.device),
            c=self.fc.out_features)
        return logp
