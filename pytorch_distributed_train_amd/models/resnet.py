"""ResNet-18/34/50/101/152 from scratch (torchvision is not installed here).

Module names follow the torchvision resnet naming (conv1, bn1, layer1..4,
<block>.convN/bnN/downsample.0/.1, fc) so state_dicts keep the reference
checkpoint layout ("same checkpoint layout"); the compute
path is NHWC with fused BN+ReLU and fused residual add+ReLU — BN modules
that are immediately followed by a ReLU carry the ReLU as a kernel epilogue
instead of a separate pass ("fused with ReLU where adjacent").

Forward contract: input NCHW fp32/any-dtype [N,3,H,W] (what the reference's
loader feeds); internally permuted once to NHWC in the compute dtype.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import functional as OF
from .layers import (BatchNorm2dNHWC, Conv2dNHWC, GlobalAvgPoolNHWC, Linear,
                     MaxPool2dNHWC, ReLUNHWC)


def conv3x3(in_c, out_c, stride=1):
    return Conv2dNHWC(in_c, out_c, 3, stride=stride, padding=1)


def conv1x1(in_c, out_c, stride=1):
    return Conv2dNHWC(in_c, out_c, 1, stride=stride, padding=0)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(inplanes, planes, stride)
        self.bn1 = BatchNorm2dNHWC(planes)
        self.bn1.fuse_relu = True
        self.relu = ReLUNHWC()  # kept for name parity; fused in practice
        self.conv2 = conv3x3(planes, planes)
        self.bn2 = BatchNorm2dNHWC(planes)
        self.bn2.fuse_relu = True  # fused with the residual join
        self.downsample = downsample
        self.stride = stride
        # residual-fork gradient fusion (GradHolder protocol): the
        # identity-path gradient (bn2 dres, or the downsample conv's dx)
        # is folded into conv1's bwd-data epilogue; the producer returns
        # None to autograd, so no fork add exists
        self._fork = OF.GradHolder()
        if downsample is not None:
            downsample[0]._dx_sink = self._fork
        else:
            self.bn2._dres_sink = self._fork

    def forward(self, x):
        if self._fork.t is not None:
            raise RuntimeError(
                "residual-fork gradient was never folded (GradHolder "
                "protocol violation — conv1 backward did not consume it)")
        identity = x
        out = self.bn1(self.conv1(x))  # ReLU fused into bn1
        if self.downsample is not None:
            identity = self.downsample(identity)
        # bn2 + residual add + ReLU in one kernel
        return self.bn2(self.conv2(out), identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv1x1(inplanes, planes)
        self.bn1 = BatchNorm2dNHWC(planes)
        self.bn1.fuse_relu = True
        self.conv2 = conv3x3(planes, planes, stride)
        self.bn2 = BatchNorm2dNHWC(planes)
        self.bn2.fuse_relu = True
        self.conv3 = conv1x1(planes, planes * self.expansion)
        self.bn3 = BatchNorm2dNHWC(planes * self.expansion)
        self.bn3.fuse_relu = True  # fused with the residual join
        self.relu = ReLUNHWC()
        self.downsample = downsample
        self.stride = stride
        self._fork = OF.GradHolder()  # see BasicBlock
        if downsample is not None:
            downsample[0]._dx_sink = self._fork
        else:
            self.bn3._dres_sink = self._fork

    def forward(self, x):
        if self._fork.t is not None:
            raise RuntimeError(
                "residual-fork gradient was never folded (GradHolder "
                "protocol violation — conv1 backward did not consume it)")
        identity = x
        out = self.bn1(self.conv1(x))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(identity)
        # bn3 + residual add + ReLU in one kernel
        return self.bn3(self.conv3(out), identity)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, compute_dtype=None):
        super().__init__()
        self.compute_dtype = compute_dtype or torch.float32
        self.inplanes = 64
        self.conv1 = Conv2dNHWC(3, 64, 7, stride=2, padding=3)
        self.bn1 = BatchNorm2dNHWC(64)
        self.bn1.fuse_relu = True
        self.relu = ReLUNHWC()
        self.maxpool = MaxPool2dNHWC(3, 2, 1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = GlobalAvgPoolNHWC()
        self.fc = Linear(512 * block.expansion, num_classes)
        # every conv feeds a BN: its fwd epilogue pre-reduces BN statistics
        for m in self.modules():
            for cname, bname in (("conv1", "bn1"), ("conv2", "bn2"),
                                 ("conv3", "bn3")):
                conv = getattr(m, cname, None)
                if isinstance(conv, Conv2dNHWC) and \
                        isinstance(getattr(m, bname, None), BatchNorm2dNHWC):
                    conv.emit_bn_stats = True
            if isinstance(m, nn.Sequential) and len(m) == 2 and \
                    isinstance(m[0], Conv2dNHWC) and \
                    isinstance(m[1], BatchNorm2dNHWC):
                m[0].emit_bn_stats = True  # downsample pairs
        # two-dy backward wiring (GradHolder protocol): block k's residual
        # producer fills blocks[k]._fork; the CONSUMER is the op that
        # produced block k's input — the last BN of block k-1 — which sums
        # it as a second dy inside its backward kernels. Block 0's input
        # comes from maxpool: its fork stays unwired (one autograd add on
        # a 56x56x64 tensor — negligible).
        blocks = [b for layer in (self.layer1, self.layer2, self.layer3,
                                  self.layer4) for b in layer]
        first = blocks[0]
        if first.downsample is not None:
            first.downsample[0]._dx_sink = None
        else:
            last_bn0 = getattr(first, "bn3", None) or first.bn2
            last_bn0._dres_sink = None
        for prev, blk in zip(blocks, blocks[1:]):
            last_bn = getattr(prev, "bn3", None) or prev.bn2
            last_bn._dy2_source = blk._fork

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                BatchNorm2dNHWC(planes * block.expansion),
            )
        layers = [block(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes))
        return nn.Sequential(*layers)

    def forward(self, x):
        # NCHW input -> NHWC compute dtype, once
        x = x.permute(0, 2, 3, 1).to(self.compute_dtype).contiguous()
        x = self.bn1(self.conv1(x))  # ReLU fused
        x = self.maxpool(x)
        x = self.layer1(x)
        x = self.layer2(x)
        x = self.layer3(x)
        x = self.layer4(x)
        x = self.avgpool(x)  # [N, C]
        return self.fc(x)


def resnet18(num_classes=1000, compute_dtype=None):
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, compute_dtype)


def resnet34(num_classes=1000, compute_dtype=None):
    return ResNet(BasicBlock, [3, 4, 6, 3], num_classes, compute_dtype)


def resnet50(num_classes=1000, compute_dtype=None):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, compute_dtype)


def resnet101(num_classes=1000, compute_dtype=None):
    return ResNet(Bottleneck, [3, 4, 23, 3], num_classes, compute_dtype)


def resnet152(num_classes=1000, compute_dtype=None):
    return ResNet(Bottleneck, [3, 8, 36, 3], num_classes, compute_dtype)
