"""ResNet family (ResNet-50 v1.5 ImageNet; ResNet-56 CIFAR).

The reference's benchmark workload is the TF-official ResNet run under TFoS
(reference ``examples/resnet/resnet_cifar_dist.py``; the synthetic-data path at
``:160-168`` is the baseline template). This is a from-scratch PyTorch
implementation of the same architectures, structured so the BN+ReLU pairs route
through the fused CDNA4 HIP kernel (``ops.FusedBNReLU``) on GPU and through
plain PyTorch on CPU.

ResNet-50 v1.5: the stride-2 3x3 conv sits in the middle of the bottleneck
(not the 1x1), matching the config every published ResNet-50 images/sec number
uses.
"""

import torch
import torch.nn as nn

from ..ops.modules import (Conv1x1, Conv3x3, FusedBN, FusedBNAddReLU,
                           FusedBNReLU, FusedMaxPool2d, StemConv7x7)


@torch.jit.ignore
def _maybe_pack_weights(model, x) -> None:
    # one batched kernel re-packs every fused-path weight when any parameter
    # changed (vs ~300 eager permute/cast launches per step)
    if x.is_cuda and x.dtype == torch.bfloat16 and model.training:
        from ..ops.packplan import ensure_packed
        ensure_packed(model, x)


def conv3x3(cin, cout, stride=1):
    if stride in (1, 2) and cin % 32 == 0 and cout % 32 == 0 and cout >= 64:
        # 3x3s (stride 1 and 2) route to the implicit-GEMM MFMA kernel
        return Conv3x3(cin, cout, stride)
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin, cout, stride=1):
    if stride == 1 or (stride == 2 and cin % 32 == 0 and cout % 32 == 0):
        # pointwise convs route to the hand-written MFMA kernels on gfx950
        # (stride 1: GEMM; stride 2: implicit-GEMM downsample conv)
        return Conv1x1(cin, cout, stride)
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, width, stride=1, downsample=None):
        super().__init__()
        self.stride = stride
        self.conv1 = conv1x1(cin, width)
        self.bnrelu1 = FusedBNReLU(width)
        self.conv2 = conv3x3(width, width, stride)  # v1.5: stride on the 3x3
        self.bnrelu2 = FusedBNReLU(width)
        self.conv3 = conv1x1(width, width * self.expansion)
        # block tail bn3 + residual-add + relu as ONE fused op
        self.bn3 = FusedBNAddReLU(width * self.expansion)
        self.downsample = downsample
        # whole-block fused path (manual backward: residual-join grad adds
        # fused into the conv dgrad epilogues): requires every conv on the
        # MFMA kernels and a (Conv1x1, FusedBN) downsample
        from ..ops.modules import Conv1x1, Conv3x3, FusedBN
        bns = [self.bnrelu1, self.bnrelu2, self.bn3] + \
            ([downsample[1]] if downsample is not None else [])
        uniform_bn = (len({b.eps for b in bns}) == 1
                      and len({b.momentum for b in bns}) == 1)
        self._block_fusable = (
            isinstance(self.conv1, Conv1x1) and isinstance(self.conv2, Conv3x3)
            and isinstance(self.conv3, Conv1x1) and cin % 32 == 0
            and uniform_bn
            and (downsample is None or
                 (isinstance(downsample[0], Conv1x1)
                  and type(downsample[1]) is FusedBN)))

    def forward(self, x):
        import os
        if (self.training and x.is_cuda and x.dtype == torch.bfloat16
                and self._block_fusable
                and os.environ.get("TFOS_FUSED_BLOCK", "on") != "off"):
            from ..ops import get_ext
            if get_ext(required=True) is not None:
                from ..ops.modules import _BottleneckFn
                x = x.contiguous(memory_format=torch.channels_last)
                bn1, bn2, bn3 = self.bnrelu1, self.bnrelu2, self.bn3
                for bn in (bn1, bn2, bn3):
                    bn.num_batches_tracked += 1
                if self.downsample is not None:
                    dconv, dbn = self.downsample[0], self.downsample[1]
                    dbn.num_batches_tracked += 1
                    dargs = (dconv.weight, dbn.weight, dbn.bias,
                             dbn.running_mean, dbn.running_var)
                else:
                    dargs = (None, None, None, None, None)
                return _BottleneckFn.apply(
                    x, self.conv1.weight, bn1.weight, bn1.bias,
                    bn1.running_mean, bn1.running_var,
                    self.conv2.weight, bn2.weight, bn2.bias,
                    bn2.running_mean, bn2.running_var,
                    self.conv3.weight, bn3.weight, bn3.bias,
                    bn3.running_mean, bn3.running_var,
                    *dargs, self.stride, bn1.momentum, bn1.eps,
                    getattr(self, "_tfos_packs", None))
        identity = x
        out = self.bnrelu1(self.conv1(x))
        out = self.bnrelu2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.bn3(self.conv3(out), identity)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, width, stride=1, downsample=None):
        super().__init__()
        self.conv1 = conv3x3(cin, width, stride)
        self.bnrelu1 = FusedBNReLU(width)
        self.conv2 = conv3x3(width, width)
        self.bn2 = FusedBNAddReLU(width)
        self.downsample = downsample

    def forward(self, x):
        identity = x
        out = self.bnrelu1(self.conv1(x))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.bn2(self.conv2(out), identity)


class _PlainBottleneck(nn.Module):
    """Scriptable clone of Bottleneck (plain torch ops, shared weights)."""

    def __init__(self, blk):
        super().__init__()
        from ..ops.modules import _bn_clone, _conv_clone
        self.conv1 = _conv_clone(blk.conv1)
        self.bn1 = _bn_clone(blk.bnrelu1)
        self.conv2 = _conv_clone(blk.conv2)
        self.bn2 = _bn_clone(blk.bnrelu2)
        self.conv3 = _conv_clone(blk.conv3)
        self.bn3 = _bn_clone(blk.bn3)
        if blk.downsample is not None:
            self.downsample = nn.Sequential(
                _conv_clone(blk.downsample[0]), _bn_clone(blk.downsample[1]))
        else:
            self.downsample = None
        self.train(blk.training)

    def forward(self, x):
        identity = x
        out = torch.relu(self.bn1(self.conv1(x)))
        out = torch.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return torch.relu(out + identity)


class _PlainBasicBlock(nn.Module):
    def __init__(self, blk):
        super().__init__()
        from ..ops.modules import _bn_clone, _conv_clone
        self.conv1 = _conv_clone(blk.conv1)
        self.bn1 = _bn_clone(blk.bnrelu1)
        self.conv2 = _conv_clone(blk.conv2)
        self.bn2 = _bn_clone(blk.bn2)
        if blk.downsample is not None:
            self.downsample = nn.Sequential(
                _conv_clone(blk.downsample[0]), _bn_clone(blk.downsample[1]))
        else:
            self.downsample = None
        self.train(blk.training)

    def forward(self, x):
        identity = x
        out = torch.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return torch.relu(out + identity)


def _keepalive_clone(cls):
    def prepare(self):
        from ..ops.modules import _SCRIPT_CLONE_KEEPALIVE
        _SCRIPT_CLONE_KEEPALIVE.append(self)
        return cls(self)
    return prepare


Bottleneck.__prepare_scriptable__ = _keepalive_clone(_PlainBottleneck)
BasicBlock.__prepare_scriptable__ = _keepalive_clone(_PlainBasicBlock)


class ResNet(nn.Module):
    def __init__(self, block, layers, num_classes=1000, cifar_stem=False,
                 base_width=64):
        super().__init__()
        self.inplanes = 16 if cifar_stem else base_width
        if cifar_stem:
            self.stem = nn.Sequential(
                conv3x3(3, self.inplanes), FusedBNReLU(self.inplanes))
            widths = [16, 32, 64]
            strides = [1, 2, 2]
        else:
            self.stem = nn.Sequential(
                StemConv7x7(3, self.inplanes),
                FusedBNReLU(self.inplanes),
                FusedMaxPool2d(3, stride=2, padding=1))
            widths = [64, 128, 256, 512]
            strides = [1, 2, 2, 2]
        stages = []
        for w, s, n in zip(widths, strides, layers):
            stages.append(self._make_stage(block, w, n, s))
        self.stages = nn.Sequential(*stages)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(widths[-1] * block.expansion, num_classes)

        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out", nonlinearity="relu")
        # zero-init last BN in each block (standard ResNet recipe)
        for m in self.modules():
            if isinstance(m, Bottleneck):
                nn.init.zeros_(m.bn3.weight)
            elif isinstance(m, BasicBlock):
                nn.init.zeros_(m.bn2.weight)

    def _make_stage(self, block, width, blocks, stride):
        downsample = None
        if stride != 1 or self.inplanes != width * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, width * block.expansion, stride),
                FusedBN(width * block.expansion))
        layers = [block(self.inplanes, width, stride, downsample)]
        self.inplanes = width * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, width))
        return nn.Sequential(*layers)

    def forward(self, x):
        if not torch.jit.is_scripting():
            _maybe_pack_weights(self, x)
        x = self.stem(x)
        x = self.stages(x)
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def resnet50(num_classes=1000):
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes=num_classes)


def resnet56_cifar(num_classes=10):
    return ResNet(BasicBlock, [9, 9, 9], num_classes=num_classes, cifar_stem=True)
