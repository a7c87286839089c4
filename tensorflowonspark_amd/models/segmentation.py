"""Segmentation models.

``unet_mobilenet`` mirrors the reference example workload — a U-Net with a
MobileNetV2-style encoder and pix2pix upsample decoder over 128x128x3 inputs
with 3 mask classes (reference ``examples/segmentation/segmentation_spark.py:68-97``).
``deeplabv3_resnet50`` is the DeepLabV3 (ASPP over a dilated ResNet-50
backbone) configuration named by BASELINE.json config 5 (bf16, 8x MI355X,
large-activation path).

Both route BN+ReLU through the fused CDNA4 kernels; widths that divide 2048
hit the fast reduction path.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.modules import ConvTranspose2dMFMA, FusedBNReLU
from .resnet import Bottleneck, ResNet, _maybe_pack_weights


class ConvBNReLU(nn.Sequential):
    def __init__(self, cin, cout, k=3, stride=1, groups=1, dilation=1):
        pad = dilation * (k - 1) // 2
        super().__init__(
            nn.Conv2d(cin, cout, k, stride, pad, groups=groups,
                      dilation=dilation, bias=False),
            FusedBNReLU(cout))


class InvertedResidual(nn.Module):
    """MobileNetV2 block: 1x1 expand -> 3x3 depthwise -> 1x1 project."""

    def __init__(self, cin, cout, stride, expand=6):
        super().__init__()
        hidden = cin * expand
        self.use_res = stride == 1 and cin == cout
        layers = []
        if expand != 1:
            layers.append(ConvBNReLU(cin, hidden, k=1))
        layers += [
            ConvBNReLU(hidden, hidden, k=3, stride=stride, groups=hidden),
            nn.Conv2d(hidden, cout, 1, bias=False),
            nn.BatchNorm2d(cout),
        ]
        self.conv = nn.Sequential(*layers)

    def forward(self, x):
        out = self.conv(x)
        return x + out if self.use_res else out


class MobileNetV2Encoder(nn.Module):
    """Down stack with taps at strides 2/4/8/16/32 (channels 16/24/32/96/320)."""

    # (expand, cout, repeats, stride)
    CFG = [(1, 16, 1, 1), (6, 24, 2, 2), (6, 32, 3, 2), (6, 64, 4, 2),
           (6, 96, 3, 1), (6, 160, 3, 2), (6, 320, 1, 1)]

    def __init__(self):
        super().__init__()
        self.stem = ConvBNReLU(3, 32, stride=2)
        stages = []
        cin = 32
        for expand, cout, reps, stride in self.CFG:
            blocks = []
            for i in range(reps):
                blocks.append(InvertedResidual(cin, cout, stride if i == 0 else 1,
                                               expand))
                cin = cout
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.ModuleList(stages)

    def forward(self, x):
        taps = []
        x = self.stem(x)
        for i, stage in enumerate(self.stages):
            x = stage(x)
            if i in (0, 1, 2, 4, 6):  # strides 2, 4, 8, 16, 32
                taps.append(x)
        return taps


class UpBlock(nn.Module):
    """pix2pix upsample: ConvTranspose2d + BN + ReLU, concat skip."""

    def __init__(self, cin, cout):
        super().__init__()
        # routes to the dilated-input implicit-GEMM MFMA kernel on GPU when
        # cin/cout are 32-multiples; falls back to the library conv otherwise
        self.up = ConvTranspose2dMFMA(cin, cout, 4, stride=2, padding=1)
        self.bnrelu = FusedBNReLU(cout)

    def forward(self, x, skip=None):
        x = self.bnrelu(self.up(x))
        if skip is not None:
            x = torch.cat([x, skip], dim=1)
        return x


class UNetMobileNet(nn.Module):
    def __init__(self, num_classes=3):
        super().__init__()
        self.encoder = MobileNetV2Encoder()
        # decoder channel plan mirrors pix2pix upsample(512/256/128/64)
        self.up1 = UpBlock(320, 512)       # 4 -> 8, concat 96 -> 608
        self.up2 = UpBlock(512 + 96, 256)  # 8 -> 16, concat 32 -> 288
        self.up3 = UpBlock(256 + 32, 128)  # 16 -> 32, concat 24 -> 152
        self.up4 = UpBlock(128 + 24, 64)   # 32 -> 64, concat 16 -> 80
        self.head = nn.ConvTranspose2d(64 + 16, num_classes, 4, stride=2,
                                       padding=1)

    def forward(self, x):
        t2, t4, t8, t16, t32 = self.encoder(x)
        x = self.up1(t32, t16)
        x = self.up2(x, t8)
        x = self.up3(x, t4)
        x = self.up4(x, t2)
        return self.head(x)


def unet_mobilenet(num_classes=3):
    return UNetMobileNet(num_classes)


# ---------------------------------------------------------------------------
# DeepLabV3 (ASPP over dilated ResNet-50)
# ---------------------------------------------------------------------------

class ASPP(nn.Module):
    def __init__(self, cin, cout=256, rates=(12, 24, 36)):
        super().__init__()
        self.branches = nn.ModuleList(
            [ConvBNReLU(cin, cout, k=1)]
            + [ConvBNReLU(cin, cout, k=3, dilation=r) for r in rates])
        self.pool = nn.Sequential(nn.AdaptiveAvgPool2d(1),
                                  nn.Conv2d(cin, cout, 1, bias=False),
                                  nn.ReLU(inplace=True))
        self.project = ConvBNReLU(cout * (2 + len(rates)), cout, k=1)

    def forward(self, x):
        size = x.shape[-2:]
        feats = [b(x) for b in self.branches]
        p = F.interpolate(self.pool(x), size=size, mode="bilinear",
                          align_corners=False)
        feats.append(p)
        return self.project(torch.cat(feats, dim=1))


class DilatedResNet50(ResNet):
    """ResNet-50 with stage-4 stride replaced by dilation (output stride 16).

    The stage-4 convs may be the routed MFMA modules (Conv3x3/Conv1x1) —
    those are converted too: the dilated 3x3 goes back to a library conv
    (the in-tree kernels do not implement filter dilation), the downsample
    1x1 just drops its stride, and the whole-block fused path is disabled
    for these blocks (it assumes undilated convs)."""

    def __init__(self):
        super().__init__(Bottleneck, [3, 4, 6, 3], num_classes=1)
        from ..ops.modules import Conv1x1, Conv3x3
        for blk in self.stages[3]:
            blk._block_fusable = False
            blk.stride = 1
            c2 = blk.conv2
            if isinstance(c2, Conv3x3) and c2.stride == 2:
                new = nn.Conv2d(c2.weight.shape[1], c2.weight.shape[0], 3,
                                stride=1, padding=2, dilation=2, bias=False)
                with torch.no_grad():
                    new.weight.copy_(c2.weight)
                blk.conv2 = new
            elif isinstance(c2, nn.Conv2d) and c2.stride == (2, 2):
                c2.stride = (1, 1)
                c2.dilation = (2, 2)
                c2.padding = (2, 2)
            ds = blk.downsample
            if ds is not None:
                d0 = ds[0]
                if isinstance(d0, Conv1x1) and d0.stride == 2:
                    d0.stride = 1
                    d0._s2_ok = False
                elif isinstance(d0, nn.Conv2d) and d0.stride == (2, 2):
                    d0.stride = (1, 1)
        del self.fc
        del self.avgpool

    def forward(self, x):
        if not torch.jit.is_scripting():
            _maybe_pack_weights(self, x)
        x = self.stem(x)
        return self.stages(x)


class DeepLabV3(nn.Module):
    def __init__(self, num_classes=21):
        super().__init__()
        self.backbone = DilatedResNet50()
        self.aspp = ASPP(2048, 256)
        self.classifier = nn.Conv2d(256, num_classes, 1)

    def forward(self, x):
        size = x.shape[-2:]
        feats = self.backbone(x)
        out = self.classifier(self.aspp(feats))
        return F.interpolate(out, size=size, mode="bilinear",
                             align_corners=False)


def deeplabv3_resnet50(num_classes=21):
    return DeepLabV3(num_classes)
