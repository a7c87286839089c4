"""MNIST models matching the reference example workloads.

``MNISTNet`` mirrors the reference Keras CNN — Conv2D(32,3,relu) → MaxPool(2) →
Flatten → Dense(64,relu) → Dense(10) (reference
``examples/mnist/keras/mnist_spark.py:13-19``). ``MNISTMLP`` is the small MLP
used for the CPU world_size=2 plumbing config (BASELINE.json config 1).
"""

import torch.nn as nn

from ..ops.modules import Conv2dIm2colMFMA, DenseMFMA


class MNISTNet(nn.Module):
    """GPU path runs entirely on in-tree kernels: the small-Cin conv as
    im2col + MFMA GEMM, the Dense layers on the MFMA GEMM (VERDICT r01
    item 6 — BASELINE config 2 previously ran on library ops)."""

    def __init__(self, num_classes=10):
        super().__init__()
        self.features = nn.Sequential(
            Conv2dIm2colMFMA(1, 32, 3), nn.ReLU(inplace=True),
            nn.MaxPool2d(2))
        self.classifier = nn.Sequential(
            nn.Flatten(),
            DenseMFMA(32 * 13 * 13, 64), nn.ReLU(inplace=True),
            DenseMFMA(64, num_classes))

    def forward(self, x):
        return self.classifier(self.features(x))


class MNISTMLP(nn.Module):
    def __init__(self, num_classes=10, hidden=128):
        super().__init__()
        self.net = nn.Sequential(
            nn.Flatten(),
            nn.Linear(784, hidden), nn.ReLU(inplace=True),
            nn.Linear(hidden, num_classes))

    def forward(self, x):
        return self.net(x)
