"""TorchScript exportability of every in-repo model (serving / pipeline
transform parity): scripting must succeed, match the eager model, and must
NOT mutate the original (torch.jit.script swaps children in-place via
__prepare_scriptable__, so export_saved_model scripts a deepcopy)."""

import torch

from tensorflowonspark_amd import TFNode
from tensorflowonspark_amd.models import (MNISTNet, resnet50, resnet56_cifar)
from tensorflowonspark_amd.models.segmentation import DeepLabV3, UNetMobileNet


def _check(model, x):
    import copy
    model.eval()
    sm = torch.jit.script(copy.deepcopy(model))
    with torch.no_grad():
        a, b = model(x), sm(x)
    assert (a - b).abs().max().item() < 1e-5


def test_resnet50_scriptable():
    _check(resnet50(num_classes=10), torch.randn(1, 3, 64, 64))


def test_resnet56_scriptable():
    _check(resnet56_cifar(), torch.randn(1, 3, 32, 32))


def test_unet_scriptable():
    _check(UNetMobileNet(), torch.randn(1, 3, 128, 128))


def test_deeplab_scriptable():
    _check(DeepLabV3(num_classes=5), torch.randn(1, 3, 64, 64))


def test_mnist_scriptable():
    _check(MNISTNet(), torch.randn(1, 1, 28, 28))


def test_export_does_not_mutate_model(tmp_path):
    from tensorflowonspark_amd.models.resnet import Bottleneck
    from tensorflowonspark_amd.ops.modules import StemConv7x7
    m = resnet50(num_classes=10)
    path = TFNode.export_saved_model(m, str(tmp_path))
    assert path.endswith("model.pt")
    # original modules untouched (scripting a deepcopy)
    assert isinstance(m.stem[0], StemConv7x7)
    assert isinstance(m.stages[0][0], Bottleneck)
    loaded = torch.jit.load(path)
    m.eval()
    x = torch.randn(1, 3, 64, 64)
    with torch.no_grad():
        assert (m(x) - loaded(x)).abs().max().item() < 1e-5
