from .mnist import MNISTNet, MNISTMLP  # noqa: F401
from .resnet import ResNet, resnet50, resnet56_cifar  # noqa: F401
from .segmentation import unet_mobilenet, deeplabv3_resnet50  # noqa: F401
