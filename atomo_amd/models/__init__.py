"""Model zoo with the reference's families.

Fresh modern-PyTorch implementations of the architectures the reference
ships in /root/reference/src/model_ops/ (lenet.py, resnet.py, vgg.py,
alexnet.py, densenet.py, fc_nn.py).  The reference's ``*Split`` variants —
manual per-layer backward interleaved with MPI sends — are NOT reimplemented
as classes: their capability (communication/computation overlap) is provided
by backward-hook driven per-layer encode+send on a side HIP stream in
``atomo_amd.parallel`` instead, which works for every model here.
"""

from .lenet import LeNet
from .fc_nn import FC_NN
from .resnet import ResNet18, ResNet34, ResNet50, ResNet101, ResNet152
from .vgg import vgg11, vgg11_bn, vgg13, vgg13_bn, vgg16, vgg16_bn, vgg19, vgg19_bn
from .alexnet import AlexNet
from .densenet import DenseNet

_ZOO = {
    "LeNet": lambda num_classes=10, in_channels=1: LeNet(num_classes, in_channels),
    # "FC" is the reference's CLI name (distributed_worker.py:147)
    "FC": lambda num_classes=10, in_channels=1: FC_NN(num_classes),
    "FC_NN": lambda num_classes=10, in_channels=1: FC_NN(num_classes),
    "ResNet18": lambda num_classes=10, in_channels=3: ResNet18(num_classes, in_channels),
    "ResNet34": lambda num_classes=10, in_channels=3: ResNet34(num_classes, in_channels),
    "ResNet50": lambda num_classes=10, in_channels=3: ResNet50(num_classes, in_channels),
    "ResNet101": lambda num_classes=10, in_channels=3: ResNet101(num_classes, in_channels),
    "ResNet152": lambda num_classes=10, in_channels=3: ResNet152(num_classes, in_channels),
    "VGG11": lambda num_classes=10, in_channels=3: vgg11_bn(num_classes, in_channels),
    "VGG13": lambda num_classes=10, in_channels=3: vgg13_bn(num_classes, in_channels),
    "VGG16": lambda num_classes=10, in_channels=3: vgg16_bn(num_classes, in_channels),
    "VGG19": lambda num_classes=10, in_channels=3: vgg19_bn(num_classes, in_channels),
    "AlexNet": lambda num_classes=10, in_channels=3: AlexNet(num_classes, in_channels),
    # reference default config: depth 190 / growth 40
    # (model_ops/densenet.py:60-116); "DenseNet40" keeps the light variant
    "DenseNet": lambda num_classes=10, in_channels=3: DenseNet(
        depth=190, growth_rate=40, num_classes=num_classes,
        in_channels=in_channels
    ),
    "DenseNet40": lambda num_classes=10, in_channels=3: DenseNet(
        depth=40, growth_rate=12, num_classes=num_classes,
        in_channels=in_channels
    ),
}


def build_model(name: str, num_classes: int = 10, in_channels: int = 3):
    """Zoo dispatch mirroring build_model (sync_replicas_master_nn.py:146-171
    / distributed_worker.py:139-164)."""
    if name not in _ZOO:
        raise ValueError(f"unknown network {name!r}; expected one of {sorted(_ZOO)}")
    return _ZOO[name](num_classes=num_classes, in_channels=in_channels)


def model_names():
    return sorted(_ZOO)


__all__ = [
    "build_model",
    "model_names",
    "LeNet",
    "FC_NN",
    "ResNet18",
    "ResNet34",
    "ResNet50",
    "ResNet101",
    "ResNet152",
    "AlexNet",
    "DenseNet",
    "vgg11",
    "vgg11_bn",
    "vgg13",
    "vgg13_bn",
    "vgg16",
    "vgg16_bn",
    "vgg19",
    "vgg19_bn",
]
