"""Model registry with the reference's --network names (distributed_nn.py:47-48,
util.py:110-224) plus the synthetic 224x224 ResNet-50."""
from .fc import FC_NN
from .lenet import LeNet
from .resnet import (
    ResNet18,
    ResNet34,
    ResNet50,
    ResNet50ImageNet,
    ResNet101,
    ResNet152,
)
from .vgg import VGG

_DATASET_SHAPE = {
    "MNIST": (1, 28, 28, 10),
    "Cifar10": (3, 32, 32, 10),
    "ImageNetSynthetic": (3, 224, 224, 1000),
}


def dataset_shape(dataset: str):
    return _DATASET_SHAPE[dataset]


def build_model(network: str, dataset: str):
    c, h, w, classes = _DATASET_SHAPE[dataset]
    name = network
    if name == "LeNet":
        return LeNet(num_classes=classes, in_channels=c)
    if name == "FC":
        return FC_NN(num_classes=classes, in_features=c * h * w)
    if name == "ResNet18":
        return ResNet18(num_classes=classes, in_channels=c)
    if name == "ResNet34":
        return ResNet34(num_classes=classes, in_channels=c)
    if name == "ResNet50":
        if dataset == "ImageNetSynthetic":
            return ResNet50ImageNet(num_classes=classes, in_channels=c)
        return ResNet50(num_classes=classes, in_channels=c)
    if name == "ResNet101":
        return ResNet101(num_classes=classes, in_channels=c)
    if name == "ResNet152":
        return ResNet152(num_classes=classes, in_channels=c)
    if name.startswith("VGG"):
        return VGG(name=name, num_classes=classes, in_channels=c)
    raise ValueError(f"unknown network {network!r}")


__all__ = ["build_model", "dataset_shape", "LeNet", "FC_NN", "VGG",
           "ResNet18", "ResNet34", "ResNet50", "ResNet101", "ResNet152",
           "ResNet50ImageNet"]
