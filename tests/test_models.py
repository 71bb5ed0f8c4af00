"""Model registry coverage: every --network x --dataset combination builds, runs a
forward pass, and produces the right logit shape."""
import pytest
import torch

from draco_amd.models import build_model, dataset_shape


@pytest.mark.parametrize("network", ["LeNet", "FC", "ResNet18", "ResNet34", "ResNet50",
                                     "ResNet101", "ResNet152", "VGG11", "VGG13",
                                     "VGG16", "VGG19"])
@pytest.mark.parametrize("dataset", ["MNIST", "Cifar10"])
def test_forward_shapes(network, dataset):
    if network == "LeNet" and dataset == "Cifar10":
        pytest.skip("LeNet geometry is 28x28 (MNIST), as in the reference")
    if network.startswith(("ResNet", "VGG")) and dataset == "MNIST":
        pytest.skip("conv stacks sized for 32x32 CIFAR input")
    c, h, w, classes = dataset_shape(dataset)
    m = build_model(network, dataset)
    x = torch.randn(2, c, h, w)
    y = m(x)
    assert y.shape == (2, classes)
    n_params = sum(p.numel() for p in m.parameters())
    assert n_params > 1000


def test_resnet50_imagenet_geometry():
    m = build_model("ResNet50", "ImageNetSynthetic")
    y = m(torch.randn(1, 3, 224, 224))
    assert y.shape == (1, 1000)
    n = sum(p.numel() for p in m.parameters())
    assert 25_000_000 < n < 26_000_000  # standard ResNet-50 ~25.6M


def test_resnet18_param_count():
    m = build_model("ResNet18", "Cifar10")
    n = sum(p.numel() for p in m.parameters())
    assert 11_000_000 < n < 11_300_000  # CIFAR ResNet-18 ~11.17M


def test_unknown_network_rejected():
    with pytest.raises(ValueError):
        build_model("AlexNet", "Cifar10")
