"""Model zoo: forward/backward shape checks (reference zoo SURVEY §2.6)."""

import pytest
import torch

from atomo_amd.models import build_model, model_names


@pytest.mark.parametrize(
    "name,in_shape,classes",
    [
        ("LeNet", (1, 28, 28), 10),
        ("FC", (1, 28, 28), 10),
        ("ResNet18", (3, 32, 32), 10),
        ("ResNet50", (3, 32, 32), 100),
        ("VGG11", (3, 32, 32), 10),
        ("DenseNet", (3, 32, 32), 10),
    ],
)
def test_forward_backward(name, in_shape, classes):
    torch.manual_seed(0)
    model = build_model(name, num_classes=classes, in_channels=in_shape[0])
    x = torch.randn(2, *in_shape)
    y = torch.randint(0, classes, (2,))
    out = model(x)
    assert out.shape == (2, classes)
    loss = torch.nn.functional.cross_entropy(out, y)
    loss.backward()
    grads = [p.grad for p in model.parameters() if p.requires_grad]
    assert all(g is not None for g in grads)
    assert any(g.abs().sum() > 0 for g in grads)


def test_alexnet_shape():
    model = build_model("AlexNet", num_classes=10)
    out = model(torch.randn(2, 3, 227, 227))
    assert out.shape == (2, 10)


def test_unknown_model():
    with pytest.raises(ValueError):
        build_model("NotAModel")


def test_zoo_covers_reference_families():
    names = set(model_names())
    for required in {"LeNet", "FC", "ResNet18", "ResNet34", "ResNet50",
                     "ResNet101", "ResNet152", "VGG11", "AlexNet", "DenseNet"}:
        assert required in names


@pytest.mark.parametrize("name", ["ResNet34", "ResNet101", "VGG16", "VGG19"])
def test_deep_zoo_forward(name):
    torch.manual_seed(0)
    model = build_model(name, num_classes=10, in_channels=3)
    out = model(torch.randn(2, 3, 32, 32))
    assert out.shape == (2, 10)
    out.sum().backward()
