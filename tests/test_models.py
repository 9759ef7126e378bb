"""Model zoo tests: parameter counts and shapes match the reference
(models.py:11-58 — CNN_MNIST 1,199,882 / CNN_CIFAR 537,610 params;
SURVEY.md §2 #16), plus the build's ResNet18 extension."""

import torch

from rlr_amd.models import CNN_MNIST, CNN_CIFAR, ResNet18, get_model


def n_params(m):
    return sum(p.numel() for p in m.parameters())


def test_cnn_mnist_param_count():
    assert n_params(CNN_MNIST()) == 1_199_882


def test_cnn_cifar_param_count():
    assert n_params(CNN_CIFAR()) == 537_610


def test_resnet18_param_count():
    n = n_params(ResNet18())
    assert 11_000_000 < n < 11_300_000  # ~11.17M


def test_registry():
    assert isinstance(get_model('fmnist'), CNN_MNIST)
    assert isinstance(get_model('fedemnist'), CNN_MNIST)
    assert isinstance(get_model('cifar10'), CNN_CIFAR)
    assert isinstance(get_model('cifar10', 'resnet18'), ResNet18)


def test_cnn_mnist_forward_shape():
    m = CNN_MNIST().eval()
    y = m(torch.randn(4, 1, 28, 28))
    assert y.shape == (4, 10)


def test_cnn_cifar_forward_shape():
    m = CNN_CIFAR().eval()
    y = m(torch.randn(4, 3, 32, 32))
    assert y.shape == (4, 10)


def test_resnet18_forward_shape():
    m = ResNet18().eval()
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)


def test_dropout_deterministic_stream():
    """Identical dropout seed -> identical training forward."""
    torch.manual_seed(0)
    m = CNN_MNIST().train()
    x = torch.randn(4, 1, 28, 28)
    m.set_dropout_seed(123)
    y1 = m(x)
    m.set_dropout_seed(123)
    y2 = m(x)
    m.set_dropout_seed(124)
    y3 = m(x)
    assert torch.equal(y1, y2)
    assert not torch.equal(y1, y3)


def test_eval_mode_no_dropout():
    torch.manual_seed(0)
    m = CNN_MNIST().eval()
    x = torch.randn(4, 1, 28, 28)
    assert torch.equal(m(x), m(x))
