"""Model registry (reference models.py:4-8, extended with ResNet18)."""

from .cnn import CNN_MNIST, CNN_CIFAR
from .resnet import ResNet18, BasicBlock


def get_model(data, model=None):
    """data -> model, as in the reference (models.py:4-8); the optional
    `model` override selects the build's extensions (e.g. 'resnet18')."""
    if model == 'resnet18':
        return ResNet18()
    if data in ('fmnist', 'fedemnist'):
        return CNN_MNIST()
    if data == 'cifar10':
        return CNN_CIFAR()
    raise ValueError(f"no model for dataset {data}")


__all__ = ['CNN_MNIST', 'CNN_CIFAR', 'ResNet18', 'BasicBlock', 'get_model']
