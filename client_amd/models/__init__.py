"""client_amd.models — model zoo for the benchmark/fixture server."""

from .resnet import ResNet50, resnet50

__all__ = ["ResNet50", "resnet50"]
