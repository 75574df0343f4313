"""client_amd.models — model zoo for the benchmark/fixture server."""

from .densenet import DenseNet121, densenet121
from .resnet import ResNet50, resnet50

__all__ = ["ResNet50", "resnet50", "DenseNet121", "densenet121"]
