from .lenet import Net  # noqa: F401
from .resnet import ResNet18  # noqa: F401
