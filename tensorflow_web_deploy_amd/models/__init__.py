from . import lenet, resnet50  # noqa: F401
