from .lenet import LeNet5  # noqa: F401
