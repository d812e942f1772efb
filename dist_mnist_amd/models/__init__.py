from .lenet import Net  # noqa: F401
