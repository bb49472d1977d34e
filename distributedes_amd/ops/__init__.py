from .reference import TorchOps  # noqa: F401
