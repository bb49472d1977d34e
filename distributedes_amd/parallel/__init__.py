from .comm import Comm  # noqa: F401
