from .normalizer import RunningNormalizer, IdentityNormalizer  # noqa: F401
from .checkpoint import save_checkpoint, load_checkpoint  # noqa: F401
from .metrics import MetricsLogger  # noqa: F401
