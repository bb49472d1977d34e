from . import noise  # noqa: F401
from .engine import TrainState, train_step  # noqa: F401
from .optim import FlatAdam, FlatSGD, make_optimizer  # noqa: F401
from .rank import centered_ranks, shape_fitness  # noqa: F401
