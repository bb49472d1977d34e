from .base import BatchedEnv  # noqa: F401
from .cartpole import CartPoleEnv  # noqa: F401
from .pong import PongEnv  # noqa: F401
from .synthetic import (Atari57SyntheticEnv, HumanoidSyntheticEnv,  # noqa: F401
                        ProcgenSyntheticEnv)

_REGISTRY = {
    "cartpole": CartPoleEnv,
    "pong": PongEnv,
    "atari57_synth": Atari57SyntheticEnv,
    "procgen_synth": ProcgenSyntheticEnv,
    "humanoid_synth": HumanoidSyntheticEnv,
}


def make_env(name: str) -> BatchedEnv:
    if name not in _REGISTRY:
        raise KeyError(f"unknown env {name!r}; have {sorted(_REGISTRY)}")
    return _REGISTRY[name]()
