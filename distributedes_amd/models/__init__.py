from .base import ParamSpec, PolicyModel  # noqa: F401
from .mlp import MLPPolicy  # noqa: F401
from .nature_cnn import NatureCNNPolicy  # noqa: F401
from .impala_cnn import ImpalaCNNPolicy  # noqa: F401


def make_model(cfg, env) -> PolicyModel:
    """Build the policy for (config, env). env supplies obs/action spaces."""
    if cfg.model == "mlp":
        return MLPPolicy(env.obs_dim, env.num_actions,
                         hidden_sizes=cfg.hidden_sizes,
                         continuous=env.continuous)
    if cfg.model == "nature_cnn":
        c, h, w = env.obs_shape
        return NatureCNNPolicy(num_actions=env.num_actions, in_ch=c, hw=h)
    if cfg.model == "impala_cnn":
        c, h, w = env.obs_shape
        return ImpalaCNNPolicy(num_actions=env.num_actions, in_ch=c, hw=h)
    raise ValueError(f"unknown model {cfg.model!r}")
