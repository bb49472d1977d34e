"""Public API: run(config) / train_step(state) / evaluate(...).

Reference-compatible surface (reconstructed and frozen since
the reference mount was empty): free functions + a Config object.
"""
from __future__ import annotations

import signal
from typing import Optional

import torch

from .config import Config
from .core.engine import TrainState
from .core.engine import train_step as _train_step
from .utils import load_checkpoint, save_checkpoint


def train_step(state: TrainState) -> dict:
    """Run exactly one generation; returns the per-generation stats dict."""
    return _train_step(state)


def run(config: Config, resume: Optional[str] = None,
        ops_backend: Optional[str] = None) -> TrainState:
    """Build everything and loop train_step() until the generation budget.

    Handles checkpoint cadence (inside train_step) and a SIGTERM hook that
    flushes one final checkpoint.  Returns the final
    TrainState.
    """
    state = TrainState(config, ops_backend=ops_backend)
    if resume:
        state.load_payload(load_checkpoint(resume))

    stop = {"flag": False}

    def _sigterm(signum, frame):
        stop["flag"] = True

    old = None
    try:
        old = signal.signal(signal.SIGTERM, _sigterm)
    except ValueError:
        pass  # not main thread

    try:
        while state.generation < config.generations and not stop["flag"]:
            train_step(state)
    finally:
        if config.checkpoint_path and state.rank == 0:
            save_checkpoint(config.checkpoint_path, state.checkpoint_payload())
        if old is not None:
            signal.signal(signal.SIGTERM, old)
        state.metrics.close()
    return state


def evaluate(checkpoint_or_state, config: Optional[Config] = None,
             episodes: int = 16, ops_backend: Optional[str] = None) -> dict:
    """Deterministic-policy evaluation: no perturbation.

    Accepts a TrainState or a checkpoint path (+ config).  Returns
    {'return_mean', 'return_max', 'ep_len_mean'}.
    """
    if isinstance(checkpoint_or_state, TrainState):
        state = checkpoint_or_state
    else:
        assert config is not None, "evaluate(path) needs the config"
        cfg = config.replace(pop=episodes if episodes % 2 == 0 else episodes + 1)
        state = TrainState(cfg, ops_backend=ops_backend)
        state.load_payload(load_checkpoint(checkpoint_or_state))

    P = episodes
    dt = (torch.bfloat16 if state.device.type == "cuda"
          and state.cfg.precision == "bf16" else torch.float32)
    stride = ((state.dim + 7) & ~7) if state._hip_rollout is not None else state.dim
    pop_w = torch.zeros(P, stride, dtype=dt, device=state.device)
    pop_w[:, :state.dim] = state.theta.to(dt)
    # scoped eval: explicit env ids (one distinct stream per episode — the
    # training slice would pair-share streams under antithetic sampling) and
    # observe=False so eval observations never enter the normalizer deltas.
    # No TrainState fields are touched.
    env_ids = torch.arange(P, dtype=torch.int64)
    fitness, env_steps, ep_len = state.rollout(pop_w, env_ids=env_ids,
                                               observe=False)
    return {
        "return_mean": float(fitness.mean()),
        "return_max": float(fitness.max()),
        "return_min": float(fitness.min()),
        "ep_len_mean": ep_len,
        "env_steps": env_steps,
    }
