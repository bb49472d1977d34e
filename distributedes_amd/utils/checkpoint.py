"""Checkpoint I/O — reference-compatible layout, atomic writes.

Layout (the reference mount was empty, so this reconstructed
shape is frozen as the contract):

    { 'network':    model.state_dict()      # conventional names fc1/conv1/...
      'optimizer':  flat-optimizer state
      'normalizer': normalizer state
      'generation': int
      'rng':        {'run_seed': int, ...}  # counters; resume continues the
                                            #  seed stream exactly
      'algo':       {'mu','sigma_vec',...}  # CEM distribution state (if CEM)
    }

Writes are atomic (tmp + os.replace) and a SIGTERM hook can flush one last
checkpoint.
"""
from __future__ import annotations

import os
import tempfile

import torch


def save_checkpoint(path: str, payload: dict) -> None:
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix=".tmp")
    try:
        with os.fdopen(fd, "wb") as f:
            torch.save(payload, f)
        os.replace(tmp, path)
    except BaseException:
        if os.path.exists(tmp):
            os.unlink(tmp)
        raise


def load_checkpoint(path: str, map_location="cpu") -> dict:
    return torch.load(path, map_location=map_location, weights_only=False)
