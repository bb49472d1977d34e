"""Fitness shaping: centered rank / z-normalize / raw.

Runs identically on every rank after the fitness all-gather, so it adds no
communication.  The GPU path has a HIP kernel twin (csrc/rank.hip) verified
against this reference.
"""
from __future__ import annotations

import torch


def centered_ranks(fitness: torch.Tensor) -> torch.Tensor:
    """OpenAI-ES centered rank transform: rank/(n-1) - 0.5 in [-0.5, 0.5].

    Ties broken by index (stable double-argsort), matching the HIP kernel.
    """
    n = fitness.numel()
    if n == 1:
        return torch.zeros_like(fitness, dtype=torch.float32)
    order = torch.argsort(fitness.to(torch.float32), stable=True)
    ranks = torch.empty(n, dtype=torch.float32, device=fitness.device)
    ranks[order] = torch.arange(n, dtype=torch.float32, device=fitness.device)
    return ranks / (n - 1) - 0.5


def normalized_fitness(fitness: torch.Tensor) -> torch.Tensor:
    f = fitness.to(torch.float32)
    std = f.std(unbiased=False)
    return (f - f.mean()) / (std + 1e-8)


def shape_fitness(fitness: torch.Tensor, mode: str) -> torch.Tensor:
    if mode == "centered":
        return centered_ranks(fitness)
    if mode == "normalized":
        return normalized_fitness(fitness)
    if mode == "raw":
        return fitness.to(torch.float32)
    raise ValueError(f"unknown rank_shaping {mode!r}")
