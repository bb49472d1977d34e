"""Batched environment interface.

All envs are vectorized over the population axis P: every member owns one env
instance; the whole batch advances per step() call.  The torch
implementations here are the reference twins of the HIP env kernels and the
CPU execution path; they are device-agnostic (tensors live wherever reset()
put them).

Fidelity policy: CartPole = exact classic dynamics; Pong = a
real simplified paddle/ball sim rendered to 84x84x4 (not ALE-bit-exact);
*Synthetic envs are dimension-faithful steppers used for throughput.
"""
from __future__ import annotations

import torch


class BatchedEnv:
    obs_shape: tuple          # e.g. (4,) or (4, 84, 84)
    num_actions: int
    continuous: bool = False

    @property
    def obs_dim(self) -> int:
        assert len(self.obs_shape) == 1
        return self.obs_shape[0]

    def reset(self, P: int, seed: int, device="cpu", env_ids=None) -> torch.Tensor:
        """Allocate state for P envs, deterministically seeded; returns obs
        [P, *obs_shape].  env_ids [P] int64 feeds the per-env hash so (a)
        results are world-size invariant (pass global indices) and (b)
        antithetic pair members can share an env stream (pass pair indices:
        mirrored sampling evaluates +eps and -eps on the same episode)."""
        raise NotImplementedError

    def step(self, actions: torch.Tensor):
        """Advance all P envs. actions: [P] int64 (discrete) or [P, A] fp32.
        Returns (obs [P,*obs_shape] fp32, reward [P] fp32, done [P] bool).
        Envs auto-reset on done (obs returned is the post-reset obs)."""
        raise NotImplementedError


def uniform_from_counter(counter: torch.Tensor, lo: float, hi: float) -> torch.Tensor:
    """Deterministic per-env uniform in [lo, hi) from an integer counter
    (device-portable, no RNG state).  All intermediates stay < 2^62 so the
    math is exact in signed int64 (torch) and uint64 (HIP) alike — the HIP
    env kernels replicate this formula bit-for-bit (csrc/common.h hash_u24)."""
    x = counter.to(torch.int64)
    x = (x ^ (x >> 17)) & 0x3FFFFFFF            # 30 bits
    x = (x * 0x9E3779B1) & 0x3FFFFFFFFFFF       # < 2^46
    x = (x ^ (x >> 13)) & 0x3FFFFFFF
    x = (x * 0x85EBCA77) & 0x3FFFFFFFFFFF
    u = ((x >> 10) & 0xFFFFFF).to(torch.float32) * (1.0 / 16777216.0)
    return lo + (hi - lo) * u
