"""Dimension-faithful synthetic steppers (SURVEY.md §7.4 fidelity policy).

No gym/ALE/MuJoCo/Procgen exists in this offline image, and BASELINE
prescribes synthetic data for the metric — these envs keep the obs/action
shapes and per-step data volumes exact while the dynamics are a cheap chaotic
latent map.  All are named *Synthetic and documented as such.

  Atari57Synthetic   obs (4,84,84) fp32, 18 discrete actions
  ProcgenSynthetic   obs (3,64,64) fp32, 15 discrete actions
  HumanoidSynthetic  obs (376,) fp32, 17 continuous actions
"""
from __future__ import annotations

import math

import torch

from .base import BatchedEnv, uniform_from_counter

_ZDIM = 8


class _LatentImageEnv(BatchedEnv):
    """Latent z[8] per env -> procedural image obs; action rotates the latent."""

    frames: int
    hw: int
    channels: int
    horizon = 1000

    def reset(self, P: int, seed: int, device="cpu", env_ids=None) -> torch.Tensor:
        self.P = P
        self.seed = seed
        self.device = torch.device(device)
        ids = (env_ids.to(self.device) if env_ids is not None
               else torch.arange(P, dtype=torch.int64, device=self.device))
        cols = [uniform_from_counter(ids * _ZDIM + j + seed * 0x51ED, -1.0, 1.0)
                for j in range(_ZDIM)]
        self.z = torch.stack(cols, dim=1)            # [P, 8]
        self.steps = torch.zeros(P, dtype=torch.int32, device=self.device)
        hh = torch.arange(self.hw, device=self.device, dtype=torch.float32)
        self._hgrid = hh.view(1, 1, self.hw, 1)
        self._wgrid = hh.view(1, 1, 1, self.hw)
        self._cidx = torch.arange(self.channels, device=self.device,
                                  dtype=torch.int64).view(1, self.channels)
        return self._render()

    def _render(self) -> torch.Tensor:
        cidx = self._cidx.expand(self.P, self.channels)
        zc0 = self.z.gather(1, cidx % _ZDIM).view(self.P, self.channels, 1, 1)
        zc1 = self.z.gather(1, (cidx + 1) % _ZDIM).view(self.P, self.channels, 1, 1)
        zc2 = self.z.gather(1, (cidx + 2) % _ZDIM).view(self.P, self.channels, 1, 1)
        pat = torch.sin(3.0 * zc0 + 0.07 * self._hgrid * (1.0 + zc1)
                        + 0.05 * self._wgrid * (1.0 + zc2))
        return 0.5 + 0.5 * pat

    def step(self, actions: torch.Tensor):
        a = actions.to(torch.float32)
        phase = (a * (2.0 * math.pi / self.num_actions)).view(-1, 1)
        zr = torch.roll(self.z, 1, dims=1)
        self.z = torch.tanh(1.3 * self.z + 0.7 * zr * torch.cos(phase)
                            + 0.15 * torch.sin(phase + zr))
        self.steps += 1
        reward = torch.sin(3.0 * self.z[:, 0] + 0.5 * a)
        done = self.steps >= self.horizon
        if done.any():
            self.steps = torch.where(done, torch.zeros_like(self.steps), self.steps)
        return self._render(), reward, done


class Atari57SyntheticEnv(_LatentImageEnv):
    obs_shape = (4, 84, 84)
    num_actions = 18
    channels = 4
    hw = 84


class ProcgenSyntheticEnv(_LatentImageEnv):
    obs_shape = (3, 64, 64)
    num_actions = 15
    channels = 3
    hw = 64


class HumanoidSyntheticEnv(BatchedEnv):
    obs_shape = (376,)
    num_actions = 17
    continuous = True
    horizon = 1000

    def reset(self, P: int, seed: int, device="cpu", env_ids=None) -> torch.Tensor:
        self.P = P
        self.device = torch.device(device)
        ids = (env_ids.to(self.device) if env_ids is not None
               else torch.arange(P, dtype=torch.int64, device=self.device))
        D = self.obs_dim
        # cheap deterministic init: one hashed column scaled by position phase
        base = uniform_from_counter(ids + seed, -0.1, 0.1).view(-1, 1)
        pos = torch.arange(D, device=self.device, dtype=torch.float32).view(1, -1)
        self.s = base * torch.cos(0.1 * pos) + 0.05 * torch.sin(0.37 * pos + base)
        self.steps = torch.zeros(P, dtype=torch.int32, device=self.device)
        return self.s.clone()

    def step(self, actions: torch.Tensor):
        a = actions.to(torch.float32)                       # [P, 17]
        D = self.obs_dim
        u = a.repeat(1, (D + a.shape[1] - 1) // a.shape[1])[:, :D]
        sr = torch.roll(self.s, 1, dims=1)
        self.s = torch.tanh(1.2 * self.s + 0.6 * sr + 0.3 * u)
        self.steps += 1
        # reward favors coordinated, bounded actuation (chaotic but smooth)
        reward = (self.s[:, :17] * a).mean(dim=1) - 0.05 * (a * a).mean(dim=1)
        done = self.steps >= self.horizon
        if done.any():
            self.steps = torch.where(done, torch.zeros_like(self.steps), self.steps)
        return self.s.clone(), reward, done
