"""CartPole-v1, exact classic dynamics, vectorized (BASELINE config #1).

Physics matches the canonical CartPole (Barto-Sutton / gym classic_control):
gravity 9.8, cart mass 1.0, pole mass 0.1, half-length 0.5, force 10, Euler
tau=0.02, termination |x|>2.4 or |theta|>12 deg, reward 1 per step, 500 cap.
Init state uniform(-0.05, 0.05) from the deterministic counter hash.
"""
from __future__ import annotations

import torch

from .base import BatchedEnv, uniform_from_counter

_GRAV = 9.8
_MC = 1.0
_MP = 0.1
_MT = _MC + _MP
_LEN = 0.5
_PML = _MP * _LEN
_FORCE = 10.0
_TAU = 0.02
_X_LIM = 2.4
_TH_LIM = 12 * 3.141592653589793 / 180.0
_MAX_STEPS = 500


class CartPoleEnv(BatchedEnv):
    obs_shape = (4,)
    num_actions = 2
    continuous = False

    def reset(self, P: int, seed: int, device="cpu", env_ids=None) -> torch.Tensor:
        self.P = P
        self.seed = seed
        self.device = torch.device(device)
        self.reset_counter = torch.zeros(P, dtype=torch.int64, device=self.device)
        base = (env_ids.to(self.device) if env_ids is not None
                else torch.arange(P, dtype=torch.int64, device=self.device))
        self._env_ids = base
        self.state = self._init_state(base * 1000 + self.reset_counter)
        self.steps = torch.zeros(P, dtype=torch.int32, device=self.device)
        return self.state.clone()

    def _init_state(self, counters: torch.Tensor) -> torch.Tensor:
        cols = []
        for j in range(4):
            c = counters * 4 + j + self.seed * 0x10001
            cols.append(uniform_from_counter(c, -0.05, 0.05))
        return torch.stack(cols, dim=1)

    def step(self, actions: torch.Tensor):
        x, x_dot, th, th_dot = self.state.unbind(dim=1)
        force = torch.where(actions.to(torch.bool), _FORCE, -_FORCE).to(torch.float32)
        costh = torch.cos(th)
        sinth = torch.sin(th)
        temp = (force + _PML * th_dot * th_dot * sinth) / _MT
        th_acc = (_GRAV * sinth - costh * temp) / (
            _LEN * (4.0 / 3.0 - _MP * costh * costh / _MT))
        x_acc = temp - _PML * th_acc * costh / _MT
        x = x + _TAU * x_dot
        x_dot = x_dot + _TAU * x_acc
        th = th + _TAU * th_dot
        th_dot = th_dot + _TAU * th_acc
        self.state = torch.stack([x, x_dot, th, th_dot], dim=1)
        self.steps += 1

        fell = (x.abs() > _X_LIM) | (th.abs() > _TH_LIM)
        done = fell | (self.steps >= _MAX_STEPS)
        reward = torch.ones(self.P, dtype=torch.float32, device=self.device)

        if done.any():
            self.reset_counter = torch.where(done, self.reset_counter + 1, self.reset_counter)
            fresh = self._init_state(self._env_ids * 1000 + self.reset_counter)
            self.state = torch.where(done.unsqueeze(1), fresh, self.state)
            self.steps = torch.where(done, torch.zeros_like(self.steps), self.steps)
        return self.state.clone(), reward, done
