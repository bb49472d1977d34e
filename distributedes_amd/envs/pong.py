"""PongSynthetic: a real simplified paddle/ball game rendered to 84x84x4.

BASELINE config #3's vectorized on-GPU env.  A genuine game (ball physics,
player paddle controlled by the policy, tracking-AI opponent, scoring), NOT
ALE-bit-exact — SURVEY.md §7.4 fidelity policy.

Geometry (units = pixels on the 84x84 court):
  player paddle: column x=80..81, height 8; opponent: x=2..3, height 8
  ball: 2x2 at (bx, by), velocity (vx, vy) with |vx| in {1.2..2.0}
  actions: a % 3 -> 0 stay, 1 up (-2 px), 2 down (+2 px)   (18-action head)
  reward: +1 when the opponent misses, -1 when the player misses
  episode: first to 5 points (either side) ends it; horizon truncates
Observation: 4-frame stack [P,4,84,84] fp32 in [0,1]; channel 3 = newest.
"""
from __future__ import annotations

import torch

from .base import BatchedEnv, uniform_from_counter

_H = 84
_PAD_H = 8.0
_PLAYER_X = 80.0
_OPP_X = 2.0
_WIN_SCORE = 5


class PongEnv(BatchedEnv):
    obs_shape = (4, 84, 84)
    num_actions = 18
    continuous = False

    def reset(self, P: int, seed: int, device="cpu", env_ids=None) -> torch.Tensor:
        self.P = P
        self.seed = seed
        self.device = torch.device(device)
        self._ids = (env_ids.to(self.device) if env_ids is not None
                     else torch.arange(P, dtype=torch.int64, device=self.device))
        self.serve_counter = torch.zeros(P, dtype=torch.int64, device=self.device)
        self.player_y = torch.full((P,), 42.0, device=self.device)
        self.opp_y = torch.full((P,), 42.0, device=self.device)
        self.score = torch.zeros(P, dtype=torch.int32, device=self.device)   # player - opponent
        self.points = torch.zeros(P, dtype=torch.int32, device=self.device)  # total points played
        self._serve(torch.ones(P, dtype=torch.bool, device=self.device))
        self.frames = torch.zeros(P, 4, _H, _H, device=self.device)
        f = self._render()
        self.frames[:] = f.unsqueeze(1)
        return self.frames.clone()

    def _serve(self, mask: torch.Tensor):
        """(Re)launch the ball for envs in mask, deterministic per serve."""
        c = self._ids * 977 + self.serve_counter * 131 + self.seed
        bx = torch.full((self.P,), 42.0, device=self.device)
        by = uniform_from_counter(c, 20.0, 64.0)
        vx_mag = uniform_from_counter(c + 1, 1.2, 2.0)
        vx_sign = torch.where(uniform_from_counter(c + 2, 0.0, 1.0) < 0.5, -1.0, 1.0)
        vy = uniform_from_counter(c + 3, -1.5, 1.5)
        if not hasattr(self, "bx"):
            self.bx, self.by = bx, by
            self.vx, self.vy = vx_mag * vx_sign, vy
        else:
            self.bx = torch.where(mask, bx, self.bx)
            self.by = torch.where(mask, by, self.by)
            self.vx = torch.where(mask, vx_mag * vx_sign, self.vx)
            self.vy = torch.where(mask, vy, self.vy)

    def step(self, actions: torch.Tensor):
        a = actions.to(torch.int64) % 3
        dy = torch.where(a == 1, -2.0, torch.where(a == 2, 2.0, 0.0)).to(torch.float32)
        self.player_y = (self.player_y + dy).clamp(_PAD_H / 2, _H - _PAD_H / 2)
        # opponent tracking AI (speed 1.5 px/step toward ball)
        delta = self.by - self.opp_y
        self.opp_y = (self.opp_y + delta.clamp(-1.5, 1.5)).clamp(_PAD_H / 2, _H - _PAD_H / 2)

        self.bx = self.bx + self.vx
        self.by = self.by + self.vy
        # bounce off top/bottom
        low = self.by < 1.0
        high = self.by > _H - 1.0
        self.by = torch.where(low, 2.0 - self.by, self.by)
        self.by = torch.where(high, 2.0 * (_H - 1.0) - self.by, self.by)
        self.vy = torch.where(low | high, -self.vy, self.vy)

        reward = torch.zeros(self.P, device=self.device)
        # player side
        at_player = self.bx >= _PLAYER_X
        hit_p = at_player & ((self.by - self.player_y).abs() <= _PAD_H / 2)
        miss_p = at_player & ~hit_p
        # reflect + English from hit position
        self.vx = torch.where(hit_p, -self.vx.abs(), self.vx)
        self.vy = torch.where(hit_p, self.vy + 0.25 * (self.by - self.player_y), self.vy)
        self.bx = torch.where(hit_p, 2.0 * _PLAYER_X - self.bx, self.bx)
        # opponent side
        at_opp = self.bx <= _OPP_X
        hit_o = at_opp & ((self.by - self.opp_y).abs() <= _PAD_H / 2)
        miss_o = at_opp & ~hit_o
        self.vx = torch.where(hit_o, self.vx.abs(), self.vx)
        self.vy = torch.where(hit_o, self.vy + 0.25 * (self.by - self.opp_y), self.vy)
        self.bx = torch.where(hit_o, 2.0 * _OPP_X - self.bx, self.bx)
        self.vy = self.vy.clamp(-2.5, 2.5)

        reward = torch.where(miss_o, reward + 1.0, reward)
        reward = torch.where(miss_p, reward - 1.0, reward)
        scored = miss_p | miss_o
        self.score = self.score + miss_o.to(torch.int32) - miss_p.to(torch.int32)
        self.points = self.points + scored.to(torch.int32)
        if scored.any():
            self.serve_counter = torch.where(scored, self.serve_counter + 1, self.serve_counter)
            self._serve(scored)

        done = (self.score.abs() >= _WIN_SCORE) | (self.points >= 2 * _WIN_SCORE + 5)
        if done.any():
            # full reset of finished games
            self.serve_counter = torch.where(done, self.serve_counter + 7, self.serve_counter)
            self.player_y = torch.where(done, torch.full_like(self.player_y, 42.0), self.player_y)
            self.opp_y = torch.where(done, torch.full_like(self.opp_y, 42.0), self.opp_y)
            self.score = torch.where(done, torch.zeros_like(self.score), self.score)
            self.points = torch.where(done, torch.zeros_like(self.points), self.points)
            self._serve(done)

        frame = self._render()
        self.frames = torch.cat([self.frames[:, 1:], frame.unsqueeze(1)], dim=1)
        if done.any():
            self.frames = torch.where(done.view(-1, 1, 1, 1),
                                      frame.unsqueeze(1).expand_as(self.frames),
                                      self.frames)
        return self.frames.clone(), reward, done

    def _render(self) -> torch.Tensor:
        """Rasterize current state -> [P, 84, 84] fp32 in [0,1]."""
        P = self.P
        dev = self.device
        ys = torch.arange(_H, device=dev, dtype=torch.float32).view(1, _H, 1)
        f = torch.zeros(P, _H, _H, device=dev)
        # paddles: columns fixed, rows within +-4 of paddle center
        prow = ((ys - self.player_y.view(-1, 1, 1)).abs() <= _PAD_H / 2)
        orow = ((ys - self.opp_y.view(-1, 1, 1)).abs() <= _PAD_H / 2)
        f[:, :, 80:82] = prow.to(torch.float32).expand(P, _H, 2)
        f[:, :, 2:4] = orow.to(torch.float32).expand(P, _H, 2)
        # ball 2x2
        bx = self.bx.round().clamp(0, _H - 2).to(torch.int64)
        by = self.by.round().clamp(0, _H - 2).to(torch.int64)
        idx = torch.arange(P, device=dev)
        for dy_ in (0, 1):
            for dx_ in (0, 1):
                f[idx, (by + dy_), (bx + dx_)] = 1.0
        return f
