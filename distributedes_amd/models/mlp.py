"""MLP policy (fc1/fc2/.../head), tanh activations.

Covers BASELINE configs #1 (CartPole 4->64->64->2, discrete argmax) and
#2 (Humanoid-dim 376->256->256->17, continuous tanh).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from .base import PolicyModel


class MLPPolicy(PolicyModel):
    def __init__(self, obs_dim: int, num_actions: int, hidden_sizes=(64, 64),
                 continuous: bool = False):
        self.obs_dim = obs_dim
        self.obs_shape = (obs_dim,)
        self.num_actions = num_actions
        self.hidden_sizes = tuple(hidden_sizes)
        self.continuous = continuous
        super().__init__()

    def template(self) -> nn.Module:
        mod = nn.Sequential()
        sizes = (self.obs_dim,) + self.hidden_sizes
        layers = {}
        for i in range(len(self.hidden_sizes)):
            layers[f"fc{i+1}"] = nn.Linear(sizes[i], sizes[i + 1])
        layers["head"] = nn.Linear(sizes[-1], self.num_actions)
        mod = nn.Module()
        for name, layer in layers.items():
            mod.add_module(name, layer)
        return mod

    def pop_forward_ref(self, flat_w: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        """flat_w [P, D] (fp32/bf16), obs [P, obs_dim] -> logits/actions:
        discrete -> int64 actions [P]; continuous -> fp32 actions [P, A]."""
        w = flat_w.to(torch.float32)
        x = obs.to(torch.float32).unsqueeze(-1)          # [P, in, 1]
        nh = len(self.hidden_sizes)
        for i in range(nh):
            W = self.spec.slice(w, f"fc{i+1}.weight")    # [P, out, in]
            b = self.spec.slice(w, f"fc{i+1}.bias")      # [P, out]
            x = torch.tanh(torch.baddbmm(b.unsqueeze(-1), W, x))
        W = self.spec.slice(w, "head.weight")
        b = self.spec.slice(w, "head.bias")
        out = torch.baddbmm(b.unsqueeze(-1), W, x).squeeze(-1)  # [P, A]
        if self.continuous:
            return torch.tanh(out)
        return out.argmax(dim=-1)
