"""Nature-CNN policy (conv1..conv3 + fc + head) for 84x84x4 frame stacks.

BASELINE configs #3/#4 (Pong / Atari-57-dim, 18 actions): conv 8x8s4x32,
4x4s2x64, 3x3s1x64, fc 3136->512, head 512->A, ReLU.  1,693,362 params
(SURVEY.md §2.5).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .base import PolicyModel


class NatureCNNPolicy(PolicyModel):
    def __init__(self, num_actions: int = 18, in_ch: int = 4, hw: int = 84):
        assert hw == 84, "Nature-CNN geometry is fixed to 84x84"
        self.obs_shape = (in_ch, hw, hw)
        self.in_ch = in_ch
        self.num_actions = num_actions
        super().__init__()

    def template(self) -> nn.Module:
        mod = nn.Module()
        mod.add_module("conv1", nn.Conv2d(self.in_ch, 32, 8, stride=4))
        mod.add_module("conv2", nn.Conv2d(32, 64, 4, stride=2))
        mod.add_module("conv3", nn.Conv2d(64, 64, 3, stride=1))
        mod.add_module("fc", nn.Linear(64 * 7 * 7, 512))
        mod.add_module("head", nn.Linear(512, self.num_actions))
        return mod

    def pop_forward_ref(self, flat_w: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        """flat_w [P, D], obs [P, C, 84, 84] float in [0,1] -> actions [P] int64.

        Per-member weights via grouped conv (groups=P) — reference only."""
        P = flat_w.shape[0]
        w = flat_w.to(torch.float32)
        x = obs.to(torch.float32).reshape(1, -1, 84, 84)          # [1, P*C, 84, 84]

        def gconv(x, wname, bname, stride):
            W = self.spec.slice(w, wname)                          # [P, Co, Ci, k, k]
            b = self.spec.slice(w, bname)                          # [P, Co]
            Co = W.shape[1]
            W = W.reshape(P * Co, *W.shape[2:])
            return F.conv2d(x, W, b.reshape(-1), stride=stride, groups=P)

        x = F.relu(gconv(x, "conv1.weight", "conv1.bias", 4))
        x = F.relu(gconv(x, "conv2.weight", "conv2.bias", 2))
        x = F.relu(gconv(x, "conv3.weight", "conv3.bias", 1))
        x = x.reshape(P, -1, 1)                                    # [P, 3136, 1]
        Wf = self.spec.slice(w, "fc.weight"); bf = self.spec.slice(w, "fc.bias")
        x = F.relu(torch.baddbmm(bf.unsqueeze(-1), Wf, x))
        Wh = self.spec.slice(w, "head.weight"); bh = self.spec.slice(w, "head.bias")
        out = torch.baddbmm(bh.unsqueeze(-1), Wh, x).squeeze(-1)   # [P, A]
        return out.argmax(dim=-1)
