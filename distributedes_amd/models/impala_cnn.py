"""IMPALA-CNN policy for 64x64x3 (Procgen-dim), BASELINE config #5.

3 stages (16/32/32 ch): conv3x3s1 -> maxpool3x3s2(pad1) -> 2 residual blocks
(relu-conv-relu-conv + skip).  64x64 -> 32 -> 16 -> 8 spatial, flatten
8*8*32=2048 -> relu -> fc 2048->256 -> relu -> head 256->A.
~622k params.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from .base import PolicyModel

_CHANNELS = (16, 32, 32)


class ImpalaCNNPolicy(PolicyModel):
    def __init__(self, num_actions: int = 15, in_ch: int = 3, hw: int = 64):
        assert hw == 64, "IMPALA-CNN geometry fixed to 64x64"
        self.obs_shape = (in_ch, hw, hw)
        self.in_ch = in_ch
        self.num_actions = num_actions
        super().__init__()

    def template(self) -> nn.Module:
        mod = nn.Module()
        cin = self.in_ch
        for s, ch in enumerate(_CHANNELS, start=1):
            mod.add_module(f"stage{s}_conv", nn.Conv2d(cin, ch, 3, padding=1))
            for r in (1, 2):
                mod.add_module(f"stage{s}_res{r}_conv1", nn.Conv2d(ch, ch, 3, padding=1))
                mod.add_module(f"stage{s}_res{r}_conv2", nn.Conv2d(ch, ch, 3, padding=1))
            cin = ch
        mod.add_module("fc", nn.Linear(8 * 8 * _CHANNELS[-1], 256))
        mod.add_module("head", nn.Linear(256, self.num_actions))
        return mod

    def pop_forward_ref(self, flat_w: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        P = flat_w.shape[0]
        w = flat_w.to(torch.float32)
        x = obs.to(torch.float32).reshape(1, -1, 64, 64)

        def gconv(x, prefix):
            W = self.spec.slice(w, prefix + ".weight")
            b = self.spec.slice(w, prefix + ".bias")
            Co = W.shape[1]
            return F.conv2d(x, W.reshape(P * Co, *W.shape[2:]), b.reshape(-1),
                            padding=1, groups=P)

        for s in (1, 2, 3):
            x = gconv(x, f"stage{s}_conv")
            x = F.max_pool2d(x, 3, stride=2, padding=1)
            for r in (1, 2):
                y = gconv(F.relu(x), f"stage{s}_res{r}_conv1")
                y = gconv(F.relu(y), f"stage{s}_res{r}_conv2")
                x = x + y
        x = F.relu(x).reshape(P, -1, 1)                       # [P, 2048, 1]
        Wf = self.spec.slice(w, "fc.weight"); bf = self.spec.slice(w, "fc.bias")
        x = F.relu(torch.baddbmm(bf.unsqueeze(-1), Wf, x))
        Wh = self.spec.slice(w, "head.weight"); bh = self.spec.slice(w, "head.bias")
        return torch.baddbmm(bh.unsqueeze(-1), Wh, x).squeeze(-1).argmax(dim=-1)
