"""Flat-parameter plumbing shared by all policy models.

ES treats a policy as one flat fp32 vector theta.  The canonical ordering is
the torch state_dict order of the template module (conventional layer names —
fc1/fc2/..., conv1..conv3 — per SURVEY.md §2.9, so checkpoints stay
reference-shaped).  The GPU population path consumes the same flat layout.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Tuple

import torch
import torch.nn as nn


@dataclass(frozen=True)
class ParamEntry:
    name: str
    shape: Tuple[int, ...]
    numel: int
    offset: int


class ParamSpec:
    """Maps a module's state_dict <-> one flat fp32 vector."""

    def __init__(self, module: nn.Module):
        self.entries: List[ParamEntry] = []
        off = 0
        for name, p in module.state_dict().items():
            n = p.numel()
            self.entries.append(ParamEntry(name, tuple(p.shape), n, off))
            off += n
        self.dim = off
        self._by_name: Dict[str, ParamEntry] = {e.name: e for e in self.entries}

    def flatten(self, state_dict: Dict[str, torch.Tensor]) -> torch.Tensor:
        out = torch.empty(self.dim, dtype=torch.float32)
        for e in self.entries:
            out[e.offset:e.offset + e.numel] = state_dict[e.name].reshape(-1).to(torch.float32)
        return out

    def unflatten(self, theta: torch.Tensor) -> Dict[str, torch.Tensor]:
        return {e.name: theta[e.offset:e.offset + e.numel].reshape(e.shape).clone()
                for e in self.entries}

    def slice(self, theta: torch.Tensor, name: str) -> torch.Tensor:
        """View of one parameter inside a flat vector (works batched:
        theta [..., dim] -> [..., *shape])."""
        e = self._by_name[name]
        return theta[..., e.offset:e.offset + e.numel].reshape(
            *theta.shape[:-1], *e.shape)


class PolicyModel:
    """Interface every policy family implements.

    - template(): nn.Module with the canonical layer names (checkpoint layout)
    - spec: ParamSpec over the template
    - pop_forward_ref(flat_w [P, D], obs [P, *obs_shape]) -> actions [P, ...]
      pure-torch population forward (reference twin of the HIP kernels);
      flat_w may be fp32 or bf16, math in fp32.
    """

    obs_shape: tuple
    num_actions: int
    continuous: bool = False

    def __init__(self):
        self.module = self.template()
        self.spec = ParamSpec(self.module)
        self.dim = self.spec.dim

    def template(self) -> nn.Module:
        raise NotImplementedError

    def init_theta(self, seed: int) -> torch.Tensor:
        g = torch.Generator().manual_seed(seed)
        mod = self.template()
        for m in mod.modules():
            if isinstance(m, (nn.Linear, nn.Conv2d)):
                nn.init.orthogonal_(m.weight, gain=2 ** 0.5, generator=g)
                if m.bias is not None:
                    nn.init.zeros_(m.bias)
        return self.spec.flatten(mod.state_dict())

    def pop_forward_ref(self, flat_w: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError
