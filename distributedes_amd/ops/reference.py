"""Pure-torch reference twins of every HIP kernel.

These are (a) the CPU execution path, (b) the eager-GPU self-baseline the
hand kernels are measured against, and (c) the oracles for kernel parity
tests.  Member-chunked so the GPU eager path never materializes pop x dim
index tensors beyond `chunk` rows.
"""
from __future__ import annotations

import torch

from ..core import noise as noise_mod


class TorchOps:
    name = "torch_ref"

    # --- K1 ---------------------------------------------------------------
    def fill_table(self, size: int, seed: int, device,
                   dtype=torch.float32) -> torch.Tensor:
        t = noise_mod.fill_noise_table_cpu(size, seed)
        return t.to(dtype).to(device)

    # --- K2: theta_i = theta + sigma * sign_i * table[off_i : off_i+D] ----
    def perturb(self, theta: torch.Tensor, table: torch.Tensor,
                offsets: torch.Tensor, signs: torch.Tensor, sigma: float,
                sigma_vec: torch.Tensor = None, out_dtype=torch.float32,
                chunk: int = 256) -> torch.Tensor:
        """offsets [P] int64 (device), signs [P] fp32 -> [P, D] out_dtype.
        sigma_vec [D] (CEM) overrides scalar sigma per-coordinate."""
        P = offsets.shape[0]
        D = theta.shape[0]
        out = torch.empty(P, D, dtype=out_dtype, device=theta.device)
        ar = torch.arange(D, device=theta.device)
        for s in range(0, P, chunk):
            e = min(s + chunk, P)
            idx = offsets[s:e].unsqueeze(1) + ar
            eps = table[idx].to(torch.float32)              # [c, D]
            eps.mul_(signs[s:e].unsqueeze(1))
            if sigma_vec is not None:
                eps.mul_(sigma_vec)
            else:
                eps.mul_(sigma)
            out[s:e] = (theta + eps).to(out_dtype)
        return out

    # --- K10: partial weighted noise sum (the ES "backward") --------------
    def weighted_noise_sum(self, table: torch.Tensor, offsets: torch.Tensor,
                           signs: torch.Tensor, weights: torch.Tensor,
                           dim: int, want_sq: bool = False, chunk: int = 256):
        """returns sum_i w_i * s_i * eps_i  [D] fp32 (and optionally
        sum_i w_i * eps_i^2 for the CEM variance refit)."""
        dev = table.device
        acc = torch.zeros(dim, dtype=torch.float32, device=dev)
        acc_sq = torch.zeros(dim, dtype=torch.float32, device=dev) if want_sq else None
        ar = torch.arange(dim, device=dev)
        P = offsets.shape[0]
        for s in range(0, P, chunk):
            e = min(s + chunk, P)
            idx = offsets[s:e].unsqueeze(1) + ar
            eps = table[idx].to(torch.float32)              # [c, D]
            w = weights[s:e].unsqueeze(1)
            acc += (eps * (w * signs[s:e].unsqueeze(1))).sum(dim=0)
            if want_sq:
                acc_sq += (eps * eps * w).sum(dim=0)
        return (acc, acc_sq) if want_sq else acc
