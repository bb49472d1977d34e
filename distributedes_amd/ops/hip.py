"""HIP kernel backend: loads the in-tree gfx950 extension.

The extension is built IN-TREE (setup.py build_ext --inplace -> _hip_ops.so
next to this file) so the .so ships with the repo snapshot to GPU boxes.
On a GPU box, ops FAIL LOUDLY if the extension is missing — no silent
eager fallback (the engine enforces this in _make_ops).
"""
from __future__ import annotations

import importlib

import torch

_ext = None
_err = None


def _load():
    global _ext, _err
    if _ext is not None or _err is not None:
        return _ext
    try:
        from . import _hip_ops  # built in-tree
        _ext = _hip_ops
    except ImportError as e:
        try:
            _ext = importlib.import_module("_hip_ops")
        except ImportError:
            _err = e
    return _ext


def hip_available() -> bool:
    return _load() is not None


def load_error():
    _load()
    return _err


class HipOps:
    """Same surface as ops.reference.TorchOps, backed by CDNA4 HIP kernels."""

    name = "hip"

    def __init__(self):
        self.ext = _load()
        if self.ext is None:
            raise RuntimeError(f"HIP extension not built: {_err}")

    # K1
    def fill_table(self, size: int, seed: int, device,
                   dtype=torch.float32) -> torch.Tensor:
        t = torch.empty(size, dtype=dtype, device=device)
        self.ext.noise_fill(t, seed)
        return t

    # K2
    def perturb(self, theta, table, offsets, signs, sigma,
                sigma_vec=None, out_dtype=torch.float32, chunk=None,
                pad_rows: bool = False, out=None):
        """pad_rows: allocate [P, pad16(D)] so every member row is 16B-aligned
        for the vectorized policy-forward kernels (pad tail is unwritten).
        out: reuse a persistent buffer (stable pointer for hipGraph replay)."""
        P = offsets.shape[0]
        D = theta.shape[0]
        stride = (D + 7) & ~7 if pad_rows else D
        if out is None or out.shape != (P, stride) or out.dtype != out_dtype:
            out = torch.empty(P, stride, dtype=out_dtype, device=theta.device)
        self.ext.perturb(out, theta, table, offsets, signs, float(sigma),
                         sigma_vec if sigma_vec is not None else
                         torch.empty(0, device=theta.device))
        return out

    # K10
    def weighted_noise_sum(self, table, offsets, signs, weights, dim,
                           want_sq=False, chunk=None):
        acc = torch.zeros(dim, dtype=torch.float32, device=table.device)
        acc_sq = (torch.zeros(dim, dtype=torch.float32, device=table.device)
                  if want_sq else torch.empty(0, device=table.device))
        self.ext.weighted_noise_sum(acc, acc_sq, table, offsets, signs,
                                    weights, bool(want_sq))
        return (acc, acc_sq) if want_sq else acc
