"""Flat-vector optimizers for the ES update (fp32 master weights).

The ES "backward" produces one flat gradient-estimate vector g (ascent
direction).  We apply weight decay as in OpenAI-ES (L2 pull toward 0 added to
the *descent* direction) and step with Adam or SGD.  The GPU path fuses this
into the weighted-sum update kernel (csrc/update.hip); this module is the
reference + CPU path, and owns the state_dict layout used by checkpoints.
"""
from __future__ import annotations

import torch


class FlatAdam:
    def __init__(self, dim: int, lr: float, beta1: float = 0.9,
                 beta2: float = 0.999, eps: float = 1e-8,
                 device="cpu", dtype=torch.float32):
        self.lr = lr
        self.beta1 = beta1
        self.beta2 = beta2
        self.eps = eps
        self.t = 0
        self.m = torch.zeros(dim, dtype=dtype, device=device)
        self.v = torch.zeros(dim, dtype=dtype, device=device)

    def step(self, theta: torch.Tensor, grad_descent: torch.Tensor) -> None:
        """In-place Adam step on theta along -grad_descent."""
        self.t += 1
        self.m.mul_(self.beta1).add_(grad_descent, alpha=1 - self.beta1)
        self.v.mul_(self.beta2).addcmul_(grad_descent, grad_descent, value=1 - self.beta2)
        bc1 = 1 - self.beta1 ** self.t
        bc2 = 1 - self.beta2 ** self.t
        step_size = self.lr / bc1
        denom = (self.v / bc2).sqrt_().add_(self.eps)
        theta.addcdiv_(self.m, denom, value=-step_size)

    def state_dict(self):
        return {"t": self.t, "m": self.m, "v": self.v, "lr": self.lr,
                "beta1": self.beta1, "beta2": self.beta2, "eps": self.eps}

    def load_state_dict(self, sd):
        self.t = int(sd["t"])
        self.m = sd["m"].to(self.m.device)
        self.v = sd["v"].to(self.v.device)
        self.lr = float(sd["lr"]); self.beta1 = float(sd["beta1"])
        self.beta2 = float(sd["beta2"]); self.eps = float(sd["eps"])


class FlatSGD:
    def __init__(self, dim: int, lr: float, momentum: float = 0.9,
                 device="cpu", dtype=torch.float32):
        self.lr = lr
        self.momentum = momentum
        self.t = 0
        self.buf = torch.zeros(dim, dtype=dtype, device=device)

    def step(self, theta: torch.Tensor, grad_descent: torch.Tensor) -> None:
        self.t += 1
        self.buf.mul_(self.momentum).add_(grad_descent)
        theta.add_(self.buf, alpha=-self.lr)

    def state_dict(self):
        return {"t": self.t, "buf": self.buf, "lr": self.lr, "momentum": self.momentum}

    def load_state_dict(self, sd):
        self.t = int(sd["t"])
        self.buf = sd["buf"].to(self.buf.device)
        self.lr = float(sd["lr"]); self.momentum = float(sd["momentum"])


def make_optimizer(cfg, dim: int, device) -> "FlatAdam | FlatSGD":
    if cfg.optimizer == "adam":
        return FlatAdam(dim, cfg.lr, cfg.adam_beta1, cfg.adam_beta2,
                        cfg.adam_eps, device=device)
    if cfg.optimizer == "sgd":
        return FlatSGD(dim, cfg.lr, device=device)
    raise ValueError(f"unknown optimizer {cfg.optimizer!r}")
