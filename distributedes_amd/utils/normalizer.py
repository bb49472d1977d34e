"""Running observation normalizer.

Welford-style accumulation in fp64 (count, sum, sumsq).  Per-generation
observations accumulate into a DELTA buffer; commit() all-reduces only the
delta and folds it into the global accumulators on every rank — so cross-rank
merge never double-counts and the global stats stay rank-identical.
Normalization uses the stats frozen at the last commit, so all members of one
generation see identical stats.
"""
from __future__ import annotations

import torch


class RunningNormalizer:
    def __init__(self, dim: int, device="cpu", clip: float = 5.0):
        self.dim = dim
        self.clip = clip
        self.device = torch.device(device)
        z = lambda *s: torch.zeros(*s, dtype=torch.float64, device=self.device)  # noqa: E731
        self.count, self.sum, self.sumsq = z(()), z(dim), z(dim)
        self._dcount, self._dsum, self._dsumsq = z(()), z(dim), z(dim)
        self._mean = torch.zeros(dim, dtype=torch.float32, device=self.device)
        self._inv_std = torch.ones(dim, dtype=torch.float32, device=self.device)

    def observe(self, obs: torch.Tensor) -> None:
        """Accumulate a batch [N, dim] into the per-generation delta."""
        x = obs.to(torch.float64)
        self._dcount += x.shape[0]
        self._dsum += x.sum(dim=0)
        self._dsumsq += (x * x).sum(dim=0)

    def commit(self, comm=None) -> None:
        """All-reduce the generation delta (C5), fold into the global stats,
        refresh the frozen (mean, inv_std)."""
        if comm is not None and getattr(comm, "is_dist", False):
            buf = torch.cat([self._dcount.reshape(1), self._dsum, self._dsumsq])
            if comm.backend != "nccl" and buf.device.type != "cpu":
                cpu = buf.cpu()
                comm.allreduce_(cpu)
                buf = cpu.to(buf.device)
            else:
                comm.allreduce_(buf)
            self.count += buf[0]
            self.sum += buf[1:1 + self.dim]
            self.sumsq += buf[1 + self.dim:]
        else:
            self.count += self._dcount
            self.sum += self._dsum
            self.sumsq += self._dsumsq
        self._dcount.zero_(); self._dsum.zero_(); self._dsumsq.zero_()
        self.refresh()

    def refresh(self) -> None:
        if float(self.count) < 2:
            return
        mean = self.sum / self.count
        var = self.sumsq / self.count - mean * mean
        var = var.clamp_min(1e-8)
        # in-place: _mean/_inv_std pointers stay stable so captured hipGraphs
        # (which bake the addresses into the MLP forward) see fresh stats
        self._mean.copy_(mean.to(torch.float32))
        self._inv_std.copy_(var.rsqrt().to(torch.float32))

    def normalize(self, obs: torch.Tensor) -> torch.Tensor:
        x = (obs.to(torch.float32) - self._mean) * self._inv_std
        return x.clamp(-self.clip, self.clip)

    def state_dict(self):
        return {"count": self.count, "sum": self.sum, "sumsq": self.sumsq,
                "clip": self.clip}

    def load_state_dict(self, sd):
        self.count = sd["count"].to(self.device)
        self.sum = sd["sum"].to(self.device)
        self.sumsq = sd["sumsq"].to(self.device)
        self.clip = float(sd.get("clip", 5.0))
        self._dcount.zero_(); self._dsum.zero_(); self._dsumsq.zero_()
        self.refresh()


class IdentityNormalizer:
    """No-op stand-in for image envs (obs already in [0,1])."""

    def observe(self, obs):
        pass

    def commit(self, comm=None):
        pass

    def refresh(self):
        pass

    def normalize(self, obs):
        return obs

    def state_dict(self):
        return {}

    def load_state_dict(self, sd):
        pass
