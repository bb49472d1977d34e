"""Distributed communication: the C1-C6 call sites.

One OS process per GPU; `torch.distributed` backend "nccl" IS RCCL on ROCm,
"gloo" is the CPU CI twin (C7).  Call sites:

  C1 broadcast(theta, 0)           - startup
  C2 broadcast(table_seed)         - startup (table regenerated locally, K1)
  C3 all_gather fitness            - per generation (latency-bound, ~KiB)
  C4 all_reduce gradient / theta   - per generation (<=6.5 MiB fp32)
  C5 all_reduce normalizer stats   - per generation
  C6 all_reduce metrics            - per generation

On GPU, C3/C4/C5/C6 run on a SIDE HIP stream so they overlap the next
generation's noise/perturb work on the main stream (BASELINE north star);
`Comm.sidestream()` is the context manager for that.  With gloo or a single
process every call degrades gracefully (no-op or same-device op).

xGMI sizing note: each MI355X has 7 p2p links x ~153 GB/s, so ring
all-reduce is single-link bound (~153 GB/s effective) — at the ES payload
(theta <= 6.5 MiB fp32) one bucket is ~74 us and bucketing below 4 MiB buys
nothing; C3 (KiB-scale) is latency-bound and prefers the default small-message
algorithm.  We therefore issue C4 as ONE all-reduce, not a bucketed stream.
"""
from __future__ import annotations

import os
from contextlib import contextmanager, nullcontext
from typing import Optional

import torch
import torch.distributed as dist


class Comm:
    def __init__(self, backend: Optional[str] = None, device: str = "cpu"):
        self.device = torch.device(device)
        self.initialized_here = False
        env_ws = int(os.environ.get("WORLD_SIZE", "1"))
        if dist.is_initialized():
            pass
        elif env_ws > 1:
            if backend is None:
                backend = "nccl" if self.device.type == "cuda" else "gloo"
            store_file = os.environ.get("DES_FILE_STORE")
            if store_file:
                # test hook: file-based rendezvous (no TCP port to race on)
                rank = int(os.environ.get("RANK", "0"))
                store = dist.FileStore(store_file, env_ws)
                dist.init_process_group(backend=backend, store=store,
                                        rank=rank, world_size=env_ws)
            else:
                if "MASTER_ADDR" not in os.environ:
                    os.environ["MASTER_ADDR"] = "127.0.0.1"
                if "MASTER_PORT" not in os.environ:
                    os.environ["MASTER_PORT"] = "29531"
                dist.init_process_group(backend=backend)
            self.initialized_here = True
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.backend = dist.get_backend() if dist.is_initialized() else None
        self._side = (torch.cuda.Stream() if self.device.type == "cuda"
                      and torch.cuda.is_available() else None)

    # --- plumbing -----------------------------------------------------------
    @property
    def is_dist(self) -> bool:
        return self.world > 1

    def shutdown(self):
        if self.initialized_here and dist.is_initialized():
            dist.destroy_process_group()

    def barrier(self):
        if self.is_dist:
            dist.barrier()

    @contextmanager
    def sidestream(self):
        """Run collectives on the side HIP stream (overlap with main-stream
        compute).  Caller must wait_stream() before consuming results; the
        returned object exposes .wait() for that."""
        if self._side is None:
            yield nullcontext()
            return
        self._side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(self._side):
            yield self
        # no sync here; join() must be called before results are consumed

    def join_sidestream(self):
        if self._side is not None:
            torch.cuda.current_stream().wait_stream(self._side)

    # --- call sites ---------------------------------------------------------
    def broadcast_theta(self, theta: torch.Tensor):               # C1
        if self.is_dist:
            dist.broadcast(theta, src=0)

    def broadcast_seed(self, seed: int) -> int:                   # C2
        if not self.is_dist:
            return seed
        t = torch.tensor([seed], dtype=torch.int64,
                         device=self.device if self.backend == "nccl" else "cpu")
        dist.broadcast(t, src=0)
        return int(t.item())

    def allgather_fitness(self, local_fit: torch.Tensor) -> torch.Tensor:  # C3
        """local_fit [local_pop] fp32 -> global [pop] fp32 in rank order.
        Member seeds are counter-derived (core.noise.member_seed), so the
        (seed, fitness) pair of SURVEY C3 needs only fitness on the wire."""
        if not self.is_dist:
            return local_fit
        if self.backend == "nccl":
            out = torch.empty(self.world * local_fit.numel(),
                              dtype=local_fit.dtype, device=local_fit.device)
            dist.all_gather_into_tensor(out, local_fit)
            return out
        src = local_fit.cpu() if local_fit.device.type != "cpu" else local_fit
        parts = [torch.empty_like(src) for _ in range(self.world)]
        dist.all_gather(parts, src)
        return torch.cat(parts).to(local_fit.device)

    def allreduce_(self, t: torch.Tensor, avg: bool = False):     # C4/C5/C6
        if self.is_dist:
            if self.backend != "nccl" and t.device.type != "cpu":
                cpu = t.cpu()
                dist.all_reduce(cpu)
                t.copy_(cpu)
            else:
                dist.all_reduce(t)
            if avg:
                t.div_(self.world)
        return t
