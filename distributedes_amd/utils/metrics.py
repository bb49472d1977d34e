"""Structured per-generation metrics: JSONL sink + console line."""
from __future__ import annotations

import json
import os
import sys
import time


class MetricsLogger:
    def __init__(self, path=None, rank: int = 0, log_every: int = 1):
        self.rank = rank
        self.log_every = max(1, log_every)
        self.path = path
        self._fh = None
        if path and rank == 0:
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
            self._fh = open(path, "a", buffering=1)
        self._t0 = time.time()

    def log(self, stats: dict) -> None:
        if self.rank != 0:
            return
        rec = dict(stats)
        rec["wall_s"] = round(time.time() - self._t0, 3)
        if self._fh:
            self._fh.write(json.dumps(rec) + "\n")
        g = rec.get("generation", -1)
        if g % self.log_every == 0:
            msg = (f"[gen {g:5d}] ret_mean={rec.get('episodic_return_mean', float('nan')):9.3f} "
                   f"ret_max={rec.get('episodic_return_max', float('nan')):9.3f} "
                   f"steps/s={rec.get('steps_per_sec', 0):,.0f}")
            print(msg, file=sys.stderr, flush=True)

    def close(self):
        if self._fh:
            self._fh.close()
            self._fh = None
