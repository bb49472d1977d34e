"""Training engine: TrainState + one-generation step for ES and CEM.

Target call stack:

    api.run(config)
      runtime init: process group, C1 broadcast theta, C2 broadcast table
                    seed, K1 local noise-table fill
      loop: train_step(state)
        K2/K4  perturb local members (materialize or fused)
        hot loop over timesteps: K7 env step -> K3/K5 policy fwd -> K8 fitness
        C3 all-gather fitness              (xGMI)
        K9 rank shaping (identical on every rank)
        K10 local weighted noise sum  -> C4 all-reduce -> Adam step
        C5 normalizer stats, C6 metrics     (xGMI)

Every rank computes the identical theta update (counter-derived seeds +
all-gathered fitness), so C4 doubles as drift insurance; a periodic
average-theta all-reduce (cfg.theta_allreduce_every) guards against any
accumulated nondeterminism.
"""
from __future__ import annotations

import logging
import math
import os
import time
from typing import Optional

import torch

from ..config import Config
from ..envs import make_env
from ..models import make_model
from ..parallel import Comm
from ..utils import (IdentityNormalizer, MetricsLogger, RunningNormalizer,
                     save_checkpoint)
from . import noise as noise_mod
from .optim import make_optimizer
from .rank import shape_fitness

log = logging.getLogger("distributedes_amd")


def _resolve_device(cfg: Config) -> torch.device:
    if cfg.device == "auto":
        dev = torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
    else:
        dev = torch.device(cfg.device)
    # under torchrun each rank must land on its own GPU BEFORE the process
    # group initializes (else every rank grabs GPU 0 and NCCL hangs/errors)
    if dev.type == "cuda" and dev.index is None:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if local_rank < torch.cuda.device_count():
            dev = torch.device("cuda", local_rank)
            torch.cuda.set_device(dev)
    return dev


def _make_ops(device: torch.device, want_hip: bool = True):
    if device.type == "cuda" and want_hip:
        from ..ops.hip import HipOps, hip_available
        if hip_available():
            return HipOps()
        raise RuntimeError(
            "CUDA device requested but the distributedes_amd HIP extension is "
            "not built — refusing to silently fall back to eager torch. "
            "Build it (python setup.py build_ext --inplace) or pass "
            "ops_backend='torch_ref' explicitly.")
    from ..ops.reference import TorchOps
    return TorchOps()


class TrainState:
    def __init__(self, cfg: Config, comm: Optional[Comm] = None,
                 ops_backend: Optional[str] = None):
        cfg.validate()
        self.cfg = cfg
        self.device = _resolve_device(cfg)
        self.comm = comm or Comm(cfg.backend, device=self.device)
        self.rank = self.comm.rank
        self.world = self.comm.world
        if cfg.pop % self.world != 0:
            raise ValueError(f"pop={cfg.pop} not divisible by world={self.world}")
        self.local_pop = cfg.pop // self.world
        self.member_base = self.rank * self.local_pop   # first global member index

        self.env = make_env(cfg.env)
        self.model = make_model(cfg, self.env)
        self.dim = self.model.dim

        # theta (ES) / mu (CEM): fp32 master, identical on every rank
        self.theta = self.model.init_theta(cfg.seed).to(self.device)
        self.comm.broadcast_theta(self.theta)                        # C1
        self.sigma_vec = None
        self.cem_noise_floor = cfg.cem_noise_floor
        if cfg.algo == "cem":
            self.sigma_vec = torch.full((self.dim,), cfg.cem_sigma_init,
                                        dtype=torch.float32, device=self.device)

        self.optimizer = make_optimizer(cfg, self.dim, self.device)

        if cfg.obs_normalize and len(self.env.obs_shape) == 1:
            self.normalizer = RunningNormalizer(self.env.obs_dim, device=self.device)
        else:
            self.normalizer = IdentityNormalizer()

        # noise table: seed broadcast (C2), filled locally (K1)
        # masked to 63 bits: the seed rides an int64 broadcast tensor (C2)
        self.table_seed = self.comm.broadcast_seed(
            noise_mod.splitmix64(cfg.seed ^ 0xE5E5E5E5) & 0x7FFFFFFFFFFFFFFF)
        want_hip = (ops_backend != "torch_ref")
        self.ops = _make_ops(self.device, want_hip) if ops_backend != "torch_ref" \
            else _make_ops(torch.device("cpu"), False)
        table_dtype = (torch.bfloat16 if cfg.noise_dtype == "bf16"
                       else torch.float32)
        self.table = self.ops.fill_table(cfg.noise_table_size, self.table_seed,
                                         self.device, dtype=table_dtype)

        self.generation = 0
        self.env_steps_total = 0
        self.metrics = MetricsLogger(cfg.metrics_path, rank=self.rank,
                                     log_every=cfg.log_every)
        self._pop_w = None          # reusable [local_pop, D] perturbed weights
        self._offs_cache = {}       # generation -> (offsets, signs) prefetch

        # GPU-native rollout path (K3/K5/K7/K8 kernels); torch loop otherwise
        self._hip_rollout = None
        if self.ops.name == "hip":
            from ..ops.rollout import HipRollout
            if HipRollout.supports(cfg, self.model):
                self._hip_rollout = HipRollout(cfg, self.env, self.model,
                                               self.ops.ext, self.device)
            else:
                log.warning(
                    "env=%s model=%s precision=%s has no HIP rollout path; "
                    "the hot loop will run eager torch ops on GPU",
                    cfg.env, type(self.model).__name__, cfg.precision)

    # ------------------------------------------------------------------ #
    def member_offsets_signs(self, generation: int):
        """Offsets/signs for the LOCAL member slice.  Results are cached one
        generation ahead: train_step derives generation g+1 on the HOST while
        the C3/C4 collectives for generation g ride the side stream."""
        cached = self._offs_cache.pop(generation, None)
        if cached is not None:
            return cached
        cfg = self.cfg
        offs, signs = noise_mod.member_offsets(
            cfg.seed, generation, cfg.pop, cfg.antithetic,
            cfg.noise_table_size, self.dim)
        lo, hi = self.member_base, self.member_base + self.local_pop
        offs_t = torch.from_numpy(offs[lo:hi]).to(self.device)
        signs_t = torch.from_numpy(signs[lo:hi]).to(self.device)
        return offs_t, signs_t

    def prefetch_offsets(self, generation: int):
        if generation not in self._offs_cache:
            self._offs_cache[generation] = self.member_offsets_signs(generation)

    def env_ids(self):
        """Global env-stream ids for local members (pair-shared if antithetic)."""
        gids = torch.arange(self.member_base, self.member_base + self.local_pop,
                            dtype=torch.int64)
        if self.cfg.antithetic:
            gids = gids // 2
        return gids

    # ------------------------------------------------------------------ #
    def rollout(self, pop_w: torch.Tensor, env_ids: Optional[torch.Tensor] = None,
                observe: bool = True):
        """Evaluate local members; returns (fitness [P] fp32, env_steps int,
        ep_len_mean float).

        env_ids: env-stream ids (default: the training member slice).
        observe=False keeps the rollout's observations OUT of the running
        normalizer deltas — evaluation must not perturb training statistics.
        """
        cfg = self.cfg
        P = pop_w.shape[0]
        if env_ids is None:
            env_ids = self.env_ids()
        gen_seed = noise_mod.splitmix64(cfg.seed ^ (self.generation * 0x9E37)) & 0x7FFFFFFF
        if self._hip_rollout is not None:
            # hipGraph capture only for the persistent training buffer:
            # transient pop_w (evaluate()) must not bake freed pointers into
            # a cached graph
            return self._hip_rollout.run(pop_w, gen_seed, env_ids,
                                         cfg.episode_steps, self.normalizer,
                                         allow_graph=pop_w is self._pop_w,
                                         observe=observe)
        if pop_w.shape[1] != self.dim:           # padded rows -> logical view
            pop_w = pop_w[:, :self.dim]
        obs = self.env.reset(P, gen_seed, device=self.device, env_ids=env_ids)
        fitness = torch.zeros(P, dtype=torch.float32, device=self.device)
        alive = torch.ones(P, dtype=torch.bool, device=self.device)
        ep_len = torch.zeros(P, dtype=torch.float32, device=self.device)
        is_state_env = len(self.env.obs_shape) == 1
        steps_done = 0
        for t in range(cfg.episode_steps):
            if is_state_env:
                if observe:
                    self.normalizer.observe(obs)
                x = self.normalizer.normalize(obs)
            else:
                x = obs
            actions = self.pop_forward(pop_w, x)
            obs, reward, done = self.env.step(actions)
            fitness += reward * alive
            ep_len += alive.to(torch.float32)
            alive &= ~done
            steps_done = t + 1
        return fitness, P * steps_done, float(ep_len.mean())

    def pop_forward(self, pop_w: torch.Tensor, obs: torch.Tensor) -> torch.Tensor:
        return self.model.pop_forward_ref(pop_w, obs)

    # ------------------------------------------------------------------ #
    def perturb_local(self, offsets, signs) -> torch.Tensor:
        cfg = self.cfg
        out_dtype = (torch.bfloat16 if (cfg.precision == "bf16"
                     and self.device.type == "cuda") else torch.float32)
        base = self.theta
        sigma_vec = self.sigma_vec if cfg.algo == "cem" else None
        kw = {}
        if self.ops.name == "hip":
            kw["pad_rows"] = self._hip_rollout is not None
            kw["out"] = self._pop_w      # stable pointer: hipGraph replay
        self._pop_w = self.ops.perturb(base, self.table, offsets, signs,
                                       cfg.sigma, sigma_vec=sigma_vec,
                                       out_dtype=out_dtype, **kw)
        return self._pop_w

    # ------------------------------------------------------------------ #
    def shape_fitness(self, global_fitness: torch.Tensor) -> torch.Tensor:
        """K9: centered-rank kernel on the HIP path (bit-identical integer
        counting), torch reference elsewhere/for other modes."""
        cfg = self.cfg
        if (cfg.rank_shaping == "centered" and self.ops.name == "hip"
                and global_fitness.is_cuda and global_fitness.numel() >= 2):
            u = torch.empty_like(global_fitness)
            self.ops.ext.centered_rank(global_fitness.contiguous(), u)
            return u
        return shape_fitness(global_fitness, cfg.rank_shaping)

    def es_update(self, global_fitness: torch.Tensor, offsets, signs):
        """Centered-rank weighted noise sum (K10) + C4 all-reduce + Adam."""
        cfg = self.cfg
        u = self.shape_fitness(global_fitness)                       # K9, all ranks
        lo, hi = self.member_base, self.member_base + self.local_pop
        w_local = u[lo:hi].to(self.device)
        partial = self.ops.weighted_noise_sum(self.table, offsets, signs,
                                              w_local, self.dim)
        self.comm.allreduce_(partial)                                # C4
        grad_ascent = partial / (cfg.pop * cfg.sigma)
        descent = cfg.weight_decay * self.theta - grad_ascent
        self.optimizer.step(self.theta, descent)
        if (cfg.theta_allreduce_every
                and (self.generation + 1) % cfg.theta_allreduce_every == 0):
            self.comm.allreduce_(self.theta, avg=True)               # drift insurance

    def cem_update(self, global_fitness: torch.Tensor, offsets, signs):
        cfg = self.cfg
        k = max(1, int(math.ceil(cfg.elite_frac * cfg.pop)))
        elite_idx = torch.topk(global_fitness, k).indices
        lo, hi = self.member_base, self.member_base + self.local_pop
        w = torch.zeros(cfg.pop, dtype=torch.float32, device=self.device)
        w[elite_idx] = 1.0
        w_local = w[lo:hi]
        s_eps, s_eps2 = self.ops.weighted_noise_sum(
            self.table, offsets, signs, w_local, self.dim, want_sq=True)
        buf = torch.cat([s_eps, s_eps2])
        self.comm.allreduce_(buf)                                    # C4 (CEM refit)
        s_eps, s_eps2 = buf[:self.dim], buf[self.dim:]
        m1 = s_eps / k
        m2 = s_eps2 / k
        var_eps = (m2 - m1 * m1).clamp_min(0.0)
        self.theta += self.sigma_vec * m1
        self.cem_noise_floor *= cfg.cem_noise_decay
        self.sigma_vec = (self.sigma_vec ** 2 * var_eps
                          + self.cem_noise_floor ** 2).sqrt()
        if (cfg.theta_allreduce_every
                and (self.generation + 1) % cfg.theta_allreduce_every == 0):
            # drift insurance for mu AND the diagonal sigma (same role as the
            # ES path's periodic parameter all-reduce)
            buf2 = torch.cat([self.theta, self.sigma_vec])
            self.comm.allreduce_(buf2, avg=True)
            self.theta = buf2[:self.dim].clone()
            self.sigma_vec = buf2[self.dim:].clone()

    # ------------------------------------------------------------------ #
    def checkpoint_payload(self) -> dict:
        net_sd = self.model.spec.unflatten(self.theta.cpu())
        # .clone() everywhere: a live snapshot must not alias tensors that the
        # continuing run mutates in place (a .cpu() of a CPU tensor is a view)
        payload = {
            "network": net_sd,
            "optimizer": {k: (v.cpu().clone() if torch.is_tensor(v) else v)
                          for k, v in self.optimizer.state_dict().items()},
            "normalizer": {k: (v.cpu().clone() if torch.is_tensor(v) else v)
                           for k, v in self.normalizer.state_dict().items()},
            "generation": self.generation,
            "rng": {"run_seed": self.cfg.seed, "table_seed": self.table_seed},
        }
        if self.cfg.algo == "cem":
            payload["algo"] = {"sigma_vec": self.sigma_vec.cpu().clone(),
                               "noise_floor": self.cem_noise_floor}
        return payload

    def load_payload(self, payload: dict) -> None:
        self.theta = self.model.spec.flatten(payload["network"]).to(self.device)
        self.optimizer.load_state_dict(payload["optimizer"])
        self.normalizer.load_state_dict(payload["normalizer"])
        self.generation = int(payload["generation"])
        rng = payload.get("rng", {})
        if rng.get("table_seed") is not None and rng["table_seed"] != self.table_seed:
            self.table_seed = int(rng["table_seed"])
            self.table = self.ops.fill_table(
                self.cfg.noise_table_size, self.table_seed, self.device,
                dtype=self.table.dtype)
        if self.cfg.algo == "cem" and "algo" in payload:
            self.sigma_vec = payload["algo"]["sigma_vec"].to(self.device)
            self.cem_noise_floor = float(payload["algo"]["noise_floor"])


# ---------------------------------------------------------------------- #
def train_step(state: TrainState) -> dict:
    """Exactly one generation.  Returns the stats dict."""
    cfg = state.cfg
    t0 = time.perf_counter()

    def _phase_mark():
        if cfg.profile_phases and state.device.type == "cuda":
            torch.cuda.synchronize()
        return time.perf_counter()

    offsets, signs = state.member_offsets_signs(state.generation)
    pop_w = state.perturb_local(offsets, signs)                      # K2
    t_perturb = _phase_mark()
    fitness, env_steps, ep_len = state.rollout(pop_w)                # K7/K3/K5/K8
    t_rollout = _phase_mark()

    # C3 rides the side HIP stream (overlap_comm) while the host derives the
    # NEXT generation's member offsets (BASELINE: collectives overlapped with
    # the next generation's noise draw)
    if cfg.overlap_comm and state.device.type == "cuda" and state.comm.is_dist:
        with state.comm.sidestream():
            global_fitness = state.comm.allgather_fitness(fitness)   # C3
        state.prefetch_offsets(state.generation + 1)
        state.comm.join_sidestream()
    else:
        global_fitness = state.comm.allgather_fitness(fitness)       # C3
        state.prefetch_offsets(state.generation + 1)
    if cfg.algo == "es":
        state.es_update(global_fitness, offsets, signs)              # K9/K10/C4
    else:
        state.cem_update(global_fitness, offsets, signs)             # K11/C4

    # C5: normalizer stats merge (delta all-reduce) + refresh
    state.normalizer.commit(state.comm)

    # C6: metric reduction
    stats_t = torch.tensor([float(env_steps)], device=state.device
                           if state.comm.backend == "nccl" else "cpu")
    state.comm.allreduce_(stats_t)
    env_steps_global = int(stats_t[0].item())

    if state.device.type == "cuda":
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    state.generation += 1
    state.env_steps_total += env_steps_global

    stats = {
        "generation": state.generation,
        "episodic_return_mean": float(global_fitness.mean()),
        "episodic_return_max": float(global_fitness.max()),
        "episodic_return_min": float(global_fitness.min()),
        "ep_len_mean": ep_len,
        "env_steps": env_steps_global,
        "env_steps_total": state.env_steps_total,
        "steps_per_sec": env_steps_global / dt,
        "gen_time_s": dt,
        "ops_backend": state.ops.name,
        "hip_graphs": bool(getattr(state._hip_rollout, "last_used_graph", False)),
    }
    ng = getattr(state.env, "n_games", 1)
    if ng > 1:
        # per-game fitness breakdown (BASELINE config #4: Atari-57 means 57
        # tasks in one population; game = env_id % n_games)
        m = torch.arange(cfg.pop, device=global_fitness.device)
        game = ((m // 2 if cfg.antithetic else m) % ng)
        sums = torch.zeros(ng, device=global_fitness.device).index_add_(
            0, game, global_fitness)
        cnts = torch.zeros(ng, device=global_fitness.device).index_add_(
            0, game, torch.ones_like(global_fitness))
        per_game = sums / cnts.clamp_min(1.0)
        stats["per_game_return_mean"] = [round(float(x), 4) for x in per_game]
    if cfg.profile_phases:
        stats["t_perturb_ms"] = (t_perturb - t0) * 1e3
        stats["t_rollout_ms"] = (t_rollout - t_perturb) * 1e3
        stats["t_update_ms"] = dt * 1e3 - stats["t_perturb_ms"] - stats["t_rollout_ms"]
    if cfg.checkpoint_path and cfg.checkpoint_every and state.rank == 0 \
            and state.generation % cfg.checkpoint_every == 0:
        save_checkpoint(cfg.checkpoint_path, state.checkpoint_payload())
    state.metrics.log(stats)
    return stats
