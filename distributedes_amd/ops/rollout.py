"""HIP rollout engine: the per-generation hot loop on GPU kernels.

Dispatches {env family} x {model family} to the csrc kernels:
    K7 env step (cartpole / pong / latent-image / humanoid)
 -> K3/K5 policy forward (MLP / Nature-CNN) with fused normalization + action
 -> K8 fitness accumulation (fused into env step)
All state lives in device buffers allocated once; the loop makes zero
host<->device transfers and no torch ops, so it is hipGraph-capturable
(engine captures a whole generation's T steps when cfg.use_hip_graphs).
"""
from __future__ import annotations

import logging

import torch

log = logging.getLogger("distributedes_amd")

from ..models import ImpalaCNNPolicy, MLPPolicy, NatureCNNPolicy
from ..models.base import ParamSpec


def mlp_layer_offsets(spec: ParamSpec, n_hidden: int):
    names = []
    for i in range(n_hidden):
        names += [f"fc{i+1}.weight", f"fc{i+1}.bias"]
    names += ["head.weight", "head.bias"]
    offs = [spec._by_name[n].offset for n in names]
    for n in names:
        e = spec._by_name[n]
        assert e.offset % 4 == 0, (n, e.offset)
    if n_hidden == 1:
        offs = offs[:2] + [0, 0] + offs[2:]
    return offs


def nature_cnn_offsets(spec: ParamSpec):
    names = ["conv1.weight", "conv1.bias", "conv2.weight", "conv2.bias",
             "conv3.weight", "conv3.bias", "fc.weight", "fc.bias",
             "head.weight", "head.bias"]
    offs = [spec._by_name[n].offset for n in names]
    assert all(o % 4 == 0 for o in offs)
    return offs


def impala_offsets(spec: ParamSpec):
    o = lambda n: spec._by_name[n].offset  # noqa: E731
    d = {}
    for sg in (1, 2, 3):
        d[f"s{sg}"] = (o(f"stage{sg}_conv.weight"), o(f"stage{sg}_conv.bias"))
        for r in (1, 2):
            d[f"s{sg}r{r}c1"] = (o(f"stage{sg}_res{r}_conv1.weight"),
                                 o(f"stage{sg}_res{r}_conv1.bias"))
            d[f"s{sg}r{r}c2"] = (o(f"stage{sg}_res{r}_conv2.weight"),
                                 o(f"stage{sg}_res{r}_conv2.bias"))
    d["fc"] = (o("fc.weight"), o("fc.bias"))
    d["head"] = (o("head.weight"), o("head.bias"))
    assert all(v % 4 == 0 for pair in d.values() for v in pair)
    return d


class HipRollout:
    """Owns device buffers + kernel sequencing for one (env, model) pair."""

    SUPPORTED_ENVS = ("cartpole", "pong", "atari57_synth", "procgen_synth",
                      "humanoid_synth")

    def __init__(self, cfg, env, model, ext, device):
        self.cfg = cfg
        self.env_name = cfg.env
        self.env = env
        self.model = model
        self.ext = ext
        self.device = device
        self.P = None
        if isinstance(model, MLPPolicy):
            self.kind = "mlp"
            self.h1 = model.hidden_sizes[0]
            self.h2 = model.hidden_sizes[1] if len(model.hidden_sizes) > 1 else 0
            assert len(model.hidden_sizes) <= 2, "HIP MLP path supports <=2 hidden"
            self.offs = mlp_layer_offsets(model.spec, len(model.hidden_sizes))
        elif isinstance(model, NatureCNNPolicy):
            self.kind = "nature_cnn"
            self.offs_h = nature_cnn_offsets(model.spec)
            self.offs_t = torch.tensor(self.offs_h, dtype=torch.int64,
                                       device=device)
        elif isinstance(model, ImpalaCNNPolicy):
            self.kind = "impala"
            self.ioffs = impala_offsets(model.spec)
        else:
            raise NotImplementedError(type(model))

    @classmethod
    def supports(cls, cfg, model) -> bool:
        if cfg.env not in cls.SUPPORTED_ENVS:
            return False
        if isinstance(model, (NatureCNNPolicy, ImpalaCNNPolicy)):
            return cfg.precision == "bf16"  # kernels consume bf16 pop weights
        return isinstance(model, MLPPolicy) and len(model.hidden_sizes) <= 2

    # ------------------------------------------------------------------ #
    def _alloc(self, P: int):
        dev = self.device
        self.P = P
        z = lambda *s, dt=torch.float32: torch.zeros(*s, dtype=dt, device=dev)  # noqa: E731
        self._empty = torch.empty(0, device=dev)
        self.seed_buf = torch.zeros(1, dtype=torch.int64, device=dev)
        # persistent env-id buffer: captured hipGraphs bake this pointer into
        # the step kernels, so it must never be reallocated between replays
        # (a fresh .to(device) tensor each generation gets freed and its block
        # reused, silently corrupting per-env RNG streams under graph replay)
        self._ids = torch.zeros(P, dtype=torch.int64, device=dev)
        self._graphs = {}
        self.fitness = z(P)
        self.alive = z(P, dt=torch.uint8)
        self.ep_len = z(P)
        self.actions_i = z(P, dt=torch.int32)
        self.actions_f = (z(P, self.env.num_actions)
                          if self.env.continuous else None)
        e = self.env_name
        if e == "cartpole":
            self.st_state = z(P, 4)
            self.st_steps = z(P, dt=torch.int32)
            self.st_rctr = z(P, dt=torch.int64)
        elif e == "pong":
            for n in ("bx", "by", "vx", "vy", "py", "oy"):
                setattr(self, "pg_" + n, z(P))
            self.pg_score = z(P, dt=torch.int32)
            self.pg_points = z(P, dt=torch.int32)
            self.pg_serve = z(P, dt=torch.int64)
            self.frames = z(P, 4, 84, 84, dt=torch.uint8)
            # phase p (= t % 4): logical channel c -> physical slot (p+c) % 4
            self.slot_maps = [torch.tensor([(p + c) % 4 for c in range(4)],
                                           dtype=torch.int32, device=dev)
                              for p in range(4)]
        elif e in ("atari57_synth", "procgen_synth"):
            self.sy_z = z(P, 8)
            self.sy_coef = z(P, 6)       # per-env game coefficients
            self.sy_steps = z(P, dt=torch.int32)
            c, hw, _ = self.env.obs_shape
            self.obs_bf16 = z(P, c, hw, hw, dt=torch.bfloat16)
            self.slot_map = torch.tensor([0, 1, 2, 3], dtype=torch.int32,
                                         device=dev)
        if self.kind == "impala":
            bf = torch.bfloat16
            self.ib64 = z(P, 16, 64, 64, dt=bf)
            self.ix32 = z(P, 16, 32, 32, dt=bf)
            self.it32 = z(P, 16, 32, 32, dt=bf)
            self.ib32 = z(P, 32, 32, 32, dt=bf)
            self.ix16 = z(P, 32, 16, 16, dt=bf)
            self.ib16 = z(P, 32, 16, 16, dt=bf)
            self.ix8 = z(P, 32, 8, 8, dt=bf)
        elif e == "humanoid_synth":
            self.hu_s = z(P, 376)
            self.hu_steps = z(P, dt=torch.int32)
        if len(self.env.obs_shape) == 1:
            # K12 two-stage observe: per-block fp32 partial buffer
            self.nz_scratch = z(256, 2, self.env.obs_dim)

    # ------------------------------------------------------------------ #
    def _reset(self, gen_seed: int, env_ids: torch.Tensor):
        e = self.env_name
        self._ids.copy_(env_ids.to(self.device))
        ids = self._ids
        if e == "cartpole":
            self.ext.cartpole_reset(self.st_state, self.st_steps, self.st_rctr,
                                    self.fitness, self.alive, self.ep_len,
                                    ids, gen_seed * 0x10001)
        elif e == "pong":
            self.ext.pong_reset(self.pg_bx, self.pg_by, self.pg_vx, self.pg_vy,
                                self.pg_py, self.pg_oy, self.pg_score,
                                self.pg_points, self.pg_serve, self.frames,
                                self.fitness, self.alive, self.ep_len, ids,
                                gen_seed)
        elif e in ("atari57_synth", "procgen_synth"):
            # host-side single source of truth for the per-game coefficients
            # (Atari-57 multi-task parameterization; classics for procgen)
            from ..envs.synthetic import game_coefs
            ng = self.env.n_games
            self.sy_coef.copy_(game_coefs(ids % ng, ng))
            self.ext.latent_reset(self.sy_z, self.sy_steps, self.fitness,
                                  self.alive, self.ep_len, ids, gen_seed)
            self.ext.latent_render(self.sy_z, self.sy_coef, self.obs_bf16)
        elif e == "humanoid_synth":
            self.ext.humanoid_reset(self.hu_s, self.hu_steps, self.fitness,
                                    self.alive, self.ep_len, ids, gen_seed)
        self._gen_seed = gen_seed
        # the step kernels read their hash seed from device memory so a
        # captured hipGraph replays correctly across generations
        self.seed_buf.fill_(gen_seed * 0x10001 if e == "cartpole" else gen_seed)

    # ------------------------------------------------------------------ #
    def _forward(self, pop_w: torch.Tensor, normalizer, slot_newest: int):
        """One policy forward: obs buffers -> action buffers."""
        if self.kind == "mlp":
            obs = self.st_state if self.env_name == "cartpole" else self.hu_s
            nmean = getattr(normalizer, "_mean", None)
            if nmean is not None and nmean.device != self.device:
                nmean = None
            if nmean is not None:
                ninv = normalizer._inv_std
                clip = normalizer.clip
            else:
                nmean = torch.empty(0, device=self.device)
                ninv = nmean
                clip = 0.0
            act = self.actions_f if self.env.continuous else self.actions_i
            self.ext.mlp_forward(pop_w, obs, nmean, ninv, clip, act,
                                 self._empty, self.h1, self.h2,
                                 self.env.num_actions, self.offs)
        elif self.kind == "nature_cnn":
            if self.env_name == "pong":
                frames = self.frames
                sm = self.slot_maps[(slot_newest + 1) % 4]
            else:
                frames = self.obs_bf16
                sm = self.slot_map
            self.ext.nature_cnn_forward(pop_w, frames, sm, self.actions_i,
                                        self._empty, self.env.num_actions,
                                        self.offs_t)
        else:  # impala
            self._impala_forward(pop_w)

    def _impala_forward(self, pop_w, logits=None):
        ext, io, E = self.ext, self.ioffs, self._empty
        conv = lambda i, o, sk, name, shape, relu, add: ext.impala_conv(  # noqa: E731
            i, o, sk if sk is not None else E, pop_w, io[name][0], io[name][1],
            shape, relu, add)
        # stage 1
        conv(self.obs_bf16, self.ib64, None, "s1", 0, False, False)
        ext.impala_pool(self.ib64, self.ix32, 16, 64)
        for r in (1, 2):
            conv(self.ix32, self.it32, None, f"s1r{r}c1", 1, True, False)
            conv(self.it32, self.ix32, self.ix32, f"s1r{r}c2", 1, True, True)
        # stages 2/3: whole-image tiles (HW <= 16) -> fused residual pairs
        # (y1 lives in LDS; no intermediate tensor, half the launches)
        conv(self.ix32, self.ib32, None, "s2", 2, False, False)
        ext.impala_pool(self.ib32, self.ix16, 32, 32)
        for r in (1, 2):
            c1, c2 = io[f"s2r{r}c1"], io[f"s2r{r}c2"]
            ext.impala_resblock(self.ix16, pop_w, c1[0], c1[1], c2[0], c2[1])
        # stage 3
        conv(self.ix16, self.ib16, None, "s3", 3, False, False)
        ext.impala_pool(self.ib16, self.ix8, 32, 16)
        for r in (1, 2):
            c1, c2 = io[f"s3r{r}c1"], io[f"s3r{r}c2"]
            ext.impala_resblock(self.ix8, pop_w, c1[0], c1[1], c2[0], c2[1])
        ext.impala_fc_head(pop_w, self.ix8.view(self.P, -1), self.actions_i,
                           logits if logits is not None else E,
                           self.env.num_actions, io["fc"][0], io["fc"][1],
                           io["head"][0], io["head"][1])

    def _env_step(self, slot: int):
        e = self.env_name
        if e == "cartpole":
            self.ext.cartpole_step(self.st_state, self.st_steps, self.st_rctr,
                                   self.actions_i, self.fitness, self.alive,
                                   self.ep_len, self._ids, self.seed_buf)
        elif e == "pong":
            self.ext.pong_step(self.pg_bx, self.pg_by, self.pg_vx, self.pg_vy,
                               self.pg_py, self.pg_oy, self.pg_score,
                               self.pg_points, self.pg_serve, self.frames,
                               slot, self.actions_i, self.fitness, self.alive,
                               self.ep_len, self._ids, self.seed_buf)
        elif e in ("atari57_synth", "procgen_synth"):
            self.ext.latent_step(self.sy_z, self.sy_coef, self.sy_steps,
                                 self.actions_i, self.fitness, self.alive,
                                 self.ep_len, self.env.num_actions,
                                 self.env.horizon)
            self.ext.latent_render(self.sy_z, self.sy_coef, self.obs_bf16)
        elif e == "humanoid_synth":
            self.ext.humanoid_step(self.hu_s, self.hu_steps, self.actions_f,
                                   self.fitness, self.alive, self.ep_len,
                                   self.env.horizon)

    def _observe(self, normalizer):
        """K12 step half: accumulate obs into the fp32 per-block partial
        scratch; the fp64 reduce runs ONCE per generation in run() (flush) —
        the count is added analytically there too (graph-capture-safe)."""
        if not hasattr(normalizer, "_dsum"):
            return
        obs = self.st_state if self.env_name == "cartpole" else self.hu_s
        self.ext.normalizer_observe(obs, self.nz_scratch)

    # ------------------------------------------------------------------ #
    def _loop(self, pop_w, episode_steps, normalizer, is_state_env,
              observe=True):
        # pong ring: reset wrote the initial frame to every slot; step t
        # writes slot t % 4 and the forward sees slots in logical order
        for t in range(episode_steps):
            slot = t % 4
            if is_state_env and observe:
                self._observe(normalizer)
            self._forward(pop_w, normalizer, slot_newest=(t - 1) % 4)
            self._env_step(slot)

    def run(self, pop_w: torch.Tensor, gen_seed: int, env_ids: torch.Tensor,
            episode_steps: int, normalizer, allow_graph: bool = True,
            observe: bool = True):
        P = pop_w.shape[0]
        if self.P != P:
            self._alloc(P)
        is_state_env = len(self.env.obs_shape) == 1
        do_observe = (is_state_env and observe
                      and hasattr(normalizer, "_dsum"))
        if do_observe:
            self.nz_scratch.zero_()
        self._reset(gen_seed, env_ids)
        use_graph = (allow_graph
                     and getattr(self.cfg, "use_hip_graphs", False)
                     and self.device.type == "cuda"
                     and not getattr(self, "_graphs_broken", False))
        if use_graph:
            key = (episode_steps, pop_w.data_ptr(), id(normalizer), observe)
            g = self._graphs.get(key)
            if g is None:
                try:
                    g = self._capture(pop_w, gen_seed, env_ids, episode_steps,
                                      normalizer, is_state_env, observe)
                    self._graphs[key] = g
                except Exception:
                    log.warning(
                        "hipGraph capture failed; falling back to eager "
                        "kernel launches for all subsequent generations",
                        exc_info=True)
                    self._graphs_broken = True
                    use_graph = False
            if use_graph:
                g.replay()
        if not use_graph:
            self._loop(pop_w, episode_steps, normalizer, is_state_env, observe)
        self.last_used_graph = use_graph
        if do_observe:
            self.ext.normalizer_observe_flush(self.nz_scratch, P,
                                              normalizer._dsum,
                                              normalizer._dsumsq)
            normalizer._dcount += P * episode_steps
        return self.fitness.clone(), P * episode_steps, float(self.ep_len.mean())

    def _capture(self, pop_w, gen_seed, env_ids, episode_steps, normalizer,
                 is_state_env, observe=True):
        """Capture one generation's kernel sequence into a hipGraph.
        The warmup pass EXECUTES a full loop (torch requirement), so env +
        normalizer-delta state is re-reset before capture; the replay then
        performs the real generation."""
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            self._loop(pop_w, episode_steps, normalizer, is_state_env, observe)
        torch.cuda.current_stream().wait_stream(s)
        # undo warmup side effects (the warmup accumulates into the observe
        # scratch; the fp64 deltas are only touched by the post-loop flush)
        self._reset(self._gen_seed, self._ids)
        if hasattr(normalizer, "_dsum"):
            self.nz_scratch.zero_()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._loop(pop_w, episode_steps, normalizer, is_state_env, observe)
        return g
