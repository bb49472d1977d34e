"""Config dataclass + named presets (the five BASELINE.json configs).

SURVEY.md §2.8: frozen Config dataclass + argparse CLI + named presets
matching the five BASELINE configs exactly.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass
from typing import Optional


@dataclass
class Config:
    # --- task ---
    env: str = "cartpole"            # cartpole | pong | atari57_synth | procgen_synth | humanoid_synth
    model: str = "mlp"               # mlp | nature_cnn | impala_cnn
    hidden_sizes: tuple = (64, 64)   # for mlp
    algo: str = "es"                 # es | cem

    # --- population / sampling ---
    pop: int = 16                    # GLOBAL population (across all ranks); even if antithetic
    sigma: float = 0.05              # perturbation std
    antithetic: bool = True          # mirrored (+eps, -eps) pairs
    rank_shaping: str = "centered"   # centered | normalized | raw

    # --- update rule (ES) ---
    lr: float = 0.01
    weight_decay: float = 0.005
    optimizer: str = "adam"          # adam | sgd
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8

    # --- CEM ---
    elite_frac: float = 0.2
    cem_sigma_init: float = 0.1
    cem_noise_floor: float = 0.01    # extra variance floor added to refit sigma^2
    cem_noise_decay: float = 0.995   # floor decays each generation

    # --- noise table ---
    noise_table_size: int = 1 << 20  # floats; GPU presets use >= 1<<28 (1 GiB fp32)
    noise_dtype: str = "fp32"        # fp32 | bf16 (table storage)

    # --- rollout ---
    episode_steps: int = 200         # horizon per generation (auto-reset inside)
    obs_normalize: bool = False      # running obs normalizer (MLP/state envs)

    # --- schedule ---
    generations: int = 100
    seed: int = 1234

    # --- execution ---
    device: str = "auto"             # auto | cuda | cpu
    backend: Optional[str] = None    # None=auto: nccl if cuda else gloo
    precision: str = "bf16"          # population forward dtype on GPU: bf16 | fp32
    overlap_comm: bool = True        # side-stream overlap of collectives (GPU)
    use_hip_graphs: bool = True      # capture the rollout loop in a hipGraph

    # --- io ---
    checkpoint_path: Optional[str] = None
    checkpoint_every: int = 0        # generations; 0 = off
    log_every: int = 1
    metrics_path: Optional[str] = None  # JSONL per-generation metrics

    # --- misc ---
    theta_allreduce_every: int = 10  # drift-insurance parameter all-reduce cadence
    profile_phases: bool = False     # per-phase timings in train_step stats
                                     # (adds device syncs)

    def replace(self, **kw) -> "Config":
        return dataclasses.replace(self, **kw)

    def validate(self) -> "Config":
        if self.antithetic and self.pop % 2 != 0:
            raise ValueError("antithetic sampling requires an even population")
        if self.algo not in ("es", "cem"):
            raise ValueError(f"unknown algo {self.algo!r}")
        if self.rank_shaping not in ("centered", "normalized", "raw"):
            raise ValueError(f"unknown rank_shaping {self.rank_shaping!r}")
        return self


# ---------------------------------------------------------------------------
# Named presets == the five BASELINE.json configs (BASELINE.json "configs").
# ---------------------------------------------------------------------------

PRESETS = {
    # 1. "CartPole-v1 ES, 2-layer MLP policy, pop=16, world_size=2 on CPU/gloo"
    "cartpole_es": Config(
        env="cartpole", model="mlp", hidden_sizes=(64, 64), algo="es",
        pop=16, sigma=0.1, lr=0.02, weight_decay=0.005,
        episode_steps=500, generations=100, obs_normalize=True,
        noise_table_size=1 << 20, device="auto",
    ),
    # 2. "Humanoid-v4 MLP ES pop=10000 (OpenAI-ES headline config) on 1 MI355X"
    "humanoid_es_10k": Config(
        env="humanoid_synth", model="mlp", hidden_sizes=(256, 256), algo="es",
        pop=10000, sigma=0.02, lr=0.01, weight_decay=0.005,
        episode_steps=1000, generations=100, obs_normalize=True,
        noise_table_size=1 << 28, device="auto",
    ),
    # 3. "Pong Nature-CNN ES pop=8192 bf16, vectorized on-GPU envs, 8x MI355X"
    "pong_es_8k": Config(
        env="pong", model="nature_cnn", algo="es",
        pop=8192, sigma=0.02, lr=0.01, weight_decay=0.005,
        episode_steps=240, generations=50, precision="bf16",
        noise_table_size=1 << 28, device="auto",
    ),
    # 4. "Atari-57 Nature-CNN ES, 1024 envs/GPU, antithetic + mirrored sampling"
    "atari57_es": Config(
        env="atari57_synth", model="nature_cnn", algo="es",
        pop=8192, sigma=0.02, lr=0.01, weight_decay=0.005,
        episode_steps=240, generations=50, precision="bf16",
        noise_table_size=1 << 28, device="auto",
    ),
    # 5. "Procgen IMPALA-CNN ES pop=32768 (288 GB HBM shared-noise-table stress)"
    "procgen_es_32k": Config(
        env="procgen_synth", model="impala_cnn", algo="es",
        pop=32768, sigma=0.02, lr=0.01, weight_decay=0.005,
        episode_steps=192, generations=50, precision="bf16",
        noise_table_size=1 << 30, device="auto",   # 4 GiB fp32 table
    ),
    # CEM twin of the plumbing config
    "cartpole_cem": Config(
        env="cartpole", model="mlp", hidden_sizes=(64, 64), algo="cem",
        pop=64, antithetic=False, elite_frac=0.25, cem_sigma_init=0.2,
        episode_steps=500, generations=100,
        noise_table_size=1 << 20, device="auto",
    ),
}


def preset(name: str, **overrides) -> Config:
    if name not in PRESETS:
        raise KeyError(f"unknown preset {name!r}; have {sorted(PRESETS)}")
    return PRESETS[name].replace(**overrides).validate()
