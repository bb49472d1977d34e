"""Engine tests: train_step (ES+CEM), learning smoke, checkpoint resume
bit-equivalence, evaluate()."""
import os

import pytest
import torch

from distributedes_amd import evaluate, preset
from distributedes_amd.core.engine import TrainState, train_step
from distributedes_amd.utils import load_checkpoint, save_checkpoint


def _mini(algo="es", **kw):
    base = "cartpole_es" if algo == "es" else "cartpole_cem"
    return preset(base, pop=8, generations=3, episode_steps=30,
                  noise_table_size=1 << 16, device="cpu", **kw)


def test_train_step_es_runs():
    st = TrainState(_mini("es"), ops_backend="torch_ref")
    s = train_step(st)
    assert s["generation"] == 1
    assert s["env_steps"] == 8 * 30
    assert "episodic_return_mean" in s


def test_train_step_cem_runs():
    st = TrainState(_mini("cem"), ops_backend="torch_ref")
    t0 = st.theta.clone()
    s = train_step(st)
    assert s["generation"] == 1
    assert not torch.equal(st.theta, t0)
    assert (st.sigma_vec > 0).all()


def test_es_learns_cartpole_quick():
    cfg = preset("cartpole_es", pop=32, generations=12, episode_steps=300,
                 noise_table_size=1 << 18, device="cpu", sigma=0.1, lr=0.05,
                 log_every=1000)
    st = TrainState(cfg, ops_backend="torch_ref")
    first = train_step(st)["episodic_return_mean"]
    best = first
    for _ in range(11):
        best = max(best, train_step(st)["episodic_return_mean"])
    assert best > first + 10, (first, best)


def test_checkpoint_resume_bit_equivalence(tmp_path):
    cfg = _mini("es", checkpoint_path=None)
    a = TrainState(cfg, ops_backend="torch_ref")
    for _ in range(2):
        train_step(a)
    p = str(tmp_path / "ck.pt")
    save_checkpoint(p, a.checkpoint_payload())
    for _ in range(2):
        train_step(a)

    b = TrainState(cfg, ops_backend="torch_ref")
    b.load_payload(load_checkpoint(p))
    assert b.generation == 2
    for _ in range(2):
        train_step(b)
    assert torch.equal(a.theta, b.theta)
    assert a.optimizer.t == b.optimizer.t
    assert torch.equal(a.optimizer.m, b.optimizer.m)


def test_checkpoint_layout_reference_shape(tmp_path):
    st = TrainState(_mini("es"), ops_backend="torch_ref")
    train_step(st)
    payload = st.checkpoint_payload()
    assert set(payload) >= {"network", "optimizer", "normalizer", "generation", "rng"}
    # conventional layer names
    assert "fc1.weight" in payload["network"]
    assert "head.bias" in payload["network"]
    # loadable into a plain torch module
    mod = st.model.template()
    mod.load_state_dict(payload["network"])


def test_evaluate():
    st = TrainState(_mini("es"), ops_backend="torch_ref")
    train_step(st)
    r = evaluate(st, episodes=4)
    assert "return_mean" in r and r["env_steps"] > 0


def test_sigterm_checkpoint(tmp_path):
    # run() flushes a final checkpoint even without checkpoint_every
    from distributedes_amd import run
    p = str(tmp_path / "final.pt")
    cfg = _mini("es", checkpoint_path=p)
    run(cfg.replace(generations=2), ops_backend="torch_ref")
    assert os.path.exists(p)
    ck = load_checkpoint(p)
    assert ck["generation"] == 2


def test_bf16_noise_table_end_to_end():
    """cfg.noise_dtype='bf16': table stored bf16 (half the memory/traffic),
    training still works and stays deterministic."""
    cfg = _mini("es").replace(noise_dtype="bf16")
    a = TrainState(cfg, ops_backend="torch_ref")
    b = TrainState(cfg, ops_backend="torch_ref")
    assert a.table.dtype == torch.bfloat16
    for _ in range(2):
        train_step(a)
        train_step(b)
    assert torch.equal(a.theta, b.theta)
    assert torch.isfinite(a.theta).all()


def test_metrics_jsonl_and_plot(tmp_path):
    """R8 parity: per-generation JSONL metrics -> SVG learning curve."""
    mp = str(tmp_path / "m.jsonl")
    cfg = _mini("es", metrics_path=mp)
    st = TrainState(cfg, ops_backend="torch_ref")
    for _ in range(3):
        train_step(st)
    st.metrics.close()
    from distributedes_amd.utils.plot import load_series, render_svg
    xs, ys = load_series(mp)
    assert xs == [1, 2, 3]
    svg_path = str(tmp_path / "c.svg")
    svg = render_svg({"ret": (xs, ys)})
    open(svg_path, "w").write(svg)
    assert "<svg" in svg and "polyline" in svg


def test_rank_shaping_modes_run():
    for mode in ("normalized", "raw"):
        st = TrainState(_mini("es", rank_shaping=mode), ops_backend="torch_ref")
        t0 = st.theta.clone()
        train_step(st)
        assert not torch.equal(st.theta, t0)
        assert torch.isfinite(st.theta).all()


def test_sgd_optimizer_runs():
    st = TrainState(_mini("es", optimizer="sgd"), ops_backend="torch_ref")
    train_step(st)
    assert st.optimizer.t == 1
    assert torch.isfinite(st.theta).all()


def test_phase_timings_in_stats():
    st = TrainState(_mini("es", profile_phases=True), ops_backend="torch_ref")
    s = train_step(st)
    assert s["t_perturb_ms"] >= 0 and s["t_rollout_ms"] > 0


def test_resume_foreign_seed_keeps_table_dtype():
    """Loading a checkpoint whose table_seed differs must rebuild the table
    in the configured noise dtype."""
    cfg = _mini("es").replace(noise_dtype="bf16")
    a = TrainState(cfg, ops_backend="torch_ref")
    train_step(a)
    payload = a.checkpoint_payload()
    payload["rng"]["table_seed"] = 12345  # pretend another run's table
    b = TrainState(cfg, ops_backend="torch_ref")
    b.load_payload(payload)
    assert b.table.dtype == torch.bfloat16
    assert b.table_seed == 12345
    train_step(b)


def test_render_filmstrip(tmp_path):
    """SURVEY §3.3 eval/render stack: rollout frames -> SVG filmstrip."""
    from distributedes_amd import run
    from distributedes_amd.utils.render import frames_to_svg, rollout_frames
    p = str(tmp_path / "ck.pt")
    cfg = preset("pong_es_8k", pop=4, episode_steps=6, generations=1,
                 noise_table_size=1 << 22, device="cpu",
                 checkpoint_path=p, log_every=10 ** 9)
    run(cfg, ops_backend="torch_ref")
    frames, rewards, actions = rollout_frames(cfg, p, n_frames=3, stride=2)
    assert len(frames) == 3 and frames[0].shape == (84, 84)
    svg = frames_to_svg(frames, rewards)
    assert "<svg" in svg


def test_atari57_per_game_fitness_reported():
    """BASELINE config #4: the engine reports per-game fitness for the
    57-game multi-task population (engine.train_step)."""
    from distributedes_amd import preset
    from distributedes_amd.core.engine import TrainState, train_step
    cfg = preset("atari57_es", pop=8, episode_steps=2, device="cpu",
                 noise_table_size=1 << 21, log_every=10 ** 9)
    st = TrainState(cfg, ops_backend="torch_ref")
    s = train_step(st)
    pg = s["per_game_return_mean"]
    assert len(pg) == 57
    # pop=8 antithetic -> env ids 0..3 -> games 0..3 populated
    assert all(pg[g] == 0.0 for g in range(4, 57))
