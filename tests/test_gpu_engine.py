"""GPU integration tier: end-to-end train_step on the HIP backend + GPU==CPU
update parity."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from distributedes_amd import preset
from distributedes_amd.core.engine import TrainState, train_step


def test_train_step_cartpole_cuda():
    cfg = preset("cartpole_es", pop=16, episode_steps=20,
                 noise_table_size=1 << 20, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    assert st.ops.name == "hip"
    s = train_step(st)
    assert s["env_steps"] == 16 * 20
    assert torch.isfinite(st.theta).all()


def test_train_step_pong_cuda_bf16():
    cfg = preset("pong_es_8k", pop=32, episode_steps=6,
                 noise_table_size=1 << 22, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    s = train_step(st)
    assert s["env_steps"] == 32 * 6
    assert torch.isfinite(st.theta).all()


def test_cem_step_cuda():
    cfg = preset("cartpole_cem", pop=32, episode_steps=10,
                 noise_table_size=1 << 20, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    train_step(st)
    assert torch.isfinite(st.sigma_vec).all()
    assert (st.sigma_vec > 0).all()


def test_update_parity_gpu_vs_cpu():
    """Identical generation-0 perturbation + injected fitness -> the ES update
    must agree between the CPU torch path and the GPU HIP path."""
    mk = lambda dev: preset("cartpole_es", pop=16, episode_steps=5,  # noqa: E731
                            noise_table_size=1 << 18, device=dev,
                            obs_normalize=False, log_every=10 ** 9)
    torch.manual_seed(0)
    fitness = torch.randn(16)

    results = {}
    for dev, backend in (("cpu", "torch_ref"), ("cuda", None)):
        st = TrainState(mk(dev), ops_backend=backend)
        offs, signs = st.member_offsets_signs(0)
        st.es_update(fitness.to(st.device), offs, signs)
        results[dev] = st.theta.cpu()
    torch.testing.assert_close(results["cpu"], results["cuda"],
                               rtol=1e-4, atol=1e-5)


def test_rollout_fitness_finite_and_varied():
    cfg = preset("pong_es_8k", pop=64, episode_steps=60,
                 noise_table_size=1 << 22, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    offs, signs = st.member_offsets_signs(0)
    pop_w = st.perturb_local(offs, signs)
    fitness, env_steps, _ = st.rollout(pop_w)
    assert torch.isfinite(fitness).all()
    assert env_steps == 64 * 60


def test_hip_graph_replay_matches_eager_across_generations():
    """Round-1 advisor HIGH finding: captured graphs baked a freeable env_id
    pointer, corrupting per-env RNG streams from the 2nd generation after
    capture.  Two identical runs — graphs on vs off — must produce the
    SAME fitness sequence over several generations (the corruption only
    shows up on replays after the first)."""
    def run(graphs):
        cfg = preset("pong_es_8k", pop=64, episode_steps=10,
                     noise_table_size=1 << 22, device="cuda",
                     use_hip_graphs=graphs, log_every=10 ** 9)
        st = TrainState(cfg)
        rets = []
        for _ in range(4):
            rets.append(train_step(st)["episodic_return_mean"])
        if graphs:
            assert st._hip_rollout.last_used_graph, "graphs must engage"
        return rets, st.theta.cpu()

    r_g, th_g = run(True)
    r_e, th_e = run(False)
    assert r_g == r_e, (r_g, r_e)
    torch.testing.assert_close(th_g, th_e, rtol=0, atol=0)


def test_evaluate_does_not_touch_training_state():
    """Round-1 advisor MEDIUM finding: evaluate() must not fold eval
    observations into the normalizer deltas nor mutate TrainState."""
    from distributedes_amd import evaluate
    cfg = preset("cartpole_es", pop=16, episode_steps=10,
                 noise_table_size=1 << 20, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    train_step(st)
    before = {
        "dsum": st.normalizer._dsum.clone(),
        "dcount": st.normalizer._dcount.clone(),
        "member_base": st.member_base,
        "local_pop": st.local_pop,
        "cfg": st.cfg,
        "theta": st.theta.clone(),
    }
    evaluate(st, episodes=8)
    assert torch.equal(before["dsum"], st.normalizer._dsum)
    assert torch.equal(before["dcount"], st.normalizer._dcount)
    assert st.member_base == before["member_base"]
    assert st.local_pop == before["local_pop"]
    assert st.cfg is before["cfg"]
    assert torch.equal(before["theta"], st.theta)
