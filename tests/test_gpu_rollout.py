"""GPU parity tier for the rollout kernels: policy forwards (K3/K5) vs the
torch reference twins, env kernels (K7) vs the torch envs, end-to-end hip
rollout vs the eager path."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from distributedes_amd import preset
from distributedes_amd.core.engine import TrainState, train_step
from distributedes_amd.envs import CartPoleEnv, PongEnv
from distributedes_amd.models import MLPPolicy, NatureCNNPolicy
from distributedes_amd.ops.rollout import mlp_layer_offsets, nature_cnn_offsets


@pytest.fixture(scope="module")
def ext():
    from distributedes_amd.ops.hip import _load
    e = _load()
    assert e is not None
    return e


def _pad(w, mult=8):
    P, D = w.shape
    Dp = (D + mult - 1) & ~(mult - 1)
    out = torch.zeros(P, Dp, dtype=w.dtype, device=w.device)
    out[:, :D] = w
    return out


# --------------------------- policy forwards ------------------------------

def test_mlp_forward_logits_match_ref(ext):
    torch.manual_seed(0)
    pol = MLPPolicy(4, 2, (64, 64))
    P = 64
    flat = (torch.randn(P, pol.dim) * 0.3).cuda()
    obs = torch.randn(P, 4).cuda()
    popw = _pad(flat)
    logits = torch.zeros(P, 2).cuda()
    acts = torch.zeros(P, dtype=torch.int32).cuda()
    offs = mlp_layer_offsets(pol.spec, 2)
    empty = torch.empty(0).cuda()
    ext.mlp_forward(popw, obs, empty, empty, 0.0, acts, logits, 64, 64, 2, offs)
    # reference logits: tanh MLP in fp32
    w = flat.cpu()
    x = obs.cpu().unsqueeze(-1)
    s = pol.spec
    x = torch.tanh(torch.baddbmm(s.slice(w, "fc1.bias").unsqueeze(-1),
                                 s.slice(w, "fc1.weight"), x))
    x = torch.tanh(torch.baddbmm(s.slice(w, "fc2.bias").unsqueeze(-1),
                                 s.slice(w, "fc2.weight"), x))
    ref = torch.baddbmm(s.slice(w, "head.bias").unsqueeze(-1),
                        s.slice(w, "head.weight"), x).squeeze(-1)
    torch.testing.assert_close(logits.cpu(), ref, rtol=1e-4, atol=1e-5)
    assert (acts.cpu() == ref.argmax(-1).to(torch.int32)).float().mean() > 0.95


def test_mlp_forward_normalizer_and_continuous(ext):
    torch.manual_seed(1)
    pol = MLPPolicy(376, 17, (256, 256), continuous=True)
    P = 16
    flat = (torch.randn(P, pol.dim) * 0.1).cuda()
    obs = (torch.randn(P, 376) * 3 + 1).cuda()
    nmean = torch.randn(376).cuda()
    ninv = torch.rand(376).cuda() + 0.5
    popw = _pad(flat)
    acts = torch.zeros(P, 17).cuda()
    offs = mlp_layer_offsets(pol.spec, 2)
    ext.mlp_forward(popw, obs, nmean, ninv, 5.0, acts, torch.empty(0).cuda(),
                    256, 256, 17, offs)
    xn = ((obs - nmean) * ninv).clamp(-5, 5)
    ref = pol.pop_forward_ref(flat, xn)
    torch.testing.assert_close(acts, ref, rtol=1e-4, atol=1e-5)


def test_nature_cnn_forward_logits_match_ref(ext):
    torch.manual_seed(2)
    pol = NatureCNNPolicy(num_actions=18)
    P = 8
    flat32 = torch.randn(P, pol.dim) * 0.05
    flat = flat32.to(torch.bfloat16).cuda()
    obs_u8 = (torch.rand(P, 4, 84, 84) < 0.15).to(torch.uint8).cuda()
    popw = _pad(flat)
    slot_map = torch.tensor([0, 1, 2, 3], dtype=torch.int32).cuda()
    acts = torch.zeros(P, dtype=torch.int32).cuda()
    logits = torch.zeros(P, 18).cuda()
    offs = torch.tensor(nature_cnn_offsets(pol.spec), dtype=torch.int64).cuda()
    ext.nature_cnn_forward(popw, obs_u8, slot_map, acts, logits, 18, offs)
    # reference: fp32 math on the same bf16-rounded weights
    ref = _nature_ref_logits(pol, flat.cpu().float(), obs_u8.cpu().float())
    # kernel stores intermediate activations in bf16 -> tolerate ~1% rel
    torch.testing.assert_close(logits.cpu(), ref, rtol=0.05, atol=0.05)
    agree = (acts.cpu() == ref.argmax(-1).to(torch.int32)).float().mean()
    assert agree > 0.7, float(agree)


def _nature_ref_logits(pol, w, obs, bf16_acts=False):
    """Grouped-conv fp32 reference.  bf16_acts=True additionally rounds the
    inter-layer activations to bf16 — modeling the kernel's per-layer
    storage dtype so each layer's rounding is budgeted explicitly (the
    matched-dtype reference of the round-1 review)."""
    import torch.nn.functional as F
    P = w.shape[0]
    s = pol.spec
    x = obs.reshape(1, -1, 84, 84)

    def rnd(t):
        return t.to(torch.bfloat16).float() if bf16_acts else t

    def gconv(x, wn, bn, stride):
        W = s.slice(w, wn)
        b = s.slice(w, bn)
        Co = W.shape[1]
        return F.conv2d(x, W.reshape(P * Co, *W.shape[2:]), b.reshape(-1),
                        stride=stride, groups=P)
    x = rnd(F.relu(gconv(x, "conv1.weight", "conv1.bias", 4)))
    x = rnd(F.relu(gconv(x, "conv2.weight", "conv2.bias", 2)))
    x = rnd(F.relu(gconv(x, "conv3.weight", "conv3.bias", 1)))
    x = x.reshape(P, -1, 1)
    x = rnd(F.relu(torch.baddbmm(s.slice(w, "fc.bias").unsqueeze(-1),
                                 s.slice(w, "fc.weight"), x)))
    return torch.baddbmm(s.slice(w, "head.bias").unsqueeze(-1),
                         s.slice(w, "head.weight"), x).squeeze(-1)


def _nature_fixed_inputs(P=8):
    """Deterministic weights/frames for the golden + matched-dtype tests."""
    g = torch.Generator().manual_seed(0xA57)
    pol = NatureCNNPolicy(num_actions=18)
    flat = (torch.randn(P, pol.dim, generator=g) * 0.05).to(torch.bfloat16)
    obs_u8 = (torch.rand(P, 4, 84, 84, generator=g) < 0.15).to(torch.uint8)
    return pol, flat, obs_u8


def _nature_kernel_logits(ext, pol, flat, obs_u8):
    P = flat.shape[0]
    popw = _pad(flat.cuda())
    slot_map = torch.tensor([0, 1, 2, 3], dtype=torch.int32).cuda()
    acts = torch.zeros(P, dtype=torch.int32).cuda()
    logits = torch.zeros(P, 18).cuda()
    offs = torch.tensor(nature_cnn_offsets(pol.spec), dtype=torch.int64).cuda()
    ext.nature_cnn_forward(popw, obs_u8.cuda(), slot_map, acts, logits, 18,
                           offs)
    return logits.cpu(), acts.cpu()


def test_nature_cnn_matched_dtype_argmax(ext):
    """Against the bf16-intermediates reference the kernel must agree on
    >=90% of argmax actions and track logits within a per-layer bf16
    error budget (round-1 review weak #3: 0.7 at fp32-ref is too loose)."""
    pol, flat, obs_u8 = _nature_fixed_inputs(P=64)
    logits, acts = _nature_kernel_logits(ext, pol, flat, obs_u8)
    ref = _nature_ref_logits(pol, flat.float(), obs_u8.float(), bf16_acts=True)
    torch.testing.assert_close(logits, ref, rtol=0.02, atol=0.02)
    agree = (acts == ref.argmax(-1).to(torch.int32)).float().mean()
    assert agree >= 0.9, float(agree)


def test_nature_cnn_golden_logits(ext):
    """Bit-exact regression pin: fixed-seed weights/frames vs a committed
    golden tensor.  Catches 1-ulp-scale kernel regressions that tolerance
    tests hide.  An INTENTIONAL numerics change must regenerate the golden:
    delete tests/golden/nature_logits_v1.pt and rerun on a GPU box (the
    test reseeds it under gpurun_out/golden/ for committing)."""
    import os
    pol, flat, obs_u8 = _nature_fixed_inputs(P=8)
    logits, acts = _nature_kernel_logits(ext, pol, flat, obs_u8)
    here = os.path.dirname(os.path.abspath(__file__))
    gpath = os.path.join(here, "golden", "nature_logits_v1.pt")
    if not os.path.exists(gpath):
        outdir = os.path.join(here, "..", "gpurun_out", "golden")
        os.makedirs(outdir, exist_ok=True)
        torch.save({"logits": logits, "acts": acts},
                   os.path.join(outdir, "nature_logits_v1.pt"))
        pytest.skip("golden seeded under gpurun_out/golden/ - commit it to "
                    "tests/golden/")
    want = torch.load(gpath)
    torch.testing.assert_close(logits, want["logits"], rtol=0, atol=0)
    assert torch.equal(acts, want["acts"])


def test_nature_cnn_pop_scale_action_distribution(ext):
    """pop=8192-shape smoke: random-weight population must produce a healthy
    action spread (no collapsed/NaN logits at bench scale)."""
    torch.manual_seed(11)
    pol = NatureCNNPolicy(num_actions=18)
    P = 8192
    flat = (torch.randn(P, pol.dim) * 0.05).to(torch.bfloat16).cuda()
    obs_u8 = (torch.rand(P, 4, 84, 84) < 0.15).to(torch.uint8).cuda()
    popw = _pad(flat)
    slot_map = torch.tensor([0, 1, 2, 3], dtype=torch.int32).cuda()
    acts = torch.zeros(P, dtype=torch.int32).cuda()
    logits = torch.zeros(P, 18).cuda()
    offs = torch.tensor(nature_cnn_offsets(pol.spec), dtype=torch.int64).cuda()
    ext.nature_cnn_forward(popw, obs_u8, slot_map, acts, logits, 18, offs)
    assert torch.isfinite(logits).all()
    counts = torch.bincount(acts.to(torch.int64), minlength=18)
    assert (counts > 0).sum() >= 12          # most actions represented
    assert counts.max() < P * 0.5            # no single-action collapse


def test_nature_cnn_slot_map_rolls_frames(ext):
    """Rotating the physical frames with the matching slot_map must give the
    identical logits (ring-buffer correctness)."""
    torch.manual_seed(3)
    pol = NatureCNNPolicy(num_actions=18)
    P = 4
    flat = (torch.randn(P, pol.dim) * 0.05).to(torch.bfloat16).cuda()
    popw = _pad(flat)
    frames = (torch.rand(P, 4, 84, 84) < 0.2).to(torch.uint8).cuda()
    offs = torch.tensor(nature_cnn_offsets(pol.spec), dtype=torch.int64).cuda()
    outs = []
    for r in range(4):
        # rolled[:, (c + r) % 4] == frames[:, c]
        rolled = torch.roll(frames, shifts=r, dims=1).contiguous()
        sm = torch.tensor([(c + r) % 4 for c in range(4)],
                          dtype=torch.int32).cuda()
        logits = torch.zeros(P, 18).cuda()
        acts = torch.zeros(P, dtype=torch.int32).cuda()
        ext.nature_cnn_forward(popw, rolled, sm, acts, logits, 18, offs)
        outs.append(logits.cpu())
    for r in (1, 2, 3):
        torch.testing.assert_close(outs[0], outs[r])


# --------------------------- env kernels ----------------------------------

def test_cartpole_kernel_matches_torch_env(ext):
    P = 16
    seed = 7
    env = CartPoleEnv()
    obs_ref = env.reset(P, seed)
    dev = "cuda"
    ids = torch.arange(P, dtype=torch.int64, device=dev)
    state = torch.zeros(P, 4, device=dev)
    steps = torch.zeros(P, dtype=torch.int32, device=dev)
    rctr = torch.zeros(P, dtype=torch.int64, device=dev)
    fit = torch.zeros(P, device=dev)
    alive = torch.ones(P, dtype=torch.uint8, device=dev)
    eplen = torch.zeros(P, device=dev)
    ext.cartpole_reset(state, steps, rctr, fit, alive, eplen, ids, seed * 0x10001)
    torch.testing.assert_close(state.cpu(), obs_ref, rtol=1e-6, atol=1e-7)
    seed_buf = torch.tensor([seed * 0x10001], dtype=torch.int64, device=dev)
    torch.manual_seed(0)
    for t in range(120):
        a = torch.randint(0, 2, (P,))
        obs_ref, r, d = env.step(a)
        ext.cartpole_step(state, steps, rctr, a.to(torch.int32).cuda(), fit,
                          alive, eplen, ids, seed_buf)
        torch.testing.assert_close(state.cpu(), obs_ref, rtol=1e-4, atol=1e-4)


def test_pong_kernel_matches_torch_env(ext):
    P = 8
    seed = 11
    env = PongEnv()
    env.reset(P, seed)
    dev = "cuda"
    ids = torch.arange(P, dtype=torch.int64, device=dev)
    z = lambda dt=torch.float32: torch.zeros(P, dtype=dt, device=dev)  # noqa
    bx, by, vx, vy, py, oy = (z() for _ in range(6))
    score, points = z(torch.int32), z(torch.int32)
    serve = z(torch.int64)
    frames = torch.zeros(P, 4, 84, 84, dtype=torch.uint8, device=dev)
    fit, alive, eplen = z(), torch.ones(P, dtype=torch.uint8, device=dev), z()
    ext.pong_reset(bx, by, vx, vy, py, oy, score, points, serve, frames, fit,
                   alive, eplen, ids, seed)
    torch.testing.assert_close(bx.cpu(), env.bx, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(by.cpu(), env.by, rtol=1e-5, atol=1e-5)
    seed_buf = torch.tensor([seed], dtype=torch.int64, device=dev)
    total_r = torch.zeros(P)
    for t in range(200):
        a = torch.full((P,), t % 3, dtype=torch.int64)
        obs_ref, r, d = env.step(a)
        ext.pong_step(bx, by, vx, vy, py, oy, score, points, serve, frames,
                      t % 4, a.to(torch.int32).cuda(), fit, alive, eplen, ids,
                      seed_buf)
        total_r += r
        torch.testing.assert_close(by.cpu(), env.by, rtol=1e-3, atol=1e-3)
        torch.testing.assert_close(py.cpu(), env.player_y, rtol=1e-4, atol=1e-4)
        # rendered newest frame == ref newest channel (allow rare 1-px diff)
        got = frames[:, t % 4].cpu().float()
        want = obs_ref[:, 3]
        mismatch = (got != want).float().mean()
        assert mismatch < 0.001, (t, float(mismatch))


def test_latent_and_humanoid_kernels_match(ext):
    from distributedes_amd.envs import Atari57SyntheticEnv, HumanoidSyntheticEnv
    from distributedes_amd.envs.synthetic import game_coefs
    P = 8
    env = Atari57SyntheticEnv()
    env.reset(P, 5)
    dev = "cuda"
    ids = torch.arange(P, dtype=torch.int64, device=dev)
    zz = torch.zeros(P, 8, device=dev)
    coef = game_coefs(ids % env.n_games, env.n_games).to(dev)
    steps = torch.zeros(P, dtype=torch.int32, device=dev)
    fit = torch.zeros(P, device=dev)
    alive = torch.ones(P, dtype=torch.uint8, device=dev)
    eplen = torch.zeros(P, device=dev)
    obs = torch.zeros(P, 4, 84, 84, dtype=torch.bfloat16, device=dev)
    ext.latent_reset(zz, steps, fit, alive, eplen, ids, 5)
    torch.testing.assert_close(zz.cpu(), env.z, rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(coef.cpu(), env.coef, rtol=0, atol=0)
    for t in range(5):
        a = torch.randint(0, 18, (P,))
        o_ref, r_ref, _ = env.step(a)
        ext.latent_step(zz, coef, steps, a.to(torch.int32).cuda(), fit, alive,
                        eplen, 18, 1000)
        ext.latent_render(zz, coef, obs)
        torch.testing.assert_close(zz.cpu(), env.z, rtol=1e-3, atol=1e-4)
        torch.testing.assert_close(obs.cpu().float(), o_ref, rtol=0.05, atol=0.05)

    henv = HumanoidSyntheticEnv()
    henv.reset(P, 3)
    s = torch.zeros(P, 376, device=dev)
    ext.humanoid_reset(s, steps, fit, alive, eplen, ids, 3)
    torch.testing.assert_close(s.cpu(), henv.s, rtol=1e-5, atol=1e-6)
    fit.zero_(); alive.fill_(1)
    hfit_ref = torch.zeros(P)
    for t in range(5):
        a = torch.randn(P, 17)
        _, r_ref, _ = henv.step(a)
        hfit_ref += r_ref
        ext.humanoid_step(s, steps, a.cuda(), fit, alive, eplen, 1000)
        torch.testing.assert_close(s.cpu(), henv.s, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(fit.cpu(), hfit_ref, rtol=1e-3, atol=1e-4)


# --------------------------- end-to-end -----------------------------------

def test_hip_rollout_pong_end_to_end():
    cfg = preset("pong_es_8k", pop=32, episode_steps=12,
                 noise_table_size=1 << 22, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    assert st._hip_rollout is not None, "hip rollout must engage for pong"
    s = train_step(st)
    assert s["env_steps"] == 32 * 12
    assert torch.isfinite(st.theta).all()


def test_hip_rollout_cartpole_matches_torch_gen1():
    mk = lambda dev, **kw: preset(  # noqa: E731
        "cartpole_es", pop=16, episode_steps=25, noise_table_size=1 << 18,
        device=dev, precision="fp32", log_every=10 ** 9, **kw)
    a = TrainState(mk("cpu"), ops_backend="torch_ref")
    sa = train_step(a)
    b = TrainState(mk("cuda"))
    assert b._hip_rollout is not None
    sb = train_step(b)
    # same seeds, fp32 weights; trajectories identical up to libm ulps
    assert abs(sa["episodic_return_mean"] - sb["episodic_return_mean"]) < 1.0
    torch.testing.assert_close(a.theta, b.theta.cpu(), rtol=2e-2, atol=2e-3)


def test_hip_rollout_humanoid_runs():
    cfg = preset("humanoid_es_10k", pop=64, episode_steps=20,
                 noise_table_size=1 << 20, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    assert st._hip_rollout is not None
    s = train_step(st)
    assert s["env_steps"] == 64 * 20
    assert float(st.normalizer.count) == 64 * 20
