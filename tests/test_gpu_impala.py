"""GPU parity for the IMPALA-CNN kernels (K6) vs the torch grouped-conv ref."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs MI355X", allow_module_level=True)

from distributedes_amd import preset
from distributedes_amd.core.engine import TrainState, train_step
from distributedes_amd.models import ImpalaCNNPolicy


def test_impala_forward_logits_match_ref():
    import torch.nn.functional as F
    from distributedes_amd.ops.hip import _load
    from distributedes_amd.ops.rollout import HipRollout
    from distributedes_amd.config import preset as mkpreset

    torch.manual_seed(0)
    pol = ImpalaCNNPolicy(num_actions=15)
    P = 4
    flat32 = torch.randn(P, pol.dim) * 0.08
    flat = flat32.to(torch.bfloat16)
    Dp = (pol.dim + 7) & ~7
    popw = torch.zeros(P, Dp, dtype=torch.bfloat16, device="cuda")
    popw[:, :pol.dim] = flat.cuda()
    obs = torch.rand(P, 3, 64, 64)

    cfg = mkpreset("procgen_es_32k", pop=P, device="cuda",
                   noise_table_size=1 << 21)
    from distributedes_amd.envs import make_env
    env = make_env("procgen_synth")
    env.obs_shape = (3, 64, 64)
    hr = HipRollout(cfg, env, pol, _load(), torch.device("cuda"))
    hr._alloc(P)
    hr.obs_bf16.copy_(obs.to(torch.bfloat16))
    logits = torch.zeros(P, 15, device="cuda")
    hr._impala_forward(popw, logits=logits)
    torch.cuda.synchronize()

    # reference in fp32 on bf16-rounded weights + bf16-rounded obs
    w = flat.float()
    obs_r = obs.to(torch.bfloat16).float()
    s = pol.spec
    x = obs_r.reshape(1, -1, 64, 64)

    def gconv(x, prefix):
        W = s.slice(w, prefix + ".weight")
        b = s.slice(w, prefix + ".bias")
        Co = W.shape[1]
        return F.conv2d(x, W.reshape(P * Co, *W.shape[2:]), b.reshape(-1),
                        padding=1, groups=P)

    for st in (1, 2, 3):
        x = gconv(x, f"stage{st}_conv")
        x = F.max_pool2d(x, 3, stride=2, padding=1)
        for r in (1, 2):
            y = gconv(F.relu(x), f"stage{st}_res{r}_conv1")
            y = gconv(F.relu(y), f"stage{st}_res{r}_conv2")
            x = x + y
    x = F.relu(x).reshape(P, -1, 1)
    xf = F.relu(torch.baddbmm(s.slice(w, "fc.bias").unsqueeze(-1),
                              s.slice(w, "fc.weight"), x))
    ref = torch.baddbmm(s.slice(w, "head.bias").unsqueeze(-1),
                        s.slice(w, "head.weight"), xf).squeeze(-1)
    torch.testing.assert_close(logits.cpu(), ref, rtol=0.08, atol=0.08)


def test_impala_end_to_end_train_step():
    cfg = preset("procgen_es_32k", pop=32, episode_steps=8,
                 noise_table_size=1 << 21, device="cuda", log_every=10 ** 9)
    st = TrainState(cfg)
    assert st._hip_rollout is not None and st._hip_rollout.kind == "impala"
    s = train_step(st)
    assert s["env_steps"] == 32 * 8
    assert torch.isfinite(st.theta).all()
