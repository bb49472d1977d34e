"""Model tests: flat<->state_dict round trip; pop_forward_ref vs per-member
module forward (independent oracle via the template nn.Module)."""
import torch

from distributedes_amd.models import ImpalaCNNPolicy, MLPPolicy, NatureCNNPolicy


def _per_member_oracle(policy, flat_w, obs):
    """Evaluate each member through a real nn.Module — the independent path."""
    outs = []
    for i in range(flat_w.shape[0]):
        mod = policy.template()
        mod.load_state_dict(policy.spec.unflatten(flat_w[i].to(torch.float32)))
        outs.append(mod)
    return outs


def test_param_spec_roundtrip():
    pol = MLPPolicy(4, 2, (8, 8))
    sd = pol.module.state_dict()
    flat = pol.spec.flatten(sd)
    sd2 = pol.spec.unflatten(flat)
    for k in sd:
        torch.testing.assert_close(sd[k].to(torch.float32), sd2[k])
    assert pol.dim == sum(v.numel() for v in sd.values())


def test_mlp_pop_forward_matches_modules():
    torch.manual_seed(0)
    pol = MLPPolicy(4, 2, (8, 8))
    P = 5
    flat = torch.randn(P, pol.dim)
    obs = torch.randn(P, 4)
    acts = pol.pop_forward_ref(flat, obs)
    mods = _per_member_oracle(pol, flat, obs)
    for i, mod in enumerate(mods):
        x = torch.tanh(mod.fc1(obs[i]))
        x = torch.tanh(mod.fc2(x))
        logits = mod.head(x)
        assert int(acts[i]) == int(logits.argmax())


def test_mlp_continuous():
    torch.manual_seed(0)
    pol = MLPPolicy(6, 3, (8,), continuous=True)
    P = 4
    flat = torch.randn(P, pol.dim)
    obs = torch.randn(P, 6)
    acts = pol.pop_forward_ref(flat, obs)
    assert acts.shape == (P, 3)
    mod = pol.template()
    mod.load_state_dict(pol.spec.unflatten(flat[2]))
    ref = torch.tanh(mod.head(torch.tanh(mod.fc1(obs[2]))))
    torch.testing.assert_close(acts[2], ref, rtol=1e-5, atol=1e-6)


def test_nature_cnn_pop_forward_matches_modules():
    torch.manual_seed(1)
    pol = NatureCNNPolicy(num_actions=18)
    assert pol.dim == 1693362
    P = 3
    flat = torch.randn(P, pol.dim) * 0.05
    obs = torch.rand(P, 4, 84, 84)
    acts = pol.pop_forward_ref(flat, obs)
    for i in range(P):
        mod = pol.template()
        mod.load_state_dict(pol.spec.unflatten(flat[i]))
        x = torch.relu(mod.conv1(obs[i:i + 1]))
        x = torch.relu(mod.conv2(x))
        x = torch.relu(mod.conv3(x))
        x = torch.relu(mod.fc(x.reshape(1, -1)))
        logits = mod.head(x)
        assert int(acts[i]) == int(logits.argmax())


def test_impala_cnn_pop_forward_matches_modules():
    torch.manual_seed(2)
    pol = ImpalaCNNPolicy(num_actions=15)
    P = 2
    flat = torch.randn(P, pol.dim) * 0.05
    obs = torch.rand(P, 3, 64, 64)
    acts = pol.pop_forward_ref(flat, obs)
    import torch.nn.functional as F
    for i in range(P):
        mod = pol.template()
        mod.load_state_dict(pol.spec.unflatten(flat[i]))
        x = obs[i:i + 1]
        for s in (1, 2, 3):
            x = getattr(mod, f"stage{s}_conv")(x)
            x = F.max_pool2d(x, 3, stride=2, padding=1)
            for r in (1, 2):
                y = getattr(mod, f"stage{s}_res{r}_conv1")(F.relu(x))
                y = getattr(mod, f"stage{s}_res{r}_conv2")(F.relu(y))
                x = x + y
        x = F.relu(getattr(mod, "fc")(F.relu(x).reshape(1, -1)))
        logits = mod.head(x)
        assert int(acts[i]) == int(logits.argmax())


def test_init_theta_deterministic():
    pol = MLPPolicy(4, 2, (8, 8))
    t1 = pol.init_theta(5)
    t2 = pol.init_theta(5)
    t3 = pol.init_theta(6)
    assert torch.equal(t1, t2)
    assert not torch.equal(t1, t3)
