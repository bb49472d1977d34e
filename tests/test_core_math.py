"""ES/CEM math vs independent oracles: rank transform, Adam, normalizer."""
import numpy as np
import torch

from distributedes_amd.core.optim import FlatAdam
from distributedes_amd.core.rank import centered_ranks, normalized_fitness
from distributedes_amd.utils.normalizer import RunningNormalizer


def test_centered_ranks_vs_numpy():
    g = np.random.default_rng(0)
    f = torch.tensor(g.normal(size=101), dtype=torch.float32)
    u = centered_ranks(f).numpy()
    order = np.argsort(f.numpy(), kind="stable")
    ranks = np.empty(101)
    ranks[order] = np.arange(101)
    np.testing.assert_allclose(u, ranks / 100 - 0.5, atol=1e-7)
    assert abs(u.sum()) < 1e-5
    assert u.min() == -0.5 and u.max() == 0.5


def test_centered_ranks_ties_stable():
    f = torch.tensor([1.0, 1.0, 0.0, 2.0])
    u = centered_ranks(f)
    # stable: first 1.0 ranks below second 1.0
    assert u[0] < u[1]
    assert u[2] == -0.5 and u[3] == 0.5


def test_normalized_fitness():
    f = torch.tensor([1.0, 2.0, 3.0, 4.0])
    u = normalized_fitness(f)
    assert abs(float(u.mean())) < 1e-6
    assert abs(float(u.std(unbiased=False)) - 1.0) < 1e-4


def test_flat_adam_matches_torch_adam():
    torch.manual_seed(0)
    dim = 257
    theta0 = torch.randn(dim)
    p = theta0.clone().requires_grad_(True)
    opt = torch.optim.Adam([p], lr=0.01, betas=(0.9, 0.999), eps=1e-8)
    mine = FlatAdam(dim, lr=0.01)
    theta = theta0.clone()
    for step in range(10):
        g = torch.randn(dim)
        opt.zero_grad()
        p.grad = g.clone()
        opt.step()
        mine.step(theta, g)
        torch.testing.assert_close(theta, p.detach(), rtol=1e-6, atol=1e-7)


def test_normalizer_merge_associative_and_correct():
    g = np.random.default_rng(1)
    a = torch.tensor(g.normal(2.0, 3.0, size=(100, 5)), dtype=torch.float32)
    b = torch.tensor(g.normal(-1.0, 0.5, size=(77, 5)), dtype=torch.float32)
    n1 = RunningNormalizer(5)
    n1.observe(a); n1.observe(b); n1.commit()
    # merged across "ranks" (sum of the three moment tensors)
    na, nb = RunningNormalizer(5), RunningNormalizer(5)
    na.observe(a); na.commit(); nb.observe(b); nb.commit()
    nm = RunningNormalizer(5)
    nm.count = na.count + nb.count
    nm.sum = na.sum + nb.sum
    nm.sumsq = na.sumsq + nb.sumsq
    nm.refresh()
    x = torch.tensor(g.normal(size=(13, 5)), dtype=torch.float32)
    torch.testing.assert_close(n1.normalize(x), nm.normalize(x), rtol=1e-6, atol=1e-6)
    # against direct mean/std
    all_ = torch.cat([a, b]).to(torch.float64)
    mean = all_.mean(0)
    std = all_.std(0, unbiased=False)
    y = ((x.to(torch.float64) - mean) / std).clamp(-5, 5).to(torch.float32)
    torch.testing.assert_close(n1.normalize(x), y, rtol=1e-4, atol=1e-5)
