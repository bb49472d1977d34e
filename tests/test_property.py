"""Hypothesis property tests: normalizer merge
associativity, centered-rank invariants, offset bounds."""
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from distributedes_amd.core.noise import member_offsets
from distributedes_amd.core.rank import centered_ranks
from distributedes_amd.utils.normalizer import RunningNormalizer


@settings(max_examples=30, deadline=None)
@given(st.lists(st.floats(-1e6, 1e6, allow_nan=False), min_size=2,
                max_size=200))
def test_centered_ranks_invariants(vals):
    f = torch.tensor(vals, dtype=torch.float32)
    u = centered_ranks(f)
    n = len(vals)
    assert abs(float(u.sum())) < 1e-4
    assert float(u.min()) == -0.5 and float(u.max()) == 0.5
    # monotone: larger fitness never gets a smaller rank weight
    # (sort the same f32 values torch saw; f64 would split f32-ties)
    order = np.argsort(f.numpy(), kind="stable")
    ranked = u.numpy()[order]
    assert (np.diff(ranked) >= -1e-7).all()


@settings(max_examples=20, deadline=None)
@given(st.integers(2, 5), st.integers(1, 40), st.integers(1, 40),
       st.integers(0, 10 ** 6))
def test_normalizer_merge_associative(dim, n1, n2, seed):
    g = np.random.default_rng(seed)
    a = torch.tensor(g.normal(size=(n1, dim)) * 3 + 1, dtype=torch.float32)
    b = torch.tensor(g.normal(size=(n2, dim)) - 2, dtype=torch.float32)
    direct = RunningNormalizer(dim)
    direct.observe(torch.cat([a, b])); direct.commit()
    split = RunningNormalizer(dim)
    split.observe(a); split.commit()
    split.observe(b); split.commit()
    x = torch.tensor(g.normal(size=(7, dim)), dtype=torch.float32)
    torch.testing.assert_close(direct.normalize(x), split.normalize(x),
                               rtol=1e-5, atol=1e-5)


@settings(max_examples=20, deadline=None)
@given(st.integers(0, 2 ** 31), st.integers(0, 1000), st.integers(2, 64),
       st.booleans())
def test_member_offsets_always_in_bounds(seed, gen, pop, anti):
    if anti and pop % 2:
        pop += 1
    T, D = 1 << 18, 5003
    offs, signs = member_offsets(seed, gen, pop, anti, T, D)
    assert (offs >= 0).all() and (offs + D <= T).all()
    assert (offs % 16 == 0).all()
    assert set(np.unique(signs)) <= {-1.0, 1.0}


@given(st.integers(0, 2**30), st.integers(2, 57))
@settings(max_examples=30, deadline=None)
def test_game_coefs_deterministic_and_in_range(seed, ng):
    """Atari-57 parameterization invariants: hashed per-game coefficients
    are pure functions of the game id, stay in their documented ranges,
    and distinct games get distinct rows (task distinctness)."""
    import torch

    from distributedes_amd.envs.synthetic import game_coefs
    g = torch.arange(ng, dtype=torch.int64)
    a = game_coefs(g, ng)
    b = game_coefs(g.clone(), ng)
    assert torch.equal(a, b)
    lo = torch.tensor([1.10, 0.40, 0.05, 1.50, 0.20, 2.00])
    hi = torch.tensor([1.60, 0.90, 0.30, 4.50, 0.80, 4.00])
    assert (a >= lo - 1e-6).all() and (a <= hi + 1e-6).all()
    assert torch.unique(a, dim=0).shape[0] == ng


@given(st.integers(0, 2**20), st.integers(1, 40))
@settings(max_examples=30, deadline=None)
def test_antithetic_pair_shares_offset_with_opposite_sign(seed, gen):
    """Mirrored sampling invariant: members 2k and 2k+1 draw the SAME
    noise slice with opposite signs (the variance-reduction contract the
    ES update and the env-stream pairing both rely on)."""
    import numpy as np

    from distributedes_amd.core.noise import member_offsets
    offs, signs = member_offsets(seed, gen, 16, True, 1 << 20, 1024)
    assert np.array_equal(offs[0::2], offs[1::2])
    assert np.all(signs[0::2] == 1.0) and np.all(signs[1::2] == -1.0)
