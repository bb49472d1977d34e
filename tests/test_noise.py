"""Noise table + seed/offset derivation unit tests (SURVEY.md §4.2 tier 1)."""
import numpy as np
import torch

from distributedes_amd.core import noise


def test_table_deterministic():
    a = noise.fill_noise_table_cpu(4096, 1234)
    b = noise.fill_noise_table_cpu(4096, 1234)
    assert torch.equal(a, b)
    c = noise.fill_noise_table_cpu(4096, 1235)
    assert not torch.equal(a, c)


def test_table_gaussian_stats():
    t = noise.fill_noise_table_cpu(1 << 20, 7)
    x = t.numpy()
    assert abs(x.mean()) < 5e-3
    assert abs(x.std() - 1.0) < 5e-3
    assert abs((x < 0).mean() - 0.5) < 2e-3
    # fourth moment of N(0,1) is 3
    assert abs((x ** 4).mean() - 3.0) < 0.05


def test_slice_regeneration_matches_table():
    seed = 42
    t = noise.fill_noise_table_cpu(8192, seed).numpy()
    offs = np.array([0, 1, 5, 4096, 8192 - 128], dtype=np.int64)
    sl = noise.noise_slice_cpu(offs, 128, seed)
    for i, off in enumerate(offs):
        np.testing.assert_array_equal(sl[i], t[off:off + 128])


def test_member_seed_determinism_and_spread():
    s1 = noise.member_seed(1, 2, 3)
    assert s1 == noise.member_seed(1, 2, 3)
    seen = {noise.member_seed(9, g, i) for g in range(8) for i in range(64)}
    assert len(seen) == 8 * 64  # no collisions in a small grid


def test_member_offsets_alignment_and_bounds():
    T, D = 1 << 20, 5003
    offs, signs = noise.member_offsets(7, 3, 64, True, T, D)
    assert offs.min() >= 0 and int(offs.max()) + D <= T
    assert (offs % 16 == 0).all()
    # antithetic: pairs share offset, alternate sign
    assert (offs[0::2] == offs[1::2]).all()
    assert (signs[0::2] == 1).all() and (signs[1::2] == -1).all()
    # different generations decorrelate
    offs2, _ = noise.member_offsets(7, 4, 64, True, T, D)
    assert (offs != offs2).any()


def test_philox_reference_vector():
    """Philox4x32-10 known-answer test (counter=0, key=0) from the Random123
    distribution: verifies the CPU oracle is real Philox, which the HIP K1
    kernel is then checked against."""
    c = [np.array([v], dtype=np.uint32) for v in (0, 0, 0, 0)]
    x0, x1, x2, x3 = noise.philox4x32_10(*c, 0, 0)
    assert [hex(int(v[0])) for v in (x0, x1, x2, x3)] == [
        "0x6627e8d5", "0xe169c58d", "0xbc57ac4c", "0x9b00dbd8"]
