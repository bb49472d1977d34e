"""Shared noise table: counter-based Philox4x32-10 N(0,1) fill + seed→offset map.

Design (BASELINE north star): one large table of standard
normals lives in HBM; member i's perturbation is the contiguous slice at a
seed-derived offset; the antithetic partner negates the same slice.  The table
is REGENERATED locally on every rank from a broadcast 64-bit seed (kernel K1),
bit-identically, so it never crosses the wire.

The CPU implementation here is the *reference oracle* for the HIP kernel
(csrc/noise.hip) and the CPU execution path for tests.  Both compute:

    element e:  block b = e >> 2, lane l = e & 3
    (x0..x3) = philox4x32_10(counter=(b, 0), key=(seed_lo, seed_hi))
    u_i = (x_i + 0.5) * 2^-32                       # in (0,1)
    z0 = sqrt(-2 ln u0) cos(2 pi u1)   z1 = sqrt(-2 ln u0) sin(2 pi u1)
    z2 = sqrt(-2 ln u2) cos(2 pi u3)   z3 = sqrt(-2 ln u2) sin(2 pi u3)
    table[e] = z_l

Cross-rank bit-determinism holds because every rank runs the same kernel with
the same seed; CPU-vs-GPU agreement is to float tolerance (libm ulp diffs).
"""
from __future__ import annotations

import numpy as np
import torch

_PHILOX_M0 = np.uint64(0xD2511F53)
_PHILOX_M1 = np.uint64(0xCD9E8D57)
_PHILOX_W0 = np.uint32(0x9E3779B9)
_PHILOX_W1 = np.uint32(0xBB67AE85)
_MASK32 = np.uint64(0xFFFFFFFF)


def philox4x32_10(c0, c1, c2, c3, key0: int, key1: int):
    """Vectorized Philox4x32-10. c0..c3: uint32 ndarrays. Returns 4 uint32 arrays."""
    c0 = c0.astype(np.uint32); c1 = c1.astype(np.uint32)
    c2 = c2.astype(np.uint32); c3 = c3.astype(np.uint32)
    k0 = np.uint32(key0 & 0xFFFFFFFF)
    k1 = np.uint32(key1 & 0xFFFFFFFF)
    for _ in range(10):
        p0 = c0.astype(np.uint64) * _PHILOX_M0
        p1 = c2.astype(np.uint64) * _PHILOX_M1
        hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
        lo0 = (p0 & _MASK32).astype(np.uint32)
        hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
        lo1 = (p1 & _MASK32).astype(np.uint32)
        c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
        k0 = np.uint32((int(k0) + int(_PHILOX_W0)) & 0xFFFFFFFF)
        k1 = np.uint32((int(k1) + int(_PHILOX_W1)) & 0xFFFFFFFF)
    return c0, c1, c2, c3


def _normals_from_blocks(blocks: np.ndarray, seed: int) -> np.ndarray:
    """blocks: uint64 array of Philox block indices -> (n,4) float32 normals."""
    n = blocks.shape[0]
    c0 = (blocks & 0xFFFFFFFF).astype(np.uint32)
    c1 = (blocks >> np.uint64(32)).astype(np.uint32)
    z = np.zeros(n, dtype=np.uint32)
    x0, x1, x2, x3 = philox4x32_10(c0, c1, z, z, seed & 0xFFFFFFFF, (seed >> 32) & 0xFFFFFFFF)
    inv = np.float32(2.0 ** -32)
    u0 = (x0.astype(np.float32) + np.float32(0.5)) * inv
    u1 = (x1.astype(np.float32) + np.float32(0.5)) * inv
    u2 = (x2.astype(np.float32) + np.float32(0.5)) * inv
    u3 = (x3.astype(np.float32) + np.float32(0.5)) * inv
    r0 = np.sqrt(np.float32(-2.0) * np.log(u0), dtype=np.float32)
    r1 = np.sqrt(np.float32(-2.0) * np.log(u2), dtype=np.float32)
    t0 = np.float32(2.0 * np.pi) * u1
    t1 = np.float32(2.0 * np.pi) * u3
    out = np.empty((n, 4), dtype=np.float32)
    out[:, 0] = r0 * np.cos(t0)
    out[:, 1] = r0 * np.sin(t0)
    out[:, 2] = r1 * np.cos(t1)
    out[:, 3] = r1 * np.sin(t1)
    return out


def fill_noise_table_cpu(size: int, seed: int) -> torch.Tensor:
    """Reference noise-table fill (oracle for the K1 HIP kernel)."""
    nblocks = (size + 3) // 4
    blocks = np.arange(nblocks, dtype=np.uint64)
    flat = _normals_from_blocks(blocks, seed).reshape(-1)[:size]
    return torch.from_numpy(flat.copy())


def noise_slice_cpu(offsets, length: int, seed: int) -> np.ndarray:
    """Regenerate table[off:off+length] per offset WITHOUT materializing the
    table (used by huge-table CPU tests).  offsets: (m,) int64."""
    offsets = np.asarray(offsets, dtype=np.int64)
    m = offsets.shape[0]
    out = np.empty((m, length), dtype=np.float32)
    for i, off in enumerate(offsets):
        b0 = off >> 2
        b1 = (off + length - 1) >> 2
        blocks = np.arange(b0, b1 + 1, dtype=np.uint64)
        vals = _normals_from_blocks(blocks, seed).reshape(-1)
        start = off - (b0 << 2)
        out[i] = vals[start:start + length]
    return out


# ---------------------------------------------------------------------------
# seed / offset derivation (splitmix64) — identical in csrc/common.h
# ---------------------------------------------------------------------------

_U64 = 0xFFFFFFFFFFFFFFFF


def splitmix64(x: int) -> int:
    x = (x + 0x9E3779B97F4A7C15) & _U64
    z = x
    z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _U64
    z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _U64
    return (z ^ (z >> 31)) & _U64


def member_seed(run_seed: int, generation: int, pair_index: int) -> int:
    """64-bit seed for antithetic pair `pair_index` of `generation`.
    Both members of a pair share the seed (mirrored sampling); with
    antithetic=False, pair_index == member index."""
    a = splitmix64((run_seed ^ (generation * 0x9E3779B97F4A7C15)) & _U64)
    return splitmix64((a ^ (pair_index * 0xBF58476D1CE4E5B9)) & _U64)


def member_offset(run_seed: int, generation: int, pair_index: int,
                  table_size: int, dim: int) -> int:
    """Offset of the pair's noise slice in the table. 16-aligned so GPU
    vectorized (float4) loads of slices stay aligned."""
    span = table_size - dim
    assert span > 0, "noise table smaller than parameter dim"
    s = member_seed(run_seed, generation, pair_index)
    return int((s % ((span // 16) + 1)) * 16) if span >= 16 else int(s % span)


def _splitmix64_np(x: np.ndarray) -> np.ndarray:
    """Vectorized splitmix64 (uint64 wraparound == the scalar version)."""
    with np.errstate(over="ignore"):
        x = (x + np.uint64(0x9E3779B97F4A7C15))
        z = x
        z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        return z ^ (z >> np.uint64(31))


def member_offsets(run_seed: int, generation: int, pop: int, antithetic: bool,
                   table_size: int, dim: int) -> tuple:
    """(offsets[pop] int64, signs[pop] float32) for the whole global population."""
    with np.errstate(over="ignore"):
        pairs = (np.arange(pop, dtype=np.uint64) // np.uint64(2)
                 if antithetic else np.arange(pop, dtype=np.uint64))
        base = np.uint64(splitmix64((run_seed ^ (generation * 0x9E3779B97F4A7C15)) & _U64))
        seeds = _splitmix64_np(np.full(pop, base, dtype=np.uint64)
                               ^ (pairs * np.uint64(0xBF58476D1CE4E5B9)))
    span = table_size - dim
    assert span > 0, "noise table smaller than parameter dim"
    if span >= 16:
        offs = ((seeds % np.uint64(span // 16 + 1)) * np.uint64(16)).astype(np.int64)
    else:
        offs = (seeds % np.uint64(span)).astype(np.int64)
    signs = np.ones(pop, dtype=np.float32)
    if antithetic:
        signs[1::2] = -1.0
    return offs, signs
