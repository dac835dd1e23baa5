"""Philox4x32-10 counter-based RNG — numpy restatement (oracle side).

This is the synthetic-input protocol defined in /root/repo/BASELINE.md
("philox-seeded, seed = 1234 + rank", mirroring the reference's
per-worker seeding `Random.seed!(1234+myid())`,
/root/reference/test/runtests.jl:23).  The algorithm is the published
Philox4x32-10 of Salmon et al. (Random123 1.09), pinned by its published
known-answer vectors (see tests/test_oracle.py::test_philox_kat).

Element mapping (the contract shared bit-exactly with the HIP kernels in
distributedarrays_jl_amd/csrc/kernels_rand.hip and the C baseline in
oracle/cpu_baseline.c):

  key = (seed & 0xffffffff, seed >> 32), counter block b:
  ctr = (b & 0xffffffff, b >> 32, 0, 0), out = philox4x32_10(ctr, key).

  f64 uniform, element i:  b = i >> 1, j = i & 1
      u53 = ((out[2j+1] << 32 | out[2j]) >> 11);  x = u53 * 2^-53
  f32 uniform, element i:  b = i >> 2, j = i & 3
      x = (out[j] >> 8) * 2^-24
  i64, element i:          b = i >> 1, j = i & 1
      x = int64(out[2j+1] << 32 | out[2j])
  f64 normal, elements 2b and 2b+1 (Box-Muller on block b):
      u1 from (out[0],out[1]) as the f64-uniform word, u2 from
      (out[2],out[3]);  r = sqrt(-2 ln(1-u1));  t = 2*pi*u2
      z_{2b} = r*cos(t), z_{2b+1} = r*sin(t)
  f32 normal: same with u1=(out[0]>>8)*2^-24, u2=(out[1]>>8)*2^-24.

Element index i is the LOCAL column-major linear index within the chunk;
each rank uses seed = 1234 + rank.
"""
import numpy as np

M0 = np.uint64(0xD2511F53)
M1 = np.uint64(0xCD9E8D57)
W0 = np.uint32(0x9E3779B9)
W1 = np.uint32(0xBB67AE85)
MASK32 = np.uint64(0xFFFFFFFF)


def philox4x32(c0, c1, c2, c3, k0, k1, rounds=10):
    """Vectorized Philox4x32; inputs uint32 arrays, returns 4 uint32 arrays."""
    c0 = np.asarray(c0, np.uint32); c1 = np.asarray(c1, np.uint32)
    c2 = np.asarray(c2, np.uint32); c3 = np.asarray(c3, np.uint32)
    k0 = np.uint32(k0); k1 = np.uint32(k1)
    for _ in range(rounds):
        p0 = M0 * c0.astype(np.uint64)
        p1 = M1 * c2.astype(np.uint64)
        hi0 = (p0 >> np.uint64(32)).astype(np.uint32)
        lo0 = (p0 & MASK32).astype(np.uint32)
        hi1 = (p1 >> np.uint64(32)).astype(np.uint32)
        lo1 = (p1 & MASK32).astype(np.uint32)
        c0, c1, c2, c3 = hi1 ^ c1 ^ k0, lo1, hi0 ^ c3 ^ k1, lo0
        k0 = np.uint32((int(k0) + int(W0)) & 0xFFFFFFFF)
        k1 = np.uint32((int(k1) + int(W1)) & 0xFFFFFFFF)
    return c0, c1, c2, c3


def _key(seed):
    seed = np.uint64(seed)
    return np.uint32(seed & MASK32), np.uint32(seed >> np.uint64(32))


def _blocks(b):
    b = np.asarray(b, np.uint64)
    return (b & MASK32).astype(np.uint32), (b >> np.uint64(32)).astype(np.uint32)


def _u64(lo, hi):
    return lo.astype(np.uint64) | (hi.astype(np.uint64) << np.uint64(32))


def fill_uniform_f64(n, seed, offset=0):
    """n uniform [0,1) doubles for elements offset..offset+n-1."""
    k0, k1 = _key(seed)
    i = np.arange(offset, offset + n, dtype=np.uint64)
    b0, b1 = _blocks(i >> np.uint64(1))
    z = np.zeros_like(b0)
    o0, o1, o2, o3 = philox4x32(b0, b1, z, z, k0, k1)
    lane = (i & np.uint64(1)).astype(np.intp)
    lo = np.where(lane == 0, o0, o2)
    hi = np.where(lane == 0, o1, o3)
    u53 = _u64(lo, hi) >> np.uint64(11)
    return u53.astype(np.float64) * (2.0 ** -53)


def fill_uniform_f32(n, seed, offset=0):
    k0, k1 = _key(seed)
    i = np.arange(offset, offset + n, dtype=np.uint64)
    b0, b1 = _blocks(i >> np.uint64(2))
    z = np.zeros_like(b0)
    outs = philox4x32(b0, b1, z, z, k0, k1)
    lane = (i & np.uint64(3)).astype(np.intp)
    o = np.select([lane == 0, lane == 1, lane == 2], outs[:3], outs[3])
    return ((o >> np.uint32(8)).astype(np.float32) * np.float32(2.0 ** -24))


def fill_int64(n, seed, offset=0):
    k0, k1 = _key(seed)
    i = np.arange(offset, offset + n, dtype=np.uint64)
    b0, b1 = _blocks(i >> np.uint64(1))
    z = np.zeros_like(b0)
    o0, o1, o2, o3 = philox4x32(b0, b1, z, z, k0, k1)
    lane = (i & np.uint64(1)).astype(np.intp)
    lo = np.where(lane == 0, o0, o2)
    hi = np.where(lane == 0, o1, o3)
    return np.ascontiguousarray(_u64(lo, hi)).view(np.int64)


def fill_normal_f64(n, seed, offset=0):
    """Box-Muller pairs; element i comes from block i>>1."""
    k0, k1 = _key(seed)
    i = np.arange(offset, offset + n, dtype=np.uint64)
    b0, b1 = _blocks(i >> np.uint64(1))
    z = np.zeros_like(b0)
    o0, o1, o2, o3 = philox4x32(b0, b1, z, z, k0, k1)
    u1 = (_u64(o0, o1) >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    u2 = (_u64(o2, o3) >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    r = np.sqrt(-2.0 * np.log1p(-u1))
    t = 2.0 * np.pi * u2
    lane = (i & np.uint64(1)).astype(np.intp)
    return np.where(lane == 0, r * np.cos(t), r * np.sin(t))


def fill_normal_f32(n, seed, offset=0):
    k0, k1 = _key(seed)
    i = np.arange(offset, offset + n, dtype=np.uint64)
    b0, b1 = _blocks(i >> np.uint64(1))
    z = np.zeros_like(b0)
    o0, o1, o2, o3 = philox4x32(b0, b1, z, z, k0, k1)
    u1 = (o0 >> np.uint32(8)).astype(np.float32) * np.float32(2.0 ** -24)
    u2 = (o1 >> np.uint32(8)).astype(np.float32) * np.float32(2.0 ** -24)
    r = np.sqrt(np.float32(-2.0) * np.log1p(-u1).astype(np.float32))
    t = np.float32(2.0 * np.pi) * u2
    lane = (i & np.uint64(1)).astype(np.intp)
    return np.where(lane == 0, r * np.cos(t), r * np.sin(t)).astype(np.float32)
