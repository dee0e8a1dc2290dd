"""Synthetic input spec shared by the engine and the tests. TEST INFRA ONLY
as code, but the SPEC here is normative: dprk_gen_pairs (device PRNG, see
dpark_amd/csrc/dprk_kernels.hip) must produce byte-identical keys/values to
these numpy mirrors — tests/test_gpu.py checks exactly that.

Spec:
  x_i        = (seed << 32) + i                (uint64 wraparound)
  key_i      = splitmix64(x_i) & (universe-1)  (universe a power of two)
               ... cast to int64 (universe 2**64 exercises negatives)
  valbits_i  = splitmix64(x_i ^ 0xDEADBEEFCAFEBABE)
  value_i    = (valbits_i >> 11) * 2**-53      (float64 uniform [0,1))
"""

from __future__ import annotations

import numpy as np

_C1 = np.uint64(0x9E3779B97F4A7C15)
_C2 = np.uint64(0xBF58476D1CE4E5B9)
_C3 = np.uint64(0x94D049BB133111EB)
_VMIX = np.uint64(0xDEADBEEFCAFEBABE)


def splitmix64(x):
    """Vectorized splitmix64 finalizer; x: uint64 ndarray."""
    old = np.seterr(over="ignore")
    try:
        z = (x + _C1).astype(np.uint64)
        z = (z ^ (z >> np.uint64(30))) * _C2
        z = (z ^ (z >> np.uint64(27))) * _C3
        return z ^ (z >> np.uint64(31))
    finally:
        np.seterr(**old)


def gen_pairs(seed: int, n: int, key_universe: int, start: int = 0):
    """CPU mirror of dprk_gen_pairs_at(dist=0 uniform): x_i = (seed<<32)
    + start + i (wraparound).  Returns (keys int64, vals float64)."""
    old = np.seterr(over="ignore")
    try:
        i = np.arange(n, dtype=np.uint64)
        x = (np.uint64(seed) << np.uint64(32)) + np.uint64(start) + i
    finally:
        np.seterr(**old)
    ks = splitmix64(x)
    if key_universe != 0 and key_universe < (1 << 64):
        ks = ks & np.uint64(key_universe - 1)
    keys = ks.view(np.int64) if ks.dtype == np.uint64 else ks
    vb = splitmix64(x ^ _VMIX)
    vals = (vb >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return keys.astype(np.int64), vals


# ---------------------------------------------------------------------------
# discrete-distribution (Zipf) generation via alias table — the table
# built HERE is normative: the engine samples the exact same uint64 table
# (dprk_set_alias_table), so device output is bit-identical to gen_zipf.
# ---------------------------------------------------------------------------

_A1 = np.uint64(0x5851F42D4C957F2D)
_A2 = np.uint64(0x14057B7EF767814F)


def build_alias_table(weights):
    """Vose alias method -> packed uint64 table[slot] = (thresh<<32)|alias.
    thresh = round(prob_scaled * 2^32), clamped to 2^32-1."""
    w = np.asarray(weights, dtype=np.float64)
    n = len(w)
    p = w / w.sum() * n          # scaled probabilities, mean 1
    alias = np.zeros(n, dtype=np.uint64)
    prob = np.ones(n, dtype=np.float64)
    # preallocated stacks (a python-list Vose loop costs ~1 us/entry; this
    # runs once per bench over a 2^24-entry table)
    idx = np.arange(n, dtype=np.int64)
    is_small = p < 1.0
    small = idx[is_small].tolist()
    large = idx[~is_small].tolist()
    p = p.copy()
    while small and large:
        s = small.pop()
        l = large.pop()
        prob[s] = p[s]
        alias[s] = l
        p[l] = (p[l] + p[s]) - 1.0
        (small if p[l] < 1.0 else large).append(l)
    for i in small + large:
        prob[i] = 1.0
    thresh = np.minimum(np.round(prob * 4294967296.0),
                        4294967295.0).astype(np.uint64)
    return (thresh << np.uint64(32)) | alias


def zipf_weights(universe, alpha=1.1):
    return 1.0 / np.power(np.arange(1, universe + 1, dtype=np.float64), alpha)


def gen_zipf(seed, n, table, universe):
    """CPU mirror of dprk k_gen_alias (dist=1)."""
    i = np.arange(n, dtype=np.uint64)
    x = (np.uint64(seed) << np.uint64(32)) + i
    u1 = splitmix64(x ^ _A1)
    # __umul64hi(u1, universe), exact for universe < 2^32:
    # floor(u1*U/2^64) = (hi(u1)*U + floor(lo(u1)*U/2^32)) >> 32
    assert universe < (1 << 32)
    U = np.uint64(universe)
    old = np.seterr(over="ignore")
    try:
        hi = u1 >> np.uint64(32)
        lo = u1 & np.uint64(0xFFFFFFFF)
        slot = (hi * U + ((lo * U) >> np.uint64(32))) >> np.uint64(32)
    finally:
        np.seterr(**old)
    e = table[slot.astype(np.int64)]
    u2 = splitmix64(x ^ _A2)
    take = (u2 >> np.uint64(32)) < (e >> np.uint64(32))
    keys = np.where(take, slot, e & np.uint64(0xFFFFFFFF)).astype(np.int64)
    vb = splitmix64(x ^ _VMIX)
    vals = (vb >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return keys, vals
