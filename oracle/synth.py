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


def gen_pairs(seed: int, n: int, key_universe: int):
    """CPU mirror of dprk_gen_pairs(dist=0 uniform). Returns (keys int64,
    vals float64)."""
    i = np.arange(n, dtype=np.uint64)
    x = (np.uint64(seed) << np.uint64(32)) + i
    ks = splitmix64(x)
    if key_universe != 0 and key_universe < (1 << 64):
        ks = ks & np.uint64(key_universe - 1)
    keys = ks.view(np.int64) if ks.dtype == np.uint64 else ks
    vb = splitmix64(x ^ _VMIX)
    vals = (vb >> np.uint64(11)).astype(np.float64) * (2.0 ** -53)
    return keys.astype(np.int64), vals
