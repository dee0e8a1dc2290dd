"""The engine's normative hash spec, host-side mirror.

splitmix64 is the GPU HashPartitioner's key mixer (device implementation:
d_splitmix64 in dpark_amd/csrc/dprk.hip).  This numpy mirror is PRODUCT
code: the multi-rank plan (distshuffle.rank_of_keys) uses it to predict
the engine's rank digits on the host.  oracle/synth.py carries its own
independent copy by design (the oracle must not share code with the
product path — tier rule: tests-only infrastructure).

Replaces dpark's portable_hash.pyx role for the engine path
[U, SURVEY.md §8(a2)]: partition LAYOUT may differ from the reference
(collected results are partition-independent; asserted by the oracle's
property tests).
"""

from __future__ import annotations

import numpy as np

_C1 = np.uint64(0x9E3779B97F4A7C15)
_C2 = np.uint64(0xBF58476D1CE4E5B9)
_C3 = np.uint64(0x94D049BB133111EB)


def splitmix64(x):
    """Vectorized splitmix64 finalizer; x: uint64 ndarray (wraparound)."""
    old = np.seterr(over="ignore")
    try:
        z = (np.asarray(x, dtype=np.uint64) + _C1).astype(np.uint64)
        z = (z ^ (z >> np.uint64(30))) * _C2
        z = (z ^ (z >> np.uint64(27))) * _C3
        return z ^ (z >> np.uint64(31))
    finally:
        np.seterr(**old)
