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


# ---------------------------------------------------------------------------
# bytes-key hash — the string-key engine path's key function (device
# implementation: k_hash_bytes in dpark_amd/csrc/dprk.hip; this numpy
# mirror is normative and byte-identical, asserted by tests/test_gpu.py).
#
# Spec: word -> 8-byte little-endian chunks (last zero-padded);
#   h = splitmix64(C1 ^ len); for each chunk c: h = splitmix64(h ^ c).
# Length in the seed keeps "a" and "a\0" distinct.
# ---------------------------------------------------------------------------

def hash_bytes_words(buf, offs):
    """Vectorized hash of n words packed in `buf` (uint8 array) at byte
    offsets `offs` (uint64, n+1 entries).  Returns uint64[n]."""
    buf = np.asarray(buf, dtype=np.uint8)
    offs = np.asarray(offs, dtype=np.uint64)
    n = len(offs) - 1
    if n <= 0:
        return np.empty(0, dtype=np.uint64)
    starts = offs[:-1]
    lens = offs[1:] - starts
    old = np.seterr(over="ignore")
    try:
        h = splitmix64(_C1 ^ lens)
        max_chunks = int((int(lens.max()) + 7) // 8) if len(lens) else 0
        # pad the buffer so out-of-range gathers read zeros
        padded = np.concatenate([buf, np.zeros(8, dtype=np.uint8)])
        shifts = (np.uint64(8) * np.arange(8, dtype=np.uint64))
        for j in range(max_chunks):
            base = starts + np.uint64(8 * j)
            # gather 8 bytes; mask bytes past each word's end to zero
            idx = base[:, None] + np.arange(8, dtype=np.uint64)[None, :]
            valid = idx < offs[1:, None]
            bytes8 = np.where(valid,
                              padded[np.minimum(idx, np.uint64(len(buf)))], 0)
            c = (bytes8.astype(np.uint64) << shifts[None, :]).sum(
                axis=1, dtype=np.uint64)
            active = lens > np.uint64(8 * j)
            h = np.where(active, splitmix64(h ^ c), h)
        return h
    finally:
        np.seterr(**old)


def hash_bytes(word: bytes) -> int:
    """Scalar reference of the same spec (slow; tests/doc)."""
    buf = np.frombuffer(word, dtype=np.uint8)
    offs = np.array([0, len(word)], dtype=np.uint64)
    return int(hash_bytes_words(buf, offs)[0])


def pack_words(words):
    """Encode a sequence of bytes/str words into (buf uint8, offs uint64)
    for dprk_put_bytes_keys / hash_bytes_words."""
    enc = [w.encode() if isinstance(w, str) else bytes(w) for w in words]
    lens = np.fromiter((len(w) for w in enc), dtype=np.uint64, count=len(enc))
    offs = np.zeros(len(enc) + 1, dtype=np.uint64)
    np.cumsum(lens, out=offs[1:])
    buf = np.frombuffer(b"".join(enc), dtype=np.uint8)
    return buf, offs
