"""String-key engine path: bytes/str-keyed pair
RDDs routed through the HIP engine via device hashing.

Replaces the CPU dict path of `TextFileRDD -> flatMap -> map ->
reduceByKey` wordcount [U, SURVEY.md §3.1] for string keys: words are
packed host-side into one bytes buffer + offsets, hashed ON DEVICE
(k_hash_bytes; spec mirrored in hashspec.hash_bytes_words), and the
64-bit hashes flow through the normal sort/segmented-reduce pipeline.
The host resolves hash -> word via EXEMPLAR rows: a second engine
pipeline groups (hash, row_index) and takes one row per unique hash, so
no host-side dict over all n words is ever built.

Hash-as-identity: two distinct words colliding in 64 bits would merge.
P(collision) ~ K^2 / 2^65 for K distinct words (~3e-6 at K = 2^24).
DPRK_VERIFY_STRING_HASH=1 verifies exactly against the numpy mirror
(used by the tests); the bound and the switch are documented here and in
DESIGN.md.
"""

from __future__ import annotations

import os

import numpy as np

from . import engine as _engine
from .hashspec import hash_bytes_words, pack_words
from .rdd import RDD, Split
from .vector import engine_required, _is_add


class StringHashCollision(RuntimeError):
    pass


def _verify_no_collisions(buf, offs, words):
    """Exact collision check via the numpy mirror (opt-in, O(n))."""
    h = hash_bytes_words(buf, offs)
    uniq_words = len({bytes(w) if not isinstance(w, (str,)) else w
                      for w in words})
    uniq_hashes = len(np.unique(h))
    if uniq_words != uniq_hashes:
        raise StringHashCollision(
            f"64-bit hash collision: {uniq_words} distinct words -> "
            f"{uniq_hashes} distinct hashes")


def engine_string_reduce(words, vals, device=0):
    """reduceByKey(+) on (bytes-key, f64) pairs through the HIP engine.
    Returns (unique_words list, sums ndarray), ordered by key hash."""
    buf, offs = pack_words(words)
    n = len(offs) - 1
    if os.environ.get("DPRK_VERIFY_STRING_HASH"):
        _verify_no_collisions(buf, offs, words)
    vals = np.ascontiguousarray(vals, dtype=np.float64)
    # pipeline A: (hash, value) -> segmented reduce -> (unique hash, sum)
    ea = _engine.Engine(device=device)
    eb = _engine.Engine(device=device)
    try:
        ea.reserve(max(n, 1), with_vals=True)
        ea.put_bytes_keys(buf, offs, vals)
        ea.reduce_by_key_sum()
        uh, sums = ea.get()
        # pipeline B: (hash, row index) -> group -> exemplar row per hash
        eb.reserve(max(n, 1), with_vals=True)
        eb.put_bytes_keys(buf, offs, np.arange(n, dtype=np.float64))
        uk, starts, counts = eb.group_by_key()
        _, sorted_rows = eb.get()
        assert np.array_equal(uh, uk), "reduce/group key sets diverged"
        exemplar = sorted_rows[starts.astype(np.int64)].astype(np.int64)
    finally:
        ea.close()
        eb.close()
    out_words = [words[i] for i in exemplar.tolist()]
    return out_words, sums


class StringPairsRDD(RDD):
    """Pair RDD with bytes/str keys and f64 values.  Iterating yields
    (word, value) tuples so every generic operator composes; reduceByKey
    with + routes to the engine's string-key path on a GPU box."""

    def __init__(self, ctx, words, vals, numSlices=None):
        super().__init__(ctx)
        self.words = list(words)
        self.vals = np.ascontiguousarray(vals, dtype=np.float64)
        assert len(self.words) == len(self.vals)
        numSlices = numSlices or ctx.defaultParallelism
        n = len(self.words)
        self._bounds = [(n * i) // numSlices for i in range(numSlices + 1)]
        self._splits = [Split(i) for i in range(numSlices)]

    def splits(self):
        return self._splits

    def compute(self, split):
        lo, hi = self._bounds[split.index], self._bounds[split.index + 1]
        return zip(self.words[lo:hi], self.vals[lo:hi].tolist())

    def reduceByKey(self, func, numSplits=None):
        if _is_add(func) and engine_required():
            out_words, sums = engine_string_reduce(self.words, self.vals)
            return StringPairsRDD(self.ctx, out_words, sums,
                                  len(self._splits))
        return super().reduceByKey(func, numSplits)

    def collectArrays(self):
        return self.words, self.vals

    def collectAsMap(self):
        return dict(zip(self.words, self.vals.tolist()))
