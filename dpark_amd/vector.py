"""Engine-eligible RDDs: int64-key / float64-value pair datasets held as
numpy arrays, routed through the HIP engine for the hot-path operators
("reduceByKey on an int64/f64 RDD routes to the engine;
everything else falls back to the M0 CPU path — that fallback IS dpark's
behavioral contract for config 1").

Routing policy (tier rule ③ — no silent fallback ON the engine path):
  - GPU present (/dev/kfd) -> the HIP engine is THE path; a missing or
    unloadable libdprk.so raises EngineUnavailable.
  - no GPU (CPU-only container) -> the generic CPU path serves the same
    API (how the reference behaves everywhere).
"""

from __future__ import annotations

import operator
import os

import numpy as np

from . import engine as _engine
from .rdd import RDD, Split


def gpu_present() -> bool:
    return os.path.exists("/dev/kfd")


def engine_required() -> bool:
    """True iff the engine MUST be used (GPU present).  Raises loudly if
    the native library is unavailable on a GPU machine."""
    if not gpu_present():
        return False
    _engine.load_library()  # raises EngineUnavailable with the cause
    return True


_ADD_FUNCS = None


def _is_add(f) -> bool:
    global _ADD_FUNCS
    if _ADD_FUNCS is None:
        _ADD_FUNCS = {operator.add, operator.iadd}
    return f in _ADD_FUNCS


class VectorPairsRDD(RDD):
    """Pair RDD backed by numpy int64 keys / float64 vals.  Iterating a
    split yields python (k, v) tuples, so every generic RDD operator
    composes; the hot-path operators below route to the engine."""

    def __init__(self, ctx, keys, vals, numSlices=None):
        super().__init__(ctx)
        self.keys = np.ascontiguousarray(keys, dtype=np.int64)
        self.vals = np.ascontiguousarray(vals, dtype=np.float64)
        assert len(self.keys) == len(self.vals)
        numSlices = numSlices or ctx.defaultParallelism
        n = len(self.keys)
        self._bounds = [(n * i) // numSlices for i in range(numSlices + 1)]
        self._splits = [Split(i) for i in range(numSlices)]

    def splits(self):
        return self._splits

    def compute(self, split):
        lo, hi = self._bounds[split.index], self._bounds[split.index + 1]
        return zip(self.keys[lo:hi].tolist(), self.vals[lo:hi].tolist())

    # -- engine-routed hot path ------------------------------------------
    def _engine_run(self, op, other=None):
        """Engine op with retry-once-on-EngineError (VERDICT r01 item 8 /
        dpark task-retry analogue on the engine path [U]): a transient
        EngineError tears the ctx down and retries on a FRESH ctx once;
        a second failure surfaces.  EngineUnavailable (no native lib on
        a GPU box) is NOT retried — that is a hard configuration error."""
        try:
            return self._engine_run_once(op, other)
        except _engine.EngineError:
            return self._engine_run_once(op, other)

    def _engine_run_once(self, op, other=None):
        eng = _engine.Engine(device=0)
        try:
            eng.reserve(max(len(self.keys), 1), with_vals=True)
            eng.put(self.keys, self.vals)
            if op == "reduce":
                eng.reduce_by_key_sum()
                k, v = eng.get()
                return VectorPairsRDD(self.ctx, k, v, len(self._splits))
            if op == "sort":
                eng.sort()
                k, v = eng.get()
                return VectorPairsRDD(self.ctx, k, v, len(self._splits))
            if op == "group":
                uk, starts, counts = eng.group_by_key()
                _, vals = eng.get()
                return uk, starts, counts, vals
            if op == "join":
                eng_b = _engine.Engine(device=0)
                try:
                    eng_b.reserve(max(len(other.keys), 1), with_vals=True)
                    eng_b.put(other.keys, other.vals)
                    jk, jva, jvb = eng.join(eng_b)
                    return jk, jva, jvb
                finally:
                    eng_b.close()
            raise ValueError(op)
        finally:
            eng.close()

    def reduceByKey(self, func, numSplits=None):
        if _is_add(func) and engine_required():
            return self._engine_run("reduce")
        return super().reduceByKey(func, numSplits)

    def sortByKey(self, key=None, reverse=False, numSplits=None):
        if key is None and not reverse and engine_required():
            return self._engine_run("sort")
        return super().sortByKey(key, reverse, numSplits)

    def groupByKey(self, numSplits=None):
        if engine_required():
            uk, starts, counts, vals = self._engine_run("group")
            return GroupedVectorRDD(self.ctx, uk, starts, counts, vals)
        return super().groupByKey(numSplits)

    def join(self, other, numSplits=None):
        if isinstance(other, VectorPairsRDD) and engine_required():
            jk, jva, jvb = self._engine_run("join", other)
            return JoinedVectorRDD(self.ctx, jk, jva, jvb)
        return super().join(other, numSplits)

    def cogroup(self, other, numSplits=None):
        """Engine-level cogroup: k -> ([vs], [ws])
        for the UNION of both key sets (CoGroupedRDD's output shape [U,
        SURVEY.md §3.3]).  Heavy work (sort + run extraction + value
        grouping) on GPU per side; the union-merge of the two SORTED
        unique-key lists (tiny vs the data) is host-side numpy."""
        if not (isinstance(other, VectorPairsRDD) and engine_required()):
            return super().cogroup(other, numSplits)
        ea = _engine.Engine(device=0)
        eb = _engine.Engine(device=0)
        try:
            ea.reserve(max(len(self.keys), 1), with_vals=True)
            ea.put(self.keys, self.vals)
            uka, sta, cta = ea.group_by_key()
            _, va = ea.get()          # values grouped in key order
            eb.reserve(max(len(other.keys), 1), with_vals=True)
            eb.put(other.keys, other.vals)
            ukb, stb, ctb = eb.group_by_key()
            _, vb = eb.get()
        finally:
            ea.close()
            eb.close()
        uk = np.union1d(uka, ukb)     # both inputs sorted: O(g)
        ia = np.searchsorted(uka, uk)
        in_a = (ia < len(uka))
        in_a[in_a] &= uka[ia[in_a]] == uk[in_a]
        ib = np.searchsorted(ukb, uk)
        in_b = (ib < len(ukb))
        in_b[in_b] &= ukb[ib[in_b]] == uk[in_b]
        return CoGroupedVectorRDD(self.ctx, uk,
                                  in_a, ia, sta, cta, va,
                                  in_b, ib, stb, ctb, vb)

    # -- array-level accessors (bulk results without tuple boxing) -------
    def collectArrays(self):
        return self.keys, self.vals

    # -- output/ingest formats ------------------
    def saveAsBinary(self, path):
        """Partitioned binary dump (checkpoint analogue of dpark's
        OutputTextFileRDD [U]): one part-NNNNN.npz per partition + a
        manifest; reload with ctx.binaryPairs(path).  Round-trip is
        bit-exact (int64 keys, float64 vals)."""
        import json as _json
        os.makedirs(path, exist_ok=True)
        for i in range(len(self._splits)):
            lo, hi = self._bounds[i], self._bounds[i + 1]
            np.savez(os.path.join(path, f"part-{i:05d}.npz"),
                     keys=self.keys[lo:hi], vals=self.vals[lo:hi])
        with open(os.path.join(path, "manifest.json"), "w") as f:
            _json.dump({"format": "dprk-pairs-v1", "n": len(self.keys),
                        "partitions": len(self._splits)}, f)
        return path

    def saveAsTextFile(self, path, ext="", overwrite=True):
        """GPU-path text dump: same one-file-per-partition layout and
        str((k, v)) line format as the generic CPU path (rdd.py),
        written array-at-a-time."""
        os.makedirs(path, exist_ok=True)
        out = []
        for i in range(len(self._splits)):
            lo, hi = self._bounds[i], self._bounds[i + 1]
            fname = os.path.join(path, "%05d%s" % (i, ext))
            with open(fname, "w") as f:
                f.writelines(f"{(k, v)}\n" for k, v in
                             zip(self.keys[lo:hi].tolist(),
                                 self.vals[lo:hi].tolist()))
            out.append(fname)
        return out


class GroupedVectorRDD(RDD):
    """groupByKey result in grouped-array layout: iterating yields
    (key, [values]) like the reference; arrays exposed directly too."""

    def __init__(self, ctx, uk, starts, counts, vals):
        super().__init__(ctx)
        self.unique_keys, self.starts, self.counts = uk, starts, counts
        self.grouped_vals = vals
        self._splits = [Split(0)]

    def splits(self):
        return self._splits

    def compute(self, split):
        for i, k in enumerate(self.unique_keys.tolist()):
            s = int(self.starts[i])
            c = int(self.counts[i])
            yield k, self.grouped_vals[s:s + c].tolist()

    def collectArrays(self):
        return (self.unique_keys, self.starts, self.counts,
                self.grouped_vals)


class CoGroupedVectorRDD(RDD):
    """cogroup result: per union key, value lists from each side (empty
    list where the key is absent — CoGroupMerger semantics [U])."""

    def __init__(self, ctx, uk, in_a, ia, sta, cta, va,
                 in_b, ib, stb, ctb, vb):
        super().__init__(ctx)
        self.unique_keys = uk
        self._a = (in_a, ia, sta, cta, va)
        self._b = (in_b, ib, stb, ctb, vb)
        self._splits = [Split(0)]

    def splits(self):
        return self._splits

    def _side(self, side, i):
        present, idx, starts, counts, vals = side
        if not present[i]:
            return []
        j = int(idx[i])
        s, c = int(starts[j]), int(counts[j])
        return vals[s:s + c].tolist()

    def compute(self, split):
        for i, k in enumerate(self.unique_keys.tolist()):
            yield k, (self._side(self._a, i), self._side(self._b, i))

    def collectArrays(self):
        """(uk, counts_a, grouped_vals_a_or_empty, counts_b, ...) in the
        flattened per-union-key layout."""
        (in_a, ia, sta, cta, va) = self._a
        (in_b, ib, stb, ctb, vb) = self._b
        ca = np.zeros(len(self.unique_keys), np.int64)
        cb = np.zeros(len(self.unique_keys), np.int64)
        if len(cta):
            ca[in_a] = cta.astype(np.int64)[ia[in_a]]
        if len(ctb):
            cb[in_b] = ctb.astype(np.int64)[ib[in_b]]
        return self.unique_keys, ca, cb


class JoinedVectorRDD(RDD):
    def __init__(self, ctx, k, va, vb):
        super().__init__(ctx)
        self.join_keys, self.va, self.vb = k, va, vb
        self._splits = [Split(0)]

    def splits(self):
        return self._splits

    def compute(self, split):
        for k, a, b in zip(self.join_keys.tolist(), self.va.tolist(),
                           self.vb.tolist()):
            yield k, (a, b)

    def collectArrays(self):
        return self.join_keys, self.va, self.vb
