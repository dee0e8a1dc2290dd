"""dpark-semantics CPU restatement (the parity oracle). TEST INFRA ONLY.

Restates, in plain Python dict/sort loops, the algorithm of dpark's hot
path as inventoried in SURVEY.md §8(a).  Each function cites the reference
symbol it follows; ALL citations are [U] (symbol-level, unverified — the
reference mount is empty; see oracle/__init__.py for the
full parity-unpinned statement).

Algorithm shape being restated:
  map side   [dpark/task.py :: ShuffleMapTask.run [U]]:
      for (k, v) in partition:
          b = partitioner.getPartition(k)          # dependency.py [U]
          if k in buckets[b]: buckets[b][k] = mergeValue(buckets[b][k], v)
          else:               buckets[b][k] = createCombiner(v)
  reduce side [dpark/shuffle.py :: Merger.merge [U]]:
      for each map output's bucket b:
          for (k, c): d[k] = mergeCombiners(d[k], c) if k in d else c

The collected result of combineByKey is independent of the partitioner and
of the order maps are merged whenever mergeValue/mergeCombiners are
commutative+associative — this is the contract that
makes GPU results comparable despite a different partition layout
(§8(a2) design freedom).  For float64 the oracle fixes a sequential
left-to-right order; the engine promises <=1e-6 relative, not bit-equal.
"""

from __future__ import annotations

import numpy as np


# ---------------------------------------------------------------------------
# hashing / partitioning
# ---------------------------------------------------------------------------

_MERSENNE61 = (1 << 61) - 1


def portable_hash_int(k: int) -> int:
    """Restatement of dpark/portable_hash.pyx :: portable_hash for ints [U].

    dpark's portable_hash reproduces CPython's hash() for int (seed
    independent): hash(n) == n reduced mod 2**61-1, with -1 mapped to -2.
    Used only to SIMULATE a dpark-like partition layout in tests; the
    engine's layout intentionally differs and collected
    results are compared layout-independently.
    """
    h = k % _MERSENNE61 if k >= 0 else -((-k) % _MERSENNE61)
    if h == -1:
        h = -2
    return h


def hash_partition(k: int, n: int) -> int:
    """dpark/dependency.py :: HashPartitioner.getPartition [U]:
    portable_hash(key) % numPartitions (Python modulo: result in [0, n))."""
    return portable_hash_int(k) % n


# ---------------------------------------------------------------------------
# combineByKey family
# ---------------------------------------------------------------------------

def combine_by_key(pairs, create_combiner, merge_value, merge_combiners,
                   n_partitions=4, n_maps=3):
    """dpark/rdd.py :: RDD.combineByKey via ShuffledRDD [U].

    Simulates the full map-side-combine -> shuffle -> reduce-side-merge
    pipeline with `n_maps` map splits and `n_partitions` reduce buckets.
    Returns {key: combiner} over all reduce buckets.
    """
    pairs = list(pairs)
    # split input into map partitions the way parallelize slices do [U]
    maps = [pairs[i::n_maps] for i in range(n_maps)]
    # map side: per-(map, bucket) dict combine [task.py::ShuffleMapTask.run [U]]
    bucket_outputs = []  # list over maps of {bucket: {k: combiner}}
    for mp in maps:
        buckets = {}
        for k, v in mp:
            b = hash_partition(k, n_partitions)
            d = buckets.setdefault(b, {})
            if k in d:
                d[k] = merge_value(d[k], v)
            else:
                d[k] = create_combiner(v)
        bucket_outputs.append(buckets)
    # reduce side: Merger.merge per reduce bucket [shuffle.py [U]]
    out = {}
    for b in range(n_partitions):
        d = {}
        for mo in bucket_outputs:
            for k, c in mo.get(b, {}).items():
                if k in d:
                    d[k] = merge_combiners(d[k], c)
                else:
                    d[k] = c
        out.update(d)
    return out


def reduce_by_key_sum(keys, vals, n_partitions=4, n_maps=3):
    """dpark/rdd.py :: RDD.reduceByKey(add) [U] == combineByKey(identity,
    add, add).  Takes int64 keys / float64 vals arrays, returns
    (unique_keys_sorted, sums) as numpy arrays.

    Output is returned key-sorted because the comparison is
    layout-independent (dict order is an implementation detail of the
    partitioner, which the contract says must not matter)."""
    keys = np.asarray(keys, dtype=np.int64)
    vals = np.asarray(vals, dtype=np.float64)
    d = combine_by_key(zip(keys.tolist(), vals.tolist()),
                       lambda v: v, lambda a, b: a + b, lambda a, b: a + b,
                       n_partitions=n_partitions, n_maps=n_maps)
    ks = np.array(sorted(d.keys()), dtype=np.int64)
    vs = np.array([d[int(k)] for k in ks], dtype=np.float64)
    return ks, vs


def group_by_key(pairs, n_partitions=4):
    """dpark/rdd.py :: RDD.groupByKey [U] == combineByKey(list-of, append,
    extend).  Returns {key: [values]} — value order within a group follows
    map order per partition; dpark does not promise a total order across
    maps, so parity tests compare per-key value MULTISETS."""
    return combine_by_key(pairs,
                          lambda v: [v],
                          lambda c, v: c + [v],
                          lambda a, b: a + b,
                          n_partitions=n_partitions)


# ---------------------------------------------------------------------------
# sortByKey
# ---------------------------------------------------------------------------

def sort_by_key(keys, vals=None, reverse=False):
    """dpark/rdd.py :: RDD.sortByKey [U]: sample -> RangePartitioner ->
    per-partition Timsort; global total order on concatenation; duplicate-
    key value order unspecified.

    Oracle contract: returns keys in total order; when vals given, returns
    a (keys, vals) pair where vals accompany their keys via a STABLE sort
    (one valid linearization — parity tests must compare duplicate-key
    value groups as multisets, not positionally)."""
    keys = np.asarray(keys, dtype=np.int64)
    if vals is None:
        out = np.sort(keys, kind="stable")
        return out[::-1].copy() if reverse else out
    vals = np.asarray(vals)
    order = np.argsort(keys, kind="stable")
    if reverse:
        order = order[::-1]
    return keys[order], vals[order]


# ---------------------------------------------------------------------------
# join / cogroup
# ---------------------------------------------------------------------------

def cogroup(pairs_a, pairs_b):
    """dpark/rdd.py :: CoGroupedRDD + dpark/shuffle.py :: CoGroupMerger [U]:
    key -> ([values from a], [values from b]); keys present in either side."""
    d = {}
    for k, v in pairs_a:
        d.setdefault(k, ([], []))[0].append(v)
    for k, w in pairs_b:
        d.setdefault(k, ([], []))[1].append(w)
    return d


def join(pairs_a, pairs_b):
    """dpark/rdd.py :: RDD.join [U] = cogroup + per-key cross product
    (SURVEY.md §3.3).  Returns list of (k, (v, w)); parity compares the
    multiset of tuples (order unspecified)."""
    d = cogroup(pairs_a, pairs_b)
    out = []
    for k, (vs, ws) in d.items():
        for v in vs:
            for w in ws:
                out.append((k, (v, w)))
    return out
