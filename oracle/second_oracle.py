"""Independent second oracle (numpy/pandas). TEST INFRA ONLY.

A vectorized implementation of the same mathematical contracts as
oracle/dpark_oracle.py, sharing NO code with it — the engine must match
BOTH.  Scales to ~1e7-1e8 elements, so it is also
the checker used at sizes where the pure-Python restatement is too slow.
"""

from __future__ import annotations

import numpy as np


def reduce_by_key_sum(keys, vals):
    """groupby(key).sum() via np.unique. Returns (unique_keys_sorted, sums)."""
    keys = np.asarray(keys, dtype=np.int64)
    vals = np.asarray(vals, dtype=np.float64)
    uk, inv = np.unique(keys, return_inverse=True)
    sums = np.zeros(len(uk), dtype=np.float64)
    np.add.at(sums, inv, vals)
    return uk, sums


def reduce_by_key_sum_pandas(keys, vals):
    """Third, pandas-based implementation used to pin the golden fixtures."""
    import pandas as pd
    df = pd.DataFrame({"k": np.asarray(keys, dtype=np.int64),
                       "v": np.asarray(vals, dtype=np.float64)})
    g = df.groupby("k", sort=True)["v"].sum()
    return g.index.to_numpy(dtype=np.int64), g.to_numpy(dtype=np.float64)


def sort_by_key(keys, vals=None, reverse=False):
    keys = np.asarray(keys, dtype=np.int64)
    if vals is None:
        out = np.sort(keys, kind="stable")
        return out[::-1].copy() if reverse else out
    vals = np.asarray(vals)
    order = np.argsort(keys, kind="stable")
    if reverse:
        order = order[::-1]
    return keys[order], vals[order]


def group_counts(keys):
    """key -> multiplicity, as (unique_keys_sorted, counts)."""
    keys = np.asarray(keys, dtype=np.int64)
    return np.unique(keys, return_counts=True)


def join_multiset(keys_a, vals_a, keys_b, vals_b):
    """Inner join as sorted arrays (k, va, vb): the canonical multiset
    representation (row order normalized by lexicographic sort) used for
    layout-independent comparison.  Vectorized via sorted runs."""
    ka = np.asarray(keys_a, dtype=np.int64)
    va = np.asarray(vals_a, dtype=np.float64)
    kb = np.asarray(keys_b, dtype=np.int64)
    vb = np.asarray(vals_b, dtype=np.float64)
    oa = np.argsort(ka, kind="stable")
    ob = np.argsort(kb, kind="stable")
    ka, va, kb, vb = ka[oa], va[oa], kb[ob], vb[ob]
    uka, starts_a, cnts_a = _runs(ka)
    ukb, starts_b, cnts_b = _runs(kb)
    common, ia, ib = np.intersect1d(uka, ukb, return_indices=True)
    out_k, out_a, out_b = [], [], []
    for key_idx in range(len(common)):
        sa, ca = starts_a[ia[key_idx]], cnts_a[ia[key_idx]]
        sb, cb = starts_b[ib[key_idx]], cnts_b[ib[key_idx]]
        avs = va[sa:sa + ca]
        bvs = vb[sb:sb + cb]
        out_k.append(np.full(ca * cb, common[key_idx], dtype=np.int64))
        out_a.append(np.repeat(avs, cb))
        out_b.append(np.tile(bvs, ca))
    if not out_k:
        return (np.empty(0, np.int64), np.empty(0, np.float64),
                np.empty(0, np.float64))
    k = np.concatenate(out_k)
    a = np.concatenate(out_a)
    b = np.concatenate(out_b)
    order = np.lexsort((b, a, k))
    return k[order], a[order], b[order]


def _runs(sorted_keys):
    """(unique_keys, run_starts, run_counts) of a sorted array."""
    if len(sorted_keys) == 0:
        return (np.empty(0, np.int64), np.empty(0, np.int64),
                np.empty(0, np.int64))
    flags = np.empty(len(sorted_keys), dtype=bool)
    flags[0] = True
    np.not_equal(sorted_keys[1:], sorted_keys[:-1], out=flags[1:])
    starts = np.flatnonzero(flags)
    counts = np.diff(np.append(starts, len(sorted_keys)))
    return sorted_keys[starts], starts, counts
