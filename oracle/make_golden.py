"""Generates tests/golden/*.npz fixtures. TEST INFRA ONLY.

Each fixture's expected output is computed by the dpark-semantics
restatement (oracle/dpark_oracle.py) AND independently by numpy/pandas
(oracle/second_oracle.py); generation FAILS unless all implementations
agree — this is what pins the oracle in the absence of reference golden
vectors (the reference mount is empty, §0).

Run:  python -m oracle.make_golden
"""

from __future__ import annotations

import os

import numpy as np

from . import dpark_oracle as dpo
from . import second_oracle as so
from .synth import gen_pairs

GOLDEN_DIR = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")


def _cases():
    # (name, keys int64, vals float64) — edge cases per SURVEY.md §8(c)(iii)
    rng = np.random.default_rng(12345)
    yield "empty", np.empty(0, np.int64), np.empty(0, np.float64)
    yield "single", np.array([7], np.int64), np.array([0.5])
    yield "all_equal", np.full(1000, -3, np.int64), rng.random(1000)
    yield "negatives", rng.integers(-50, 50, 2000).astype(np.int64), rng.random(2000)
    yield ("extremes",
           np.array([np.iinfo(np.int64).min, np.iinfo(np.int64).max, 0, -1,
                     1, np.iinfo(np.int64).min, np.iinfo(np.int64).max],
                    np.int64),
           rng.random(7))
    k, v = gen_pairs(seed=42, n=10000, key_universe=1 << 8)
    yield "synth_dense_dups", k, v          # heavy duplicates
    k, v = gen_pairs(seed=7, n=10000, key_universe=1 << 64)
    yield "synth_full64", k, v              # full 64-bit incl. negatives
    k, v = gen_pairs(seed=11, n=5000, key_universe=1 << 24)
    yield "synth_cfg2_shape", k, v          # config-2-shaped small case


def main():
    os.makedirs(GOLDEN_DIR, exist_ok=True)
    for name, keys, vals in _cases():
        # reduceByKey(sum)
        uk1, s1 = dpo.reduce_by_key_sum(keys, vals)
        uk2, s2 = so.reduce_by_key_sum(keys, vals)
        uk3, s3 = so.reduce_by_key_sum_pandas(keys, vals)
        assert np.array_equal(uk1, uk2) and np.array_equal(uk2, uk3), name
        assert np.allclose(s1, s2, rtol=1e-12, atol=0) or len(s1) == 0, name
        assert np.allclose(s2, s3, rtol=1e-12, atol=0) or len(s2) == 0, name
        # sortByKey (keys only — duplicate value order is unspecified)
        sk1 = dpo.sort_by_key(keys)
        sk2 = so.sort_by_key(keys)
        assert np.array_equal(sk1, sk2), name
        # groupByKey counts
        gk, gc = so.group_counts(keys)
        # join with itself on a thinned copy (keeps product size bounded)
        jb_k, jb_v = keys[: len(keys) // 2], vals[: len(vals) // 2]
        jk, jva, jvb = so.join_multiset(keys, vals, jb_k, jb_v)
        if len(keys) <= 2000:  # cross-check the pure-python join on small cases
            jj = dpo.join(zip(keys.tolist(), vals.tolist()),
                          zip(jb_k.tolist(), jb_v.tolist()))
            assert len(jj) == len(jk), name
        np.savez(os.path.join(GOLDEN_DIR, f"{name}.npz"),
                 keys=keys, vals=vals,
                 rbk_keys=uk1, rbk_sums=s2,  # second-oracle sums = canonical
                 sorted_keys=sk1,
                 group_keys=gk, group_counts=gc,
                 join_b_n=np.int64(len(jb_k)),
                 join_keys=jk, join_va=jva, join_vb=jvb)
        print(f"golden: {name}: n={len(keys)} uniques={len(uk1)} join={len(jk)}")
    print("all golden fixtures agree across oracle implementations")


if __name__ == "__main__":
    main()
