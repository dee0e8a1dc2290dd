"""TEST INFRASTRUCTURE ONLY — the parity oracle for the dpark hot path.

This package is a CPU restatement of douban/dpark's shuffle/sort/group-by
semantics (`reduceByKey` / `combineByKey` / `groupByKey` / `sortByKey` /
`join` / `cogroup`).  It exists solely as the *checker* for the HIP engine
and as the reported CPU baseline leg of bench.py.  It is NOT the product
path: nothing under dpark_amd/ may import or fall back to this package at
run time, and the HIP engine fails loudly when its extension is missing.

PARITY STATUS: **parity unpinned at the reference-file level.**
`/root/reference` contained zero files at survey time and still does
(re-verified at build time).  No golden vectors, tests or
source of douban/dpark exist on this machine and nothing is installable
(no network).  All reference citations below are therefore symbol-level
reconstructions from training knowledge of douban/dpark (~v0.5.x), tagged
[U] per SURVEY.md §0.  Parity is pinned instead by:
  (i)  the mathematical contracts of the operations — reduceByKey ≡
       groupby-fold of a commutative+associative op (partition-layout
       independent); sortByKey ≡ total order on keys; join ≡ per-key
       cross product;
  (ii) an independent second oracle (numpy/pandas, oracle/second_oracle.py)
       that every result must ALSO match;
  (iii) hypothesis property tests over adversarial inputs.
If /root/reference is ever populated, re-pin per SURVEY.md §9.

Only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this package.
"""

from .dpark_oracle import (
    combine_by_key,
    reduce_by_key_sum,
    group_by_key,
    sort_by_key,
    join,
    cogroup,
    portable_hash_int,
)
