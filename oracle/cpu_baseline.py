"""dpark-semantics CPU baseline (the timed `cpu_baseline` leg of bench.py).
TEST INFRA / REPORTING ONLY — never the product path.

Times the SAME algorithm shape as dpark's hot path:
per-worker hash-partitioned per-bucket dict combine (map side) followed by
dict merge (reduce side), on `cores` multiprocessing workers.  Labeled
"dpark-semantics CPU restatement" in reports — the reference binary is
absent from this machine, so this restatement stands in for
it; kind="port".

Run on a BOUNDED sample (default sized to ~10-30 s of CPU work) and scaled
to pairs/sec; bench.py reports {value, unit, cores, kind, sample}.
"""

from __future__ import annotations

import multiprocessing as mp
import os
import time

import numpy as np


def _map_side(args):
    """One map task: partition + per-bucket dict combine.
    [dpark/task.py :: ShuffleMapTask.run [U], dict loop of SURVEY.md §3.1]"""
    keys, vals, n_buckets = args
    buckets = [dict() for _ in range(n_buckets)]
    # portable_hash(int) % n == k % n for int64-range keys (Python modulo)
    for k, v in zip(keys.tolist(), vals.tolist()):
        d = buckets[k % n_buckets]
        if k in d:
            d[k] += v
        else:
            d[k] = v
    return buckets


def reduce_by_key_sum_timed(keys, vals, cores=None):
    """Returns (n_unique, elapsed_seconds, cores). Timed region covers map
    side + reduce side, matching the engine's timed region (input already
    materialized)."""
    cores = cores or os.cpu_count()
    keys = np.asarray(keys, dtype=np.int64)
    vals = np.asarray(vals, dtype=np.float64)
    n = len(keys)
    slices = [(keys[i::cores], vals[i::cores], cores) for i in range(cores)]
    t0 = time.perf_counter()
    if cores == 1:
        map_outputs = [_map_side(slices[0])]
    else:
        with mp.Pool(cores) as pool:
            map_outputs = pool.map(_map_side, slices)
    # reduce side: Merger dict-merge per bucket [dpark/shuffle.py::Merger [U]]
    n_unique = 0
    for b in range(cores):
        d = {}
        for mo in map_outputs:
            for k, c in mo[b].items():
                if k in d:
                    d[k] += c
                else:
                    d[k] = c
        n_unique += len(d)
    elapsed = time.perf_counter() - t0
    return n_unique, elapsed, cores


def measure(n_sample: int, seed: int, key_universe: int, cores=None):
    """Generate a bounded sample per the synth spec and time the restatement.
    Returns dict ready for bench.py's cpu_baseline field."""
    from .synth import gen_pairs
    keys, vals = gen_pairs(seed, n_sample, key_universe)
    n_unique, secs, used = reduce_by_key_sum_timed(keys, vals, cores)
    return {
        "value": round(n_sample / secs / 1e6, 3),   # M KV-pairs/sec
        "unit": "M KV-pairs/sec",
        "cores": used,
        "kind": "port",
        "sample": ("dpark-semantics restatement (per-bucket dict combine + "
                   f"dict merge) on {n_sample} pairs of the bench workload "
                   f"(seed {seed}, universe 2^{key_universe.bit_length()-1}), "
                   f"{used} procs"),
        "n_unique": int(n_unique),
        "elapsed_s": round(secs, 3),
    }
