"""Output/ingest formats: partitioned binary
dump + reload round trip, and the GPU-path text dump matching the
generic CPU path's format.  CPU-only (numpy-backed RDDs)."""

import os

import numpy as np

from dpark_amd import DparkContext


def test_binary_roundtrip_bit_exact(tmp_path):
    ctx = DparkContext("local[2]")
    rng = np.random.default_rng(8)
    keys = rng.integers(-(1 << 62), 1 << 62, 10_000).astype(np.int64)
    vals = rng.random(10_000)
    vals[0], vals[1], vals[2] = np.nan, np.inf, -np.inf
    r = ctx.vectorPairs(keys, vals, 4)
    p = str(tmp_path / "dump")
    r.saveAsBinary(p)
    assert os.path.exists(os.path.join(p, "manifest.json"))
    assert len([f for f in os.listdir(p) if f.startswith("part-")]) == 4
    r2 = ctx.binaryPairs(p)
    k2, v2 = r2.collectArrays()
    assert np.array_equal(keys, k2)
    assert np.array_equal(vals.view(np.uint64), v2.view(np.uint64))  # bitwise

    # empty dataset round trip
    e = ctx.vectorPairs(np.empty(0, np.int64), np.empty(0), 2)
    p2 = str(tmp_path / "empty")
    e.saveAsBinary(p2)
    k3, v3 = ctx.binaryPairs(p2).collectArrays()
    assert len(k3) == 0 and len(v3) == 0


def test_text_dump_matches_cpu_path_format(tmp_path):
    ctx = DparkContext("local[2]")
    keys = np.array([3, -1, 5, 7], dtype=np.int64)
    vals = np.array([0.5, 2.25, -1.0, 3.75])
    vec = ctx.vectorPairs(keys, vals, 2)
    p_vec = str(tmp_path / "vec")
    vec.saveAsTextFile(p_vec)
    # the generic CPU path on the same tuples
    cpu = ctx.parallelize(list(zip(keys.tolist(), vals.tolist())), 2)
    p_cpu = str(tmp_path / "cpu")
    cpu.saveAsTextFile(p_cpu)
    for i in range(2):
        with open(os.path.join(p_vec, f"{i:05d}")) as f:
            got = f.read()
        with open(os.path.join(p_cpu, f"{i:05d}")) as f:
            want = f.read()
        assert got == want
