"""API-layer tests: the DparkContext/RDD surface in local mode against the
oracle — the analogue of dpark's tests/test_rdd.py philosophy (SURVEY.md
§4: local mode IS the test double; no mock scheduler)."""

import operator
import os

import numpy as np
import pytest

from dpark_amd import DparkContext
from oracle import dpark_oracle as dpo
from oracle import second_oracle as so


@pytest.fixture
def ctx():
    c = DparkContext("local[4]")
    yield c
    c.stop()


def test_parallelize_collect(ctx):
    data = list(range(100))
    assert sorted(ctx.parallelize(data, 7).collect()) == data
    assert ctx.parallelize(data, 7).count() == 100


def test_map_filter_flatmap_union(ctx):
    r = ctx.parallelize(range(10), 3)
    assert sorted(r.map(lambda x: x * 2).collect()) == [x * 2 for x in range(10)]
    assert sorted(r.filter(lambda x: x % 2 == 0).collect()) == [0, 2, 4, 6, 8]
    assert sorted(r.flatMap(lambda x: [x, x]).collect()) == sorted(
        [x for x in range(10) for _ in range(2)])
    u = r.union(ctx.parallelize(range(5), 2))
    assert sorted(u.collect()) == sorted(list(range(10)) + list(range(5)))


def test_reduce_by_key_vs_oracle(ctx):
    rng = np.random.default_rng(3)
    keys = rng.integers(-20, 20, 500).astype(np.int64)
    vals = rng.random(500)
    pairs = list(zip(keys.tolist(), vals.tolist()))
    got = dict(ctx.parallelize(pairs, 5).reduceByKey(operator.add, 4).collect())
    uk, s = so.reduce_by_key_sum(keys, vals)
    assert sorted(got.keys()) == uk.tolist()
    np.testing.assert_allclose([got[int(k)] for k in uk], s, rtol=1e-9)


def test_group_by_key(ctx):
    pairs = [(1, "a"), (2, "b"), (1, "c"), (3, "d"), (1, "e")]
    got = {k: sorted(v) for k, v in
           ctx.parallelize(pairs, 3).groupByKey(2).collect()}
    want = {k: sorted(v) for k, v in dpo.group_by_key(pairs).items()}
    assert got == want


def test_sort_by_key(ctx):
    rng = np.random.default_rng(5)
    keys = rng.integers(-1000, 1000, 400).astype(np.int64)
    pairs = [(int(k), i) for i, k in enumerate(keys)]
    got = ctx.parallelize(pairs, 5).sortByKey(numSplits=4).collect()
    got_keys = [k for k, _ in got]
    assert got_keys == sorted(keys.tolist())
    rev = ctx.parallelize(pairs, 5).sortByKey(reverse=True, numSplits=4).collect()
    assert [k for k, _ in rev] == sorted(keys.tolist(), reverse=True)


def test_join_cogroup(ctx):
    a = [(1, "x"), (2, "y"), (1, "z"), (4, "w")]
    b = [(1, 10), (2, 20), (2, 30), (5, 50)]
    got = sorted(ctx.parallelize(a, 3).join(ctx.parallelize(b, 2), 4).collect())
    want = sorted(dpo.join(a, b))
    assert got == want
    cg = dict(ctx.parallelize(a, 3).cogroup(ctx.parallelize(b, 2), 3).collect())
    assert sorted(cg[1][0]) == ["x", "z"] and cg[1][1] == [10]
    assert cg[5] == ([], [50])


def test_config1_wordcount(ctx, tmp_path):
    """Config 1 (BASELINE.json): word-count reduceByKey on generated text,
    dpark local[4], CPU only — the plumbing acceptance check. Scaled-down
    text here (full 1M-line run lives in bench via --config1)."""
    rng = np.random.default_rng(1)
    words = [f"w{i:04d}" for i in range(200)]
    path = tmp_path / "text.txt"
    lines = [" ".join(rng.choice(words, 8)) for _ in range(2000)]
    path.write_text("\n".join(lines) + "\n")

    counts = dict(
        ctx.textFile(str(path), 4)
           .flatMap(lambda line: line.split())
           .map(lambda w: (w, 1))
           .reduceByKey(operator.add, 4)
           .collect())
    # independent python count
    want = {}
    for line in lines:
        for w in line.split():
            want[w] = want.get(w, 0) + 1
    assert counts == want


def test_text_file_split_boundaries(ctx, tmp_path):
    path = tmp_path / "lines.txt"
    lines = [f"line{i}" for i in range(1000)]
    path.write_text("\n".join(lines) + "\n")
    got = ctx.textFile(str(path), 7).collect()
    assert sorted(got) == sorted(lines)
    assert len(got) == len(lines)


def test_save_as_text_file(ctx, tmp_path):
    out = tmp_path / "out"
    ctx.parallelize(range(20), 3).saveAsTextFile(str(out))
    files = sorted(os.listdir(out))
    assert len(files) == 3
    read = []
    for f in files:
        read.extend(int(line) for line in (out / f).read_text().split())
    assert sorted(read) == list(range(20))


def test_accumulator_broadcast(ctx):
    acc = ctx.accumulator(0)
    bc = ctx.broadcast({"x": 2})
    ctx.parallelize(range(10), 2).map(
        lambda v: acc.add(v * bc.value["x"]) or v).collect()
    assert acc.value == sum(range(10)) * 2


def test_vector_pairs_cpu_path(ctx):
    """VectorPairsRDD on a no-GPU machine serves the same API via the CPU
    scheduler (reference behavior); results match the oracle."""
    import operator
    from dpark_amd.vector import VectorPairsRDD, gpu_present
    rng = np.random.default_rng(9)
    keys = rng.integers(-30, 30, 400).astype(np.int64)
    vals = rng.random(400)
    r = ctx.vectorPairs(keys, vals, 4)
    assert isinstance(r, VectorPairsRDD)
    got = dict(r.reduceByKey(operator.add, 4).collect())
    uk, s = so.reduce_by_key_sum(keys, vals)
    assert sorted(got.keys()) == uk.tolist()
    np.testing.assert_allclose([got[int(k)] for k in uk], s,
                               rtol=1e-6 if gpu_present() else 1e-9)
    # sortByKey via either path: total order on keys
    sk = [k for k, _ in r.sortByKey(numSplits=4).collect()]
    assert sk == sorted(keys.tolist())
    # join with itself
    j = sorted(r.join(ctx.vectorPairs(keys[:100], vals[:100], 2)).collect())
    want = sorted(dpo.join(zip(keys.tolist(), vals.tolist()),
                           zip(keys[:100].tolist(), vals[:100].tolist())))
    assert len(j) == len(want)
    gb = dict(r.groupByKey(3).collect())
    og = dpo.group_by_key(zip(keys.tolist(), vals.tolist()))
    assert set(gb) == set(og)
    for k in og:
        np.testing.assert_allclose(sorted(gb[k]), sorted(og[k]), rtol=1e-9)


def test_process_mode_scheduler():
    """dpark -m process analogue: result tasks in forked workers
    (cloudpickle closures), results identical to thread mode."""
    ctxp = DparkContext("process[2]")
    pairs = [(i % 17, float(i)) for i in range(2000)]
    got = dict(ctxp.parallelize(pairs, 4)
               .reduceByKey(operator.add, 3).collect())
    want = {k: v for k, v in
            zip(*so.reduce_by_key_sum(np.array([k for k, _ in pairs]),
                                      np.array([v for _, v in pairs])))}
    assert set(got) == set(int(k) for k in want)
    for k, v in want.items():
        np.testing.assert_allclose(got[int(k)], v, rtol=1e-9)
    # chained transforms over the shuffle boundary in workers
    out = sorted(ctxp.parallelize(pairs, 3)
                 .reduceByKey(operator.add, 2)
                 .map(lambda kv: (kv[0], kv[1] * 2))
                 .collect())
    assert len(out) == 17
    ctxp.stop()
