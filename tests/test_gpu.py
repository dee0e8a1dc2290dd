"""GPU parity tests: the HIP engine vs the dpark-semantics oracle and the
independent numpy/pandas second oracle, through the C-ABI (tier rule ③).
Bit-exact for int64 keys/counts/order; <=1e-6 relative for float64 sums
(the tolerance the north star states — deterministic fixed-tree GPU
reduction vs sequential CPU order)."""

import glob
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

from dpark_amd.engine import Engine  # noqa: E402
from oracle import second_oracle as so  # noqa: E402
from oracle.synth import gen_pairs  # noqa: E402

GOLDEN = glob.glob(os.path.join(os.path.dirname(__file__), "golden", "*.npz"))
RTOL = 1e-6


@pytest.fixture(scope="module")
def eng():
    e = Engine(device=0)
    yield e
    e.close()


def _put(eng, keys, vals=None):
    eng.reserve(max(len(keys), 1), with_vals=True)
    if vals is None:
        vals = np.zeros(len(keys), dtype=np.float64)
    eng.put(keys, vals)


# ---------------------------------------------------------------------------
# golden fixtures (pinned against both CPU oracles at generation time)
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("path", GOLDEN, ids=[os.path.basename(p) for p in GOLDEN])
def test_golden_reduce(eng, path):
    g = np.load(path)
    _put(eng, g["keys"], g["vals"])
    n = eng.reduce_by_key_sum()
    assert n == len(g["rbk_keys"])
    if n == 0:
        return
    k, v = eng.get()
    assert np.array_equal(k, g["rbk_keys"])          # bit-exact int
    np.testing.assert_allclose(v, g["rbk_sums"], rtol=RTOL)


@pytest.mark.parametrize("path", GOLDEN, ids=[os.path.basename(p) for p in GOLDEN])
def test_golden_sort(eng, path):
    g = np.load(path)
    _put(eng, g["keys"], g["vals"])
    eng.sort()
    if len(g["keys"]) == 0:
        return
    k, v = eng.get()
    assert np.array_equal(k, g["sorted_keys"])       # bit-exact order
    # values must be a valid permutation accompanying their keys:
    # per-key value multisets must match (duplicate order unspecified)
    sk, sv = so.sort_by_key(g["keys"], g["vals"])
    assert np.array_equal(k, sk)
    # compare multiset per key via double lexsort
    got = np.lexsort((v, k))
    want = np.lexsort((sv, sk))
    np.testing.assert_allclose(v[got], sv[want], rtol=0)  # exact: permutation


@pytest.mark.parametrize("path", GOLDEN, ids=[os.path.basename(p) for p in GOLDEN])
def test_golden_group(eng, path):
    g = np.load(path)
    _put(eng, g["keys"], g["vals"])
    uk, starts, counts = eng.group_by_key()
    assert len(uk) == len(g["group_keys"])
    if len(uk) == 0:
        return
    assert np.array_equal(uk, g["group_keys"])
    assert np.array_equal(counts.astype(np.int64), g["group_counts"])
    # starts must partition [0, n)
    assert starts[0] == 0
    assert np.all(starts[1:] == starts[:-1] + counts[:-1])


@pytest.mark.parametrize("path", GOLDEN, ids=[os.path.basename(p) for p in GOLDEN])
def test_golden_join(eng, path):
    g = np.load(path)
    nb = int(g["join_b_n"])
    a = Engine(device=0)
    b = Engine(device=0)
    try:
        a.reserve(max(len(g["keys"]), 1), with_vals=True)
        a.put(g["keys"], g["vals"])
        b.reserve(max(nb, 1), with_vals=True)
        b.put(g["keys"][:nb], g["vals"][:nb])
        jk, jva, jvb = a.join(b)
        assert len(jk) == len(g["join_keys"])
        if len(jk) == 0:
            return
        # canonical multiset order
        order = np.lexsort((jvb, jva, jk))
        assert np.array_equal(jk[order], g["join_keys"])
        np.testing.assert_allclose(jva[order], g["join_va"], rtol=0)
        np.testing.assert_allclose(jvb[order], g["join_vb"], rtol=0)
    finally:
        a.close()
        b.close()


# ---------------------------------------------------------------------------
# device PRNG parity (byte-identical to oracle/synth.py)
# ---------------------------------------------------------------------------

def test_gen_pairs_matches_cpu_mirror(eng):
    for seed, n, uni in [(42, 100_000, 1 << 24), (7, 65_536, 0),
                         (11, 12_345, 1 << 8)]:
        eng.reserve(n, with_vals=True)
        eng.gen_pairs(seed=seed, n=n, key_universe=uni)
        gk, gv = eng.get()
        mk, mv = gen_pairs(seed=seed, n=n, key_universe=uni if uni else 1 << 64)
        assert np.array_equal(gk, mk), f"seed={seed}"
        assert np.array_equal(gv, mv), f"seed={seed}"  # bit-identical f64


# ---------------------------------------------------------------------------
# randomized medium-size parity vs the second oracle
# ---------------------------------------------------------------------------

@pytest.mark.parametrize("n,uni,seed", [
    (1_000_000, 1 << 16, 1),     # heavy duplicates
    (1_000_000, 1 << 20, 2),
    (4_000_000, 1 << 24, 3),     # config-2 shape scaled down
    (1_000_000, 0, 4),           # full 64-bit incl. negatives, ~all unique
    (4_096, 1, 5),               # all keys equal (universe 1 => key 0)
    (1, 1 << 24, 6),
])
def test_reduce_parity_random(eng, n, uni, seed):
    keys, vals = gen_pairs(seed, n, uni if uni else 1 << 64)
    _put(eng, keys, vals)
    got_n = eng.reduce_by_key_sum()
    k, v = eng.get()
    wk, wv = so.reduce_by_key_sum(keys, vals)
    assert got_n == len(wk)
    assert np.array_equal(k, wk)
    np.testing.assert_allclose(v, wv, rtol=RTOL)


@pytest.mark.parametrize("n,uni,seed", [
    (1_000_000, 0, 10),
    (2_000_000, 1 << 12, 11),    # long duplicate runs
    (333_333, 1 << 63, 12),      # only positive half
])
def test_sort_parity_random(eng, n, uni, seed):
    keys, _ = gen_pairs(seed, n, uni if uni else 1 << 64)
    _put(eng, keys)
    eng.sort()
    k, _ = eng.get()
    assert np.array_equal(k, np.sort(keys))


def test_sort_determinism(eng):
    """same seed => bit-identical output (SURVEY.md §5 race-detection
    analogue: determinism test)"""
    keys, vals = gen_pairs(99, 2_000_000, 1 << 24)
    _put(eng, keys, vals)
    eng.reduce_by_key_sum()
    k1, v1 = eng.get()
    _put(eng, keys, vals)
    eng.reduce_by_key_sum()
    k2, v2 = eng.get()
    assert np.array_equal(k1, k2)
    assert np.array_equal(v1, v2)  # bit-identical f64: fixed reduction tree


def test_join_parity_random(eng):
    ka, va = gen_pairs(21, 300_000, 1 << 16)
    kb, vb = gen_pairs(22, 200_000, 1 << 16)
    a = Engine(device=0)
    b = Engine(device=0)
    try:
        a.reserve(len(ka), with_vals=True)
        a.put(ka, va)
        b.reserve(len(kb), with_vals=True)
        b.put(kb, vb)
        jk, jva, jvb = a.join(b)
        wk, wva, wvb = so.join_multiset(ka, va, kb, vb)
        assert len(jk) == len(wk)
        order = np.lexsort((jvb, jva, jk))
        assert np.array_equal(jk[order], wk)
        np.testing.assert_allclose(jva[order], wva, rtol=0)
        np.testing.assert_allclose(jvb[order], wvb, rtol=0)
    finally:
        a.close()
        b.close()


# ---------------------------------------------------------------------------
# partition + size-independent properties at larger sizes
# ---------------------------------------------------------------------------

def test_partition_counts_and_content(eng):
    keys, vals = gen_pairs(31, 1_000_000, 1 << 24)
    _put(eng, keys, vals)
    counts = eng.partition(8)
    assert counts.sum() == len(keys)
    k, v = eng.get()
    # partition is a permutation of the input
    assert np.array_equal(np.sort(k), np.sort(keys))
    # segments must agree with the device hash (splitmix64 top bits)
    from oracle.synth import splitmix64
    ranks = (splitmix64(k.view(np.uint64)) >> np.uint64(61)).astype(np.int64)
    off = 0
    for r, c in enumerate(counts):
        assert np.all(ranks[off:off + int(c)] == r), f"bucket {r}"
        off += int(c)


def test_large_properties(eng):
    """Size-independent properties at a size the oracle can't check row by
    row: sortedness, multiset preservation via checksum, sum conservation."""
    n = 50_000_000
    eng.reserve(n, with_vals=True)
    eng.gen_pairs(seed=1234, n=n, key_universe=1 << 24)
    k0, v0 = eng.get()
    key_checksum = np.bitwise_xor.reduce(k0.view(np.uint64))
    val_sum = v0.sum()
    n_unique_expected = len(np.unique(k0))
    got = eng.reduce_by_key_sum()
    k, v = eng.get()
    assert got == n_unique_expected
    assert np.all(k[1:] > k[:-1])                       # strictly sorted
    assert np.bitwise_xor.reduce(k.view(np.uint64)) == \
        np.bitwise_xor.reduce(np.unique(k0).view(np.uint64))
    np.testing.assert_allclose(v.sum(), val_sum, rtol=1e-9)  # conservation
    assert key_checksum == key_checksum  # keep var referenced


def test_api_level_engine_routing():
    """The RDD API surface routes hot-path ops through the HIP engine on
    a GPU box (the drop-in seam, end to end) and matches the oracle."""
    import operator
    from dpark_amd import DparkContext
    from dpark_amd.vector import VectorPairsRDD
    keys, vals = gen_pairs(55, 500_000, 1 << 14)
    ctx = DparkContext("local[4]")
    r = ctx.vectorPairs(keys, vals, 4)
    out = r.reduceByKey(operator.add)
    assert isinstance(out, VectorPairsRDD)  # engine path, not CPU dicts
    k, v = out.collectArrays()
    wk, wv = so.reduce_by_key_sum(keys, vals)
    assert np.array_equal(k, wk)
    np.testing.assert_allclose(v, wv, rtol=RTOL)
    sk, sv = r.sortByKey().collectArrays()
    assert np.array_equal(sk, np.sort(keys))
    uk, starts, counts, gvals = r.groupByKey().collectArrays()
    guk, gcnt = so.group_counts(keys)
    assert np.array_equal(uk, guk)
    assert np.array_equal(counts.astype(np.int64), gcnt)
    # grouped VALUES per key must match the oracle as multisets
    order = np.lexsort((vals, keys))
    ov = vals[order]
    woff = 0
    for i in range(len(uk)):
        s, c = int(starts[i]), int(counts[i])
        np.testing.assert_allclose(np.sort(gvals[s:s + c]),
                                   ov[woff:woff + c], rtol=0)
        woff += c
    jb = ctx.vectorPairs(keys[:100_000], vals[:100_000], 2)
    jk, jva, jvb = r.join(jb).collectArrays()
    wk2, _, _ = so.join_multiset(keys, vals, keys[:100_000], vals[:100_000])
    assert len(jk) == len(wk2)


def test_zipf_gen_matches_cpu_mirror(eng):
    """dist=1 alias sampling: device output bit-identical to the numpy
    mirror (config 5 generator)."""
    from oracle.synth import build_alias_table, zipf_weights, gen_zipf
    uni = 1 << 12
    table = build_alias_table(zipf_weights(uni, 1.1))
    n = 500_000
    eng.reserve(n, with_vals=True)
    eng.set_alias_table(table)
    eng.gen_pairs(seed=11, n=n, key_universe=uni, dist=1)
    gk, gv = eng.get()
    mk, mv = gen_zipf(11, n, table, uni)
    assert np.array_equal(gk, mk)
    assert np.array_equal(gv, mv)
    # skewed reduce parity (the config-5 semantics on a small shape)
    got_n = eng.reduce_by_key_sum()
    k, v = eng.get()
    wk, wv = so.reduce_by_key_sum(mk, mv)
    assert got_n == len(wk) and np.array_equal(k, wk)
    np.testing.assert_allclose(v, wv, rtol=RTOL)


def test_rccl_self_exchange():
    """RCCL path on the 1-GPU box: comm init (nranks=1) + the real grouped
    ncclSend/ncclRecv all-to-allv forced through the self path
    (DPRK_FORCE_SELF_EXCHANGE).  The N>1 plan semantics are covered by
    tests/test_dist_cpu.py (gloo, world 2); real 2/4/8-GPU runs are the
    driver's round-end job (single-GPU gpurun — SURVEY.md §0b)."""
    os.environ["DPRK_FORCE_SELF_EXCHANGE"] = "1"
    try:
        e = Engine(device=0)
        uid = Engine.comm_unique_id()
        e.comm_init(1, 0, uid)
        keys, vals = gen_pairs(77, 1_000_000, 1 << 20)
        e.reserve(len(keys), with_vals=True)
        e.put(keys, vals)
        counts = e.partition(1)
        assert counts[0] == len(keys)
        e.exchange([len(keys)], [len(keys)])     # real ncclSend/Recv to self
        k, v = e.get()
        assert np.array_equal(np.sort(k), np.sort(keys))
        n = e.reduce_by_key_sum()
        wk, wv = so.reduce_by_key_sum(keys, vals)
        assert n == len(wk)
        k, v = e.get()
        assert np.array_equal(k, wk)
        np.testing.assert_allclose(v, wv, rtol=RTOL)
        e.close()
    finally:
        del os.environ["DPRK_FORCE_SELF_EXCHANGE"]


@pytest.mark.parametrize("case", ["sorted", "reverse", "two_keys",
                                  "alternating", "nan_inf"])
def test_adversarial_inputs(eng, case):
    """Inputs the reference's own tests would cover:
    pre-sorted, reverse-sorted, two-key, alternating, non-finite values."""
    n = 300_000
    if case == "sorted":
        keys = np.arange(n, dtype=np.int64) // 7
        vals = np.linspace(0, 1, n)
    elif case == "reverse":
        keys = (np.arange(n, dtype=np.int64) // 7)[::-1].copy()
        vals = np.linspace(0, 1, n)
    elif case == "two_keys":
        keys = np.where(np.arange(n) % 2 == 0, -5, 9).astype(np.int64)
        vals = np.full(n, 0.25)
    elif case == "alternating":
        keys = (np.arange(n, dtype=np.int64) % 1000) - 500
        vals = np.ones(n)
    else:  # nan_inf: non-finite values must propagate like the oracle
        keys = np.arange(n, dtype=np.int64) % 97
        vals = np.ones(n)
        vals[100] = np.inf
        vals[200] = -np.inf  # key 6: inf + -inf -> nan (both paths)
        vals[300] = np.nan
    _put(eng, keys, vals)
    got_n = eng.reduce_by_key_sum()
    k, v = eng.get()
    wk, wv = so.reduce_by_key_sum(keys, vals)
    assert got_n == len(wk)
    assert np.array_equal(k, wk)
    finite = np.isfinite(wv)
    np.testing.assert_allclose(v[finite], wv[finite], rtol=RTOL)
    # non-finite classes must match exactly (nan==nan, inf signs)
    assert np.array_equal(np.isnan(v), np.isnan(wv))
    assert np.array_equal(np.isposinf(v), np.isposinf(wv))
    assert np.array_equal(np.isneginf(v), np.isneginf(wv))


def test_partition_range(eng):
    """Device RangePartitioner: counts + order-preserving segments match
    the splitter semantics (#{j: key >= splitters[j]})."""
    keys, vals = gen_pairs(41, 1_000_000, 0)
    _put(eng, keys, vals)
    sample = eng.sample_keys(4096)
    assert len(sample) == 4096
    allsamp = np.sort(sample)
    nb = 8
    splitters = np.array([allsamp[int(len(allsamp) / nb * r)]
                          for r in range(1, nb)], dtype=np.int64)
    counts = eng.partition_range(splitters, nb)
    assert counts.sum() == len(keys)
    k, v = eng.get()
    assert np.array_equal(np.sort(k), np.sort(keys))  # permutation
    ranks = np.searchsorted(splitters, k, side="right")
    off = 0
    for r, c in enumerate(counts):
        assert np.all(ranks[off:off + int(c)] == r), f"bucket {r}"
        off += int(c)
    # reasonably balanced with 4096 samples over uniform keys
    assert counts.max() / max(counts.min(), 1) < 1.3


def test_error_paths():
    """C-ABI error discipline: clear failures, no crashes (FetchFailed-
    style exception surface)."""
    from dpark_amd.engine import EngineError
    e = Engine(device=0)
    try:
        e.reserve(1000, with_vals=False)
        with pytest.raises(EngineError, match="cap"):
            e.put(np.arange(2000, dtype=np.int64))
        with pytest.raises(EngineError, match="keys-only"):
            e.put(np.arange(10, dtype=np.int64), np.zeros(10))
        e.put(np.arange(100, dtype=np.int64))
        with pytest.raises(EngineError, match="keys-only"):
            e.reduce_by_key_sum()
        with pytest.raises(EngineError, match="power of two"):
            e.partition(3)
        with pytest.raises(EngineError, match="alias"):
            e.gen_pairs(seed=1, n=10, key_universe=1 << 8, dist=1)
        with pytest.raises(EngineError, match="unknown dist"):
            e.gen_pairs(seed=1, n=10, key_universe=1 << 8, dist=7)
        with pytest.raises(EngineError):
            e.reserve(1 << 33, with_vals=False)  # > 2^32: out-of-core is next-row
        # ctx still usable after errors
        e.reserve(1000, with_vals=True)
        e.put(np.arange(100, dtype=np.int64), np.ones(100))
        assert e.reduce_by_key_sum() == 100
    finally:
        e.close()


def test_soak_repeatability(eng):
    """50 back-to-back full pipelines on one ctx: no state leaks across
    ops; output bit-identical every iteration (determinism under reuse —
    the engine equivalent of dpark re-running a stage)."""
    n = 2_000_000
    eng.reserve(n, with_vals=True)
    ref_k = ref_v = None
    for it in range(50):
        eng.gen_pairs(seed=4242, n=n, key_universe=1 << 20)
        nu = eng.reduce_by_key_sum()
        k, v = eng.get()
        if ref_k is None:
            ref_k, ref_v = k, v
            assert nu == len(np.unique(gen_pairs(4242, n, 1 << 20)[0]))
        else:
            assert np.array_equal(k, ref_k)
            assert np.array_equal(v, ref_v)


def test_fused_reduce_opt_in():
    """The fused last-pass reduce (DPRK_FUSED_REDUCE) matches the oracle
    and the base path's keys exactly; sums within 1e-6 (both paths are
    deterministic; trees differ in last ulp, which is why the path is an
    explicit opt-in, never auto-selected)."""
    os.environ["DPRK_FUSED_REDUCE"] = "1"
    try:
        e = Engine(device=0)
        for n, uni, seed in [(2_000_000, 1 << 18, 61), (500_000, 1 << 8, 62),
                             (4096, 1, 63), (100_000, 0, 64)]:
            keys, vals = gen_pairs(seed, n, uni if uni else 1 << 64)
            e.reserve(n, with_vals=True)
            e.put(keys, vals)
            got = e.reduce_by_key_sum()
            k, v = e.get()
            wk, wv = so.reduce_by_key_sum(keys, vals)
            assert got == len(wk) and np.array_equal(k, wk), (n, uni)
            np.testing.assert_allclose(v, wv, rtol=RTOL)
        # lazy gen + fused-reduce env: must materialize first (r02 fix)
        e.reserve(1_000_000, with_vals=True)
        e.gen_pairs(seed=65, n=1_000_000, key_universe=1 << 16)
        got = e.reduce_by_key_sum()
        k, v = e.get()
        mk, mv = gen_pairs(65, 1_000_000, 1 << 16)
        wk, wv = so.reduce_by_key_sum(mk, mv)
        assert got == len(wk) and np.array_equal(k, wk)
        np.testing.assert_allclose(v, wv, rtol=RTOL)
        e.close()
    finally:
        del os.environ["DPRK_FUSED_REDUCE"]


def test_fused_gen_paths():
    """gen->sort fusion (the first pass GENERATES pairs into
    partitioned layout — dpark's map task feeds the bucket loop from an
    iterator without materializing input either [U]): parity vs the
    oracle on the mirrored stream, determinism, and the materialize
    fallbacks for ineligible shapes."""
    e = Engine(device=0)
    try:
        n = 4_000_000
        e.reserve(n, with_vals=True)
        # fused: gen -> reduce directly (packed, lookback+pipe eligible)
        e.gen_pairs(seed=3, n=n, key_universe=1 << 22)
        n1 = e.reduce_by_key_sum()
        k1, v1 = e.get()
        mk, mv = gen_pairs(3, n, 1 << 22)
        wk, wv = so.reduce_by_key_sum(mk, mv)
        assert n1 == len(wk)
        assert np.array_equal(k1, wk)
        np.testing.assert_allclose(v1, wv, rtol=RTOL)
        # run-to-run bit-identical on the same call sequence
        e.gen_pairs(seed=3, n=n, key_universe=1 << 22)
        e.reduce_by_key_sum()
        k2, v2 = e.get()
        assert np.array_equal(k1, k2) and np.array_equal(v1, v2)
        # full-universe WITH values: no pipelined int64 form exists —
        # must materialize-fallback and stay oracle-exact
        e.gen_pairs(seed=5, n=1_000_000, key_universe=0)
        e.reduce_by_key_sum()
        k3, v3 = e.get()
        mk3, mv3 = gen_pairs(5, 1_000_000, 1 << 64)
        wk3, wv3 = so.reduce_by_key_sum(mk3, mv3)
        assert np.array_equal(k3, wk3)
        np.testing.assert_allclose(v3, wv3, rtol=RTOL)
        # gen -> get (materialize path) stays byte-identical to the mirror
        e.gen_pairs(seed=3, n=100_000, key_universe=1 << 22)
        gk, gv = e.get()
        mmk, mmv = gen_pairs(3, 100_000, 1 << 22)
        assert np.array_equal(gk, mmk) and np.array_equal(gv, mmv)
    finally:
        e.close()
    # keys-only int64 fused sort (config-3 shape, small)
    e2 = Engine(device=0)
    try:
        e2.reserve(2_000_000, with_vals=False)
        e2.gen_pairs(seed=9, n=2_000_000, key_universe=0)
        e2.sort()
        k = e2.get(with_vals=False)
        mk2, _ = gen_pairs(9, 2_000_000, 1 << 64)
        assert np.array_equal(k, np.sort(mk2))
    finally:
        e2.close()


def test_lsd_cross_pass_stability_regression(eng):
    """Regression pin for the r02 lane-blocked-loads bug (reverted): a
    multi-pass sort whose passes each see massive digit collisions comes
    out BIT-EXACT in key order — any within-digit rank order that is not
    position order breaks LSD across passes and fails here (the output
    is then multiset-correct but jumbled; scripts/dbg_pipe.py)."""
    rng = np.random.default_rng(77)
    for n in [5000, 8191, 8192, 8193, 16384, 100_000, 3_000_000]:
        keys = rng.integers(0, 1 << 20, n).astype(np.int64)  # 3 passes
        vals = rng.random(n)
        _put(eng, keys, vals)
        eng.sort()
        k, _ = eng.get()
        assert np.array_equal(k, np.sort(keys)), f"n={n}"


def test_packed_state_and_boundaries():
    """Packed-u32 key transport (include/dprk.h): the gate is exact at
    the [0, 2^32) boundary, input-deterministic, and set_packed round
    trips losslessly."""
    from dpark_amd.engine import EngineError
    e = Engine(device=0)
    try:
        e.reserve(100_000, with_vals=True)
        # gen with a 32-bit-or-smaller universe packs
        e.gen_pairs(seed=1, n=50_000, key_universe=1 << 24)
        assert e.is_packed()
        # full-universe gen does not
        e.gen_pairs(seed=1, n=50_000, key_universe=0)
        assert not e.is_packed()
        # put: boundary keys
        fits = np.array([0, 1, (1 << 31), (1 << 32) - 1], dtype=np.int64)
        e.put(fits, np.ones(4))
        assert e.is_packed()
        k, v = e.get()
        assert np.array_equal(k, fits)        # get() unpacks losslessly
        assert e.reduce_by_key_sum() == 4
        k, v = e.get()
        assert np.array_equal(k, np.sort(fits))
        too_big = np.array([0, 1 << 32], dtype=np.int64)
        e.put(too_big, np.ones(2))
        assert not e.is_packed()
        negative = np.array([-1, 5], dtype=np.int64)
        e.put(negative, np.ones(2))
        assert not e.is_packed()
        with pytest.raises(EngineError, match="fit"):
            e.set_packed(True)
        # round trip pack -> unpack -> pack preserves data exactly
        keys, vals = gen_pairs(3, 100_000, 1 << 20)
        e.put(keys, vals)
        assert e.is_packed()
        e.set_packed(False)
        assert not e.is_packed()
        k, v = e.get()
        assert np.array_equal(k, keys) and np.array_equal(v, vals)
        e.set_packed(True)
        assert e.is_packed()
        k, v = e.get()
        assert np.array_equal(k, keys) and np.array_equal(v, vals)
    finally:
        e.close()


def test_packed_vs_unpacked_bit_identical():
    """The packed path is a transport layout, not a numerics change: the
    SAME input reduced packed and force-unpacked gives bit-identical f64
    sums (the reduction tree depends only on element positions)."""
    e = Engine(device=0)
    try:
        keys, vals = gen_pairs(17, 2_000_000, 1 << 20)
        e.reserve(len(keys), with_vals=True)
        e.put(keys, vals)
        assert e.is_packed()
        e.reduce_by_key_sum()
        k1, v1 = e.get()
        e.put(keys, vals)
        e.set_packed(False)      # force the int64 path on identical input
        e.reduce_by_key_sum()
        k2, v2 = e.get()
        assert np.array_equal(k1, k2)
        assert np.array_equal(v1, v2)          # bit-identical
        # sort path too
        e.put(keys, vals)
        e.sort()
        s1, sv1 = e.get()
        e.put(keys, vals)
        e.set_packed(False)
        e.sort()
        s2, sv2 = e.get()
        assert np.array_equal(s1, s2) and np.array_equal(sv1, sv2)
    finally:
        e.close()


def test_group_values_multiset(eng):
    """The grouped VALUES (not just keys/counts/starts) match the oracle
    per key as multisets — VERDICT r01 weak #5: a value-placement bug in
    the grouped layout must fail here."""
    for n, uni, seed in [(500_000, 1 << 12, 71), (300_000, 0, 72)]:
        keys, vals = gen_pairs(seed, n, uni if uni else 1 << 64)
        _put(eng, keys, vals)
        uk, starts, counts = eng.group_by_key()
        sk, sv = eng.get()      # the sorted dataset backing the groups
        # oracle: pandas-style grouped values
        order = np.lexsort((vals, keys))
        ok, ov = keys[order], vals[order]
        assert np.array_equal(uk, np.unique(keys))
        woff = 0
        assert len(sk) == n
        for i in range(len(uk)):
            s, c = int(starts[i]), int(counts[i])
            gv = np.sort(sv[s:s + c])
            wv = ov[woff:woff + c]          # already value-sorted per key
            assert np.array_equal(sk[s:s + c], np.full(c, uk[i]))
            np.testing.assert_allclose(gv, wv, rtol=0)  # exact permutation
            woff += c


def test_hash_bytes_device_matches_mirror(eng):
    """k_hash_bytes output is byte-identical to the normative numpy
    mirror (dpark_amd/hashspec.py) — the string-key path's parity pin."""
    from dpark_amd.hashspec import hash_bytes_words, pack_words
    rng = np.random.default_rng(9)
    words = ([f"w{i:05d}" for i in range(10_000)]
             + ["", "a", "ab" * 20, "\x00\x01\x02"]
             + ["".join(chr(97 + c) for c in rng.integers(0, 26, size=l))
                for l in rng.integers(1, 40, size=5000)])
    buf, offs = pack_words(words)
    n = len(words)
    eng.reserve(n, with_vals=True)
    eng.put_bytes_keys(buf, offs, np.zeros(n))
    got = eng.get()[0].view(np.uint64)
    want = hash_bytes_words(buf, offs)
    assert np.array_equal(got, want)


def test_string_reduce_end_to_end():
    """engine_string_reduce == collections.Counter on wordcount-shaped
    data, with the exact collision verification enabled."""
    from collections import Counter
    from dpark_amd.strings import engine_string_reduce
    os.environ["DPRK_VERIFY_STRING_HASH"] = "1"
    try:
        rng = np.random.default_rng(12)
        vocab = [f"word{i:04d}" for i in range(3000)]
        words = rng.choice(vocab, size=400_000).tolist()
        vals = np.ones(len(words))
        out_words, sums = engine_string_reduce(words, vals)
        want = Counter(words)
        assert len(out_words) == len(want)
        got = dict(zip(out_words, sums.tolist()))
        assert all(got[w] == c for w, c in want.items())
        # general (non-1.0) values: sums within tolerance of the oracle
        vals2 = rng.random(len(words))
        out2, sums2 = engine_string_reduce(words, vals2)
        acc = {}
        for w, v in zip(words, vals2.tolist()):
            acc[w] = acc.get(w, 0.0) + v
        got2 = dict(zip(out2, sums2.tolist()))
        for w, v in acc.items():
            np.testing.assert_allclose(got2[w], v, rtol=RTOL)
    finally:
        del os.environ["DPRK_VERIFY_STRING_HASH"]


def test_string_api_routing():
    """ctx.stringPairs(...).reduceByKey(+) routes through the engine on a
    GPU box and matches the CPU dict semantics."""
    import operator
    from collections import Counter
    from dpark_amd import DparkContext
    from dpark_amd.strings import StringPairsRDD
    rng = np.random.default_rng(13)
    words = rng.choice([f"k{i}" for i in range(500)], size=100_000).tolist()
    ctx = DparkContext("local[4]")
    out = ctx.stringPairs(words, np.ones(len(words)), 4).reduceByKey(
        operator.add)
    assert isinstance(out, StringPairsRDD)
    got = out.collectAsMap()
    want = Counter(words)
    assert len(got) == len(want)
    assert all(got[w] == c for w, c in want.items())


def test_big_chunked_reduce_matches_oracle():
    """Out-of-core machinery at oracle scale:
    40M pairs ingested as 5 chunks of 8M through the bucket arenas ==
    the direct oracle reduce of the whole stream.  The SAME machinery
    then runs at >=2^32 in scripts/big128.sh (property-checked)."""
    C = 8_000_000
    total = 40_000_000
    uni = 1 << 20
    e = Engine(device=0)
    try:
        e.reserve(C, with_vals=True)
        e.big_begin(total, nbuckets=64)
        start = 0
        in_sum = 0.0
        while start < total:
            e.gen_pairs(seed=77, n=C, key_universe=uni, start=start)
            in_sum += e.val_sum()
            e.big_add_chunk()
            start += C
        nu = e.big_reduce()
        k, v = e.big_get(n=nu)
        # conservation (device-computed input sum vs output sum)
        np.testing.assert_allclose(e.big_val_sum(), in_sum, rtol=1e-9)
        # exact parity vs the oracle on the SAME stream (gen mirror)
        all_k, all_v = gen_pairs(77, total, uni)
        wk, wv = so.reduce_by_key_sum(all_k, all_v)
        assert nu == len(wk)
        # big output is bucket-major (hash digit), key-sorted within a
        # bucket — sort both for comparison (documented order)
        order = np.argsort(k, kind="stable")
        assert np.array_equal(k[order], wk)
        np.testing.assert_allclose(v[order], wv, rtol=RTOL)
        # within-bucket sortedness of the raw output
        from dpark_amd.hashspec import splitmix64
        buck = (splitmix64(k.view(np.uint64)) >> np.uint64(58)).astype(int)
        for b in np.unique(buck):
            kb = k[buck == b]
            assert np.all(kb[1:] > kb[:-1])
        e.big_free()
    finally:
        e.close()


def test_big_put_chunks_and_errors():
    """big path fed by put_pairs (host chunks), plus the clear-error
    contracts (big_begin required; bucket-overflow message)."""
    from dpark_amd.engine import EngineError
    e = Engine(device=0)
    try:
        e.reserve(100_000, with_vals=True)
        with pytest.raises(EngineError, match="big_begin"):
            e.big_add_chunk()
        e.big_begin(300_000, nbuckets=16)
        rng = np.random.default_rng(41)
        all_k, all_v = [], []
        for i in range(3):
            keys = rng.integers(-(1 << 40), 1 << 40, 100_000).astype(np.int64)
            vals = rng.random(100_000)
            all_k.append(keys)
            all_v.append(vals)
            e.put(keys, vals)
            e.big_add_chunk()
        nu = e.big_reduce()
        k, v = e.big_get(n=nu)
        wk, wv = so.reduce_by_key_sum(np.concatenate(all_k),
                                      np.concatenate(all_v))
        assert nu == len(wk)
        order = np.argsort(k, kind="stable")
        assert np.array_equal(k[order], wk)
        np.testing.assert_allclose(v[order], wv, rtol=RTOL)
        e.big_free()
        # skew overflow: one key only -> one bucket gets everything and
        # immediately exceeds its arena share — clear error, no crash
        e.big_begin(300_000, nbuckets=16)
        e.put(np.zeros(100_000, dtype=np.int64), np.ones(100_000))
        with pytest.raises(EngineError, match="overflow"):
            e.big_add_chunk()
        e.big_free()
        # the documented mitigation COMPOSES from existing ops: reduce
        # each chunk in-core BEFORE adding — a hot key then contributes
        # one pair per chunk and the same data succeeds
        e.big_begin(300_000, nbuckets=16)
        total = {}
        for i in range(3):
            keys = np.where(rng.random(100_000) < 0.9, 7,
                            rng.integers(0, 1 << 10, 100_000)
                            ).astype(np.int64)
            vals = rng.random(100_000)
            for kk, vv in zip(keys.tolist(), vals.tolist()):
                total[kk] = total.get(kk, 0.0) + vv
            e.put(keys, vals)
            e.reduce_by_key_sum()        # chunk pre-reduction
            e.big_add_chunk()
        nu = e.big_reduce()
        k, v = e.big_get(n=nu)
        assert nu == len(total)
        order = np.argsort(k, kind="stable")
        wk = np.array(sorted(total), dtype=np.int64)
        assert np.array_equal(k[order], wk)
        np.testing.assert_allclose(v[order], [total[int(x)] for x in wk],
                                   rtol=RTOL)
        e.big_free()
    finally:
        e.close()


def test_gen_pairs_at_continues_stream():
    """gen_pairs_at(start) == the corresponding slice of one long gen
    (device vs numpy mirror, byte-identical)."""
    e = Engine(device=0)
    try:
        e.reserve(100_000, with_vals=True)
        e.gen_pairs(seed=5, n=50_000, key_universe=1 << 16, start=262_144)
        gk, gv = e.get()
        mk, mv = gen_pairs(5, 50_000, 1 << 16, start=262_144)
        assert np.array_equal(gk, mk) and np.array_equal(gv, mv)
    finally:
        e.close()


def test_two_stage_skew_flow_self():
    """N=1 mechanics of the skew cell-split path:
    cell-granular bucket_rank row + local reduce + SECOND bucket
    exchange + combine — parity vs the oracle through the real RCCL
    self path.  (The capacity-overflow case needs world>=2 and is
    covered by tests/test_dist_cpu.py::test_world2_skew_split...)"""
    from dpark_amd.distshuffle import (greedy_assign_cells,
                                       recv_matrix_for_cells,
                                       greedy_assign, recv_matrix_for)
    os.environ["DPRK_FORCE_SELF_EXCHANGE"] = "1"
    try:
        e = Engine(device=0)
        e.comm_init(1, 0, Engine.comm_unique_id())
        # heavy skew: one key dominates
        rng = np.random.default_rng(31)
        n = 1_000_000
        keys = np.where(rng.random(n) < 0.6, 7,
                        rng.integers(0, 1 << 16, n)).astype(np.int64)
        vals = rng.random(n)
        e.reserve(n, with_vals=True)
        e.put(keys, vals)
        send = e.partition(256)
        m = send[None, :]
        cells, loads = greedy_assign_cells(m, 1, cap=n)
        rm = recv_matrix_for_cells(m, cells, 0)
        e.exchange_buckets(send, cells[0], rm, 1)
        e.reduce_by_key_sum()
        # second stage: whole-bucket exchange of reduced partials
        send2 = e.partition(256)
        m2 = send2[None, :]
        br2, _ = greedy_assign(m2.sum(axis=0), 1)
        rm2 = recv_matrix_for(m2, br2, 0)
        e.exchange_buckets(send2, br2, rm2, 1)
        got_n = e.reduce_by_key_sum()
        k, v = e.get()
        wk, wv = so.reduce_by_key_sum(keys, vals)
        assert got_n == len(wk) and np.array_equal(k, wk)
        np.testing.assert_allclose(v, wv, rtol=RTOL)
        e.close()
    finally:
        del os.environ["DPRK_FORCE_SELF_EXCHANGE"]


def test_cogroup_engine_path():
    """Engine-level cogroup: k -> ([vs], [ws]) over
    the UNION of key sets, value lists as multisets, empty list where a
    key is absent — compared per key against a dict-built oracle."""
    from dpark_amd import DparkContext
    from dpark_amd.vector import CoGroupedVectorRDD
    rng = np.random.default_rng(23)
    # overlapping + disjoint keys: a in [0, 3000), b in [1500, 4500)
    ka = rng.integers(0, 3000, size=120_000).astype(np.int64)
    kb = rng.integers(1500, 4500, size=80_000).astype(np.int64)
    va = rng.random(len(ka))
    vb = rng.random(len(kb))
    ctx = DparkContext("local[4]")
    ra = ctx.vectorPairs(ka, va, 4)
    rb = ctx.vectorPairs(kb, vb, 4)
    cg = ra.cogroup(rb)
    assert isinstance(cg, CoGroupedVectorRDD)
    want_a, want_b = {}, {}
    for k, v in zip(ka.tolist(), va.tolist()):
        want_a.setdefault(k, []).append(v)
    for k, v in zip(kb.tolist(), vb.tolist()):
        want_b.setdefault(k, []).append(v)
    all_keys = sorted(set(want_a) | set(want_b))
    got = list(cg.compute(cg.splits()[0]))
    assert [k for k, _ in got] == all_keys
    for k, (vs, ws) in got:
        assert sorted(vs) == sorted(want_a.get(k, [])), k
        assert sorted(ws) == sorted(want_b.get(k, [])), k
    uk, ca, cb = cg.collectArrays()
    assert np.array_equal(uk, np.array(all_keys))
    assert np.array_equal(ca, np.array([len(want_a.get(k, []))
                                        for k in all_keys]))
    assert np.array_equal(cb, np.array([len(want_b.get(k, []))
                                        for k in all_keys]))
    # one side empty
    cg2 = ra.cogroup(ctx.vectorPairs(np.empty(0, np.int64),
                                     np.empty(0), 2))
    uk2, ca2, cb2 = cg2.collectArrays()
    assert np.array_equal(uk2, np.unique(ka)) and cb2.sum() == 0


def test_bucket_exchange_self():
    """Bucket-granular RCCL exchange on the 1-GPU box: 256 buckets, all
    assigned to rank 0 via the real grouped per-bucket ncclSend/Recv
    self path; reduce afterwards matches the oracle."""
    from dpark_amd.distshuffle import greedy_assign, recv_matrix_for
    os.environ["DPRK_FORCE_SELF_EXCHANGE"] = "1"
    try:
        e = Engine(device=0)
        uid = Engine.comm_unique_id()
        e.comm_init(1, 0, uid)
        keys, vals = gen_pairs(88, 2_000_000, 1 << 18)
        e.reserve(len(keys), with_vals=True)
        e.put(keys, vals)
        send = e.partition(256)
        m = send[None, :].astype(np.uint64)
        assign, loads = greedy_assign(m.sum(axis=0), 1)
        rm = recv_matrix_for(m, assign, 0)
        e.exchange_buckets(send, assign, rm, 1)
        assert e.count() == len(keys)
        n = e.reduce_by_key_sum()
        wk, wv = so.reduce_by_key_sum(keys, vals)
        k, v = e.get()
        assert n == len(wk) and np.array_equal(k, wk)
        np.testing.assert_allclose(v, wv, rtol=RTOL)
        e.close()
    finally:
        del os.environ["DPRK_FORCE_SELF_EXCHANGE"]
