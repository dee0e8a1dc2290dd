"""String-key path, CPU side: the normative bytes-hash mirror
(hashspec.hash_bytes_words) and the StringPairsRDD CPU fallback
(SURVEY.md §8(f) rank 1)."""

import operator
from collections import Counter

import numpy as np

from dpark_amd.hashspec import hash_bytes, hash_bytes_words, pack_words


def test_pack_words_roundtrip():
    words = ["foo", "bar", "", "a", "longer-word-here", "foo"]
    buf, offs = pack_words(words)
    assert len(offs) == len(words) + 1
    for i, w in enumerate(words):
        assert bytes(buf[int(offs[i]):int(offs[i + 1])]).decode() == w


def test_hash_bytes_properties():
    words = [f"w{i:05d}" for i in range(5000)] + ["", "a", "ab", "abc"]
    buf, offs = pack_words(words)
    h = hash_bytes_words(buf, offs)
    assert len(np.unique(h[:5000])) == 5000          # no collisions here
    # deterministic
    assert np.array_equal(h, hash_bytes_words(buf, offs))
    # scalar reference agrees
    for i in [0, 1, 4999, 5000, 5003]:
        w = words[i].encode()
        assert int(h[i]) == hash_bytes(w)
    # length in the seed: "a" and "a\x00" (same chunk after pad) differ
    b2, o2 = pack_words([b"a", b"a\x00"])
    h2 = hash_bytes_words(b2, o2)
    assert h2[0] != h2[1]
    # words longer than one chunk
    b3, o3 = pack_words([b"x" * 7, b"x" * 8, b"x" * 9, b"x" * 25])
    h3 = hash_bytes_words(b3, o3)
    assert len(np.unique(h3)) == 4
    assert int(h3[3]) == hash_bytes(b"x" * 25)


def test_string_rdd_cpu_fallback():
    """No GPU in this container: StringPairsRDD.reduceByKey runs the
    generic CPU path (that IS dpark's behavior off the engine path)."""
    from dpark_amd import DparkContext
    rng = np.random.default_rng(3)
    words = rng.choice([f"t{i}" for i in range(50)], size=2000).tolist()
    ctx = DparkContext("local[2]")
    r = ctx.stringPairs(words, np.ones(len(words)), 4)
    got = dict(r.reduceByKey(operator.add).collect())
    want = Counter(words)
    assert {k: int(v) for k, v in got.items()} == dict(want)
