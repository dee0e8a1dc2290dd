"""ctypes binding to libdprk.so — the drop-in boundary (include/dprk.h).

Loads the engine's C-ABI and exposes a numpy-facing Engine class.  On a
machine WITH a GPU the HIP path is the only path: a missing/unbuildable
extension raises EngineUnavailable loudly (no silent CPU fallback —
tier rule ③).  CPU-only containers (this build box) can import this
module and introspect symbols; calling compute ops without a GPU fails
in HIP with a clear error.
"""

from __future__ import annotations

import ctypes
import os

import numpy as np

_HERE = os.path.dirname(os.path.abspath(__file__))
# DPRK_SO overrides the library path (A/B harness for compile-knob
# variants, e.g. TILE-size builds); default is the in-tree build.
_SO = os.environ.get("DPRK_SO") or os.path.join(_HERE, "libdprk.so")

_lib = None
_load_error = None


class EngineUnavailable(RuntimeError):
    pass


class EngineError(RuntimeError):
    pass


class DprkStats(ctypes.Structure):
    _fields_ = [
        ("ms_gen", ctypes.c_double), ("ms_hist", ctypes.c_double),
        ("ms_tilehist", ctypes.c_double), ("ms_scan", ctypes.c_double),
        ("ms_scatter", ctypes.c_double), ("ms_reduce", ctypes.c_double),
        ("ms_runext", ctypes.c_double), ("ms_join", ctypes.c_double),
        ("ms_exchange", ctypes.c_double), ("ms_other", ctypes.c_double),
        ("n_gen", ctypes.c_uint64), ("n_hist", ctypes.c_uint64),
        ("n_tilehist", ctypes.c_uint64), ("n_scan", ctypes.c_uint64),
        ("n_scatter", ctypes.c_uint64), ("n_reduce", ctypes.c_uint64),
        ("n_runext", ctypes.c_uint64), ("n_join", ctypes.c_uint64),
        ("n_exchange", ctypes.c_uint64), ("n_other", ctypes.c_uint64),
        ("bytes_scatter", ctypes.c_uint64), ("bytes_total", ctypes.c_uint64),
        ("sort_passes", ctypes.c_uint64),
    ]

    def as_dict(self):
        return {name: getattr(self, name) for name, _ in self._fields_}


# every symbol include/dprk.h declares — tests/test_cabi.py asserts these
SYMBOLS = [
    "dprk_init", "dprk_free", "dprk_last_error", "dprk_build_info",
    "dprk_reserve", "dprk_gen_pairs", "dprk_set_alias_table",
    "dprk_put_pairs", "dprk_get_pairs", "dprk_put_bytes_keys",
    "dprk_gen_pairs_at", "dprk_val_sum",
    "dprk_big_begin", "dprk_big_add_chunk", "dprk_big_reduce",
    "dprk_big_get", "dprk_big_val_sum", "dprk_big_free",
    "dprk_count", "dprk_is_packed", "dprk_set_packed",
    "dprk_partition", "dprk_partition_range",
    "dprk_sample_keys", "dprk_sort_pairs",
    "dprk_reduce_by_key_sum", "dprk_group_by_key", "dprk_get_groups",
    "dprk_join", "dprk_get_join", "dprk_comm_unique_id", "dprk_comm_init",
    "dprk_exchange", "dprk_exchange_buckets", "dprk_stats",
    "dprk_stats_reset",
]


def load_library():
    """Load (once) and signature-check libdprk.so. Raises EngineUnavailable
    with the underlying cause if the native engine cannot be loaded."""
    global _lib, _load_error
    if _lib is not None:
        return _lib
    if _load_error is not None:
        raise EngineUnavailable(_load_error)
    if not os.path.exists(_SO):
        _load_error = (f"libdprk.so not found at {_SO}; build it with "
                       f"`python __graft_entry__.py build` (hipcc, gfx950)")
        raise EngineUnavailable(_load_error)
    try:
        lib = ctypes.CDLL(_SO)
    except OSError as e:
        _load_error = f"failed to load {_SO}: {e}"
        raise EngineUnavailable(_load_error)
    for s in SYMBOLS:
        if not hasattr(lib, s):
            _load_error = f"{_SO} missing symbol {s}"
            raise EngineUnavailable(_load_error)
    u64 = ctypes.c_uint64
    p = ctypes.c_void_p
    lib.dprk_last_error.restype = ctypes.c_char_p
    lib.dprk_build_info.restype = ctypes.c_char_p
    lib.dprk_init.argtypes = [ctypes.POINTER(p), ctypes.c_int]
    lib.dprk_free.argtypes = [p]
    lib.dprk_free.restype = None
    lib.dprk_reserve.argtypes = [p, u64, ctypes.c_int]
    lib.dprk_gen_pairs.argtypes = [p, u64, u64, ctypes.c_int, u64]
    lib.dprk_set_alias_table.argtypes = [p, ctypes.c_void_p, u64]
    lib.dprk_put_pairs.argtypes = [p, ctypes.c_void_p, ctypes.c_void_p, u64]
    lib.dprk_get_pairs.argtypes = [p, ctypes.c_void_p, ctypes.c_void_p, u64,
                                   ctypes.POINTER(u64)]
    lib.dprk_put_bytes_keys.argtypes = [p, ctypes.c_void_p, u64,
                                        ctypes.c_void_p, u64, ctypes.c_void_p]
    lib.dprk_gen_pairs_at.argtypes = [p, u64, u64, u64, ctypes.c_int, u64]
    lib.dprk_val_sum.argtypes = [p, ctypes.POINTER(ctypes.c_double)]
    lib.dprk_big_begin.argtypes = [p, u64, ctypes.c_int]
    lib.dprk_big_add_chunk.argtypes = [p]
    lib.dprk_big_reduce.argtypes = [p, ctypes.POINTER(u64)]
    lib.dprk_big_get.argtypes = [p, ctypes.c_void_p, ctypes.c_void_p, u64,
                                 ctypes.POINTER(u64)]
    lib.dprk_big_val_sum.argtypes = [p, ctypes.POINTER(ctypes.c_double)]
    lib.dprk_big_free.argtypes = [p]
    lib.dprk_count.argtypes = [p, ctypes.POINTER(u64)]
    lib.dprk_is_packed.argtypes = [p, ctypes.POINTER(ctypes.c_int)]
    lib.dprk_set_packed.argtypes = [p, ctypes.c_int]
    lib.dprk_partition.argtypes = [p, ctypes.c_int, ctypes.POINTER(u64)]
    lib.dprk_partition_range.argtypes = [p, ctypes.c_void_p, ctypes.c_int,
                                         ctypes.POINTER(u64)]
    lib.dprk_sample_keys.argtypes = [p, u64, ctypes.c_void_p]
    lib.dprk_sort_pairs.argtypes = [p]
    lib.dprk_reduce_by_key_sum.argtypes = [p, ctypes.POINTER(u64)]
    lib.dprk_group_by_key.argtypes = [p, ctypes.POINTER(u64)]
    lib.dprk_get_groups.argtypes = [p, ctypes.c_void_p, ctypes.c_void_p,
                                    ctypes.c_void_p, u64, ctypes.POINTER(u64)]
    lib.dprk_join.argtypes = [p, p, ctypes.POINTER(u64)]
    lib.dprk_get_join.argtypes = [p, ctypes.c_void_p, ctypes.c_void_p,
                                  ctypes.c_void_p, u64, ctypes.POINTER(u64)]
    lib.dprk_comm_unique_id.argtypes = [ctypes.c_char_p]
    lib.dprk_comm_init.argtypes = [p, ctypes.c_int, ctypes.c_int,
                                   ctypes.c_char_p]
    lib.dprk_exchange.argtypes = [p, ctypes.POINTER(u64), ctypes.POINTER(u64),
                                  ctypes.c_int]
    lib.dprk_exchange_buckets.argtypes = [
        p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p,
        ctypes.c_int]
    lib.dprk_stats.argtypes = [p, ctypes.POINTER(DprkStats)]
    lib.dprk_stats_reset.argtypes = [p]
    _lib = lib
    return lib


def _check(lib, rc, what):
    if rc != 0:
        raise EngineError(f"{what}: {lib.dprk_last_error().decode()}")


class Engine:
    """One GPU's engine context (numpy in/out, device memory engine-owned)."""

    def __init__(self, device=0):
        self._lib = load_library()
        self._ctx = ctypes.c_void_p()
        _check(self._lib, self._lib.dprk_init(ctypes.byref(self._ctx), device),
               "dprk_init")
        self.device = device

    def close(self):
        if self._ctx:
            self._lib.dprk_free(self._ctx)
            self._ctx = ctypes.c_void_p()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    # -- data ------------------------------------------------------------
    def reserve(self, max_n, with_vals=True):
        _check(self._lib, self._lib.dprk_reserve(self._ctx, max_n,
                                                 1 if with_vals else 0),
               "dprk_reserve")

    def gen_pairs(self, seed, n, key_universe=0, dist=0, start=0):
        _check(self._lib,
               self._lib.dprk_gen_pairs_at(self._ctx, seed, start, n, dist,
                                           key_universe),
               "dprk_gen_pairs_at")

    def val_sum(self):
        s = ctypes.c_double()
        _check(self._lib, self._lib.dprk_val_sum(self._ctx, ctypes.byref(s)),
               "dprk_val_sum")
        return s.value

    # -- out-of-core (n >= 2^32) -----------------------------------------
    def big_begin(self, total_cap, nbuckets=256):
        _check(self._lib, self._lib.dprk_big_begin(self._ctx, total_cap,
                                                   nbuckets),
               "dprk_big_begin")

    def big_add_chunk(self):
        _check(self._lib, self._lib.dprk_big_add_chunk(self._ctx),
               "dprk_big_add_chunk")

    def big_reduce(self):
        n = ctypes.c_uint64()
        _check(self._lib, self._lib.dprk_big_reduce(self._ctx,
                                                    ctypes.byref(n)),
               "dprk_big_reduce")
        return n.value

    def big_get(self, n, with_vals=True):
        """n = the count returned by big_reduce()."""
        keys = np.empty(n, dtype=np.int64)
        vals = np.empty(n, dtype=np.float64) if with_vals else None
        nout = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_big_get(
                   self._ctx, keys.ctypes.data_as(ctypes.c_void_p),
                   vals.ctypes.data_as(ctypes.c_void_p) if with_vals else None,
                   n, ctypes.byref(nout)),
               "dprk_big_get")
        return (keys, vals) if with_vals else keys

    def big_val_sum(self):
        s = ctypes.c_double()
        _check(self._lib, self._lib.dprk_big_val_sum(self._ctx,
                                                     ctypes.byref(s)),
               "dprk_big_val_sum")
        return s.value

    def big_free(self):
        _check(self._lib, self._lib.dprk_big_free(self._ctx),
               "dprk_big_free")

    def set_alias_table(self, table):
        import numpy as _np
        table = _np.ascontiguousarray(table, dtype=_np.uint64)
        _check(self._lib,
               self._lib.dprk_set_alias_table(
                   self._ctx, table.ctypes.data_as(ctypes.c_void_p),
                   len(table)),
               "dprk_set_alias_table")

    def put(self, keys, vals=None):
        keys = np.ascontiguousarray(keys, dtype=np.int64)
        vp = None
        if vals is not None:
            vals = np.ascontiguousarray(vals, dtype=np.float64)
            assert len(vals) == len(keys)
            vp = vals.ctypes.data_as(ctypes.c_void_p)
        _check(self._lib,
               self._lib.dprk_put_pairs(self._ctx,
                                        keys.ctypes.data_as(ctypes.c_void_p),
                                        vp, len(keys)),
               "dprk_put_pairs")

    def put_bytes_keys(self, buf, offs, vals=None):
        """String-key path: device-hash n words (packed bytes `buf`,
        offsets `offs` of n+1 entries) into the keys; vals alongside.
        Spec + numpy mirror: dpark_amd/hashspec.py hash_bytes_words."""
        buf = np.ascontiguousarray(buf, dtype=np.uint8)
        offs = np.ascontiguousarray(offs, dtype=np.uint64)
        n = len(offs) - 1
        vp = None
        if vals is not None:
            vals = np.ascontiguousarray(vals, dtype=np.float64)
            assert len(vals) == n
            vp = vals.ctypes.data_as(ctypes.c_void_p)
        _check(self._lib,
               self._lib.dprk_put_bytes_keys(
                   self._ctx, buf.ctypes.data_as(ctypes.c_void_p), len(buf),
                   offs.ctypes.data_as(ctypes.c_void_p), n, vp),
               "dprk_put_bytes_keys")

    def get(self, with_vals=True):
        n = self.count()
        keys = np.empty(n, dtype=np.int64)
        vals = np.empty(n, dtype=np.float64) if with_vals else None
        nout = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_get_pairs(
                   self._ctx, keys.ctypes.data_as(ctypes.c_void_p),
                   vals.ctypes.data_as(ctypes.c_void_p) if with_vals else None,
                   n, ctypes.byref(nout)),
               "dprk_get_pairs")
        return (keys, vals) if with_vals else keys

    def count(self):
        n = ctypes.c_uint64()
        _check(self._lib, self._lib.dprk_count(self._ctx, ctypes.byref(n)),
               "dprk_count")
        return n.value

    def is_packed(self):
        """Whether keys currently travel as packed u32 (include/dprk.h)."""
        v = ctypes.c_int()
        _check(self._lib, self._lib.dprk_is_packed(self._ctx,
                                                   ctypes.byref(v)),
               "dprk_is_packed")
        return bool(v.value)

    def set_packed(self, want):
        _check(self._lib, self._lib.dprk_set_packed(self._ctx,
                                                    1 if want else 0),
               "dprk_set_packed")

    def sync_packed(self, dist):
        """Multi-rank exchange agreement: all ranks must ship the same key
        width.  All-reduce (MIN) of the local packed flag over the gloo
        control plane; ranks packed while any peer is not unpack."""
        import torch
        t = torch.tensor([1 if self.is_packed() else 0])
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        agreed = bool(int(t[0]))
        if self.is_packed() != agreed:
            self.set_packed(agreed)
        return agreed

    # -- ops -------------------------------------------------------------
    def partition(self, nbuckets):
        counts = (ctypes.c_uint64 * nbuckets)()
        _check(self._lib, self._lib.dprk_partition(self._ctx, nbuckets, counts),
               "dprk_partition")
        return np.array(counts, dtype=np.uint64)

    def sample_keys(self, n_samples):
        out = np.empty(n_samples, dtype=np.int64)
        _check(self._lib,
               self._lib.dprk_sample_keys(self._ctx, n_samples,
                                          out.ctypes.data_as(ctypes.c_void_p)),
               "dprk_sample_keys")
        return out

    def partition_range(self, splitters, nbuckets):
        splitters = np.ascontiguousarray(splitters, dtype=np.int64)
        assert len(splitters) == nbuckets - 1
        counts = (ctypes.c_uint64 * nbuckets)()
        _check(self._lib,
               self._lib.dprk_partition_range(
                   self._ctx, splitters.ctypes.data_as(ctypes.c_void_p),
                   nbuckets, counts),
               "dprk_partition_range")
        return np.array(counts, dtype=np.uint64)

    def sort(self):
        _check(self._lib, self._lib.dprk_sort_pairs(self._ctx),
               "dprk_sort_pairs")

    def reduce_by_key_sum(self):
        n = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_reduce_by_key_sum(self._ctx, ctypes.byref(n)),
               "dprk_reduce_by_key_sum")
        return n.value

    def group_by_key(self):
        n = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_group_by_key(self._ctx, ctypes.byref(n)),
               "dprk_group_by_key")
        g = n.value
        uk = np.empty(g, dtype=np.int64)
        st = np.empty(g, dtype=np.uint64)
        cn = np.empty(g, dtype=np.uint64)
        nout = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_get_groups(
                   self._ctx, uk.ctypes.data_as(ctypes.c_void_p),
                   st.ctypes.data_as(ctypes.c_void_p),
                   cn.ctypes.data_as(ctypes.c_void_p), g, ctypes.byref(nout)),
               "dprk_get_groups")
        return uk, st, cn

    def join(self, other):
        n = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_join(self._ctx, other._ctx, ctypes.byref(n)),
               "dprk_join")
        j = n.value
        k = np.empty(j, dtype=np.int64)
        va = np.empty(j, dtype=np.float64)
        vb = np.empty(j, dtype=np.float64)
        nout = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_get_join(
                   self._ctx, k.ctypes.data_as(ctypes.c_void_p),
                   va.ctypes.data_as(ctypes.c_void_p),
                   vb.ctypes.data_as(ctypes.c_void_p), j, ctypes.byref(nout)),
               "dprk_get_join")
        return k, va, vb

    def join_count(self, other):
        """join without fetching the (possibly huge) result."""
        n = ctypes.c_uint64()
        _check(self._lib,
               self._lib.dprk_join(self._ctx, other._ctx, ctypes.byref(n)),
               "dprk_join")
        return n.value

    # -- comm ------------------------------------------------------------
    @staticmethod
    def comm_unique_id():
        lib = load_library()
        buf = ctypes.create_string_buffer(128)
        _check(lib, lib.dprk_comm_unique_id(buf), "dprk_comm_unique_id")
        return buf.raw

    def comm_init(self, nranks, rank, uid: bytes):
        assert len(uid) == 128
        _check(self._lib,
               self._lib.dprk_comm_init(self._ctx, nranks, rank, uid),
               "dprk_comm_init")

    def exchange_buckets(self, bucket_counts, bucket_rank, recv_matrix,
                         nranks):
        bc = np.ascontiguousarray(bucket_counts, dtype=np.uint64)
        br = np.ascontiguousarray(bucket_rank, dtype=np.int32)
        rm = np.ascontiguousarray(recv_matrix, dtype=np.uint64)
        assert rm.size == nranks * len(bc)
        _check(self._lib,
               self._lib.dprk_exchange_buckets(
                   self._ctx, bc.ctypes.data_as(ctypes.c_void_p),
                   br.ctypes.data_as(ctypes.c_void_p), len(bc),
                   rm.ctypes.data_as(ctypes.c_void_p), nranks),
               "dprk_exchange_buckets")

    def exchange(self, send_counts, recv_counts):
        nr = len(send_counts)
        sc = (ctypes.c_uint64 * nr)(*[int(x) for x in send_counts])
        rc = (ctypes.c_uint64 * nr)(*[int(x) for x in recv_counts])
        _check(self._lib, self._lib.dprk_exchange(self._ctx, sc, rc, nr),
               "dprk_exchange")

    # -- stats -----------------------------------------------------------
    def stats(self):
        s = DprkStats()
        _check(self._lib, self._lib.dprk_stats(self._ctx, ctypes.byref(s)),
               "dprk_stats")
        return s.as_dict()

    def stats_reset(self):
        _check(self._lib, self._lib.dprk_stats_reset(self._ctx),
               "dprk_stats_reset")

    def build_info(self):
        return self._lib.dprk_build_info().decode()
