"""SimEngine — CPU control-plane simulator of the dprk C-ABI surface.

TEST INFRASTRUCTURE ONLY: bench.py substitutes this for Engine ONLY when
DPRK_BENCH_SIM=1 is set (tests/test_bench_dist_cpu.py uses it to exercise
the exact torchrun launch line the driver uses for multi-GPU scaling,
with no GPU).  It is NEVER a compute fallback: the real engine path
raises EngineUnavailable when libdprk.so is missing on a GPU machine, and
bench marks its output "data": "SIMULATED-no-gpu" so a sim run can never
be mistaken for a measurement.
"""

from __future__ import annotations

import numpy as np

from oracle.synth import gen_pairs  # sim-only import: mirrors dprk_gen_pairs
from .distshuffle import rank_of_keys


class SimEngine:
    def __init__(self, device=0):
        self.device = device
        self.keys = np.empty(0, np.int64)
        self.vals = np.empty(0, np.float64)
        self._cap = 0
        self._with_vals = True

    def close(self):
        pass

    def reserve(self, max_n, with_vals=True):
        self._cap = max_n
        self._with_vals = with_vals

    def gen_pairs(self, seed, n, key_universe=0, dist=0, start=0):
        assert n <= self._cap, "gen: n > reserved cap"
        k, v = gen_pairs(seed, n, key_universe if key_universe else 1 << 64,
                         start=start)
        self.keys, self.vals = k, v

    def set_alias_table(self, table):
        self._table = np.asarray(table, dtype=np.uint64)

    def is_packed(self):
        if len(self.keys) == 0:
            return False
        return bool(self.keys.min() >= 0 and self.keys.max() < (1 << 32))

    def set_packed(self, want):
        if want and not self.is_packed():
            raise RuntimeError("set_packed: keys do not fit [0, 2^32)")

    def sync_packed(self, dist):
        import torch
        t = torch.tensor([1 if self.is_packed() else 0])
        dist.all_reduce(t, op=dist.ReduceOp.MIN)
        return bool(int(t[0]))

    def partition(self, nbuckets):
        r = rank_of_keys(self.keys, nbuckets)
        order = np.argsort(r, kind="stable")
        self.keys = self.keys[order]
        self.vals = self.vals[order]
        return np.bincount(r, minlength=nbuckets).astype(np.uint64)

    def sample_keys(self, n_samples):
        stride = max(1, len(self.keys) // n_samples)
        return self.keys[::stride][:n_samples].copy()

    def partition_range(self, splitters, nbuckets):
        r = np.searchsorted(np.asarray(splitters, np.int64), self.keys,
                            side="right")
        order = np.argsort(r, kind="stable")
        self.keys = self.keys[order]
        self.vals = self.vals[order]
        return np.bincount(r, minlength=nbuckets).astype(np.uint64)

    def exchange_buckets(self, bucket_counts, bucket_rank, recv_matrix,
                         nranks):
        assert sum(int(x) for x in bucket_counts) == len(self.keys)
        rm = np.asarray(recv_matrix, dtype=np.uint64)
        assert rm.shape[0] == nranks
        assert int(rm.sum()) <= self._cap, "exchange_buckets: recv > cap"
        # keep local shard (control-plane sim)

    def exchange(self, send_counts, recv_counts):
        # single-process sim cannot move data between ranks; the CONTROL
        # bookkeeping (what the driver's launch exercises) is checked:
        assert sum(int(x) for x in send_counts) == len(self.keys)
        n_recv = sum(int(x) for x in recv_counts)
        assert n_recv <= self._cap, "exchange: recv > cap"
        # keep the local shard (self-exchange equivalent)

    def sort(self):
        order = np.argsort(self.keys, kind="stable")
        self.keys = self.keys[order]
        self.vals = self.vals[order]

    def reduce_by_key_sum(self):
        uk, inv = np.unique(self.keys, return_inverse=True)
        sums = np.zeros(len(uk))
        np.add.at(sums, inv, self.vals)
        self.keys, self.vals = uk, sums
        return len(uk)

    def join_count(self, other):
        common = np.intersect1d(np.unique(self.keys), np.unique(other.keys))
        return int(len(common))  # control-flow only

    def get(self, with_vals=True):
        return (self.keys.copy(), self.vals.copy()) if with_vals \
            else self.keys.copy()

    def count(self):
        return len(self.keys)

    def stats(self):
        z = {f"ms_{k}": 0.0 for k in
             ["gen", "hist", "tilehist", "scan", "scatter", "reduce",
              "runext", "join", "exchange", "other"]}
        z.update({f"n_{k}": 0 for k in
                  ["gen", "hist", "tilehist", "scan", "scatter", "reduce",
                   "runext", "join", "exchange", "other"]})
        z.update({"bytes_scatter": 0, "bytes_total": 0, "sort_passes": 0})
        return z

    def stats_reset(self):
        pass

    @staticmethod
    def comm_unique_id():
        return b"\x00" * 128

    def comm_init(self, nranks, rank, uid):
        assert len(uid) == 128
