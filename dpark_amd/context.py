"""DparkContext — mirrors dpark/context.py :: DparkContext [U]
(reference mount empty, §0 — citations symbol-level).

Master selection collapses to local modes on this single-node target:
  'local'        — threaded LocalScheduler (dpark -m local, default)
  'local[N]'     — N worker threads (config 1 uses local[4])
Mesos / process modes are out of scope; the
multi-GPU path is one process per GPU via dpark_amd.engine +
dpark_amd.distshuffle, not a cluster scheduler.
"""

from __future__ import annotations

import re

from .rdd import (LocalScheduler, MultiProcessScheduler, ParallelCollection,
                  ShuffleStore, TextFileRDD, UnionRDD)


class Broadcast:
    """dpark/broadcast.py :: TreeBroadcast [U] degenerates to a value
    holder on a single node (no zmq tree needed)."""

    def __init__(self, value):
        self.value = value


class Accumulator:
    """dpark/accumulator.py :: Accumulator [U] — single-process variant."""

    def __init__(self, value=0, op=None):
        self.value = value
        self._op = op or (lambda a, b: a + b)

    def add(self, v):
        self.value = self._op(self.value, v)

    __iadd__ = None  # use .add(), matching dpark's accum += in tasks [U?]


class DparkContext:
    def __init__(self, master="local", parallel=None):
        m = re.match(r"(local|process)\[(\d+)\]", master or "local")
        mode = m.group(1) if m else (master or "local")
        workers = int(m.group(2)) if m else 4
        self.master = master
        self.defaultParallelism = parallel or workers
        self.shuffle_store = ShuffleStore()
        cls = MultiProcessScheduler if mode == "process" else LocalScheduler
        self.scheduler = cls(self, workers=workers)
        self._stopped = False

    # -- RDD constructors -------------------------------------------------
    def parallelize(self, data, numSlices=None):
        return ParallelCollection(self, data,
                                  numSlices or self.defaultParallelism)

    makeRDD = parallelize

    def textFile(self, path, numSplits=None):
        return TextFileRDD(self, path, numSplits)

    def vectorPairs(self, keys, vals, numSlices=None):
        """Engine-eligible pair RDD (numpy int64 keys / float64 vals):
        the hot-path operators route to the HIP engine on a GPU machine
        (dpark_amd/vector.py)."""
        from .vector import VectorPairsRDD
        return VectorPairsRDD(self, keys, vals, numSlices)

    def binaryPairs(self, path):
        """Reload a VectorPairsRDD dumped by saveAsBinary (partitioned
        npz shards + manifest; bit-exact round trip)."""
        import glob
        import json as _json
        from .vector import VectorPairsRDD
        with open(f"{path}/manifest.json") as f:
            man = _json.load(f)
        assert man.get("format") == "dprk-pairs-v1", man
        parts = sorted(glob.glob(f"{path}/part-*.npz"))
        assert len(parts) == man["partitions"], (len(parts), man)
        ks, vs = [], []
        for p in parts:
            z = __import__("numpy").load(p)
            ks.append(z["keys"])
            vs.append(z["vals"])
        import numpy as np
        keys = np.concatenate(ks) if ks else np.empty(0, np.int64)
        vals = np.concatenate(vs) if vs else np.empty(0, np.float64)
        assert len(keys) == man["n"]
        return VectorPairsRDD(self, keys, vals, max(1, len(parts)))

    def stringPairs(self, words, vals, numSlices=None):
        """Engine-eligible pair RDD with bytes/str keys: reduceByKey(+)
        routes through the device-hash string path (dpark_amd/strings.py,
        SURVEY.md §8(f) rank 1)."""
        from .strings import StringPairsRDD
        return StringPairsRDD(self, words, vals, numSlices)

    def union(self, rdds):
        return UnionRDD(self, list(rdds))

    def broadcast(self, value):
        return Broadcast(value)

    def accumulator(self, value=0, op=None):
        return Accumulator(value, op)

    # -- job running ------------------------------------------------------
    def runJob(self, rdd, func):
        """dpark/context.py :: DparkContext.runJob -> DAGScheduler.runJob
        [U]; returns one result per partition."""
        return self.scheduler.run_job(rdd, func)

    def runJob_collect(self, rdd):
        out = []
        for part in self.runJob(rdd, list):
            out.extend(part)
        return out

    def stop(self):
        self._stopped = True

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.stop()
