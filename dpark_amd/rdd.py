"""RDD surface + local scheduler, mirroring douban/dpark's API for the
kept hot path (all reference citations [U]
symbol-level — the reference mount is empty).

The lazy-DAG / stage-cut semantics restated here (map-side combine per
bucket, reduce-side merge — dpark/rdd.py :: ShuffledRDD, dpark/task.py ::
ShuffleMapTask.run, dpark/shuffle.py :: Merger [U]) define the behavioral
contract the HIP engine must match; the CPU execution below is the
product's behavior for generic Python-object RDDs (as in dpark), NOT a
fallback for the engine path — engine-eligible ops on a GPU box run the
HIP path or raise.
"""

from __future__ import annotations

import bisect
import itertools
import operator
import os
import threading
from concurrent.futures import ThreadPoolExecutor

from .portable_hash import portable_hash


# ---------------------------------------------------------------------------
# shuffle contract  [dpark/dependency.py [U]]
# ---------------------------------------------------------------------------

class Aggregator:
    """dpark/dependency.py :: Aggregator [U]."""

    def __init__(self, createCombiner, mergeValue, mergeCombiners):
        self.createCombiner = createCombiner
        self.mergeValue = mergeValue
        self.mergeCombiners = mergeCombiners


class Partitioner:
    @property
    def numPartitions(self):
        raise NotImplementedError

    def getPartition(self, key):
        raise NotImplementedError


class HashPartitioner(Partitioner):
    """dpark/dependency.py :: HashPartitioner [U]: portable_hash(key) % n."""

    def __init__(self, partitions):
        self.partitions = max(1, partitions)

    @property
    def numPartitions(self):
        return self.partitions

    def getPartition(self, key):
        return portable_hash(key) % self.partitions

    def __eq__(self, other):
        return (isinstance(other, HashPartitioner)
                and other.partitions == self.partitions)


class RangePartitioner(Partitioner):
    """dpark/dependency.py :: RangePartitioner [U]: bisect over sorted
    splitter keys; used by sortByKey."""

    def __init__(self, splitters, reverse=False):
        self.splitters = sorted(splitters)
        self.reverse = reverse

    @property
    def numPartitions(self):
        return len(self.splitters) + 1

    def getPartition(self, key):
        p = bisect.bisect_left(self.splitters, key)
        return (len(self.splitters) - p) if self.reverse else p

    def __eq__(self, other):
        return (isinstance(other, RangePartitioner)
                and other.splitters == self.splitters
                and other.reverse == self.reverse)


class Dependency:
    def __init__(self, rdd):
        self.rdd = rdd


class NarrowDependency(Dependency):
    def getParents(self, split):
        return [split]


class OneToOneDependency(NarrowDependency):
    pass


class ShuffleDependency(Dependency):
    """dpark/dependency.py :: ShuffleDependency [U] — the stage cut."""

    _next_shuffle_id = itertools.count(1)

    def __init__(self, rdd, aggregator, partitioner):
        super().__init__(rdd)
        self.shuffleId = next(self._next_shuffle_id)
        self.aggregator = aggregator
        self.partitioner = partitioner


# ---------------------------------------------------------------------------
# in-process shuffle store (single node: dpark's LocalFileShuffle +
# MapOutputTracker degenerate to a dict — SURVEY.md §2 row 7)
# ---------------------------------------------------------------------------

MAX_TASK_FAILURES = 4   # dpark/job.py retry budget [U]


class FetchFailed(Exception):
    """dpark/shuffle.py :: FetchFailed analogue [U]: a reduce task could
    not read a map output; the scheduler invalidates that shuffle and
    resubmits the parent map stage before retrying the reduce task."""

    def __init__(self, shuffle_id):
        super().__init__(f"fetch failed for shuffle {shuffle_id}")
        self.shuffle_id = shuffle_id


class ShuffleStore:
    def __init__(self):
        self._outputs = {}  # shuffleId -> {mapId: {bucket: items}}
        self._complete = set()

    def invalidate(self, shuffle_id):
        """Drop a shuffle's outputs (FetchFailed recovery: the map stage
        re-runs — dpark resubmits the parent stage [U])."""
        self._outputs.pop(shuffle_id, None)
        self._complete.discard(shuffle_id)

    def put(self, shuffle_id, map_id, buckets):
        self._outputs.setdefault(shuffle_id, {})[map_id] = buckets

    def mark_complete(self, shuffle_id):
        self._complete.add(shuffle_id)

    def fetch(self, shuffle_id, reduce_id):
        """dpark/shuffle.py :: SimpleShuffleFetcher.fetch [U]: yields each
        map output's bucket for this reduce partition, in map order."""
        maps = self._outputs.get(shuffle_id, {})
        for map_id in sorted(maps):
            yield maps[map_id].get(reduce_id, [])

    def has(self, shuffle_id):
        return shuffle_id in self._complete


# ---------------------------------------------------------------------------
# RDD  [dpark/rdd.py [U]]
# ---------------------------------------------------------------------------

# set in worker processes: a {shuffleId: {mapId: {bucket: items}}} slice
# shipped with each task (dpark ships file URIs; single node ships data)
_CHILD_STORE = None


class RDD:
    _next_id = itertools.count(1)

    def __init__(self, ctx):
        self.ctx = ctx
        self.id = next(self._next_id)
        self.dependencies = []
        self.partitioner = None

    def __getstate__(self):
        # process mode (dpark -m process [U]): the task closure is pickled
        # to a forked worker; the context (scheduler, locks) stays in the
        # driver — dpark/serialize.py :: dump_closure analogue
        d = dict(self.__dict__)
        d["ctx"] = None
        return d

    # -- to be provided by subclasses ------------------------------------
    def splits(self):
        raise NotImplementedError

    def compute(self, split):
        raise NotImplementedError

    def iterator(self, split):
        return self.compute(split)

    @property
    def numSplits(self):
        return len(self.splits())

    # -- transformations (lazy) ------------------------------------------
    def map(self, f):
        return MappedRDD(self, f)

    def flatMap(self, f):
        return FlatMappedRDD(self, f)

    def filter(self, f):
        return FilteredRDD(self, f)

    def mapPartitions(self, f):
        return MapPartitionsRDD(self, f)

    def mapValue(self, f):
        return MappedValuesRDD(self, f)

    mapValues = mapValue  # dpark names it mapValue; Spark-style alias kept

    def union(self, other):
        return UnionRDD(self.ctx, [self, other])

    def glom(self):
        return MapPartitionsRDD(self, lambda it: [list(it)])

    # -- shuffle transformations -----------------------------------------
    def combineByKey(self, aggregator, numSplits=None):
        numSplits = numSplits or self.ctx.defaultParallelism
        return ShuffledRDD(self, aggregator, HashPartitioner(numSplits))

    def reduceByKey(self, func, numSplits=None):
        """dpark/rdd.py :: RDD.reduceByKey [U] =
        combineByKey(Aggregator(identity, func, func))."""
        agg = Aggregator(lambda v: v, func, func)
        return self.combineByKey(agg, numSplits)

    def groupByKey(self, numSplits=None):
        agg = Aggregator(lambda v: [v],
                         lambda c, v: c + [v],
                         lambda a, b: a + b)
        return self.combineByKey(agg, numSplits)

    def sortByKey(self, key=None, reverse=False, numSplits=None):
        """dpark/rdd.py :: RDD.sortByKey [U]: sample ->
        splitters -> RangePartitioner shuffle -> per-partition sort.
        Output: partition i's keys <= partition i+1's, each sorted =>
        global total order on concatenation; duplicate order unspecified."""
        numSplits = numSplits or self.ctx.defaultParallelism
        keyf = key or (lambda x: x)
        sample = [keyf(x[0]) for x in self.ctx.runJob_collect(self)[::
                  max(1, self._estimate_count() // 1000 or 1)]]
        sample.sort(reverse=reverse)
        n = min(numSplits, max(1, len(sample)))
        if n <= 1 or len(sample) < 2:
            splitters = []
        else:
            step = len(sample) / n
            splitters = [sample[int(step * i)] for i in range(1, n)]
            splitters = sorted(set(splitters), reverse=False)
        part = RangePartitioner(splitters, reverse=reverse)
        list_agg = Aggregator(lambda v: [v],
                              lambda c, v: c + [v],
                              lambda a, b: a + b)
        shuffled = ShuffledRDD(self._key_by(keyf), list_agg, part)

        def sort_partition(it):
            items = sorted(it, key=operator.itemgetter(0), reverse=reverse)
            for _, pairs in items:
                for kv in pairs:
                    yield kv
        return MapPartitionsRDD(shuffled, sort_partition)

    def _key_by(self, keyf):
        if keyf is None:
            return self
        return MappedRDD(self, lambda kv: (keyf(kv[0]), kv))

    def _estimate_count(self):
        return 1000

    def cogroup(self, other, numSplits=None):
        numSplits = numSplits or self.ctx.defaultParallelism
        return CoGroupedRDD([self, other], HashPartitioner(numSplits))

    def join(self, other, numSplits=None):
        """dpark/rdd.py :: RDD.join [U] = cogroup + per-key cross product
        (SURVEY.md §3.3)."""
        def product(kv):
            k, (vs, ws) = kv
            return [(k, (v, w)) for v in vs for w in ws]
        return self.cogroup(other, numSplits).flatMap(product)

    # -- actions ----------------------------------------------------------
    def collect(self):
        return self.ctx.runJob_collect(self)

    def collectAsMap(self):
        return dict(self.collect())

    def count(self):
        return sum(self.ctx.runJob(self, lambda it: sum(1 for _ in it)))

    def reduce(self, f):
        def reducePartition(it):
            acc = None
            first = True
            for x in it:
                acc = x if first else f(acc, x)
                first = False
            return [] if first else [acc]
        parts = [p for p in self.ctx.runJob(self, reducePartition) if p]
        vals = [p[0] for p in parts]
        if not vals:
            raise ValueError("reduce of empty RDD")
        acc = vals[0]
        for v in vals[1:]:
            acc = f(acc, v)
        return acc

    def take(self, n):
        out = []
        for split in self.splits():
            for x in self.iterator(split):
                out.append(x)
                if len(out) >= n:
                    return out
        return out

    def first(self):
        t = self.take(1)
        return t[0] if t else None

    def saveAsTextFile(self, path, ext="", overwrite=True):
        """dpark/rdd.py :: OutputTextFileRDD [U]: one file per partition."""
        os.makedirs(path, exist_ok=True)

        def write(idx_it):
            idx, it = idx_it
            fname = os.path.join(path, "%05d%s" % (idx, ext))
            with open(fname, "w") as f:
                for line in it:
                    f.write(str(line))
                    f.write("\n")
            return fname
        return [write((i, self.iterator(s)))
                for i, s in enumerate(self.splits())]


class Split:
    def __init__(self, index):
        self.index = index


class ParallelCollection(RDD):
    """dpark/rdd.py :: ParallelCollection [U] (parallelize/makeRDD)."""

    def __init__(self, ctx, data, numSlices):
        super().__init__(ctx)
        data = list(data)
        numSlices = max(1, numSlices)
        n = len(data)
        self._slices = [data[(n * i) // numSlices:(n * (i + 1)) // numSlices]
                        for i in range(numSlices)]
        self._splits = [Split(i) for i in range(numSlices)]

    def splits(self):
        return self._splits

    def compute(self, split):
        return iter(self._slices[split.index])


class TextFileRDD(RDD):
    """dpark/rdd.py :: TextFileRDD [U] — local POSIX files, split by byte
    ranges aligned to line boundaries (MooseFS locality is out of scope,
    SURVEY.md §2 row 10)."""

    def __init__(self, ctx, path, numSplits=None):
        super().__init__(ctx)
        self.path = path
        size = os.path.getsize(path)
        numSplits = numSplits or ctx.defaultParallelism
        self.chunk = max(1, size // numSplits)
        n = (size + self.chunk - 1) // self.chunk if size else 1
        self._splits = [Split(i) for i in range(n)]

    def splits(self):
        return self._splits

    def compute(self, split):
        with open(self.path, "rb") as f:
            start = split.index * self.chunk
            end = start + self.chunk
            if start > 0:
                f.seek(start - 1)
                b = f.read(1)
                if b != b"\n":
                    f.readline()  # skip partial line (owned by prev split)
            else:
                f.seek(0)
            # a line belongs to the split containing its FIRST byte
            while f.tell() < end:
                line = f.readline()
                if not line:
                    break
                yield line.decode("utf-8", "replace").rstrip("\n")


class MappedRDD(RDD):
    def __init__(self, prev, f):
        super().__init__(prev.ctx)
        self.prev, self.f = prev, f
        self.dependencies = [OneToOneDependency(prev)]

    def splits(self):
        return self.prev.splits()

    def compute(self, split):
        return map(self.f, self.prev.iterator(split))


class FlatMappedRDD(MappedRDD):
    def compute(self, split):
        for x in self.prev.iterator(split):
            for y in self.f(x):
                yield y


class FilteredRDD(MappedRDD):
    def compute(self, split):
        return filter(self.f, self.prev.iterator(split))


class MapPartitionsRDD(MappedRDD):
    def compute(self, split):
        return iter(self.f(self.prev.iterator(split)))


class MappedValuesRDD(MappedRDD):
    def compute(self, split):
        f = self.f
        return ((k, f(v)) for k, v in self.prev.iterator(split))


class UnionSplit(Split):
    def __init__(self, index, rdd, split):
        super().__init__(index)
        self.rdd, self.split = rdd, split


class UnionRDD(RDD):
    def __init__(self, ctx, rdds):
        super().__init__(ctx)
        self.rdds = rdds
        self._splits = [UnionSplit(i, rdd, sp)
                        for i, (rdd, sp) in enumerate(
                            (rdd, sp) for rdd in rdds for sp in rdd.splits())]
        self.dependencies = [OneToOneDependency(r) for r in rdds]

    def splits(self):
        return self._splits

    def compute(self, split):
        return split.rdd.iterator(split.split)


class ShuffledRDD(RDD):
    """dpark/rdd.py :: ShuffledRDD [U]. compute() = reduce side: fetch each
    map's bucket, Merger dict-merge with mergeCombiners."""

    def __init__(self, parent, aggregator, partitioner):
        super().__init__(parent.ctx)
        self.parent = parent
        self.aggregator = aggregator
        self.partitioner = partitioner
        self.dep = ShuffleDependency(parent, aggregator, partitioner)
        self.dependencies = [self.dep]
        self._splits = [Split(i) for i in range(partitioner.numPartitions)]

    def splits(self):
        return self._splits

    def compute(self, split):
        if self.ctx is not None:
            self.ctx.scheduler.ensure_shuffle(self.dep)
            store = self.ctx.shuffle_store
        else:
            store = _CHILD_STORE   # worker process: parent pre-materialized
        merge_combiners = self.aggregator.mergeCombiners
        d = {}
        for bucket in store.fetch(self.dep.shuffleId, split.index):
            for k, c in bucket:
                if k in d:
                    d[k] = merge_combiners(d[k], c)
                else:
                    d[k] = c
        return iter(d.items())


class CoGroupedRDD(RDD):
    """dpark/rdd.py :: CoGroupedRDD + dpark/shuffle.py :: CoGroupMerger [U]:
    key -> tuple of value lists, one per parent."""

    def __init__(self, rdds, partitioner):
        super().__init__(rdds[0].ctx)
        self.rdds = rdds
        self.partitioner = partitioner
        list_agg = Aggregator(lambda v: [v],
                              lambda c, v: c + [v],
                              lambda a, b: a + b)
        self.deps = []
        for rdd in rdds:
            if rdd.partitioner == partitioner:
                self.deps.append(OneToOneDependency(rdd))
            else:
                self.deps.append(ShuffleDependency(rdd, list_agg, partitioner))
        self.dependencies = self.deps
        self._splits = [Split(i) for i in range(partitioner.numPartitions)]

    def splits(self):
        return self._splits

    def compute(self, split):
        nrdds = len(self.rdds)
        d = {}
        for i, dep in enumerate(self.deps):
            if isinstance(dep, ShuffleDependency):
                if self.ctx is not None:
                    self.ctx.scheduler.ensure_shuffle(dep)
                    store = self.ctx.shuffle_store
                else:
                    store = _CHILD_STORE
                for bucket in store.fetch(dep.shuffleId, split.index):
                    for k, vs in bucket:
                        if k not in d:
                            d[k] = tuple([] for _ in range(nrdds))
                        d[k][i].extend(vs)
            else:
                for k, v in dep.rdd.iterator(split):
                    if k not in d:
                        d[k] = tuple([] for _ in range(nrdds))
                    d[k][i].append(v)
        return iter(d.items())


# ---------------------------------------------------------------------------
# local scheduler  [dpark/schedule.py :: LocalScheduler /
# MultiProcessScheduler, dpark/task.py :: ShuffleMapTask / ResultTask [U]]
# ---------------------------------------------------------------------------

def _collect_shuffle_slices(rdd, split_index, store, seen=None):
    # The bucket data a worker task needs: for every ShuffleDependency
    # reachable through NARROW deps of rdd, its maps' buckets for this
    # reduce split (the single-node analogue of shipping map-output URIs
    # to the fetcher [U]).
    seen = seen if seen is not None else set()
    if rdd.id in seen:
        return {}
    seen.add(rdd.id)
    out = {}
    for dep in rdd.dependencies:
        if isinstance(dep, ShuffleDependency):
            maps = store._outputs.get(dep.shuffleId, {})
            out[dep.shuffleId] = {
                m: {split_index: maps[m].get(split_index, [])}
                for m in maps}
        else:
            out.update(_collect_shuffle_slices(dep.rdd, split_index, store,
                                               seen))
    return out


def _process_worker(payload):
    # Runs in a forked worker: rebuild the task closure, install the
    # shipped shuffle slice, compute one partition.
    import cloudpickle
    global _CHILD_STORE
    func, rdd, split, slices = cloudpickle.loads(payload)
    st = ShuffleStore()
    for sid, maps in slices.items():
        for m, buckets in maps.items():
            st.put(sid, m, buckets)
        st.mark_complete(sid)
    _CHILD_STORE = st
    # child-side retry (dpark/job.py budget [U]); FetchFailed cannot
    # occur here — the shuffle slice is shipped with the task
    last = None
    for _ in range(MAX_TASK_FAILURES):
        try:
            return func(rdd.iterator(split))
        except Exception as e:  # noqa: BLE001
            last = e
    raise RuntimeError(
        f"process task failed after {MAX_TASK_FAILURES} attempts") from last


class LocalScheduler:
    """Stage-cutting scheduler: cuts at ShuffleDependency, runs one
    ShuffleMapTask per parent split (map-side combine per bucket), stores
    outputs in the in-process ShuffleStore, then runs result tasks.
    `workers` threads (dpark local[n]); GIL-bound like reference dpark's
    threaded mode [U]."""

    def __init__(self, ctx, workers=4):
        self.ctx = ctx
        self.workers = max(1, workers)
        self._shuffle_lock = threading.RLock()

    def _attempt(self, run, desc):
        """dpark/job.py retry semantics [U]: <= MAX_TASK_FAILURES
        attempts per task.  FetchFailed invalidates the shuffle so the
        next attempt re-materializes the parent map stage (dpark's
        stage-resubmit on FetchFailed [U]); other exceptions simply
        retry.  After the budget: a clear terminal error."""
        last = None
        for attempt in range(MAX_TASK_FAILURES):
            try:
                return run()
            except FetchFailed as e:
                last = e
                with self._shuffle_lock:
                    self.ctx.shuffle_store.invalidate(e.shuffle_id)
            except Exception as e:  # noqa: BLE001 — dpark retries any task error
                last = e
        raise RuntimeError(
            f"task {desc} failed after {MAX_TASK_FAILURES} attempts: "
            f"{type(last).__name__}: {last}") from last

    def ensure_shuffle(self, dep: ShuffleDependency):
        """Materialize a shuffle's map outputs exactly once; concurrent
        reduce tasks (threads of run_job) serialize here — the map-output
        'tracker' of dpark/tracker.py [U] degenerates to this lock +
        in-process store on a single node."""
        store = self.ctx.shuffle_store
        if store.has(dep.shuffleId):
            return
        with self._shuffle_lock:
            self._ensure_shuffle_locked(dep)

    def _ensure_shuffle_locked(self, dep: ShuffleDependency):
        store = self.ctx.shuffle_store
        if store.has(dep.shuffleId):
            return
        # pre-materialize ancestor shuffles from THIS thread (RLock
        # reentrant) so pool worker threads never block on the lock —
        # the DAGScheduler's parent-stage-first order of SURVEY.md §3.1
        self._materialize_ancestors(dep.rdd)
        parent, agg, part = dep.rdd, dep.aggregator, dep.partitioner

        def shuffle_map_task(map_id, split):
            """dpark/task.py :: ShuffleMapTask.run [U]: the bucket loop."""
            n = part.numPartitions
            buckets = [dict() for _ in range(n)]
            for k, v in parent.iterator(split):
                d = buckets[part.getPartition(k)]
                if k in d:
                    d[k] = agg.mergeValue(d[k], v)
                else:
                    d[k] = agg.createCombiner(v)
            return map_id, {b: list(buckets[b].items())
                            for b in range(n) if buckets[b]}

        splits = parent.splits()
        with ThreadPoolExecutor(self.workers) as pool:
            for map_id, out in pool.map(
                    lambda args: self._attempt(
                        lambda: shuffle_map_task(*args),
                        f"ShuffleMapTask({dep.shuffleId}, {args[0]})"),
                    list(enumerate(splits))):
                store.put(dep.shuffleId, map_id, out)
        store.mark_complete(dep.shuffleId)

    def _materialize_ancestors(self, rdd, seen=None):
        seen = seen if seen is not None else set()
        if rdd.id in seen:
            return
        seen.add(rdd.id)
        for d in rdd.dependencies:
            if isinstance(d, ShuffleDependency):
                self._materialize_ancestors(d.rdd, seen)
                self._ensure_shuffle_locked(d)
            else:
                self._materialize_ancestors(d.rdd, seen)

    def run_job(self, rdd, func):
        splits = rdd.splits()
        with self._shuffle_lock:
            self._materialize_ancestors(rdd)
        return self._run_result_tasks(rdd, func, splits)

    def _run_result_tasks(self, rdd, func, splits):
        # materialize shuffle deps bottom-up first (single-thread walk
        # avoids re-entrant pools); compute() calls ensure_shuffle lazily
        # as well, so this is an optimization not a correctness need.
        # Each result task retries per dpark/job.py [U] (_attempt); a
        # FetchFailed re-materializes the invalidated shuffle lazily on
        # the retry's iterator() call.
        with ThreadPoolExecutor(self.workers) as pool:
            return list(pool.map(
                lambda s: self._attempt(lambda: func(rdd.iterator(s)),
                                        f"ResultTask({rdd.id}, {s.index})"),
                splits))


class MultiProcessScheduler(LocalScheduler):
    """dpark/schedule.py :: MultiProcessScheduler [U] (-m process):
    result tasks fork to worker PROCESSES (closures via cloudpickle, the
    dump_closure analogue), escaping the GIL for CPU-bound python code.
    Shuffle map stages run in the driver's thread pool (cheap dict work);
    each shipped result task carries the shuffle buckets it will fetch --
    the single-node stand-in for dpark's HTTP fetch of map outputs [U].
    Accumulator updates do not propagate from workers (documented limit;
    dpark ships them back with task results [U])."""

    def _run_result_tasks(self, rdd, func, splits):
        import multiprocessing
        import cloudpickle
        store = self.ctx.shuffle_store
        payloads = [
            cloudpickle.dumps(
                (func, rdd, s, _collect_shuffle_slices(rdd, s.index, store)))
            for s in splits]
        procs = min(self.workers, max(1, len(payloads)))
        with multiprocessing.get_context("spawn").Pool(procs) as pool:
            return pool.map(_process_worker, payloads)
