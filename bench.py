#!/usr/bin/env python3
"""bench.py — headline benchmark: M KV-pairs/sec through the reduceByKey
shuffle (BASELINE.json metric), on the engine's HIP path.

Workload (config.workload): BASELINE.json configs[1] — "reduceByKey on
100M synthetic int64->f64 pairs, 1 MI355X" — the largest single-GPU
config the metric is quoted on (configs[0] is the CPU plumbing check, run
via --config1).  One STEP = one pass of the hot path over one resident
batch: device-side input regeneration (same seed; ~2% of step, included —
reported value is conservative; pipeline-only ms also reported) followed
by partition(nranks) -> [RCCL all-to-allv if N>1] -> LSD radix sort ->
segmented reduce.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
  N>1 is launched by the driver via torch.distributed.run (one rank per
  GPU over RCCL); rank/world size read from the env.  N=1 needs no torch.
Prints ONE JSON line from rank 0.
"""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

N_PAIRS_PER_GPU = 100_000_000
KEY_UNIVERSE = 1 << 24
SEED = 42
HBM_PEAK_SPEC = 8000.0      # GB/s (MI355X_MICROARCH.md spec)
HBM_PEAK_MEASURED = 6290.0  # GB/s measured float4 copy (same doc)


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def run_config1():
    """BASELINE.json configs[0]: word-count reduceByKey on 1M-line text,
    dpark local[4], CPU only (plumbing check, not the headline metric)."""
    import operator
    import numpy as np
    from dpark_amd import DparkContext
    rng = np.random.default_rng(1)
    words = [f"w{i:05d}" for i in range(10_000)]
    path = "/tmp/dprk_config1.txt"
    t0 = time.perf_counter()
    with open(path, "w") as f:
        for _ in range(100):
            block = rng.choice(words, size=(10_000, 10))
            f.write("\n".join(" ".join(row) for row in block) + "\n")
    log(f"config1: wrote 1M lines in {time.perf_counter()-t0:.1f}s")
    ctx = DparkContext("local[4]")
    t0 = time.perf_counter()
    counts = (ctx.textFile(path, 16)
              .flatMap(lambda line: line.split())
              .map(lambda w: (w, 1))
              .reduceByKey(operator.add, 4)
              .collectAsMap())
    dt = time.perf_counter() - t0
    assert len(counts) == 10_000
    assert sum(counts.values()) == 10_000_000
    print(json.dumps({"config": 1, "mode": "cpu-plumbing", "lines": 1_000_000,
                      "elapsed_s": round(dt, 2), "distinct_words": len(counts),
                      "ok": True}))


def run_wordcount(args):
    """String-key engine leg: config-1 wordcount
    through the device-hash path.  One step = H2D + k_hash_bytes + sort +
    segmented reduce + exemplar grouping (the H2D and hash are part of
    the step — stated in config; the CPU comparator is --config1)."""
    from collections import Counter
    from dpark_amd.engine import Engine
    from dpark_amd.hashspec import pack_words

    rng = np.random.default_rng(1)
    vocab = [f"w{i:05d}" for i in range(10_000)]
    n_lines = args.n_pairs or 1_000_000
    log(f"wordcount: generating {n_lines} lines ...")
    blocks = [rng.choice(vocab, size=(min(10_000, n_lines), 10))
              for _ in range(max(1, n_lines // 10_000))]
    words = [w for b in blocks for w in b.ravel().tolist()]
    n = len(words)
    buf, offs = pack_words(words)
    vals = np.ones(n, dtype=np.float64)
    rowids = np.arange(n, dtype=np.float64)
    log(f"wordcount: {n} words, {len(buf)} bytes packed")

    ea = Engine(device=0)
    eb = Engine(device=0)
    ea.reserve(n, with_vals=True)
    eb.reserve(n, with_vals=True)

    def step():
        ea.put_bytes_keys(buf, offs, vals)
        ea.reduce_by_key_sum()
        uh, sums = ea.get()
        eb.put_bytes_keys(buf, offs, rowids)
        uk, starts, counts = eb.group_by_key()
        _, sorted_rows = eb.get()
        exemplar = sorted_rows[starts.astype(np.int64)].astype(np.int64)
        return uh, sums, exemplar

    for _ in range(args.warmup):
        uh, sums, exemplar = step()
    # parity vs the CPU dict path (dpark semantics), untimed
    want = Counter(words)
    got = {words[i]: s for i, s in zip(exemplar.tolist(), sums.tolist())}
    assert len(got) == len(want) == 10_000, (len(got), len(want))
    assert all(got[w] == c for w, c in want.items()), "wordcount mismatch"
    log("wordcount: parity vs CPU Counter green")

    ea.stats_reset()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    elapsed = time.perf_counter() - t0
    s = ea.stats()
    value = n * args.steps / elapsed / 1e6
    roofline = None
    if s["n_scatter"] > 0 and s["ms_scatter"] > 0:
        achieved = s["bytes_scatter"] / s["ms_scatter"] * 1e-6
        roofline = {"bound": "hbm", "achieved": round(achieved, 1),
                    "peak": HBM_PEAK_SPEC, "unit": "GB/s",
                    "frac": round(achieved / HBM_PEAK_SPEC, 4),
                    "frac_vs_measured_copy":
                        round(achieved / HBM_PEAK_MEASURED, 4),
                    "traffic": None, "kernel": "k_scatter",
                    "ms_per_launch": round(s["ms_scatter"] / s["n_scatter"],
                                           4),
                    "launches": int(s["n_scatter"])}
    print(json.dumps({
        "metric": "M words/sec through string-key reduceByKey (wordcount)",
        "value": round(value, 2), "unit": "M words/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1000 / args.steps, 3),
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "bytes-key+f64", "data": "synthetic",
        "config": {"workload": "wordcount-engine-string-path "
                               "(BASELINE.json configs[0] semantics on GPU; "
                               "step INCLUDES H2D + device hash + exemplar "
                               "grouping)",
                   "op": "wordcount", "n_words": n,
                   "distinct_words": 10_000,
                   "parallelism": "single-gpu",
                   "engine": "libdprk.so (HIP gfx950)"},
        "roofline": roofline, "cpu_baseline": None}))
    ea.close()
    eb.close()


def run_bigreduce(args):
    """Out-of-core reduceByKey: n >= 2^32 pairs
    through the chunked bucket-arena path (default 8B pairs = 128 GB of
    int64+f64 input, keys uniform over 2^26).  One step = the whole job:
    chunked device generation + ingest (hash partition into arenas) +
    per-bucket sort/reduce.  Conservation (device input sum vs output
    sum) asserted every step."""
    from dpark_amd.engine import Engine
    n_total = args.n_pairs or 8_000_000_000
    CHUNK = 1 << 28
    UNI = 1 << 26
    eng = Engine(device=0)
    eng.reserve(CHUNK, with_vals=True)
    log(f"bigreduce: {n_total} pairs ({n_total * 16 / 2**30:.0f} GiB of "
        f"int64+f64 input) in chunks of {CHUNK}")

    def job():
        # 64 buckets: fill ~n/64 stays under the chunk cap while per-
        # bucket fixed costs (launch trains + syncs) amortize 4x better
        # than 256
        eng.big_begin(n_total, nbuckets=64)
        start = 0
        in_sum = 0.0
        while start < n_total:
            take = min(CHUNK, n_total - start)
            eng.gen_pairs(seed=SEED, n=take, key_universe=UNI, start=start)
            in_sum += eng.val_sum()   # conservation leg (8 B/pair read)
            eng.big_add_chunk()
            start += take
        nu = eng.big_reduce()
        out_sum = eng.big_val_sum()
        assert abs(out_sum - in_sum) <= 1e-9 * abs(in_sum), \
            (out_sum, in_sum)
        return nu   # arenas retained: big_begin reuses them next job

    nu = None
    for _ in range(args.warmup):
        nu = job()
    log(f"bigreduce: warmup done, {nu} uniques")
    eng.stats_reset()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        nu = job()
    elapsed = time.perf_counter() - t0
    s = eng.stats()
    log("engine stats over timed region: " + json.dumps(
        {k: round(v, 3) if isinstance(v, float) else v
         for k, v in s.items()}))
    value = n_total * args.steps / elapsed / 1e6
    roofline = None
    if s["n_scatter"] > 0 and s["ms_scatter"] > 0:
        achieved = s["bytes_scatter"] / s["ms_scatter"] * 1e-6
        roofline = {"bound": "hbm", "achieved": round(achieved, 1),
                    "peak": HBM_PEAK_SPEC, "unit": "GB/s",
                    "frac": round(achieved / HBM_PEAK_SPEC, 4),
                    "frac_vs_measured_copy":
                        round(achieved / HBM_PEAK_MEASURED, 4),
                    "traffic": None, "kernel": "k_scatter",
                    "ms_per_launch": round(s["ms_scatter"] / s["n_scatter"],
                                           4),
                    "launches": int(s["n_scatter"])}
    print(json.dumps({
        "metric": "M KV-pairs/sec through out-of-core reduceByKey",
        "value": round(value, 2), "unit": "M KV-pairs/s", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": round(elapsed * 1000 / args.steps, 3),
        "higher_is_better": True, "scaling": "weak", "vs_baseline": None,
        "dtype": "int64+f64", "data": "synthetic",
        "config": {"workload": f"out-of-core-reduceByKey-{n_total}-pairs-"
                               f"2^26-universe (>=2^32 row, SURVEY.md "
                               f"§8(f)2; step includes chunked gen + "
                               f"ingest + conservation check)",
                   "op": "bigreduce", "n_pairs_per_gpu": n_total,
                   "chunk": CHUNK, "key_universe": "2^26",
                   "n_uniques": int(nu), "seed": SEED,
                   "parallelism": "single-gpu",
                   "engine": "libdprk.so (HIP gfx950)"},
        "roofline": roofline, "cpu_baseline": None}))
    eng.big_free()
    eng.close()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--n-pairs", type=int, default=None,
                    help="pairs per GPU (default: workload-specific)")
    ap.add_argument("--op", choices=["reduce", "sort", "join", "zipf",
                                     "wordcount", "bigreduce"],
                    default="reduce",
                    help="reduce=config 2 (headline), sort=config 3 "
                         "(1B keys-only LSD sort), join=config 4 shape, "
                         "zipf=config 5 (1B Zipf(1.1) skewed reduceByKey); "
                         "wordcount=config 1 semantics on the GPU "
                         "string-key path; bigreduce=out-of-core >=2^32 "
                         "pairs (both N=1 only)")
    ap.add_argument("--config1", action="store_true",
                    help="run the CPU plumbing check instead")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()
    if args.config1:
        run_config1()
        return
    if args.op == "wordcount":
        assert args.gpus == 1, "wordcount leg is single-GPU"
        run_wordcount(args)
        return
    if args.op == "bigreduce":
        assert args.gpus == 1, "bigreduce leg is single-GPU"
        run_bigreduce(args)
        return

    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        # Self-bootstrap: plain `python bench.py
        # --gpus N` relaunches itself under torch.distributed.run, one
        # rank per GPU, so the driver's literal command measures N GPUs
        # instead of silently clamping to 1.
        # port range disjoint from the fixed test ports (295xx/296xx)
        port = 25000 + (os.getpid() % 2000)
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={args.gpus}", "--master-addr=127.0.0.1",
               f"--master-port={port}",
               os.path.abspath(__file__)] + sys.argv[1:]
        log(f"bench: bootstrapping {args.gpus} ranks via torch.distributed.run")
        os.execv(sys.executable, cmd)

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    nranks = max(world, 1)
    if args.gpus != nranks and world == 1:
        nranks = 1  # torchrun-launched with WORLD_SIZE=1: honest N=1

    dist = None
    torch = None
    if nranks > 1:
        import torch as _torch
        import torch.distributed as _dist
        torch = _torch
        dist = _dist
        dist.init_process_group(backend="gloo")  # control plane only

    if os.environ.get("DPRK_BENCH_SIM"):
        # CPU control-plane dry run (tests only): exercises the exact
        # torchrun launch path with no GPU; output is marked SIMULATED.
        from dpark_amd.testing import SimEngine as Engine
    else:
        from dpark_amd.engine import Engine  # fails loudly if .so missing

    defaults = {"reduce": N_PAIRS_PER_GPU, "sort": 1_000_000_000,
                "join": 500_000_000, "zipf": 1_000_000_000}
    n_local = args.n_pairs or defaults[args.op]
    with_vals = args.op != "sort"
    eng = Engine(device=local_rank)
    # multi-rank: a rank receives a hash-dependent share — reserve slack
    # (uniform keys: ~+0.03%; generous 4% margin. Heavy zipf skew can
    # exceed any margin — hot-bucket splitting is a next-row item,
    # SURVEY.md §7 M4 — dprk_exchange then fails with a clear error.)
    cap = n_local if nranks == 1 else int(n_local * 1.04) + 4096
    eng.reserve(cap, with_vals=with_vals)
    if args.op == "zipf":
        from oracle.synth import build_alias_table, zipf_weights
        log("building Zipf(1.1) alias table over 2^24 keys ...")
        t0 = time.perf_counter()
        eng.set_alias_table(build_alias_table(zipf_weights(1 << 24, 1.1)))
        log(f"alias table built+uploaded in {time.perf_counter()-t0:.1f}s")
    eng_b = None
    if args.op == "join":
        eng_b = Engine(device=local_rank)
        eng_b.reserve(n_local, with_vals=True)

    if nranks > 1:
        # RCCL init: rank 0 creates the unique id, gloo broadcasts it
        if rank == 0:
            uid = Engine.comm_unique_id()
            t = torch.tensor(list(uid), dtype=torch.uint8)
        else:
            t = torch.zeros(128, dtype=torch.uint8)
        dist.broadcast(t, src=0)
        eng.comm_init(nranks, rank, bytes(t.tolist()))

    def barrier():
        if dist is not None:
            dist.barrier()

    def exchange_if_dist(e):
        # Balanced bucket exchange: partition into 256
        # hash buckets, all-gather the count matrix, deterministically
        # greedy-assign buckets to ranks (hot buckets stay whole but get
        # a rank largely to themselves), then one bucket-granular RCCL
        # all-to-allv.  Skew beyond whole-bucket balancing (SURVEY.md §7
        # M4): the plan degrades to (source, bucket) CELL granularity —
        # split buckets leave PARTIAL sums on several ranks, and the
        # caller runs a second (small) exchange + combine after the
        # local reduce (returns True when that is needed).
        if nranks <= 1:
            return False
        from dpark_amd.distshuffle import (plan_exchange, recv_matrix_for,
                                           recv_matrix_for_cells)
        e.sync_packed(dist)  # all ranks must ship the same key width
        nbuckets = 256
        send = e.partition(nbuckets)
        counts = torch.tensor(send.astype("int64"))
        mat = [torch.zeros(nbuckets, dtype=torch.int64)
               for _ in range(nranks)]
        dist.all_gather(mat, counts)              # count matrix (tiny)
        m = np.stack([r.numpy() for r in mat]).astype(np.uint64)
        kind, assign, loads = plan_exchange(m, nranks, cap)
        imbalance["max"] = max(imbalance.get("max", 0.0),
                               max(loads) / (sum(loads) / nranks))
        if kind == "whole":
            rm = recv_matrix_for(m, assign, rank)
            e.exchange_buckets(send, assign, rm, nranks)
            return False
        rm = recv_matrix_for_cells(m, assign, rank)
        e.exchange_buckets(send, assign[rank], rm, nranks)
        return True

    def second_combine(e):
        # after the local reduce of a cell-split exchange: whole-bucket
        # exchange of the (small) reduced partials + one more combine
        from dpark_amd.distshuffle import greedy_assign, recv_matrix_for
        nbuckets = 256
        send = e.partition(nbuckets)
        counts = torch.tensor(send.astype("int64"))
        mat = [torch.zeros(nbuckets, dtype=torch.int64)
               for _ in range(nranks)]
        dist.all_gather(mat, counts)
        m = np.stack([r.numpy() for r in mat]).astype(np.uint64)
        bucket_rank, _ = greedy_assign(m.sum(axis=0), nranks)
        rm = recv_matrix_for(m, bucket_rank, rank)
        e.exchange_buckets(send, bucket_rank, rm, nranks)
        return e.reduce_by_key_sum()

    imbalance = {}

    def step(i):
        # regenerate the resident input (seed fixed per rank; deterministic)
        if args.op == "reduce":
            eng.gen_pairs(seed=SEED + rank, n=n_local,
                          key_universe=KEY_UNIVERSE)
            split = exchange_if_dist(eng)
            n_u = eng.reduce_by_key_sum()
            return second_combine(eng) if split else n_u
        if args.op == "sort":
            # config 3: 1B int64 keys uniform over full 64 bits, keys only
            eng.gen_pairs(seed=7 + rank, n=n_local, key_universe=0)
            if nranks > 1:
                # distributed sortByKey: sampled range splitters (dpark
                # RangePartitioner semantics) -> all-to-allv -> local sort
                eng.sync_packed(dist)
                sample = eng.sample_keys(4096)
                obj = [None] * nranks
                dist.all_gather_object(obj, sample)
                allsamp = np.sort(np.concatenate(obj))
                step_w = len(allsamp) / nranks
                splitters = np.array(
                    [allsamp[int(step_w * r)] for r in range(1, nranks)],
                    dtype=np.int64)
                send = eng.partition_range(splitters, nranks)
                counts = torch.tensor(send.astype("int64"))
                mat = [torch.zeros(nranks, dtype=torch.int64)
                       for _ in range(nranks)]
                dist.all_gather(mat, counts)
                recv = [int(mat[r][rank]) for r in range(nranks)]
                eng.exchange([int(x) for x in send], recv)
            eng.sort()
            return n_local
        if args.op == "zipf":
            # config 5: Zipf(alpha=1.1) skewed keys over 2^24
            eng.gen_pairs(seed=11 + rank, n=n_local, key_universe=1 << 24,
                          dist=1)
            split = exchange_if_dist(eng)
            n_u = eng.reduce_by_key_sum()
            return second_combine(eng) if split else n_u
        # config 4 shape: two sides, keys from 2^26 universe, co-partition
        # (shared hash digits) -> sort both -> merge join.  BOTH sides
        # must use ONE bucket->rank map (key co-location); the plan is
        # built from the summed count matrices, whole buckets only (a
        # cell split would break co-partitioning).
        eng.gen_pairs(seed=13 + rank, n=n_local, key_universe=1 << 26)
        eng_b.gen_pairs(seed=17 + rank, n=n_local, key_universe=1 << 26)
        if nranks > 1:
            from dpark_amd.distshuffle import greedy_assign, recv_matrix_for
            eng.sync_packed(dist)
            eng_b.sync_packed(dist)
            nbuckets = 256
            mats = []
            sends = []
            for e in (eng, eng_b):
                send = e.partition(nbuckets)
                sends.append(send)
                counts = torch.tensor(send.astype("int64"))
                mat = [torch.zeros(nbuckets, dtype=torch.int64)
                       for _ in range(nranks)]
                dist.all_gather(mat, counts)
                mats.append(np.stack([r.numpy() for r in mat])
                            .astype(np.uint64))
            bucket_rank, loads = greedy_assign(mats[0].sum(axis=0)
                                               + mats[1].sum(axis=0), nranks)
            imbalance["max"] = max(imbalance.get("max", 0.0),
                                   max(loads) / (sum(loads) / nranks))
            for e, m, send in zip((eng, eng_b), mats, sends):
                rm = recv_matrix_for(m, bucket_rank, rank)
                e.exchange_buckets(send, bucket_rank, rm, nranks)
        return eng.join_count(eng_b)

    # warmup
    nu = None
    for i in range(args.warmup):
        nu = step(i)
    log(f"rank {rank}: warmup done, {nu} uniques")

    eng.stats_reset()
    barrier()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    # engine ops are synchronous on their stream (each drains via
    # hipStreamSynchronize), so the device is idle here; barrier then
    # max-over-ranks wall time
    barrier()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist is not None:
        t = torch.tensor([elapsed])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    s = eng.stats()
    log("engine stats over timed region: " + json.dumps(
        {k: round(v, 3) if isinstance(v, float) else v
         for k, v in s.items()}))
    if rank != 0:
        return

    total_pairs = n_local * nranks * args.steps
    if args.op == "join":
        total_pairs *= 2  # pairs processed across both sides
    value = total_pairs / elapsed / 1e6
    ms_per_step = elapsed * 1000.0 / args.steps
    gen_ms = s["ms_gen"] / args.steps
    pipeline_ms = ms_per_step - gen_ms

    # roofline of the dominant kernel (scatter): ALGORITHMIC bytes per
    # launch / avg HIP-event launch duration (16 B
    # read + 16 B write per pair per scatter launch)
    roofline = None
    if s["n_scatter"] > 0 and s["ms_scatter"] > 0:
        achieved = s["bytes_scatter"] / s["ms_scatter"] * 1e-6  # GB/s
        roofline = {
            "bound": "hbm",
            "achieved": round(achieved, 1),
            "peak": HBM_PEAK_SPEC,
            "unit": "GB/s",
            "frac": round(achieved / HBM_PEAK_SPEC, 4),
            "frac_vs_measured_copy": round(achieved / HBM_PEAK_MEASURED, 4),
            "traffic": None,  # PMC counters live in profiles/ (separate
                              # rocprofv3 --pmc runs)
            "kernel": "k_scatter",
            "ms_per_launch": round(s["ms_scatter"] / s["n_scatter"], 4),
            "launches": int(s["n_scatter"]),
        }

    cpu_baseline = None
    if not args.skip_cpu_baseline:
        from oracle import cpu_baseline as cb
        log("cpu baseline: dpark-semantics restatement on host cores ...")
        cpu_baseline = cb.measure(n_sample=8_000_000, seed=SEED,
                                  key_universe=KEY_UNIVERSE)

    out = {
        "metric": {"reduce": "M KV-pairs/sec through reduceByKey shuffle",
                   "sort": "M keys/sec through sortByKey (LSD radix)",
                   "join": "M input KV-pairs/sec through join",
                   "zipf": "M KV-pairs/sec through skewed reduceByKey"}[args.op],
        "value": round(value, 2),
        "unit": {"reduce": "M KV-pairs/s", "sort": "M keys/s",
                 "join": "M input KV-pairs/s",
                 "zipf": "M KV-pairs/s"}[args.op],
        "n_gpus": nranks,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published reference number
        "dtype": "int64+f64",
        "data": ("SIMULATED-no-gpu" if os.environ.get("DPRK_BENCH_SIM") else "synthetic"),
        "config": {
            "workload": {
                "reduce": "reduceByKey-100M-int64-f64-uniform-2^24-per-gpu "
                          "(BASELINE.json configs[1])",
                "sort": "sortByKey-1B-int64-keys-full64 "
                        "(BASELINE.json configs[2])",
                "join": "join-2x500M-int64-f64-universe-2^26 "
                        "(BASELINE.json configs[3] shape)",
                "zipf": "reduceByKey-1B-zipf1.1-2^24 "
                        "(BASELINE.json configs[4])",
            }[args.op],
            "op": args.op,
            "n_pairs_per_gpu": n_local,
            "key_universe": {"reduce": "2^24", "sort": "2^64",
                             "join": "2^26", "zipf": "2^24"}[args.op],
            "seed": SEED,
            "parallelism": f"dp{nranks}" if nranks > 1 else "single-gpu",
            "step_includes_gen_ms": round(gen_ms, 3),
            "pipeline_ms_per_step": round(pipeline_ms, 3),
            "sort_passes": int(s["sort_passes"]),
            "exchange_imbalance": (round(imbalance["max"], 4)
                                   if imbalance else None),
            "engine": "libdprk.so (HIP gfx950)",
        },
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
