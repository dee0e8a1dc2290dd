"""Multi-process (world_size 2) CPU tests of the exchange-plan semantics:
gloo backend, numpy payloads — the N>1 path minus the GPU kernels
(gpurun has one GPU, so this is where the distributed
logic is kept correct-by-construction)."""

import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from dpark_amd.distshuffle import rank_of_keys, recv_counts  # noqa: E402


def test_rank_function_properties():
    from oracle.synth import gen_pairs
    keys, _ = gen_pairs(3, 100_000, 1 << 24)
    for nranks in (1, 2, 4, 8):
        r = rank_of_keys(keys, nranks)
        assert r.min() >= 0 and r.max() < nranks
        # deterministic
        assert np.array_equal(r, rank_of_keys(keys, nranks))
        # same key -> same rank (the property reduceByKey correctness
        # rests on: all copies of a key land on one rank)
        uk, inv = np.unique(keys, return_inverse=True)
        per_key = rank_of_keys(uk, nranks)
        assert np.array_equal(r, per_key[inv])
        # roughly balanced for uniform keys
        if nranks > 1:
            counts = np.bincount(r, minlength=nranks)
            assert counts.max() / counts.min() < 1.05


def test_recv_counts_matrix():
    cm = [[5, 3], [2, 7]]
    assert recv_counts(cm, 0) == [5, 2]
    assert recv_counts(cm, 1) == [3, 7]


def _dist_worker(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world)
        import torch.distributed as dist
        from dpark_amd.distshuffle import exchange_with_torch_dist
        from oracle.synth import gen_pairs
        from oracle import second_oracle as so
        import torch

        dist.init_process_group(backend="gloo", rank=rank, world_size=world)
        n_local = 50_000
        keys, vals = gen_pairs(seed=100 + rank, n=n_local,
                               key_universe=1 << 12)
        rk, rv = exchange_with_torch_dist(dist, keys, vals, world, rank)
        # every received key must belong to this rank
        assert np.all(rank_of_keys(rk, world) == rank)
        # local reduce of received shard
        uk, sums = so.reduce_by_key_sum(rk, rv)
        # gather all shards' results and compare with the global oracle
        obj = [None] * world
        dist.all_gather_object(obj, (uk, sums))
        if rank == 0:
            gk = np.concatenate([o[0] for o in obj])
            gs = np.concatenate([o[1] for o in obj])
            order = np.argsort(gk)
            gk, gs = gk[order], gs[order]
            all_k = np.concatenate([gen_pairs(100 + r, n_local, 1 << 12)[0]
                                    for r in range(world)])
            all_v = np.concatenate([gen_pairs(100 + r, n_local, 1 << 12)[1]
                                    for r in range(world)])
            wk, ws = so.reduce_by_key_sum(all_k, all_v)
            assert np.array_equal(gk, wk)
            np.testing.assert_allclose(gs, ws, rtol=1e-9)
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


@pytest.mark.timeout(180)
def test_world2_exchange_reduce_matches_oracle():
    torch = pytest.importorskip("torch")
    assert torch  # noqa
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29511
    procs = [ctx.Process(target=_dist_worker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=170)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, errs
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]


def test_greedy_assign_balances_hot_bucket():
    """a zipf-hot bucket (just under a rank's mean
    share) is isolated on one rank and the rest pack around it."""
    from dpark_amd.distshuffle import greedy_assign
    n = 1_000_000_000
    sizes = np.full(256, (n - int(0.12 * n)) // 255, dtype=np.uint64)
    sizes[7] = int(0.12 * n)                      # 12% hot bucket
    assign, loads = greedy_assign(sizes, 8)
    assert len(set(int(a) for a in assign)) == 8
    mean = sum(loads) / 8
    assert max(loads) / mean < 1.05               # balanced despite skew
    # deterministic
    a2, _ = greedy_assign(sizes, 8)
    assert np.array_equal(assign, a2)
    # naive rank=bucket%8 for comparison would be imbalanced:
    naive = np.zeros(8)
    for b, sz in enumerate(sizes):
        naive[b % 8] += int(sz)
    assert max(naive) / (sum(naive) / 8) > 1.8


def test_recv_matrix_partition():
    from dpark_amd.distshuffle import greedy_assign, recv_matrix_for
    rng = np.random.default_rng(4)
    m = rng.integers(0, 1000, size=(4, 64)).astype(np.uint64)
    assign, _ = greedy_assign(m.sum(axis=0), 4)
    got = sum(int(recv_matrix_for(m, assign, me).sum()) for me in range(4))
    assert got == int(m.sum())                    # every element delivered


# ---------------------------------------------------------------------------
# skew-safe plan (SURVEY.md §7 M4 / VERDICT r01 item 4)
# ---------------------------------------------------------------------------

def test_plan_exchange_whole_vs_cells():
    from dpark_amd.distshuffle import (greedy_assign_cells, plan_exchange,
                                       recv_matrix_for_cells)
    # uniform: whole-bucket plan
    m = np.full((4, 64), 100, dtype=np.uint64)
    kind, assign, loads = plan_exchange(m, 4, cap=10_000)
    assert kind == "whole" and max(loads) <= 10_000
    # one bucket holds 70% of everything -> whole-bucket greedy cannot
    # fit a rank: the plan degrades to cells
    m2 = np.full((4, 64), 10, dtype=np.uint64)
    m2[:, 5] = 2000                               # hot bucket, all sources
    total = int(m2.sum())
    cap = total // 4 + 600                        # no rank can hold bucket 5
    kind2, cells, loads2 = plan_exchange(m2, 4, cap)
    assert kind2 == "cells"
    assert cells.shape == m2.shape
    assert max(loads2) <= cap
    # every cell delivered exactly once
    got = sum(int(recv_matrix_for_cells(m2, cells, me).sum())
              for me in range(4))
    assert got == total
    # hot bucket actually split across ranks
    assert len(set(int(x) for x in cells[:, 5])) > 1
    # deterministic
    k3, c3, _ = plan_exchange(m2, 4, cap)
    assert k3 == "cells" and np.array_equal(cells, c3)
    # true capacity exhaustion raises
    with pytest.raises(ValueError, match="capacity"):
        greedy_assign_cells(m2, 4, cap=total // 8)


def _skew_worker(rank, world, port, fail_q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch
        import torch.distributed as dist
        from dpark_amd.hashspec import splitmix64
        from dpark_amd.distshuffle import (plan_exchange, greedy_assign,
                                           recv_matrix_for)
        from oracle import second_oracle as so

        dist.init_process_group(backend="gloo", rank=rank, world_size=world)
        nbuckets = 256

        def bucket_of(keys):
            mix = splitmix64(np.asarray(keys, np.int64).view(np.uint64))
            return (mix >> np.uint64(56)).astype(np.int64)   # top 8 bits

        def bucketize(keys, vals):
            b = bucket_of(keys)
            order = np.argsort(b, kind="stable")
            k, v = keys[order], vals[order]
            counts = np.bincount(b, minlength=nbuckets).astype(np.uint64)
            offs = np.concatenate([[0], np.cumsum(counts)]).astype(np.int64)
            return k, v, counts, offs

        def allgather_m(counts):
            mat = [torch.zeros(nbuckets, dtype=torch.int64)
                   for _ in range(world)]
            dist.all_gather(mat, torch.tensor(counts.astype("int64")))
            return np.stack([r.numpy() for r in mat]).astype(np.uint64)

        def move(keys, vals, dest_of_bucket):
            """gloo object exchange of bucket slices (test-only mirror of
            dprk_exchange_buckets)."""
            k, v, counts, offs = bucketize(keys, vals)
            outbound = []
            for me in range(world):
                sel = [b for b in range(nbuckets)
                       if dest_of_bucket[b] == me and counts[b]]
                if sel:
                    outbound.append((
                        np.concatenate([k[offs[b]:offs[b + 1]] for b in sel]),
                        np.concatenate([v[offs[b]:offs[b + 1]] for b in sel])))
                else:
                    outbound.append((np.empty(0, np.int64),
                                     np.empty(0, np.float64)))
            gathered = [None] * world
            dist.all_gather_object(gathered, outbound)
            rk = np.concatenate([gathered[s][rank][0] for s in range(world)])
            rv = np.concatenate([gathered[s][rank][1] for s in range(world)])
            return rk, rv

        # adversarial skew: ONE key is ~75% of the whole dataset
        n_local = 40_000
        rng = np.random.default_rng(500 + rank)
        hot = 42
        keys = np.where(rng.random(n_local) < 0.75, hot,
                        rng.integers(0, 1 << 20, n_local)).astype(np.int64)
        vals = rng.random(n_local)
        # cap: above total/world (feasible) but BELOW the global hot
        # bucket (~75% of 2*n_local) so whole-bucket assignment fails
        cap = int(n_local * 1.05)
        _, _, counts, _ = bucketize(keys, vals)
        m = allgather_m(counts)
        kind, assign, loads = plan_exchange(m, world, cap)
        assert kind == "cells", "test must exercise the split path"
        assert max(loads) <= cap
        rk, rv = move(keys, vals, assign[rank])
        assert len(rk) <= cap
        # local (partial) reduce
        uk, sums = so.reduce_by_key_sum(rk, rv)
        # second combine: whole-bucket exchange of the reduced partials
        _, _, c2, _ = bucketize(uk, sums)
        m2 = allgather_m(c2)
        br2, _ = greedy_assign(m2.sum(axis=0), world)
        fk, fv = move(uk, sums, br2)
        fk2, fs2 = so.reduce_by_key_sum(fk, fv)
        # gather and compare against the global oracle
        obj = [None] * world
        dist.all_gather_object(obj, (fk2, fs2))
        if rank == 0:
            gk = np.concatenate([o[0] for o in obj])
            gs = np.concatenate([o[1] for o in obj])
            assert len(gk) == len(np.unique(gk)), "key split across ranks"
            order = np.argsort(gk)
            gk, gs = gk[order], gs[order]
            all_k, all_v = [], []
            for r in range(world):
                rng_r = np.random.default_rng(500 + r)
                kk = np.where(rng_r.random(n_local) < 0.75, hot,
                              rng_r.integers(0, 1 << 20, n_local)
                              ).astype(np.int64)
                all_k.append(kk)
                all_v.append(rng_r.random(n_local))
            wk, ws = so.reduce_by_key_sum(np.concatenate(all_k),
                                          np.concatenate(all_v))
            assert np.array_equal(gk, wk)
            np.testing.assert_allclose(gs, ws, rtol=1e-9)
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # pragma: no cover
        fail_q.put(f"rank {rank}: {type(e).__name__}: {e}")
        raise


@pytest.mark.timeout(180)
def test_world2_skew_split_two_stage_reduce():
    """Adversarial skew (one key = 75% of the data, hot bucket > any
    rank's capacity): the cell-split plan + second combine matches the
    global oracle — the case r01 errored on."""
    torch = pytest.importorskip("torch")
    assert torch  # noqa
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_skew_worker, args=(r, 2, 29513, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=170)
    errs = []
    while not q.empty():
        errs.append(q.get())
    assert not errs, errs
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
