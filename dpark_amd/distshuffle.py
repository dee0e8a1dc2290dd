"""Multi-rank shuffle plan — the host-side half of the exchange
(SURVEY.md §8(e)).  The data path is dprk_partition + dprk_exchange (RCCL
all-to-allv inside the engine); this module holds the pure-plan logic
(rank function mirror, count matrix -> recv counts) so the N>1 semantics
are testable on CPU with gloo (tests/test_dist_cpu.py), world_size > 1,
no GPU.
"""

from __future__ import annotations

import numpy as np


def rank_of_keys(keys, nranks: int):
    """Numpy mirror of the engine's rank digit: TOP log2(nranks) bits of
    splitmix64(key) (d_digit mode 1 in dpark_amd/csrc/dprk.hip).  The GPU
    HashPartitioner — layout intentionally differs from reference
    portable_hash."""
    from dpark_amd.hashspec import splitmix64
    assert nranks >= 1 and (nranks & (nranks - 1)) == 0
    if nranks == 1:
        return np.zeros(len(keys), dtype=np.int64)
    bits = nranks.bit_length() - 1
    mix = splitmix64(np.asarray(keys, dtype=np.int64).view(np.uint64))
    return (mix >> np.uint64(64 - bits)).astype(np.int64)


def recv_counts(count_matrix, rank: int):
    """count_matrix[r] = rank r's send counts (length nranks).  Rank r
    receives column r."""
    return [int(count_matrix[src][rank]) for src in range(len(count_matrix))]


def exchange_with_torch_dist(dist, keys, vals, nranks, rank):
    """CPU/gloo reference exchange used by the world_size>1 CPU tests:
    partition locally by rank_of_keys, all-to-allv via isend/irecv pairs
    (gloo has no all_to_all).  Returns the received (keys, vals)."""
    import torch
    r = rank_of_keys(keys, nranks)
    order = np.argsort(r, kind="stable")
    keys, vals = keys[order], vals[order]
    send = np.bincount(r, minlength=nranks)
    cm = [torch.zeros(nranks, dtype=torch.int64) for _ in range(nranks)]
    dist.all_gather(cm, torch.tensor(send, dtype=torch.int64))
    recv = recv_counts(cm, rank)
    offs = np.concatenate([[0], np.cumsum(send)]).astype(np.int64)
    out_k, out_v = [], []
    reqs = []
    send_bufs = []
    for peer in range(nranks):
        kb = torch.tensor(keys[offs[peer]:offs[peer + 1]], dtype=torch.int64)
        vb = torch.tensor(vals[offs[peer]:offs[peer + 1]],
                          dtype=torch.float64)
        send_bufs.append((kb, vb))
    recv_bufs = [(torch.zeros(recv[p], dtype=torch.int64),
                  torch.zeros(recv[p], dtype=torch.float64))
                 for p in range(nranks)]
    for peer in range(nranks):
        if peer == rank:
            continue
        if len(send_bufs[peer][0]):
            reqs.append(dist.isend(send_bufs[peer][0], dst=peer, tag=0))
            reqs.append(dist.isend(send_bufs[peer][1], dst=peer, tag=1))
        if recv[peer]:
            reqs.append(dist.irecv(recv_bufs[peer][0], src=peer, tag=0))
            reqs.append(dist.irecv(recv_bufs[peer][1], src=peer, tag=1))
    for q in reqs:
        q.wait()
    recv_bufs[rank] = send_bufs[rank]
    for p in range(nranks):
        out_k.append(recv_bufs[p][0].numpy())
        out_v.append(recv_bufs[p][1].numpy())
    return np.concatenate(out_k), np.concatenate(out_v)


def greedy_assign(bucket_totals, nranks: int):
    """Deterministic greedy balancing: buckets sorted by global size
    (descending, ties by index) each go to the currently lightest rank
    (ties by rank id).  Identical on every rank given the all-gathered
    count matrix — the SURVEY.md §8(e) hot-bucket mitigation: a hot
    bucket stays whole, the rest pack around it."""
    totals = np.asarray(bucket_totals, dtype=np.uint64)
    order = np.lexsort((np.arange(len(totals)), -totals.astype(np.int64)))
    load = [0] * nranks
    assign = np.zeros(len(totals), dtype=np.int32)
    for b in order:
        r = min(range(nranks), key=lambda x: (load[x], x))
        assign[b] = r
        load[r] += int(totals[b])
    return assign, load


def recv_matrix_for(count_matrix, bucket_rank, me: int):
    """count_matrix[s][b] = rank s's local size of bucket b.  Returns the
    [nranks, nbuckets] receive matrix for rank `me` (zeros elsewhere)."""
    m = np.asarray(count_matrix, dtype=np.uint64)
    mask = (np.asarray(bucket_rank) == me)
    return m * mask[None, :].astype(np.uint64)


# ---------------------------------------------------------------------------
# skew-safe exchange plan (SURVEY.md §7 M4 / VERDICT r01 item 4): when a
# single bucket exceeds a rank's recv capacity, split it at (source,
# bucket) CELL granularity — dprk_exchange_buckets already supports a
# per-source bucket->rank map, so no engine change is needed.  Keys of a
# split bucket then hold PARTIAL sums on several ranks; the caller runs a
# second (small) bucket exchange + combine over the locally-reduced data
# (two_stage_needed tells it whether to).
# ---------------------------------------------------------------------------

def greedy_assign_cells(count_matrix, nranks: int, cap: int):
    """Deterministic cell-granular balancing: cells (s, b) sorted by size
    descending (ties by (b, s)) each go to the lightest rank with room.
    Returns (cellassign [nranks, nbuckets] int32, loads).  Raises if the
    total exceeds nranks*cap (true capacity exhaustion)."""
    m = np.asarray(count_matrix, dtype=np.uint64)
    nr, nb = m.shape
    assert nr == nranks
    total = int(m.sum())
    if total > nranks * cap:
        raise ValueError(f"dataset {total} exceeds total capacity "
                         f"{nranks * cap}")
    s_idx, b_idx = np.meshgrid(np.arange(nr), np.arange(nb), indexing="ij")
    flat = np.stack([m.ravel().astype(np.int64), b_idx.ravel(),
                     s_idx.ravel()], axis=1)
    order = np.lexsort((flat[:, 2], flat[:, 1], -flat[:, 0]))
    load = [0] * nranks
    assign = np.zeros((nr, nb), dtype=np.int32)
    for row in order:
        sz, b, s = (int(flat[row, 0]), int(flat[row, 1]), int(flat[row, 2]))
        if sz == 0:
            continue
        cands = sorted(range(nranks), key=lambda x: (load[x], x))
        placed = False
        for r in cands:
            if load[r] + sz <= cap:
                assign[s, b] = r
                load[r] += sz
                placed = True
                break
        if not placed:
            raise ValueError(f"cell ({s},{b}) of {sz} does not fit any "
                             f"rank (cap {cap}, loads {load})")
    return assign, load


def plan_exchange(count_matrix, nranks: int, cap: int):
    """Choose the exchange plan from the all-gathered count matrix.
    Returns ("whole", bucket_rank[nb], loads) when whole-bucket greedy
    fits every rank under cap (the normal path — every key lands whole,
    no second combine), else ("cells", cellassign[nr, nb], loads) — the
    caller must run a second combine (split buckets leave partial sums
    on several ranks).  Deterministic: a pure function of (m, nranks,
    cap) identical on every rank."""
    m = np.asarray(count_matrix, dtype=np.uint64)
    bucket_rank, loads = greedy_assign(m.sum(axis=0), nranks)
    if max(loads) <= cap:
        return "whole", bucket_rank, loads
    cellassign, loads = greedy_assign_cells(m, nranks, cap)
    return "cells", cellassign, loads


def recv_matrix_for_cells(count_matrix, cellassign, me: int):
    m = np.asarray(count_matrix, dtype=np.uint64)
    return m * (np.asarray(cellassign) == me).astype(np.uint64)
