/* dprk — MI355X-native shuffle/sort/group-by engine C-ABI.
 *
 * This boundary replaces douban/dpark's shuffle seam (all citations [U]
 * symbol-level; the reference mount is empty — SURVEY.md §0, §8(b)):
 *   - dpark/task.py :: ShuffleMapTask.run bucket loop  -> dprk_partition
 *   - dpark/shuffle.py :: LocalFileShuffle write + ShuffleFetcher.fetch
 *     (HTTP/file shuffle exchange)                     -> dprk_exchange
 *       (one RCCL all-to-allv over xGMI)
 *   - dpark/shuffle.py :: Merger / sort-based external merger
 *     (reduce-side combine)                            -> dprk_sort_pairs +
 *       dprk_reduce_by_key_sum (LSD radix sort + segmented reduce)
 *   - dpark/rdd.py :: RDD.sortByKey per-partition sort -> dprk_sort_pairs
 *   - dpark/rdd.py :: RDD.join / CoGroupedRDD + CoGroupMerger
 *                                                      -> dprk_join
 *   - dpark/dependency.py :: HashPartitioner/portable_hash
 *     -> splitmix64 digits device-side (partition layout intentionally
 *        differs; collected results are layout-independent, SURVEY.md
 *        §8(a2))
 *
 * Conventions: every function returns int status (0 = OK); on failure
 * dprk_last_error() describes the most recent error (per-process).  Caller
 * owns host buffers (numpy); the engine owns all device memory.  One
 * dprk_ctx per GPU; a ctx is not thread-safe; ops are synchronous on the
 * ctx's private HIP stream.  Loaded via ctypes from dpark_amd/engine.py;
 * the reference-side binding a dpark maintainer would add is shown in
 * INTEGRATION.md.  No torch types anywhere.
 */

#ifndef DPRK_H
#define DPRK_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct dprk_ctx dprk_ctx;

/* ---- lifecycle ------------------------------------------------------- */
int dprk_init(dprk_ctx **out, int device_id);
void dprk_free(dprk_ctx *ctx);
const char *dprk_last_error(void);
/* engine/version marker so the host can assert the native path is live */
const char *dprk_build_info(void);

/* Reserve device capacity for up to max_n pairs (with_vals=0: keys only).
 * Must be called before put/gen; re-reservation allowed. */
int dprk_reserve(dprk_ctx *ctx, uint64_t max_n, int with_vals);

/* ---- data in/out ----------------------------------------------------- */
/* Device-side PRNG per the synth spec (oracle/synth.py is the CPU mirror):
 * dist 0 = uniform over key_universe (power of two; 0 => full 64-bit),
 * dist 1 = alias-table sampling (zipf etc.; dprk_set_alias_table).
 */
int dprk_gen_pairs(dprk_ctx *ctx, uint64_t seed, uint64_t n, int dist,
                   uint64_t key_universe);
/* dist 1 (zipf etc.) samples a caller-provided alias table (built host
 * side — oracle/synth.py build_alias_table is the normative builder, so
 * the CPU mirror is bit-identical): table[slot] = (thresh<<32)|alias. */
int dprk_set_alias_table(dprk_ctx *ctx, const uint64_t *table,
                         uint64_t universe);
int dprk_put_pairs(dprk_ctx *ctx, const int64_t *keys, const double *vals,
                   uint64_t n); /* vals may be NULL for keys-only */
int dprk_get_pairs(dprk_ctx *ctx, int64_t *keys, double *vals, uint64_t cap,
                   uint64_t *n_out); /* D2H of the current dataset */
/* String-key path (SURVEY.md §8(f) rank 1 — TextFileRDD -> wordcount
 * [U]): device-hash n_words bytes keys (packed buffer + n_words+1 byte
 * offsets; spec mirrored in dpark_amd/hashspec.py) into the dataset's
 * int64 keys, vals uploaded alongside.  The hashes then flow through
 * sort/reduce/group; hash->word resolution is host-side via exemplar
 * rows (dpark_amd/strings.py documents the collision bound). */
int dprk_put_bytes_keys(dprk_ctx *ctx, const uint8_t *buf, uint64_t nbytes,
                        const uint64_t *offs, uint64_t n_words,
                        const double *vals);
/* current dataset size */
int dprk_count(dprk_ctx *ctx, uint64_t *n_out);

/* Packed-u32 key transport (internal layout, not a semantics change):
 * when every key of the dataset lies in [0, 2^32) the engine stores and
 * moves keys as uint32 (25% fewer bytes per radix pass; detected
 * input-deterministically at gen/put time).  is_packed reads the state;
 * set_packed forces it (want=0 always legal; want=1 errors if any key
 * does not fit).  Multi-rank: all ranks must agree on the key width
 * before dprk_exchange* — the host layer all-reduces the flags and
 * aligns disagreeing ranks (Engine.sync_packed). */
int dprk_is_packed(dprk_ctx *ctx, int *out);
int dprk_set_packed(dprk_ctx *ctx, int want);

/* gen with an index offset into the synthetic stream (x_i = (seed<<32)
 * + start + i): out-of-core chunks continue ONE stream; start=0 ==
 * dprk_gen_pairs. */
int dprk_gen_pairs_at(dprk_ctx *ctx, uint64_t seed, uint64_t start,
                      uint64_t n, int dist, uint64_t key_universe);
/* deterministic f64 sum of the current values (conservation checks) */
int dprk_val_sum(dprk_ctx *ctx, double *out);

/* ---- out-of-core reduceByKey (n >= 2^32) ----
 * Chunks (< 2^32 each, via gen_pairs_at/put_pairs) are hash-partitioned
 * into device bucket ARENAS (every key lands wholly in one bucket);
 * big_reduce then runs the normal in-core sort + segmented reduce per
 * bucket, compacting (unique key, f64 sum) outputs forward.  Output
 * order: bucket-major (hash digit), key-sorted within a bucket.
 * dpark's external sorted-run Merger is the semantic precedent [U]. */
int dprk_big_begin(dprk_ctx *ctx, uint64_t total_cap, int nbuckets);
int dprk_big_add_chunk(dprk_ctx *ctx);   /* consumes the current dataset */
int dprk_big_reduce(dprk_ctx *ctx, uint64_t *n_out);
int dprk_big_get(dprk_ctx *ctx, int64_t *keys, double *vals, uint64_t cap,
                 uint64_t *n_out);
int dprk_big_val_sum(dprk_ctx *ctx, double *out);
int dprk_big_free(dprk_ctx *ctx);

/* ---- the hot path ---------------------------------------------------- */
/* Radix-partition the current dataset into nbuckets contiguous segments by
 * the TOP bits of splitmix64(key) (the GPU HashPartitioner — SURVEY.md
 * §8(a2)); bucket_counts (host, nbuckets entries) receives the segment
 * sizes.  nbuckets must be a power of two <= 256. */
int dprk_partition(dprk_ctx *ctx, int nbuckets, uint64_t *bucket_counts);

/* Range partition by nbuckets-1 sorted splitters (bucket r = #{j : key >=
 * splitters[j]}) — the engine RangePartitioner (dpark/dependency.py ::
 * RangePartitioner [U]); order-preserving across ranks, used by the
 * distributed sortByKey (splitters from dprk_sample_keys + host merge,
 * mirroring dpark's driver-side sampling). */
int dprk_partition_range(dprk_ctx *ctx, const int64_t *splitters,
                         int nbuckets, uint64_t *bucket_counts);
/* Evenly strided sample of the current keys (D2H). */
int dprk_sample_keys(dprk_ctx *ctx, uint64_t n_samples, int64_t *out_host);

/* LSD radix sort of the current dataset by key, ascending (int64 order),
 * values carried.  Passes over constant digit positions are skipped (the
 * pass count actually executed is reported via dprk_stats). */
int dprk_sort_pairs(dprk_ctx *ctx);

/* reduceByKey(+): sort (if needed) + segmented reduce.  Replaces the
 * current dataset with (unique_keys_sorted, per-key float64 sums); n_out
 * receives the number of unique keys.  Deterministic fixed-tree float64
 * reduction: run-to-run bit-identical; vs a sequential CPU oracle the
 * contract is <=1e-6 relative. */
int dprk_reduce_by_key_sum(dprk_ctx *ctx, uint64_t *n_out);

/* groupByKey structure: sort (if needed), then expose grouped layout:
 * unique keys, per-key counts and start offsets into the (sorted) values
 * array.  D2H accessors below. */
int dprk_group_by_key(dprk_ctx *ctx, uint64_t *n_groups_out);
int dprk_get_groups(dprk_ctx *ctx, int64_t *unique_keys, uint64_t *starts,
                    uint64_t *counts, uint64_t cap, uint64_t *n_out);

/* Inner join of ctx_a's and ctx_b's datasets (same device): sorts both,
 * then per-key cross product (dpark join semantics — SURVEY.md §3.3).
 * Result (k, va, vb) is held by ctx_a; fetch with dprk_get_join. */
int dprk_join(dprk_ctx *ctx_a, dprk_ctx *ctx_b, uint64_t *n_out);
int dprk_get_join(dprk_ctx *ctx, int64_t *keys, double *va, double *vb,
                  uint64_t cap, uint64_t *n_out);

/* ---- multi-GPU exchange (RCCL over xGMI) ------------------------------ */
/* 128-byte opaque ncclUniqueId; rank 0 creates it, host control plane
 * distributes it (e.g. torch.distributed gloo broadcast — plumbing only) */
int dprk_comm_unique_id(char uid[128]);
int dprk_comm_init(dprk_ctx *ctx, int nranks, int rank, const char uid[128]);
/* All-to-allv of the current dataset, whose rank segments must have been
 * laid out by dprk_partition(ctx, nranks): one grouped ncclSend/ncclRecv
 * per peer (pairwise-direct over xGMI links — SURVEY.md §5).  send_counts/
 * recv_counts are host arrays of nranks entries (recv side from the
 * host-allgathered count matrix).  Replaces the current dataset with the
 * received one. */
int dprk_exchange(dprk_ctx *ctx, const uint64_t *send_counts,
                  const uint64_t *recv_counts, int nranks);
/* Bucket-granular all-to-allv under an arbitrary bucket->rank map (after
 * dprk_partition(ctx, nbuckets)): lets the host balance skewed buckets
 * across ranks (SURVEY.md §8(e) hot buckets) — each bucket stays whole,
 * so reduceByKey needs no second combine.  recv_matrix[s*nbuckets+b] =
 * elements source rank s sends this rank from bucket b. */
int dprk_exchange_buckets(dprk_ctx *ctx, const uint64_t *bucket_counts,
                          const int *bucket_rank, int nbuckets,
                          const uint64_t *recv_matrix, int nranks);

/* ---- measurement ------------------------------------------------------ */
typedef struct {
  /* accumulated HIP-event time (ms) and launch counts per kernel class,
   * plus algorithmic bytes attributed */
  double ms_gen, ms_hist, ms_tilehist, ms_scan, ms_scatter, ms_reduce,
      ms_runext, ms_join, ms_exchange, ms_other;
  uint64_t n_gen, n_hist, n_tilehist, n_scan, n_scatter, n_reduce, n_runext,
      n_join, n_exchange, n_other;
  uint64_t bytes_scatter;      /* algorithmic bytes over scatter launches */
  uint64_t bytes_total;        /* algorithmic bytes over the whole op mix */
  uint64_t sort_passes;        /* digit passes actually executed (last sort) */
} dprk_stats_t;

int dprk_stats(dprk_ctx *ctx, dprk_stats_t *out);
int dprk_stats_reset(dprk_ctx *ctx);

#ifdef __cplusplus
}
#endif

#endif /* DPRK_H */
