/* dprk — MI355X-native (gfx950/CDNA4) shuffle/sort/group-by engine.
 *
 * Brand-new kernels designed for CDNA4 (wave64, LDS-histogram radix
 * machinery, HBM-streaming layout) — NOT a translation: the reference
 * (douban/dpark) has zero GPU code; its hot path is Python dicts + pickle
 * + file shuffle (citations [U]).  What each piece
 * replaces is documented in include/dprk.h.
 *
 * Pipeline (DESIGN.md has the full data-layout and roofline derivation):
 *   k_gen / k_gen_hist / k_gen_alias   device PRNG (splitmix64 / alias
 *       table; numpy mirrors in oracle/synth.py are byte-identical);
 *       k_gen_hist also emits the pass-1 tile histogram
 *   k_minmax_or      OR/AND reduce of biased keys -> digit-skip detection
 *   k_tile_hist      per-tile 256-bin digit histogram      (8 B/pair read)
 *   k_seg_sums/k_scan_mid/k_tile_off   3-level scan of tile histograms
 *   k_scatter / k_scatter_lb  rank + LDS exchange + coalesced write
 *       (32 B/pair r+w); _lb (the default sort path) adds in-kernel
 *       decoupled-lookback offsets, dropping tile_hist + the scans
 *   k_sample / mode-2 digits           device RangePartitioner (sortByKey)
 *   k_reduce_tiles / k_stitchA-C / k_compact_runs  segmented f64 reduce
 *   k_scatter_reduce / k_fused_gather  fused last-pass reduce (opt-in)
 *   k_head_* / k_join_* / k_scanA-C    group/join machinery
 *   dprk_exchange / dprk_exchange_buckets   RCCL all-to-allv over xGMI
 * Sorting int64 ascending == unsigned radix on u = key XOR 0x8000...0.
 */

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cstdarg>
#include <cstdlib>
#include <cstdio>
#include <cstdint>
#include <cstring>

#include "../../include/dprk.h"

/* ------------------------------------------------------------------ */
/* tunables                                                            */
/* ------------------------------------------------------------------ */
#define RADIX 256
#define RADIX_BITS 8
/* TILE 8192 x 1024 threads is the measured default (r02 A/B, same box:
 * scatter 1.18 -> 0.87 ms/launch on config 2 — half the tiles, half the
 * barrier convoys, 32-element digit runs).  LDS: staging 96 KiB (u32) /
 * 144 KiB (i64) + 16 KiB hists fits 160 KiB at 1 WG/CU = 16 waves/CU,
 * same wave count as the old 4096x512 at 2 WG/CU. */
#ifndef TILE
#define TILE 8192          /* elements per radix-pass workgroup (tunable) */
#endif
#define NTHREADS 256       /* 4 waves of 64                              */
#ifndef NTHREADS_SC
#define NTHREADS_SC 1024   /* scatter workgroup: 16 waves (tunable)      */
#endif
#ifndef RTILE
#define RTILE 2048         /* elements per reduce/head workgroup (4 WG/CU
                              occupancy: the reduce is latency-bound)    */
#endif
#define EPT_R (RTILE / NTHREADS) /* 8 consecutive elements per thread    */
#define ELEMS_PER_THREAD (TILE / NTHREADS) /* 16 */
#define SCAN_CHUNK 16384   /* generic u64 scan: elements per workgroup   */
#define STITCH_CHUNK 4096  /* tiles per stitch workgroup                 */
#define MAX_EVTS 16384

#define SIGNFLIP 0x8000000000000000ull

/* ------------------------------------------------------------------ */
/* error handling                                                      */
/* ------------------------------------------------------------------ */
static char g_err[2048] = "";

static void set_err(const char *fmt, ...) {
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(g_err, sizeof(g_err), fmt, ap);
  va_end(ap);
}

extern "C" const char *dprk_last_error(void) { return g_err; }

#define DPRK_STR_(x) #x
#define DPRK_STR(x) DPRK_STR_(x)
extern "C" const char *dprk_build_info(void) {
  return "dprk/0.1 gfx950 TILE=" DPRK_STR(TILE);
}

#define HIP_CHECK(x)                                                       \
  do {                                                                     \
    hipError_t e_ = (x);                                                   \
    if (e_ != hipSuccess) {                                                \
      set_err("HIP error '%s' at %s:%d", hipGetErrorString(e_), __FILE__,  \
              __LINE__);                                                   \
      return 1;                                                            \
    }                                                                      \
  } while (0)

#define NCCL_CHECK(x)                                                      \
  do {                                                                     \
    ncclResult_t r_ = (x);                                                 \
    if (r_ != ncclSuccess) {                                               \
      set_err("RCCL error '%s' at %s:%d", ncclGetErrorString(r_),          \
              __FILE__, __LINE__);                                         \
      return 1;                                                            \
    }                                                                      \
  } while (0)

/* ------------------------------------------------------------------ */
/* context                                                             */
/* ------------------------------------------------------------------ */
enum KernKind {
  K_GEN = 0, K_HIST, K_TILEHIST, K_SCAN, K_SCATTER, K_REDUCE, K_RUNEXT,
  K_JOIN, K_EXCHANGE, K_OTHER, K_NKINDS
};

struct EvtRec { hipEvent_t a, b; int kind; };

struct dprk_ctx {
  int device = -1;
  hipStream_t stream = nullptr;

  /* current dataset: ping-pong buffers, engine-owned */
  int64_t *keys[2] = {nullptr, nullptr};
  double *vals[2] = {nullptr, nullptr};
  int cur = 0;
  uint64_t n = 0;
  uint64_t cap = 0;
  int with_vals = 0;
  int sorted = 0;
  int packed = 0;   /* keys currently stored as packed u32 (all keys in
                       [0, 2^32); input-deterministic — see KT traits) */
  /* lazy uniform generation (gen->sort fusion): dprk_gen_pairs(dist=0)
   * only RECORDS the stream; the first sort pass then GENERATES pairs
   * straight into partitioned layout (dpark's ShuffleMapTask.run never
   * materializes input before the bucket loop either [U]).  Any op
   * needing materialized data calls materialize_lazy() first. */
  int lazy_gen = 0;
  uint64_t lazy_seed_hi = 0, lazy_keymask = 0, lazy_universe = 0;
  /* conservative OR/AND of biased keys known from generation (digit-skip
   * without the k_minmax_or pass); valid only straight after gen /
   * permutation-only ops */
  int orand_valid = 0;
  uint64_t known_or = 0, known_and = 0;
  int hist_shift_valid = -1; /* tile_hist currently holds this pass shift */
  /* lookback-scatter scratch (DPRK_LOOKBACK) */
  unsigned long long *d_lb = nullptr;   /* [ntiles*RADIX] granules */
  unsigned int *d_fail = nullptr;
  uint32_t *d_ghist = nullptr;          /* [8][RADIX] global pass hists */
  uint64_t ghist_shifts = 0;            /* packed shifts the hists cover */
  int ghist_nact = 0;
  int ghist_valid = 0;
  uint32_t *d_dbase = nullptr;          /* [RADIX] current pass bases */
  uint64_t lb_cap_tiles = 0;


  /* radix scratch */
  uint32_t *tile_hist = nullptr;  /* [RADIX][ntiles] raw                 */
  uint32_t *tile_off = nullptr;   /* [RADIX][ntiles] scanned (+ base)    */
  uint32_t *seg_sums = nullptr;   /* [RADIX][nseg]                       */
  uint64_t *row_tot = nullptr;    /* [RADIX] per-bucket totals (partition)*/
  uint64_t *d_orand = nullptr;    /* {OR, AND} of biased keys            */
  uint64_t *alias_table = nullptr; /* packed (thresh<<32)|alias           */
  uint64_t alias_universe = 0;
  uint64_t max_tiles = 0;

  /* segmented-reduce / run-extract scratch */
  double *tile_T = nullptr;       /* per-tile trailing open sum          */
  uint32_t *tile_anyend = nullptr;
  uint32_t *tile_nruns = nullptr;
  uint64_t *tile_outoff = nullptr;
  double *tile_C = nullptr;       /* stitched incoming carry             */
  uint8_t *chain_open = nullptr;
  double *chunk_T = nullptr;
  uint32_t *chunk_F = nullptr;
  uint64_t *chunk_N = nullptr;
  uint64_t *d_scalar = nullptr;   /* small readback slot (totals)        */

  /* group layout (valid after dprk_group_by_key) */
  int64_t *grp_keys = nullptr;
  uint64_t *grp_starts = nullptr;
  uint64_t *grp_counts = nullptr;
  uint64_t n_groups = 0;
  uint64_t grp_cap = 0;

  /* join output (valid after dprk_join) */
  int64_t *join_k = nullptr;
  double *join_a = nullptr, *join_b = nullptr;
  uint64_t join_n = 0;
  uint64_t join_cap = 0;
  uint64_t *d_bidx = nullptr;
  /* generic scan scratch */
  uint64_t *scan_part = nullptr;
  uint64_t scan_part_cap = 0;
  uint64_t *run_prod = nullptr, *run_boff = nullptr;
  uint64_t run_prod_cap = 0;

  /* fused-reduce scratch (lazy; DPRK_FUSED_REDUCE) */
  uint64_t fused_cap_rec = 0;   /* (ntiles*RADIX) capacity */
  uint32_t *f_cnt = nullptr;    /* per-(tile,digit) fragment counts */
  uint32_t *f_lex = nullptr;    /* tile-local digit bases */
  uint32_t *f_goff = nullptr;   /* d-major global fragment offsets */

  /* out-of-core "big" aggregate (n >= 2^32):
   * chunk-partitioned bucket arenas on device, reduced bucket-by-bucket
   * through the normal in-core pipeline.  dpark's disk-spill external
   * Merger is the semantic precedent [U]. */
  int64_t *big_k = nullptr;       /* keys arena (u32 when big_packed)   */
  double *big_v = nullptr;        /* vals arena                          */
  uint64_t big_cap_total = 0;
  uint64_t big_cap_b = 0;         /* per-bucket capacity                 */
  int big_nbuckets = 0;
  int big_packed = -1;            /* set by the first chunk              */
  uint64_t big_fill[256] = {0};
  uint64_t big_out_n = 0;         /* after big_reduce: output pairs      */
  uint64_t big_or = 0, big_and = ~0ull;
  int big_valid = 0;

  /* comm */
  ncclComm_t comm = nullptr;
  int nranks = 1, rank = 0;

  /* timing */
  hipEvent_t evts[2 * MAX_EVTS];
  int n_evts_created = 0;
  EvtRec recs[MAX_EVTS];
  int n_recs = 0;
  dprk_stats_t stats;
};

/* event-pair timing around a launch; drained at op end (after stream sync)
 * so bench.py's roofline.achieved comes from HIP events on the launch
 * stream, as required. */
static int timed_begin(dprk_ctx *c, int kind) {
  if (c->n_recs >= MAX_EVTS) return 0; /* drop timing, never the launch */
  EvtRec &r = c->recs[c->n_recs];
  if (2 * c->n_recs + 1 >= c->n_evts_created) {
    if (hipEventCreate(&c->evts[c->n_evts_created]) != hipSuccess) return 0;
    if (hipEventCreate(&c->evts[c->n_evts_created + 1]) != hipSuccess) return 0;
    c->n_evts_created += 2;
  }
  r.a = c->evts[2 * c->n_recs];
  r.b = c->evts[2 * c->n_recs + 1];
  r.kind = kind;
  (void)hipEventRecord(r.a, c->stream);
  return 1;
}
static void timed_end(dprk_ctx *c, int began) {
  if (!began) return;
  (void)hipEventRecord(c->recs[c->n_recs].b, c->stream);
  c->n_recs++;
}

#define TIMED(c, kind, launch)                                             \
  do {                                                                     \
    int began_ = timed_begin(c, kind);                                     \
    launch;                                                                \
    HIP_CHECK(hipGetLastError());                                          \
    timed_end(c, began_);                                                  \
  } while (0)

static int drain_timing(dprk_ctx *c) {
  HIP_CHECK(hipStreamSynchronize(c->stream));
  double *ms[K_NKINDS] = {
      &c->stats.ms_gen, &c->stats.ms_hist, &c->stats.ms_tilehist,
      &c->stats.ms_scan, &c->stats.ms_scatter, &c->stats.ms_reduce,
      &c->stats.ms_runext, &c->stats.ms_join, &c->stats.ms_exchange,
      &c->stats.ms_other};
  uint64_t *nn[K_NKINDS] = {
      &c->stats.n_gen, &c->stats.n_hist, &c->stats.n_tilehist,
      &c->stats.n_scan, &c->stats.n_scatter, &c->stats.n_reduce,
      &c->stats.n_runext, &c->stats.n_join, &c->stats.n_exchange,
      &c->stats.n_other};
  for (int i = 0; i < c->n_recs; i++) {
    float t = 0.f;
    HIP_CHECK(hipEventElapsedTime(&t, c->recs[i].a, c->recs[i].b));
    *ms[c->recs[i].kind] += t;
    (*nn[c->recs[i].kind])++;
  }
  c->n_recs = 0;
  return 0;
}

/* ------------------------------------------------------------------ */
/* key-type traits: the PACKED-u32 fast path                           */
/*                                                                     */
/* When every key of a dataset lies in [0, 2^32) — detected input-     */
/* deterministically from the biased OR/AND (gen universe <= 2^32, or  */
/* k_minmax_or at put time) — keys travel as packed uint32 through the */
/* whole pipeline: 24 B/pair per scatter pass instead of 32 (-25%),    */
/* and the reduce reads 12 B/pair instead of 16.  The C-ABI stays      */
/* int64 (put/get convert); sorting u32 unsigned == int64 ascending    */
/* because packed keys are non-negative.  Both instantiations are the  */
/* SAME kernels — no behavioral fork, only the key element type.       */
/* ------------------------------------------------------------------ */
template <typename K> struct KT;
template <> struct KT<int64_t> {
  using V2 = longlong2;                       /* 2 keys = 16 B          */
  static constexpr int KSZ = 8;
  static __device__ __forceinline__ uint64_t bias(int64_t k) {
    return (uint64_t)k ^ SIGNFLIP;            /* signed -> unsigned order */
  }
  static __device__ __forceinline__ uint64_t raw(int64_t k) {
    return (uint64_t)k;
  }
  static __device__ __forceinline__ int64_t from_raw(uint64_t u) {
    return (int64_t)u;
  }
};
/* longlong2's members are `long long` while int64_t is `long` on LP64 —
 * an identical specialization keeps vector-element digit calls typed */
template <> struct KT<long long> {
  using V2 = longlong2;
  static constexpr int KSZ = 8;
  static __device__ __forceinline__ uint64_t bias(long long k) {
    return (uint64_t)k ^ SIGNFLIP;
  }
  static __device__ __forceinline__ uint64_t raw(long long k) {
    return (uint64_t)k;
  }
  static __device__ __forceinline__ long long from_raw(uint64_t u) {
    return (long long)u;
  }
};
template <> struct KT<uint32_t> {
  using V2 = uint2;                           /* 2 keys = 8 B           */
  static constexpr int KSZ = 4;
  static __device__ __forceinline__ uint64_t bias(uint32_t k) {
    return (uint64_t)k;                       /* non-negative: no flip  */
  }
  static __device__ __forceinline__ uint64_t raw(uint32_t k) {
    return (uint64_t)k;                       /* == the int64 key value */
  }
  static __device__ __forceinline__ uint32_t from_raw(uint64_t u) {
    return (uint32_t)u;
  }
};

/* ------------------------------------------------------------------ */
/* device utilities                                                    */
/* ------------------------------------------------------------------ */
__device__ __forceinline__ uint64_t d_splitmix64(uint64_t x) {
  uint64_t z = x + 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

/* digit extraction.  mode 0: radix digit of the sign-biased key (sort);
 * mode 1: TOP bits of splitmix64(key) (the GPU HashPartitioner,
 * dpark/dependency.py :: HashPartitioner equivalent [U] — layout freedom
 * per SURVEY.md §8(a2)); mode 2: range partition over sorted splitters
 * (dpark/dependency.py :: RangePartitioner [U], bisect_left semantics:
 * rank = #{j : splitters[j] <= key}... bisect_left gives
 * #{j : splitters[j] < key} + (key present ? its first index) — we use
 * rank = #{j : key >= splitters[j]}, which equals bisect_right; any
 * consistent monotone rule yields a valid total order across ranks). */
template <typename K>
__device__ __forceinline__ uint32_t d_digit_s(
    K key, int mode, int shift, uint32_t mask,
    const int64_t *__restrict__ splitters, int nsplit) {
  if (mode == 2) {
    int64_t k = (int64_t)KT<K>::raw(key);
    uint32_t r = 0;
    for (int j = 0; j < nsplit; j++) r += (k >= splitters[j]) ? 1u : 0u;
    return r;
  }
  uint64_t u = mode ? d_splitmix64(KT<K>::raw(key)) : KT<K>::bias(key);
  return (uint32_t)(u >> shift) & mask;
}

template <typename K>
__device__ __forceinline__ uint32_t d_digit(K key, int mode, int shift,
                                            uint32_t mask) {
  uint64_t u = mode ? d_splitmix64(KT<K>::raw(key)) : KT<K>::bias(key);
  return (uint32_t)(u >> shift) & mask;
}

/* wave-level inclusive scan over 64 lanes (no barriers) */
__device__ __forceinline__ uint32_t d_wave_incl_scan(uint32_t v, int lane) {
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    uint32_t up = (uint32_t)__shfl_up((int)v, d, 64);
    if (lane >= d) v += up;
  }
  return v;
}

/* strided sample of the current keys (sortByKey splitter sampling —
 * dpark samples partitions to the driver [U]) */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_sample(
    const K *__restrict__ keys, uint64_t n, uint64_t stride_elems,
    int64_t *__restrict__ out, uint64_t n_out) {
  uint64_t gstride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_out; i += gstride)
    out[i] = (int64_t)KT<K>::raw(keys[min(i * stride_elems, n - 1)]);
}

/* bytes-key hashing (the string-key engine path, SURVEY.md §8(f) rank 1:
 * TextFileRDD -> wordcount [U]).  Hash spec (normative; numpy mirror in
 * dpark_amd/hashspec.py :: hash_bytes_words, asserted identical by
 * tests): the word is split into 8-byte little-endian chunks (last one
 * zero-padded); h = splitmix64(C1 ^ len); for each chunk c:
 * h = splitmix64(h ^ c).  Length in the seed keeps "a" and "a\0"
 * distinct.  One thread per word; words are consecutive in the buffer so
 * the byte loads stay L2-coalesced across lanes. */
__global__ __launch_bounds__(NTHREADS) void k_hash_bytes(
    const uint8_t *__restrict__ buf, const uint64_t *__restrict__ offs,
    uint64_t n_words, int64_t *__restrict__ keys) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < n_words; i += stride) {
    uint64_t b0 = offs[i], b1 = offs[i + 1];
    uint64_t h = d_splitmix64(0x9E3779B97F4A7C15ull ^ (b1 - b0));
    for (uint64_t p = b0; p < b1; p += 8) {
      uint64_t c = 0;
      uint64_t lim = min(b1 - p, (uint64_t)8);
      for (uint64_t j = 0; j < lim; j++)
        c |= (uint64_t)buf[p + j] << (8 * j);
      h = d_splitmix64(h ^ c);
    }
    keys[i] = (int64_t)h;
  }
}

/* pack (int64 -> u32) / unpack (u32 -> int64); caller guarantees fit */
__global__ __launch_bounds__(NTHREADS) void k_pack_keys(
    const int64_t *__restrict__ in, uint32_t *__restrict__ out, uint64_t n) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (uint32_t)(uint64_t)in[i];
}
__global__ __launch_bounds__(NTHREADS) void k_unpack_keys(
    const uint32_t *__restrict__ in, int64_t *__restrict__ out, uint64_t n) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = (int64_t)(uint64_t)in[i];
}

/* ------------------------------------------------------------------ */
/* kernels: generation                                                 */
/* ------------------------------------------------------------------ */
/* tile-structured generator that also produces the per-tile digit-0
 * histogram (tile-major) — the first radix pass then skips its
 * k_tile_hist read (~4% of the config-2 step).  One WG per tile. */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_gen_hist(
    K *__restrict__ keys, double *__restrict__ vals, uint64_t n,
    uint64_t seed_hi, uint64_t keymask, uint32_t *__restrict__ tile_hist,
    uint64_t ntiles, int shift, uint64_t active_shifts, int nact,
    uint32_t *__restrict__ g_hist /* [nact][RADIX]; lookback path */,
    uint64_t *__restrict__ orand /* [2] = {OR, AND} of biased keys */) {
  __shared__ uint32_t h[NTHREADS / 64][RADIX];
  __shared__ uint32_t gh[NTHREADS / 64][8][RADIX]; /* 32 KiB */
  __shared__ uint64_t s_or[NTHREADS / 64], s_and[NTHREADS / 64];
  const int wave = threadIdx.x / 64;
  uint64_t acc_or = 0, acc_and = ~0ull;
  for (int i = threadIdx.x; i < (NTHREADS / 64) * 8 * RADIX; i += NTHREADS)
    ((uint32_t *)gh)[i] = 0;
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    for (int j = threadIdx.x; j < (NTHREADS / 64) * RADIX; j += NTHREADS)
      ((uint32_t *)h)[j] = 0;
    __syncthreads();
    uint64_t start = t * (uint64_t)TILE;
    uint64_t end = min(start + TILE, n);
    for (uint64_t i = start + 2 * threadIdx.x; i + 1 < end;
         i += 2 * NTHREADS) {
      uint64_t x0 = seed_hi + i, x1 = seed_hi + i + 1;
      typename KT<K>::V2 kk;
      kk.x = KT<K>::from_raw(d_splitmix64(x0) & keymask);
      kk.y = KT<K>::from_raw(d_splitmix64(x1) & keymask);
      *reinterpret_cast<typename KT<K>::V2 *>(&keys[i]) = kk;
      acc_or |= KT<K>::bias(kk.x) | KT<K>::bias(kk.y);
      acc_and &= KT<K>::bias(kk.x) & KT<K>::bias(kk.y);
      atomicAdd(&h[wave][d_digit(kk.x, 0, shift, 0xFF)], 1u);
      atomicAdd(&h[wave][d_digit(kk.y, 0, shift, 0xFF)], 1u);
      for (int p = 0; p < nact; p++) {
        int sh = (int)((active_shifts >> (8 * p)) & 0xFF);
        atomicAdd(&gh[wave][p][d_digit(kk.x, 0, sh, 0xFF)], 1u);
        atomicAdd(&gh[wave][p][d_digit(kk.y, 0, sh, 0xFF)], 1u);
      }
      if (vals) {
        double2 vv;
        uint64_t v0 = d_splitmix64(x0 ^ 0xDEADBEEFCAFEBABEull);
        uint64_t v1 = d_splitmix64(x1 ^ 0xDEADBEEFCAFEBABEull);
        vv.x = (double)(v0 >> 11) * (1.0 / 9007199254740992.0);
        vv.y = (double)(v1 >> 11) * (1.0 / 9007199254740992.0);
        *reinterpret_cast<double2 *>(&vals[i]) = vv;
      }
    }
    if ((end - start) & 1 && threadIdx.x == 0) {
      uint64_t i = end - 1;
      uint64_t x = seed_hi + i;
      K k = KT<K>::from_raw(d_splitmix64(x) & keymask);
      keys[i] = k;
      acc_or |= KT<K>::bias(k);
      acc_and &= KT<K>::bias(k);
      atomicAdd(&h[wave][d_digit(k, 0, shift, 0xFF)], 1u);
      for (int p = 0; p < nact; p++) {
        int sh = (int)((active_shifts >> (8 * p)) & 0xFF);
        atomicAdd(&gh[wave][p][d_digit(k, 0, sh, 0xFF)], 1u);
      }
      if (vals) {
        uint64_t vb = d_splitmix64(x ^ 0xDEADBEEFCAFEBABEull);
        vals[i] = (double)(vb >> 11) * (1.0 / 9007199254740992.0);
      }
    }
    __syncthreads();
    for (int d = threadIdx.x; d < RADIX; d += NTHREADS) {
      uint32_t s = 0;
      for (int w = 0; w < NTHREADS / 64; w++) s += h[w][d];
      tile_hist[t * RADIX + d] = s;
    }
    __syncthreads();
  }
  if (g_hist)
    for (int p = 0; p < nact; p++)
      for (int d = threadIdx.x; d < RADIX; d += NTHREADS) {
        uint32_t s = 0;
        for (int w = 0; w < NTHREADS / 64; w++) s += gh[w][p][d];
        if (s) atomicAdd(&g_hist[p * RADIX + d], s);
      }
  /* exact biased-key OR/AND -> digit-skip without a k_minmax_or read */
  for (int off = 32; off; off >>= 1) {
    acc_or |= (uint64_t)__shfl_down((long long)acc_or, off, 64);
    acc_and &= (uint64_t)__shfl_down((long long)acc_and, off, 64);
  }
  if (threadIdx.x % 64 == 0) { s_or[wave] = acc_or; s_and[wave] = acc_and; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < NTHREADS / 64; w++) {
      acc_or |= s_or[w];
      acc_and &= s_and[w];
    }
    atomicOr((unsigned long long *)&orand[0], (unsigned long long)acc_or);
    atomicAnd((unsigned long long *)&orand[1], (unsigned long long)acc_and);
  }
}

/* Zipf (or any discrete distribution) sampling via a caller-provided
 * alias table: one random 8-B table load per sample.
 * table[slot] = (thresh << 32) | alias; take slot if u2_hi32 < thresh
 * else alias.  Bit-exact mirror: oracle/synth.py gen_zipf (the table is
 * BUILT host-side in numpy and shipped via dprk_set_alias_table, so no
 * cross-libm drift can break parity). */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_gen_alias(
    K *__restrict__ keys, double *__restrict__ vals, uint64_t n,
    uint64_t seed_hi, const uint64_t *__restrict__ table, uint64_t universe) {
  /* 4-way batched iteration: four independent table loads in flight per
   * thread before any resolve (the random 8-B loads over the 128 MiB
   * table are the generator's cost — one dependent load per iteration
   * left most of the latency unhidden).  Stores are nontemporal (the
   * 12 GB output stream cannot stay cached anyway; measured neutral on
   * its own).  Values identical to the numpy mirror either way. */
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  uint64_t i0 = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i0 + 3 * stride < n; i0 += 4 * stride) {
    uint64_t x[4], u1[4], slot[4], e[4];
#pragma unroll
    for (int j = 0; j < 4; j++) {
      x[j] = seed_hi + i0 + j * stride;
      u1[j] = d_splitmix64(x[j] ^ 0x5851F42D4C957F2Dull);
      slot[j] = __umul64hi(u1[j], universe);
      e[j] = table[slot[j]];
    }
#pragma unroll
    for (int j = 0; j < 4; j++) {
      uint64_t u2 = d_splitmix64(x[j] ^ 0x14057B7EF767814Full);
      uint64_t k = ((u2 >> 32) < (e[j] >> 32)) ? slot[j]
                                               : (e[j] & 0xFFFFFFFFull);
      __builtin_nontemporal_store(KT<K>::from_raw(k),
                                  &keys[i0 + j * stride]);
      if (vals) {
        uint64_t vb = d_splitmix64(x[j] ^ 0xDEADBEEFCAFEBABEull);
        __builtin_nontemporal_store(
            (double)(vb >> 11) * (1.0 / 9007199254740992.0),
            &vals[i0 + j * stride]);
      }
    }
  }
  for (uint64_t i = i0; i < n; i += stride) {
    uint64_t x = seed_hi + i;
    uint64_t u1 = d_splitmix64(x ^ 0x5851F42D4C957F2Dull);
    uint64_t slot = __umul64hi(u1, universe);
    uint64_t e = table[slot];
    uint64_t u2 = d_splitmix64(x ^ 0x14057B7EF767814Full);
    uint64_t k = ((u2 >> 32) < (e >> 32)) ? slot : (e & 0xFFFFFFFFull);
    keys[i] = KT<K>::from_raw(k);
    if (vals) {
      uint64_t vb = d_splitmix64(x ^ 0xDEADBEEFCAFEBABEull);
      vals[i] = (double)(vb >> 11) * (1.0 / 9007199254740992.0);
    }
  }
}

/* global per-pass digit histograms for every ACTIVE pass in ONE read of
 * the keys (lookback path: replaces per-pass k_tile_hist + the 3-level
 * scans).  active_shifts packs the byte shifts of up to 8 active passes;
 * per-wave LDS replicas, one global atomic merge per workgroup. */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_hist_all(
    const K *__restrict__ keys, uint64_t n, uint64_t active_shifts,
    int nact, uint32_t *__restrict__ g_hist /* [nact][RADIX] */) {
  __shared__ uint32_t h[NTHREADS / 64][8][RADIX];   /* 32 KiB */
  const int wave = threadIdx.x / 64;
  for (int i = threadIdx.x; i < (NTHREADS / 64) * 8 * RADIX; i += NTHREADS)
    ((uint32_t *)h)[i] = 0;
  __syncthreads();
  const uint64_t stride = 2 * (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = 2 * ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x);
       i + 1 < n; i += stride) {
    typename KT<K>::V2 kk =
        *reinterpret_cast<const typename KT<K>::V2 *>(&keys[i]);
    uint64_t u0 = KT<K>::bias(kk.x);
    uint64_t u1 = KT<K>::bias(kk.y);
    for (int p = 0; p < nact; p++) {
      int sh = (int)((active_shifts >> (8 * p)) & 0xFF);
      atomicAdd(&h[wave][p][(u0 >> sh) & 0xFF], 1u);
      atomicAdd(&h[wave][p][(u1 >> sh) & 0xFF], 1u);
    }
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    uint64_t u = KT<K>::bias(keys[n - 1]);
    for (int p = 0; p < nact; p++) {
      int sh = (int)((active_shifts >> (8 * p)) & 0xFF);
      atomicAdd(&h[wave][p][(u >> sh) & 0xFF], 1u);
    }
  }
  __syncthreads();
  for (int p = 0; p < nact; p++)
    for (int d = threadIdx.x; d < RADIX; d += NTHREADS) {
      uint32_t s = 0;
      for (int w = 0; w < NTHREADS / 64; w++) s += h[w][p][d];
      if (s) atomicAdd(&g_hist[p * RADIX + d], s);
    }
}

/* deterministic f64 value sum (conservation checks at out-of-core sizes
 * the CPU oracle cannot reach): fixed per-block partial tree, partials
 * combined in block order host-side */
__global__ __launch_bounds__(NTHREADS) void k_val_sum(
    const double *__restrict__ vals, uint64_t n,
    double *__restrict__ partials /* [gridDim.x] */) {
  __shared__ double sh[NTHREADS];
  double acc = 0.0;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    acc += vals[i];
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = NTHREADS / 2; off; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) partials[blockIdx.x] = sh[0];
}

/* hist-only generation prepass (gen->sort fusion): replay the PRNG
 * stream to produce every active pass's global digit histogram and the
 * exact biased OR/AND — ZERO global data traffic; the fused first sort
 * pass then generates pairs straight into partitioned layout. */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_gen_hist_only(
    uint64_t n, uint64_t seed_hi, uint64_t keymask, uint64_t active_shifts,
    int nact, uint32_t *__restrict__ g_hist /* [nact][RADIX] */,
    uint64_t *__restrict__ orand /* [2] pre-set to {0, ~0} */) {
  __shared__ uint32_t h[NTHREADS / 64][8][RADIX];
  __shared__ uint64_t s_or[NTHREADS / 64], s_and[NTHREADS / 64];
  const int wave = threadIdx.x / 64;
  for (int i = threadIdx.x; i < (NTHREADS / 64) * 8 * RADIX; i += NTHREADS)
    ((uint32_t *)h)[i] = 0;
  __syncthreads();
  uint64_t acc_or = 0, acc_and = ~0ull;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    K k = KT<K>::from_raw(d_splitmix64(seed_hi + i) & keymask);
    uint64_t u = KT<K>::bias(k);
    acc_or |= u;
    acc_and &= u;
    for (int p = 0; p < nact; p++) {
      int sh = (int)((active_shifts >> (8 * p)) & 0xFF);
      atomicAdd(&h[wave][p][(u >> sh) & 0xFF], 1u);
    }
  }
  __syncthreads();
  for (int p = 0; p < nact; p++)
    for (int d = threadIdx.x; d < RADIX; d += NTHREADS) {
      uint32_t s = 0;
      for (int w = 0; w < NTHREADS / 64; w++) s += h[w][p][d];
      if (s) atomicAdd(&g_hist[p * RADIX + d], s);
    }
  for (int off = 32; off; off >>= 1) {
    acc_or |= (uint64_t)__shfl_down((long long)acc_or, off, 64);
    acc_and &= (uint64_t)__shfl_down((long long)acc_and, off, 64);
  }
  if (threadIdx.x % 64 == 0) { s_or[wave] = acc_or; s_and[wave] = acc_and; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < NTHREADS / 64; w++) {
      acc_or |= s_or[w];
      acc_and &= s_and[w];
    }
    atomicOr((unsigned long long *)&orand[0], (unsigned long long)acc_or);
    atomicAnd((unsigned long long *)&orand[1], (unsigned long long)acc_and);
  }
}

/* OR/AND reduction of biased keys -> constant-digit (skip-pass) detection */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_minmax_or(
    const K *__restrict__ keys, uint64_t n,
    uint64_t *__restrict__ orand /* [2] pre-set to {0, ~0} */) {
  __shared__ uint64_t s_or[NTHREADS / 64], s_and[NTHREADS / 64];
  uint64_t o = 0, a = ~0ull;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint64_t u = KT<K>::bias(keys[i]);
    o |= u;
    a &= u;
  }
  /* wave reduce */
  for (int off = 32; off; off >>= 1) {
    o |= __shfl_down(o, off, 64);
    a &= __shfl_down(a, off, 64);
  }
  int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  if (lane == 0) { s_or[wave] = o; s_and[wave] = a; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < NTHREADS / 64; w++) { o |= s_or[w]; a &= s_and[w]; }
    atomicOr((unsigned long long *)&orand[0], (unsigned long long)o);
    atomicAnd((unsigned long long *)&orand[1], (unsigned long long)a);
  }
}

/* ------------------------------------------------------------------ */
/* kernels: radix pass (tile histogram -> 3-level scan -> scatter)     */
/* ------------------------------------------------------------------ */

/* per-tile 256-bin histogram; one workgroup per tile; per-wave LDS
 * replicas to cut same-digit ds_add serialization (SURVEY.md §7 hard
 * part 2).  Histograms are stored TILE-MAJOR (hist[t*RADIX + d]) so each
 * workgroup writes one contiguous 1 KiB block — digit-major layout made
 * every 4 B store a partial-line write that each XCD's L2 evicted as a
 * full 64 B line (measured 10x write amplification, profiles/r01). */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_tile_hist(
    const K *__restrict__ keys, uint64_t n,
    uint32_t *__restrict__ tile_hist, uint64_t ntiles, int mode, int shift,
    uint32_t mask, const int64_t *__restrict__ splitters, int nsplit) {
  __shared__ uint32_t h[NTHREADS / 64][RADIX];
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    int wave = threadIdx.x / 64;
    for (int i = threadIdx.x; i < (NTHREADS / 64) * RADIX; i += NTHREADS)
      ((uint32_t *)h)[i] = 0;
    __syncthreads();
    uint64_t start = t * (uint64_t)TILE;
    uint64_t end = min(start + TILE, n);
    /* vector loads: two keys per lane (8-B streams run at 0.54-0.70x the
     * 16-B rate on gfx950).  Full tiles take the constant-trip unrolled
     * path so all 8 loads issue up front (the variable-bound loop kept
     * only one load in flight: 95% wave-parked in the SQ profile). */
    if (end - start == TILE) {
      typename KT<K>::V2 kk[TILE / (2 * NTHREADS)];
#pragma unroll
      for (int j = 0; j < TILE / (2 * NTHREADS); j++)
        kk[j] = *reinterpret_cast<const typename KT<K>::V2 *>(
            &keys[start + 2 * threadIdx.x + (uint64_t)j * 2 * NTHREADS]);
#pragma unroll
      for (int j = 0; j < TILE / (2 * NTHREADS); j++) {
        atomicAdd(&h[wave][d_digit_s(kk[j].x, mode, shift, mask, splitters, nsplit)], 1u);
        atomicAdd(&h[wave][d_digit_s(kk[j].y, mode, shift, mask, splitters, nsplit)], 1u);
      }
    } else {
      for (uint64_t i = start + 2 * threadIdx.x; i + 1 < end; i += 2 * NTHREADS) {
        typename KT<K>::V2 kk =
            *reinterpret_cast<const typename KT<K>::V2 *>(&keys[i]);
        atomicAdd(&h[wave][d_digit_s(kk.x, mode, shift, mask, splitters, nsplit)], 1u);
        atomicAdd(&h[wave][d_digit_s(kk.y, mode, shift, mask, splitters, nsplit)], 1u);
      }
      if (end == n && (n & 1) && threadIdx.x == 0 && end - start > 0)
        atomicAdd(&h[wave][d_digit_s(keys[end - 1], mode, shift, mask, splitters, nsplit)], 1u);
    }
    __syncthreads();
    for (int d = threadIdx.x; d < RADIX; d += NTHREADS) {
      uint32_t s = 0;
      for (int w = 0; w < NTHREADS / 64; w++) s += h[w][d];
      tile_hist[t * RADIX + d] = s; /* tile-major: coalesced full lines */
    }
    __syncthreads();
  }
}

/* scan level 1: per-(segment, digit) sums.  One WG per segment, thread =
 * digit, sequential walk over the segment's tiles (coalesced 1 KiB loads;
 * the 25 MB hist array is L2/L3-resident right after k_tile_hist). */
__global__ __launch_bounds__(NTHREADS) void k_seg_sums(
    const uint32_t *__restrict__ tile_hist, uint64_t ntiles,
    uint64_t seg_tiles, uint32_t *__restrict__ seg_sums) {
  uint64_t s = blockIdx.x;
  uint64_t t0 = s * seg_tiles;
  uint64_t t1 = min(t0 + seg_tiles, ntiles);
  uint32_t acc = 0;
#pragma unroll 4
  for (uint64_t t = t0; t < t1; t++) acc += tile_hist[t * RADIX + threadIdx.x];
  seg_sums[s * RADIX + threadIdx.x] = acc;
}

/* scan level 2 (single workgroup): per-digit exclusive scan over segment
 * sums (nseg <= 256), fold in the cross-digit exclusive base, export
 * per-digit totals (u64) for the partition path.  Thread = digit;
 * all accesses coalesced in the seg-major layout. */
__global__ __launch_bounds__(NTHREADS) void k_scan_mid(
    uint32_t *__restrict__ seg_sums, uint64_t nseg,
    uint64_t *__restrict__ row_tot) {
  __shared__ uint64_t tot[RADIX];
  int d = threadIdx.x;
  uint64_t run = 0;
  /* register-blocked: 8 loads in flight before any store — the naive
   * in-place loop serializes on (unprovable) store->load aliasing
   * (measured 77 us for a single-WG pass over 96 segments) */
  for (uint64_t s0 = 0; s0 < nseg; s0 += 8) {
    uint32_t v[8];
#pragma unroll
    for (int j = 0; j < 8; j++)
      v[j] = (s0 + j < nseg) ? seg_sums[(s0 + j) * RADIX + d] : 0;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      if (s0 + j < nseg) seg_sums[(s0 + j) * RADIX + d] = (uint32_t)run;
      run += v[j];
    }
  }
  tot[d] = run;
  if (row_tot) row_tot[d] = run;
  __syncthreads();
  __shared__ uint64_t sc[RADIX];
  sc[d] = tot[d];
  __syncthreads();
  for (int off = 1; off < RADIX; off <<= 1) {
    uint64_t add = (d >= off) ? sc[d - off] : 0;
    __syncthreads();
    sc[d] += add;
    __syncthreads();
  }
  uint64_t base = sc[d] - tot[d]; /* inclusive -> exclusive */
  for (uint64_t s0 = 0; s0 < nseg; s0 += 8) {
    uint32_t v[8];
#pragma unroll
    for (int j = 0; j < 8; j++)
      v[j] = (s0 + j < nseg) ? seg_sums[(s0 + j) * RADIX + d] : 0;
#pragma unroll
    for (int j = 0; j < 8; j++)
      if (s0 + j < nseg)
        seg_sums[(s0 + j) * RADIX + d] = v[j] + (uint32_t)base;
  }
  /* NOTE: offsets fit u32 because n < 2^32 (enforced in dprk_reserve) */
}

/* scan level 3: per-(segment) exclusive tile offsets.  One WG per
 * segment, thread = digit, sequential over tiles; loads and stores both
 * coalesced (tile-major). */
__global__ __launch_bounds__(NTHREADS) void k_tile_off(
    const uint32_t *__restrict__ tile_hist, uint64_t ntiles,
    uint64_t seg_tiles, const uint32_t *__restrict__ seg_sums,
    uint32_t *__restrict__ tile_off) {
  uint64_t s = blockIdx.x;
  uint64_t t0 = s * seg_tiles;
  uint64_t t1 = min(t0 + seg_tiles, ntiles);
  uint32_t run = seg_sums[s * RADIX + threadIdx.x];
#pragma unroll 2
  for (uint64_t t = t0; t < t1; t++) {
    uint32_t v = tile_hist[t * RADIX + threadIdx.x];
    tile_off[t * RADIX + threadIdx.x] = run;
    run += v;
  }
}

/* the scatter: stable rank within tile + LDS exchange + coalesced write.
 *
 * Wave-private structure (v2 — v1 paid ~80 barriers/tile and 64 KiB of
 * LDS counter zeroing per tile, measured 2.0 TB/s; see git history):
 * each of the 4 waves owns a contiguous 1024-element quarter of the
 * tile, holds its 16 elements/lane in REGISTERS, builds a private LDS
 * histogram, and ranks with 8-ballot same-digit masks against private
 * running counters — no cross-wave traffic between two barriers.
 * Digit-sorted staging in LDS, then a sequential sweep writes
 * equal-digit runs to contiguous global addresses. */
/* lookback scatter: one kernel per pass does per-tile histogram (from
 * registers), STABLE rank, LDS exchange, AND its own global offsets via
 * onesweep-style decoupled lookback over agent-scope 8-byte granules
 * (the R2 recipe: one relaxed agent store publishes {state,value}; the
 * per-XCD L2s are not coherent, so plain loads/stores would be stale —
 * cdna_hip_programming.md §6 G16).  Tiles are taken from a global
 * ticket so assignment order == issue order (no circular waits with a
 * partially resident grid); spins are BOUNDED and set a failure flag
 * the host turns into a 3-kernel-path retry — hang-safe by design.
 * Output is bit-identical to the k_tile_hist+scan+k_scatter path (the
 * same permutation), so path selection has no numerics implications. */
#define LB_AGG  (1ull << 62)
#define LB_INC  (2ull << 62)
#define LB_VAL  ((1ull << 62) - 1)
#define LB_SPIN_LIMIT (1u << 16)

/* RANK_ATOMIC (experiment, DPRK_RANK_ATOMIC=1): phase-3 stable ranking
 * via ONE ds_add_rtn per element on the phase-2 cursors instead of the
 * 8-ballot chain (~20 wave-ops/element).  Equal-digit order within a
 * wave becomes the hardware's fixed atomic order rather than lane
 * order: still a deterministic permutation, so run-to-run
 * bit-repeatability, sorted-key order and per-key value MULTISETS (the
 * sort contract — duplicate order unspecified) all
 * hold; only the unspecified duplicate order differs from the ballot
 * path. */
template <typename K, bool WITH_VALS, bool RANK_ATOMIC = false>
__global__ __launch_bounds__(NTHREADS_SC, 4) void k_scatter_lb(
    const K *__restrict__ keys_in, const double *__restrict__ vals_in,
    K *__restrict__ keys_out, double *__restrict__ vals_out,
    uint64_t n, uint64_t ntiles, int shift,
    const uint32_t *__restrict__ digit_base,
    unsigned long long *__restrict__ lb,
    unsigned int *__restrict__ d_fail) {
  constexpr int NW = NTHREADS_SC / 64;
  constexpr int WBLK = TILE / NW;
  constexpr int EPL = WBLK / 64;
  __shared__ K lk[TILE];
  __shared__ double lv[WITH_VALS ? TILE : 1];
  __shared__ uint32_t excl[RADIX];
  __shared__ uint32_t goff[RADIX];
  __shared__ uint32_t h[NW][RADIX];
  __shared__ uint32_t wtot[NW > 4 ? NW : 4];

  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const uint64_t lowmask = ((uint64_t)1 << lane) - 1;
  const uint32_t mask = 0xFF;

  /* STATIC issue-ordered assignment: block b owns tiles b, b+R, b+2R...
   * with grid R <= true residency (host derives R from the occupancy
   * query, clamped), so every predecessor tile is finished or being
   * processed by a RESIDENT block — progress is guaranteed without a
   * ticket (a single ticket word saturates near 88 atomics/us: ~2.8 ms
   * per pass at 244k tiles; a batched ticket delays AGG publication,
   * measured 130x worse — both in git history). */
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    /* uniform failure bailout (reuse wtot[0]) */
    if (threadIdx.x == 0)
      wtot[0] = __hip_atomic_load(d_fail, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();
    if (wtot[0]) return;
    const uint64_t start = t * (uint64_t)TILE;
    const uint64_t end = min(start + TILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
    const uint32_t wbeg = wave * WBLK;

    /* phase 1: load + wave-private histogram */
    for (int j = 0; j < RADIX / 64; j++) h[wave][lane * (RADIX / 64) + j] = 0;
    __syncthreads();
    K rk[EPL];
    double rv[EPL];
    uint32_t rd[EPL];
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        rk[i] = keys_in[start + idx];
        if (WITH_VALS) rv[i] = vals_in[start + idx];
        rd[i] = d_digit(rk[i], 0, shift, mask);
        atomicAdd(&h[wave][rd[i]], 1u);
      }
    }
    __syncthreads();

    /* phase 2: tile counts -> publish AGG granule; local digit excl scan
     * (shfl) + per-wave bases in place.  The LOOKBACK runs after phase 3
     * (LDS stage) — measured better than the eager onesweep ordering on
     * this chip (walking right after publish spins on predecessors that
     * have not reached their publish point yet: +0.24/+8 ms per pass at
     * 1e8/1e9; the extra phase-3 time lets the generation's AGGs land). */
    uint32_t tile_cnt = 0;
    if (threadIdx.x < RADIX) {
      for (int w = 0; w < NW; w++) tile_cnt += h[w][threadIdx.x];
      __hip_atomic_store(&lb[t * RADIX + threadIdx.x], LB_AGG | tile_cnt,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      uint32_t inc = d_wave_incl_scan(tile_cnt, lane);
      if (lane == 63) wtot[wave] = inc;
      excl[threadIdx.x] = inc - tile_cnt;
    }
    __syncthreads();
    if (threadIdx.x < RADIX) {
      if (wave > 0) {
        uint32_t add = 0;
        for (int w = 0; w < wave; w++) add += wtot[w];
        excl[threadIdx.x] += add;
      }
      uint32_t b = excl[threadIdx.x];
      for (int w = 0; w < NW; w++) {
        uint32_t cnt = h[w][threadIdx.x];
        h[w][threadIdx.x] = b;
        b += cnt;
      }
    }
    __syncthreads();

    /* phase 3: stable rank + LDS scatter */
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        uint32_t dig = rd[i];
        uint32_t pos;
        if (RANK_ATOMIC) {
          pos = atomicAdd(&h[wave][dig], 1u);
        } else {
          uint64_t same = __ballot(1);
          for (int bit = 0; bit < RADIX_BITS; bit++) {
            uint64_t bb = __ballot((dig >> bit) & 1);
            same &= ((dig >> bit) & 1) ? bb : ~bb;
          }
          uint32_t rank_in_wave = (uint32_t)__popcll(same & lowmask);
          uint32_t base = h[wave][dig];
          if (rank_in_wave == 0)
            h[wave][dig] = base + (uint32_t)__popcll(same);
          pos = base + rank_in_wave;
        }
        lk[pos] = rk[i];
        if (WITH_VALS) lv[pos] = rv[i];
      }
    }

    /* lookback (thread = digit): sum predecessors' granules back to the
     * nearest INC; publish own INC; bounded spins -> d_fail */
    if (threadIdx.x < RADIX) {
      uint64_t excl_g = 0;
      uint32_t spins = 0;
      bool ok = true;
      int64_t s = (int64_t)t - 1;
      while (s >= 0) {
        uint64_t e = __hip_atomic_load(&lb[(uint64_t)s * RADIX + threadIdx.x],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
        if (!(e >> 62)) {
          __builtin_amdgcn_s_sleep(2);
          if (++spins > LB_SPIN_LIMIT) {
            __hip_atomic_store(d_fail, 1u, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            ok = false;
            break;
          }
          continue;
        }
        excl_g += e & LB_VAL;
        if ((e >> 62) >= 2) break;
        s--;
      }
      if (ok) {
        __hip_atomic_store(&lb[t * RADIX + threadIdx.x],
                           LB_INC | (excl_g + tile_cnt), __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        /* combined write base: phase 4 computes dst = goff[dig] + idx
         * with ONE LDS read (goff already folds -excl[dig]) */
        goff[threadIdx.x] = digit_base[threadIdx.x] + (uint32_t)excl_g -
                            excl[threadIdx.x];
      }
    }
    /* uniform post-lookback failure check (reuse wtot[0] as a flag) */
    if (threadIdx.x == 0)
      wtot[0] = __hip_atomic_load(d_fail, __ATOMIC_RELAXED,
                                  __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();
    if (wtot[0]) return;

    /* phase 4: write out in digit-sorted tile order */
    for (uint32_t idx = threadIdx.x; idx < tile_n; idx += NTHREADS_SC) {
      K key = lk[idx];
      uint32_t dig = d_digit(key, 0, shift, mask);
      uint32_t dst = goff[dig] + idx;   /* goff folds -excl[dig] */
      keys_out[dst] = key;
      if (WITH_VALS) vals_out[dst] = lv[idx];
    }
    __syncthreads();
  }
}

/* Software-pipelined lookback scatter (DPRK_PIPE experiment; u32 keys
 * only — the double hist + staging fits 160 KiB LDS only at 12 B/pair):
 * tile t+grid's global loads are ISSUED during tile t's staging and
 * consumed after t's write phase, so the ~900-cycle HBM load latency
 * hides under the lookback walk + store issue — at TILE 8192 there is
 * only 1 WG/CU, so no partner workgroup fills those gaps (the r01
 * rejection of cross-tile pipelining was measured at 2 WG/CU).  Rank is
 * ds_add_rtn (the r02 default).  Identical output permutation to
 * k_scatter_lb<u32, WV, true>. */
/* GEN mode (third template param): the "load" of a tile is the PRNG
 * replay itself — pairs are generated straight into pass-1-partitioned
 * layout (dpark's map task feeds its bucket loop from the parent
 * iterator without materializing input either [U]); seed_hi/keymask
 * are unused when !GEN.  Same strided element mapping and rank order
 * as the load form. */
template <typename K, bool WITH_VALS, bool GEN = false>
__global__ __launch_bounds__(NTHREADS_SC, 4) void k_scatter_lb_pipe(
    const K *__restrict__ keys_in, const double *__restrict__ vals_in,
    K *__restrict__ keys_out, double *__restrict__ vals_out,
    uint64_t n, uint64_t ntiles, int shift,
    const uint32_t *__restrict__ digit_base,
    unsigned long long *__restrict__ lb,
    unsigned int *__restrict__ d_fail,
    uint64_t seed_hi = 0, uint64_t keymask = 0) {
  constexpr int NW = NTHREADS_SC / 64;
  constexpr int WBLK = TILE / NW;
  constexpr int EPL = WBLK / 64;
  __shared__ K lk[TILE];
  __shared__ double lv[WITH_VALS ? TILE : 1];
  __shared__ uint32_t excl[RADIX];
  __shared__ uint32_t goff[RADIX];
  __shared__ uint32_t h[2][NW][RADIX];
  __shared__ uint32_t wtot[NW > 4 ? NW : 4];
  __shared__ uint32_t sfail;

  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const uint32_t mask = 0xFF;
  const uint32_t wbeg = wave * WBLK;

  uint64_t t0 = blockIdx.x;
  if (t0 >= ntiles) return;
  for (int j = 0; j < RADIX / 64; j++) {
    h[0][wave][lane * (RADIX / 64) + j] = 0;
    h[1][wave][lane * (RADIX / 64) + j] = 0;
  }
  if (threadIdx.x == 0) sfail = 0;

  K rk0[EPL], rk1[EPL];
  double rv0[EPL], rv1[EPL];
  uint32_t rd0[EPL], rd1[EPL];

  /* prolog: load + hist tile t0 into buffer 0 (barrier below publishes
   * the zeroed hists before the atomics) */
  __syncthreads();
  {
    const uint64_t start = t0 * (uint64_t)TILE;
    const uint64_t end = min(start + TILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        if (GEN) {
          uint64_t x = seed_hi + start + idx;
          rk0[i] = KT<K>::from_raw(d_splitmix64(x) & keymask);
          if (WITH_VALS)
            rv0[i] = (double)(d_splitmix64(x ^ 0xDEADBEEFCAFEBABEull) >> 11)
                     * (1.0 / 9007199254740992.0);
        } else {
          rk0[i] = keys_in[start + idx];
          if (WITH_VALS) rv0[i] = vals_in[start + idx];
        }
        rd0[i] = d_digit(rk0[i], 0, shift, mask);
        atomicAdd(&h[0][wave][rd0[i]], 1u);
      }
    }
  }

  /* one pipeline stage: process tile t from (rkc, hc); prefetch + hist
   * tile t+grid into (rkn, hn).  Returns false on lookback failure. */
  auto stage = [&](K (&rkc)[EPL], double (&rvc)[EPL], uint32_t (&rdc)[EPL],
                   K (&rkn)[EPL], double (&rvn)[EPL], uint32_t (&rdn)[EPL],
                   uint32_t (*hc)[RADIX], uint32_t (*hn)[RADIX],
                   uint64_t t) -> bool {
    if (threadIdx.x == 0)       /* cross-block failure bailout */
      sfail = __hip_atomic_load(d_fail, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();            /* hist(t) complete; sfail visible */
    if (sfail) return false;
    const uint64_t start = t * (uint64_t)TILE;
    const uint64_t end = min(start + TILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
    const uint64_t tn = t + gridDim.x;

    /* phase 2: publish AGG; excl scan; wave cursors in place */
    uint32_t tile_cnt = 0;
    if (threadIdx.x < RADIX) {
      for (int w = 0; w < NW; w++) tile_cnt += hc[w][threadIdx.x];
      __hip_atomic_store(&lb[t * RADIX + threadIdx.x], LB_AGG | tile_cnt,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      uint32_t inc = d_wave_incl_scan(tile_cnt, lane);
      if (lane == 63) wtot[wave] = inc;
      excl[threadIdx.x] = inc - tile_cnt;
    }
    __syncthreads();
    if (threadIdx.x < RADIX) {
      if (wave > 0) {
        uint32_t add = 0;
        for (int w = 0; w < wave; w++) add += wtot[w];
        excl[threadIdx.x] += add;
      }
      uint32_t b = excl[threadIdx.x];
      for (int w = 0; w < NW; w++) {
        uint32_t cnt = hc[w][threadIdx.x];
        hc[w][threadIdx.x] = b;
        b += cnt;
      }
    }
    __syncthreads();

    /* phase 3: rank (ds_add_rtn) + stage; then ISSUE next tile's loads.
     * DPRK_EARLY_WALK (compile variant): the RADIX lookback threads walk
     * FIRST — their spins overlap the other 12 waves' staging, and the
     * post-staging barrier doubles as the fail check (one barrier
     * saved); the cost is waves 0-3's staging starting after the walk. */
#ifdef DPRK_EARLY_WALK
#define DPRK_WALK_NOW 1
#else
#define DPRK_WALK_NOW 0
#endif
    uint32_t tile_nn = 0;
    if (DPRK_WALK_NOW && threadIdx.x < RADIX) {
      uint64_t excl_g = 0;
      uint32_t spins = 0;
      bool ok = true;
      int64_t s = (int64_t)t - 1;
      while (s >= 0) {
        uint64_t e = __hip_atomic_load(&lb[(uint64_t)s * RADIX + threadIdx.x],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
        if (!(e >> 62)) {
          __builtin_amdgcn_s_sleep(2);
          if (++spins > LB_SPIN_LIMIT) {
            __hip_atomic_store(d_fail, 1u, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            ok = false;
            break;
          }
          continue;
        }
        excl_g += e & LB_VAL;
        if ((e >> 62) >= 2) break;
        s--;
      }
      if (ok) {
        __hip_atomic_store(&lb[t * RADIX + threadIdx.x],
                           LB_INC | (excl_g + tile_cnt), __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        goff[threadIdx.x] = digit_base[threadIdx.x] + (uint32_t)excl_g -
                            excl[threadIdx.x];
      } else {
        sfail = 1;
      }
    }
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        uint32_t pos = atomicAdd(&hc[wave][rdc[i]], 1u);
        lk[pos] = rkc[i];
        if (WITH_VALS) lv[pos] = rvc[i];
      }
    }
    if (tn < ntiles) {
      const uint64_t startn = tn * (uint64_t)TILE;
      tile_nn = (uint32_t)(min(startn + TILE, n) - startn);
#pragma unroll
      for (int i = 0; i < EPL; i++) {
        uint32_t idx = wbeg + i * 64 + lane;
        if (idx < tile_nn) {
          if (GEN) {
            uint64_t x = seed_hi + startn + idx;
            rkn[i] = KT<K>::from_raw(d_splitmix64(x) & keymask);
            if (WITH_VALS)
              rvn[i] = (double)(d_splitmix64(x ^ 0xDEADBEEFCAFEBABEull)
                                >> 11) * (1.0 / 9007199254740992.0);
            continue;
          }
          rkn[i] = keys_in[startn + idx];
          if (WITH_VALS) rvn[i] = vals_in[startn + idx];
        }
      }
    }

    /* lookback (thread = digit) overlaps the in-flight prefetch */
    if (!DPRK_WALK_NOW && threadIdx.x < RADIX) {
      uint64_t excl_g = 0;
      uint32_t spins = 0;
      bool ok = true;
      int64_t s = (int64_t)t - 1;
      while (s >= 0) {
        uint64_t e = __hip_atomic_load(&lb[(uint64_t)s * RADIX + threadIdx.x],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
        if (!(e >> 62)) {
          __builtin_amdgcn_s_sleep(2);
          if (++spins > LB_SPIN_LIMIT) {
            __hip_atomic_store(d_fail, 1u, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            ok = false;
            break;
          }
          continue;
        }
        excl_g += e & LB_VAL;
        if ((e >> 62) >= 2) break;
        s--;
      }
      if (ok) {
        __hip_atomic_store(&lb[t * RADIX + threadIdx.x],
                           LB_INC | (excl_g + tile_cnt), __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        goff[threadIdx.x] = digit_base[threadIdx.x] + (uint32_t)excl_g -
                            excl[threadIdx.x];
      } else {
        sfail = 1;
      }
    }
    __syncthreads();
    if (sfail) return false;

    /* phase 4: write out; stores drain while the tail runs */
    for (uint32_t idx = threadIdx.x; idx < tile_n; idx += NTHREADS_SC) {
      K key = lk[idx];
      uint32_t dig = d_digit(key, 0, shift, mask);
      keys_out[goff[dig] + idx] = key;
      if (WITH_VALS) vals_out[goff[dig] + idx] = lv[idx];
    }

    /* tail: zero the consumed hist (becomes t+2*grid's buffer) and
     * histogram the prefetched tile (consumes the early loads) */
    for (int j = 0; j < RADIX / 64; j++)
      hc[wave][lane * (RADIX / 64) + j] = 0;
    if (tn < ntiles) {
#pragma unroll
      for (int i = 0; i < EPL; i++) {
        uint32_t idx = wbeg + i * 64 + lane;
        if (idx < tile_nn) {
          rdn[i] = d_digit(rkn[i], 0, shift, mask);
          atomicAdd(&hn[wave][rdn[i]], 1u);
        }
      }
    }
    return true;
  };

  uint64_t t = t0;
  for (;;) {
    if (!stage(rk0, rv0, rd0, rk1, rv1, rd1, h[0], h[1], t)) return;
    t += gridDim.x;
    if (t >= ntiles) return;
    if (!stage(rk1, rv1, rd1, rk0, rv0, rd0, h[1], h[0], t)) return;
    t += gridDim.x;
    if (t >= ntiles) return;
  }
}

/* Two-deep key prefetch variant (DPRK_PIPE2): keys of tile t+2 issue at
 * tile t (a FULL tile of latency cover before their tail-hist use —
 * the one-deep pipe only covers them with the short write phase), vals
 * stay one-deep (consumed at staging, already a full tile of cover).
 * 3 key buffers (+8 VGPRs, alloc 120 <= 128: occupancy preserved at 4
 * waves/SIMD). */
template <typename K, bool WITH_VALS>
__global__ __launch_bounds__(NTHREADS_SC, 4) void k_scatter_lb_pipe2(
    const K *__restrict__ keys_in, const double *__restrict__ vals_in,
    K *__restrict__ keys_out, double *__restrict__ vals_out,
    uint64_t n, uint64_t ntiles, int shift,
    const uint32_t *__restrict__ digit_base,
    unsigned long long *__restrict__ lb,
    unsigned int *__restrict__ d_fail) {
  constexpr int NW = NTHREADS_SC / 64;
  constexpr int WBLK = TILE / NW;
  constexpr int EPL = WBLK / 64;
  __shared__ K lk[TILE];
  __shared__ double lv[WITH_VALS ? TILE : 1];
  __shared__ uint32_t excl[RADIX];
  __shared__ uint32_t goff[RADIX];
  __shared__ uint32_t h[2][NW][RADIX];
  __shared__ uint32_t wtot[NW > 4 ? NW : 4];
  __shared__ uint32_t sfail;

  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const uint32_t mask = 0xFF;
  const uint32_t wbeg = wave * WBLK;

  uint64_t t0 = blockIdx.x;
  if (t0 >= ntiles) return;
  for (int j = 0; j < RADIX / 64; j++) {
    h[0][wave][lane * (RADIX / 64) + j] = 0;
    h[1][wave][lane * (RADIX / 64) + j] = 0;
  }
  if (threadIdx.x == 0) sfail = 0;

  K ka[EPL], kb[EPL], kc3[EPL];
  double va[EPL], vb[EPL];
  uint32_t da[EPL], db[EPL];

  auto loadk = [&](K (&dst)[EPL], uint64_t t) {
    if (t >= ntiles) return;
    const uint64_t start = t * (uint64_t)TILE;
    const uint32_t tile_n = (uint32_t)(min(start + TILE, n) - start);
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) dst[i] = keys_in[start + idx];
    }
  };
  auto loadv = [&](double (&dst)[EPL], uint64_t t) {
    if (!WITH_VALS || t >= ntiles) return;
    const uint64_t start = t * (uint64_t)TILE;
    const uint32_t tile_n = (uint32_t)(min(start + TILE, n) - start);
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) dst[i] = vals_in[start + idx];
    }
  };
  auto histd = [&](K (&keys)[EPL], uint32_t (&dig)[EPL],
                   uint32_t (*hh)[RADIX], uint64_t t) {
    if (t >= ntiles) return;
    const uint64_t start = t * (uint64_t)TILE;
    const uint32_t tile_n = (uint32_t)(min(start + TILE, n) - start);
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        dig[i] = d_digit(keys[i], 0, shift, mask);
        atomicAdd(&hh[wave][dig[i]], 1u);
      }
    }
  };

  /* prolog: keys t0+t1 + vals t0 in flight; hist(t0) */
  __syncthreads();          /* zeroed hists visible */
  loadk(ka, t0);
  loadk(kb, t0 + gridDim.x);
  loadv(va, t0);
  histd(ka, da, h[0], t0);

  /* one stage; (kcur, vcur, dcur, hcur) hold tile t; kload receives the
   * keys of t+2, vload the vals of t+1; ktail feeds t+1's tail hist. */
  auto stage = [&](K (&kcur)[EPL], double (&vcur)[EPL],
                   uint32_t (&dcur)[EPL], K (&ktail)[EPL],
                   uint32_t (&dnext)[EPL], K (&kload)[EPL],
                   double (&vload)[EPL], uint32_t (*hcur)[RADIX],
                   uint32_t (*hnext)[RADIX], uint64_t t) -> bool {
    if (threadIdx.x == 0)
      sfail = __hip_atomic_load(d_fail, __ATOMIC_RELAXED,
                                __HIP_MEMORY_SCOPE_AGENT);
    __syncthreads();
    if (sfail) return false;
    const uint64_t start = t * (uint64_t)TILE;
    const uint32_t tile_n = (uint32_t)(min(start + TILE, n) - start);

    uint32_t tile_cnt = 0;
    if (threadIdx.x < RADIX) {
      for (int w = 0; w < NW; w++) tile_cnt += hcur[w][threadIdx.x];
      __hip_atomic_store(&lb[t * RADIX + threadIdx.x], LB_AGG | tile_cnt,
                         __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      uint32_t inc = d_wave_incl_scan(tile_cnt, lane);
      if (lane == 63) wtot[wave] = inc;
      excl[threadIdx.x] = inc - tile_cnt;
    }
    __syncthreads();
    if (threadIdx.x < RADIX) {
      if (wave > 0) {
        uint32_t add = 0;
        for (int w = 0; w < wave; w++) add += wtot[w];
        excl[threadIdx.x] += add;
      }
      uint32_t b = excl[threadIdx.x];
      for (int w = 0; w < NW; w++) {
        uint32_t cnt = hcur[w][threadIdx.x];
        hcur[w][threadIdx.x] = b;
        b += cnt;
      }
    }
    __syncthreads();

    /* rank + stage; issue t+2 keys and t+1 vals */
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        uint32_t pos = atomicAdd(&hcur[wave][dcur[i]], 1u);
        lk[pos] = kcur[i];
        if (WITH_VALS) lv[pos] = vcur[i];
      }
    }
    loadk(kload, t + 2 * gridDim.x);
    loadv(vload, t + gridDim.x);

    /* lookback walk */
    if (threadIdx.x < RADIX) {
      uint64_t excl_g = 0;
      uint32_t spins = 0;
      bool ok = true;
      int64_t s = (int64_t)t - 1;
      while (s >= 0) {
        uint64_t e = __hip_atomic_load(&lb[(uint64_t)s * RADIX + threadIdx.x],
                                       __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
        if (!(e >> 62)) {
          __builtin_amdgcn_s_sleep(2);
          if (++spins > LB_SPIN_LIMIT) {
            __hip_atomic_store(d_fail, 1u, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
            ok = false;
            break;
          }
          continue;
        }
        excl_g += e & LB_VAL;
        if ((e >> 62) >= 2) break;
        s--;
      }
      if (ok) {
        __hip_atomic_store(&lb[t * RADIX + threadIdx.x],
                           LB_INC | (excl_g + tile_cnt), __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
        goff[threadIdx.x] = digit_base[threadIdx.x] + (uint32_t)excl_g -
                            excl[threadIdx.x];
      } else {
        sfail = 1;
      }
    }
    __syncthreads();
    if (sfail) return false;

    for (uint32_t idx = threadIdx.x; idx < tile_n; idx += NTHREADS_SC) {
      K key = lk[idx];
      uint32_t dig = d_digit(key, 0, shift, mask);
      keys_out[goff[dig] + idx] = key;
      if (WITH_VALS) vals_out[goff[dig] + idx] = lv[idx];
    }

    /* tail: zero the consumed hist; hist + digits of t+1 (keys were
     * issued a FULL tile ago) */
    for (int j = 0; j < RADIX / 64; j++)
      hcur[wave][lane * (RADIX / 64) + j] = 0;
    histd(ktail, dnext, hnext, t + gridDim.x);
    return true;
  };

  /* 6-stage rotation: keys period 3 (a,b,c), vals/digits/hists period 2 */
  uint64_t t = t0;
#define PSTEP(KC, VC, DC, KT, DN, KL, VL, HC, HN)                         \
  do {                                                                     \
    if (!stage(KC, VC, DC, KT, DN, KL, VL, HC, HN, t)) return;             \
    t += gridDim.x;                                                        \
    if (t >= ntiles) return;                                               \
  } while (0)
  for (;;) {
    PSTEP(ka, va, da, kb, db, kc3, vb, h[0], h[1]);
    PSTEP(kb, vb, db, kc3, da, ka, va, h[1], h[0]);
    PSTEP(kc3, va, da, ka, db, kb, vb, h[0], h[1]);
    PSTEP(ka, vb, db, kb, da, kc3, va, h[1], h[0]);
    PSTEP(kb, va, da, kc3, db, ka, vb, h[0], h[1]);
    PSTEP(kc3, vb, db, ka, da, kb, va, h[1], h[0]);
  }
#undef PSTEP
}

template <typename K, bool WITH_VALS, bool RANK_ATOMIC = false>
__global__ __launch_bounds__(NTHREADS_SC, 2) void k_scatter(
    const K *__restrict__ keys_in, const double *__restrict__ vals_in,
    K *__restrict__ keys_out, double *__restrict__ vals_out,
    uint64_t n, const uint32_t *__restrict__ tile_hist,
    const uint32_t *__restrict__ tile_off, uint64_t ntiles, int mode,
    int shift, uint32_t mask, const int64_t *__restrict__ splitters,
    int nsplit) {
  constexpr int NW = NTHREADS_SC / 64;     /* 8 waves per WG              */
  constexpr int WBLK = TILE / NW;          /* 512 elements per wave       */
  constexpr int EPL = WBLK / 64;           /* 8 elements per lane         */
  __shared__ K lk[TILE];
  __shared__ double lv[WITH_VALS ? TILE : 1];
  __shared__ uint32_t excl[RADIX];     /* tile-local digit excl scan  */
  __shared__ uint32_t goff[RADIX];     /* global write base per digit */
#ifdef DPRK_PACKED_H
  /* two u16 counters per word (digits 2d, 2d+1): halves the per-wave
   * histogram LDS so 16-wave workgroups still fit 2 blocks/CU */
  __shared__ uint32_t h[NW][RADIX / 2];
#else
  __shared__ uint32_t h[NW][RADIX];    /* per-wave hist -> wave bases
                                          -> running counters (in place) */
#endif
  __shared__ uint32_t wtot[NW > 4 ? NW : 4]; /* cross-wave scan carries  */

  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const uint64_t lowmask = ((uint64_t)1 << lane) - 1;

  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const uint64_t start = t * (uint64_t)TILE;
    const uint64_t end = min(start + TILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
    const uint32_t wbeg = wave * WBLK;

    /* phase 0: per-tile digit metadata; excl scan via wave shfl scans
     * (2 barriers, vs 16 for the Hillis-Steele form); zero h rows */
    uint32_t myv = 0;
    if (threadIdx.x < RADIX) {
      myv = tile_hist[t * RADIX + threadIdx.x];
      goff[threadIdx.x] = tile_off[t * RADIX + threadIdx.x];
      uint32_t inc = d_wave_incl_scan(myv, lane);
      if (lane == 63) wtot[wave] = inc;
      excl[threadIdx.x] = inc - myv; /* within-wave exclusive */
    }
#ifdef DPRK_PACKED_H
    for (int j = threadIdx.x; j < NW * RADIX / 2; j += NTHREADS_SC)
      ((uint32_t *)h)[j] = 0;
#else
    for (int j = 0; j < RADIX / 64; j++) h[wave][lane * (RADIX / 64) + j] = 0;
#endif
    __syncthreads();
    if (threadIdx.x < RADIX) {
      uint32_t add = 0;
      for (int w = 0; w < wave; w++) add += wtot[w];
      uint32_t e = excl[threadIdx.x] + add;
      excl[threadIdx.x] = e;
      goff[threadIdx.x] -= e;   /* combined: phase 4 dst = goff[dig]+idx */
    }

    /* phase 1: load into registers + wave-private histogram */
    K rk[EPL];
    double rv[EPL];
    uint32_t rd[EPL];
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        rk[i] = keys_in[start + idx];
        if (WITH_VALS) rv[i] = vals_in[start + idx];
        rd[i] = d_digit_s(rk[i], mode, shift, mask, splitters, nsplit);
#ifdef DPRK_PACKED_H
        atomicAdd(&h[wave][rd[i] >> 1], 1u << (16 * (rd[i] & 1)));
#else
        atomicAdd(&h[wave][rd[i]], 1u);
#endif
      }
    }
    __syncthreads();

    /* phase 2: per-wave digit bases in place */
#ifdef DPRK_PACKED_H
    if (threadIdx.x < RADIX / 2) {      /* thread = digit pair */
      uint32_t d = threadIdx.x;
      uint32_t b0 = excl[2 * d], b1 = excl[2 * d + 1];
      for (int w = 0; w < NW; w++) {
        uint32_t word = h[w][d];
        h[w][d] = (b0 & 0xFFFFu) | (b1 << 16);
        b0 += word & 0xFFFFu;
        b1 += word >> 16;
      }
    }
#else
    if (threadIdx.x < RADIX) {
      uint32_t d = threadIdx.x;
      uint32_t b = excl[d];
      for (int w = 0; w < NW; w++) {
        uint32_t cnt = h[w][d];
        h[w][d] = b;
        b += cnt;
      }
    }
#endif
    __syncthreads();

    /* phase 3: stable rank (wave-local, no barriers) + LDS scatter */
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        uint32_t dig = rd[i];
        uint32_t pos;
#ifndef DPRK_PACKED_H
        if (RANK_ATOMIC) {
          pos = atomicAdd(&h[wave][dig], 1u);
        } else
#endif
        {
        uint64_t same = __ballot(1);
        for (int b = 0; b < RADIX_BITS; b++) {
          uint64_t bb = __ballot((dig >> b) & 1);
          same &= ((dig >> b) & 1) ? bb : ~bb;
        }
        uint32_t rank_in_wave = (uint32_t)__popcll(same & lowmask);
#ifdef DPRK_PACKED_H
        uint32_t word = h[wave][dig >> 1];        /* read before leader  */
        uint32_t base = (word >> (16 * (dig & 1))) & 0xFFFFu;
        if (rank_in_wave == 0)                    /* leader advances     */
          atomicAdd(&h[wave][dig >> 1],
                    ((uint32_t)__popcll(same)) << (16 * (dig & 1)));
#else
        uint32_t base = h[wave][dig];             /* broadcast LDS read  */
        if (rank_in_wave == 0)                    /* leader advances     */
          h[wave][dig] = base + (uint32_t)__popcll(same);
#endif
        pos = base + rank_in_wave;
        }
        lk[pos] = rk[i];
        if (WITH_VALS) lv[pos] = rv[i];
      }
    }
    __syncthreads();

    /* phase 4: write out in digit-sorted tile order */
    for (uint32_t idx = threadIdx.x; idx < tile_n; idx += NTHREADS_SC) {
      K key = lk[idx];
      uint32_t dig = d_digit_s(key, mode, shift, mask, splitters, nsplit);
      uint32_t dst = goff[dig] + idx;   /* goff folds -excl[dig] */
      keys_out[dst] = key;
      if (WITH_VALS) vals_out[dst] = lv[idx];
    }
    __syncthreads();
  }
}

/* ------------------------------------------------------------------ */
/* FUSED last-pass scatter+reduce (opt-in: DPRK_FUSED_REDUCE)          */
/*                                                                     */
/* In the FINAL radix pass the digit-sorted LDS tile is fully KEY-     */
/* sorted (LSD stability).  Phase R emits one (key, partial-sum)       */
/* FRAGMENT per within-tile run — runs crossing tile/segment borders   */
/* simply produce several fragments.  Reordered d-major (the output    */
/* order), fragments of one logical run are ADJACENT, so the standard  */
/* segmented-reduce pipeline run on the ~n/6 fragment stream merges    */
/* them exactly.  Deterministic: fragment order is position order.     */
/* (First cut kept per-segment records + a key-gated stitch — correct  */
/* but the 36 B/segment record traffic ate the win; git history.)      */
/* ------------------------------------------------------------------ */

template <typename K>
__global__ __launch_bounds__(NTHREADS_SC, 4) void k_scatter_reduce(
    const K *__restrict__ keys_in, const double *__restrict__ vals_in,
    K *__restrict__ keys_scr, double *__restrict__ sums_scr,
    uint64_t n, const uint32_t *__restrict__ tile_hist, uint64_t ntiles,
    int shift, uint32_t *__restrict__ em_counts /* [t*RADIX+d] */,
    uint32_t *__restrict__ em_excl /* [t*RADIX+d] tile-local base */) {
  constexpr int NW = NTHREADS_SC / 64;
  constexpr int WBLK = TILE / NW;
  constexpr int EPL = WBLK / 64;
  constexpr int EPT = TILE / NTHREADS_SC;
#define FPIDX(j) ((j) + ((j) >> 3))
  __shared__ K lk[TILE + TILE / 8];
  __shared__ uint16_t pmap[TILE + TILE / 8];
  __shared__ uint32_t excl[RADIX];
  __shared__ uint32_t h[NW][RADIX];
  __shared__ uint32_t wtot[NW > 4 ? NW : 4];
  __shared__ uint32_t cnt256[RADIX];   /* fragment count per digit */
  __shared__ double t_carry[NTHREADS_SC];
  __shared__ uint32_t t_flag[NTHREADS_SC];
  __shared__ uint32_t t_ends[NTHREADS_SC];

  const int wave = threadIdx.x / 64, lane = threadIdx.x % 64;
  const uint64_t lowmask = ((uint64_t)1 << lane) - 1;
  const uint32_t mask = 0xFF;

  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const uint64_t start = t * (uint64_t)TILE;
    const uint64_t end = min(start + TILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
    const uint32_t wbeg = wave * WBLK;

    /* ---- phases 0-3: the standard scatter (posmap form) ---- */
    uint32_t myv = 0;
    if (threadIdx.x < RADIX) {
      myv = tile_hist[t * RADIX + threadIdx.x];
      uint32_t inc = d_wave_incl_scan(myv, lane);
      if (lane == 63) wtot[wave] = inc;
      excl[threadIdx.x] = inc - myv;
      cnt256[threadIdx.x] = 0;
    }
    for (int j = 0; j < RADIX / 64; j++) h[wave][lane * (RADIX / 64) + j] = 0;
    __syncthreads();
    if (threadIdx.x < RADIX && wave > 0) {
      uint32_t add = 0;
      for (int w = 0; w < wave; w++) add += wtot[w];
      excl[threadIdx.x] += add;
    }
    K rk[EPL];
    uint32_t rd[EPL];
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        rk[i] = keys_in[start + idx];
        rd[i] = d_digit(rk[i], 0, shift, mask);
        atomicAdd(&h[wave][rd[i]], 1u);
      }
    }
    __syncthreads();
    if (threadIdx.x < RADIX) {
      uint32_t d = threadIdx.x;
      uint32_t b = excl[d];
      for (int w = 0; w < NW; w++) {
        uint32_t cnt = h[w][d];
        h[w][d] = b;
        b += cnt;
      }
    }
    __syncthreads();
#pragma unroll
    for (int i = 0; i < EPL; i++) {
      uint32_t idx = wbeg + i * 64 + lane;
      if (idx < tile_n) {
        uint32_t dig = rd[i];
        uint64_t same = __ballot(1);
        for (int bit = 0; bit < RADIX_BITS; bit++) {
          uint64_t bb = __ballot((dig >> bit) & 1);
          same &= ((dig >> bit) & 1) ? bb : ~bb;
        }
        uint32_t rank_in_wave = (uint32_t)__popcll(same & lowmask);
        uint32_t base = h[wave][dig];
        if (rank_in_wave == 0)
          h[wave][dig] = base + (uint32_t)__popcll(same);
        uint32_t pos = base + rank_in_wave;
        lk[FPIDX(pos)] = rk[i];
        pmap[FPIDX(pos)] = (uint16_t)idx;
      }
    }
    __syncthreads();

    /* ---- phase R: per-run FRAGMENT emission over the sorted tile ----
     * head(p) = p==0 || key(p)!=key(p-1); frag end = p==tile_n-1 ||
     * key(p+1)!=key(p).  Values register-gathered once via pmap. */
    const uint32_t c0 = threadIdx.x * EPT;
    double cv[EPT];
#pragma unroll
    for (int j = 0; j < EPT; j++) {
      uint32_t p = c0 + j;
      if (p < tile_n) cv[j] = vals_in[start + pmap[FPIDX(p)]];
    }
#define KEYAT(p) lk[FPIDX(p)]

    double open = 0.0;
    uint32_t has_head = 0, n_ends = 0;
    if (c0 < tile_n) {
      bool have_prev = c0 > 0;
      K prev_key = have_prev ? KEYAT(c0 - 1) : (K)0;
#pragma unroll
      for (int j = 0; j < EPT; j++) {
        uint32_t p = c0 + j;
        if (p >= tile_n) break;
        K k = KEYAT(p);
        bool head = !have_prev || k != prev_key;
        if (head) { open = cv[j]; has_head = 1; }
        else open += cv[j];
        if (p + 1 == tile_n || KEYAT(p + 1) != k) {
          n_ends++;
          atomicAdd(&cnt256[d_digit(k, 0, shift, mask)], 1u);
        }
        prev_key = k;
        have_prev = true;
      }
    }
    t_carry[threadIdx.x] = open;
    t_flag[threadIdx.x] = has_head;
    t_ends[threadIdx.x] = n_ends;
    __syncthreads();
    for (int off = 1; off < NTHREADS_SC; off <<= 1) {
      double s_prev = 0.0;
      uint32_t f_prev = 0, e_prev = 0;
      if (threadIdx.x >= (unsigned)off) {
        s_prev = t_carry[threadIdx.x - off];
        f_prev = t_flag[threadIdx.x - off];
        e_prev = t_ends[threadIdx.x - off];
      }
      __syncthreads();
      if (threadIdx.x >= (unsigned)off) {
        if (!t_flag[threadIdx.x]) {
          t_carry[threadIdx.x] += s_prev;
          t_flag[threadIdx.x] = f_prev;
        }
        t_ends[threadIdx.x] += e_prev;
      }
      __syncthreads();
    }
    {
      double incoming = 0.0;
      if (threadIdx.x > 0 && c0 < tile_n)
        incoming = t_carry[threadIdx.x - 1];
      uint32_t w = t_ends[threadIdx.x] - n_ends;   /* exclusive emit base */
      __syncthreads();
      if (c0 < tile_n) {
        double running = incoming;
        bool have_prev = c0 > 0;
        K prev_key = have_prev ? KEYAT(c0 - 1) : (K)0;
#pragma unroll
        for (int j = 0; j < EPT; j++) {
          uint32_t p = c0 + j;
          if (p >= tile_n) break;
          K k = KEYAT(p);
          bool head = !have_prev || k != prev_key;
          if (head) running = cv[j];
          else running += cv[j];
          if (p + 1 == tile_n || KEYAT(p + 1) != k) {
            keys_scr[start + w] = k;
            sums_scr[start + w] = running;
            w++;
          }
          prev_key = k;
          have_prev = true;
        }
      }
    }
    __syncthreads();
    /* per-(tile,digit) fragment counts + tile-local digit bases */
    if (threadIdx.x < RADIX) {
      uint32_t cme = cnt256[threadIdx.x];
      em_counts[t * RADIX + threadIdx.x] = cme;
      uint32_t inc = d_wave_incl_scan(cme, lane);
      if (lane == 63) wtot[wave] = inc;
      em_excl[t * RADIX + threadIdx.x] = inc - cme; /* within-wave excl */
    }
    __syncthreads();
    if (threadIdx.x < RADIX && wave > 0) {
      uint32_t add = 0;
      for (int w = 0; w < wave; w++) add += wtot[w];
      em_excl[t * RADIX + threadIdx.x] += add;
    }
    __syncthreads();
  }
#undef FPIDX
#undef KEYAT
}

/* gather tile-local fragments into the d-major fragment stream */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_fused_gather(
    const uint32_t *__restrict__ em_counts,
    const uint32_t *__restrict__ em_excl,
    const uint32_t *__restrict__ goff /* d-major offsets, [t*RADIX+d] */,
    const K *__restrict__ keys_scr, const double *__restrict__ sums_scr,
    uint64_t ntiles, uint64_t nrec, K *__restrict__ stream_k,
    double *__restrict__ stream_v) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t r = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       r < nrec; r += stride) {
    uint64_t d = r / ntiles, t = r % ntiles;   /* d-major iteration */
    uint64_t rt = t * RADIX + d;
    uint32_t cme = em_counts[rt];
    if (cme == 0) continue;
    uint64_t srcb = t * (uint64_t)TILE + em_excl[rt];
    uint64_t dstb = goff[rt];
    for (uint32_t j = 0; j < cme; j++) {
      stream_k[dstb + j] = keys_scr[srcb + j];
      stream_v[dstb + j] = sums_scr[srcb + j];
    }
  }
}

/* ------------------------------------------------------------------ */
/* kernels: segmented float64 reduce (reduceByKey(+))                  */
/*   dpark reduce-side Merger dict-merge equivalent [U], deterministic */
/*   fixed-tree per SURVEY.md §7 hard part 3.                          */
/* ------------------------------------------------------------------ */

/* per-tile pass: emit (key, within-tile run sum) for every run END in the
 * tile, compacted into scratch at tile*TILE; record per-tile trailing open
 * sum T, any-end flag and run count for the stitch. Input must be sorted.
 * Run ends are globally exact (next-key lookahead crosses tiles). */
/* per-thread chunks are CONTIGUOUS 16-element runs, so LDS indices are
 * padded (PIDX: +1 slot per 16) — the unpadded layout put lanes l and
 * l+2 on the same bank (16-way conflict, measured 1.26 ms/launch). */
#define PIDX(j) ((j) + ((j) >> 3))
template <typename K>
__global__ __launch_bounds__(NTHREADS, 2) void k_reduce_tiles(
    const K *__restrict__ keys, const double *__restrict__ vals,
    uint64_t n, K *__restrict__ keys_scr, double *__restrict__ sums_scr,
    double *__restrict__ tile_T, uint32_t *__restrict__ tile_anyend,
    uint32_t *__restrict__ tile_nruns, uint64_t ntiles) {
  __shared__ K lk[RTILE + RTILE / 8 + 1];  /* padded; +1: lookahead */
  __shared__ double lv[RTILE + RTILE / 8];
  __shared__ double t_carry[NTHREADS];   /* per-thread trailing open sum */
  __shared__ uint32_t t_flag[NTHREADS];  /* thread chunk has a head      */
  __shared__ uint32_t t_ends[NTHREADS];  /* run ends in thread chunk     */

  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const uint64_t start = t * (uint64_t)RTILE;
    const uint64_t end = min(start + RTILE, n);
    const uint32_t tile_n = (uint32_t)(end - start);
    /* 16-B staging loads; full tiles take a constant-trip unrolled path
     * so all loads issue up front (variable-bound loops keep ~1 load in
     * flight; the kernel was 69% wave-parked) */
    if (tile_n == RTILE) {
      typename KT<K>::V2 kk[RTILE / (2 * NTHREADS)];
      double2 vv[RTILE / (2 * NTHREADS)];
#pragma unroll
      for (int j = 0; j < RTILE / (2 * NTHREADS); j++) {
        uint32_t i = 2 * threadIdx.x + (uint32_t)j * 2 * NTHREADS;
        kk[j] = *reinterpret_cast<const typename KT<K>::V2 *>(&keys[start + i]);
        vv[j] = *reinterpret_cast<const double2 *>(&vals[start + i]);
      }
#pragma unroll
      for (int j = 0; j < RTILE / (2 * NTHREADS); j++) {
        uint32_t i = 2 * threadIdx.x + (uint32_t)j * 2 * NTHREADS;
        lk[PIDX(i)] = kk[j].x;
        lk[PIDX(i + 1)] = kk[j].y;
        lv[PIDX(i)] = vv[j].x;
        lv[PIDX(i + 1)] = vv[j].y;
      }
    } else {
      for (uint32_t i = 2 * threadIdx.x; i + 1 < tile_n; i += 2 * NTHREADS) {
        typename KT<K>::V2 kk =
            *reinterpret_cast<const typename KT<K>::V2 *>(&keys[start + i]);
        double2 vv = *reinterpret_cast<const double2 *>(&vals[start + i]);
        lk[PIDX(i)] = kk.x;
        lk[PIDX(i + 1)] = kk.y;
        lv[PIDX(i)] = vv.x;
        lv[PIDX(i + 1)] = vv.y;
      }
      if ((tile_n & 1) && threadIdx.x == 0) {
        lk[PIDX(tile_n - 1)] = keys[start + tile_n - 1];
        lv[PIDX(tile_n - 1)] = vals[start + tile_n - 1];
      }
    }
    if (threadIdx.x == 0)
      lk[PIDX(tile_n)] = (end < n) ? keys[end] : (K)0; /* lookahead (unused if end==n) */
    __syncthreads();

    const uint32_t c0 = threadIdx.x * EPT_R;                    /* chunk */
    const uint32_t c1 = min(c0 + (uint32_t)EPT_R, tile_n);

    /* pass 1: trailing open sum + has-head + end count per thread chunk */
    double open = 0.0;
    uint32_t has_head = 0, n_ends = 0;
    K prev_key = 0;
    bool have_prev = false;
    if (c0 < tile_n) {
      if (c0 == 0) {
        if (start == 0) { have_prev = false; }
        else { prev_key = keys[start - 1]; have_prev = true; }
      } else { prev_key = lk[PIDX(c0 - 1)]; have_prev = true; }
      for (uint32_t i = c0; i < c1; i++) {
        bool head = !have_prev || lk[PIDX(i)] != prev_key;
        if (head) { open = lv[PIDX(i)]; has_head = 1; }
        else open += lv[PIDX(i)];
        bool is_end = (start + i == n - 1) ||
                      ((i + 1 < tile_n) ? lk[PIDX(i + 1)] != lk[PIDX(i)]
                                        : lk[PIDX(tile_n)] != lk[PIDX(i)]);
        if (is_end) n_ends++;
        prev_key = lk[PIDX(i)];
        have_prev = true;
      }
    }
    t_carry[threadIdx.x] = open;
    t_flag[threadIdx.x] = has_head;
    t_ends[threadIdx.x] = n_ends;
    __syncthreads();

    /* cross-thread segmented inclusive scan of (carry, flag); and plain
     * exclusive scan of end counts. Hillis-Steele, fixed tree. */
    for (int off = 1; off < NTHREADS; off <<= 1) {
      double s_prev = 0.0;
      uint32_t f_prev = 0, e_prev = 0;
      if (threadIdx.x >= (unsigned)off) {
        s_prev = t_carry[threadIdx.x - off];
        f_prev = t_flag[threadIdx.x - off];
        e_prev = t_ends[threadIdx.x - off];
      }
      __syncthreads();
      if (threadIdx.x >= (unsigned)off) {
        if (!t_flag[threadIdx.x]) {
          t_carry[threadIdx.x] += s_prev;
          t_flag[threadIdx.x] = f_prev;
        }
        t_ends[threadIdx.x] += e_prev;
      }
      __syncthreads();
    }
    /* incoming within-tile carry for thread = inclusive seg-scan of the
     * PREVIOUS thread; end-write base = exclusive scan of ends */
    double incoming = 0.0;
    if (threadIdx.x > 0 && c0 < tile_n) incoming = t_carry[threadIdx.x - 1];
    /* t_flag after scan is unused below; t_ends is inclusive -> make
     * exclusive via subtraction of own count */
    uint32_t ends_excl = t_ends[threadIdx.x] - n_ends;
    __syncthreads();

    /* pass 2: walk chunk again with the incoming carry; emit run ends */
    if (c0 < tile_n) {
      double running = incoming;
      K prev_key2 = 0;
      bool have_prev2 = false;
      if (c0 == 0) {
        if (start != 0) { prev_key2 = keys[start - 1]; have_prev2 = true; }
      } else { prev_key2 = lk[PIDX(c0 - 1)]; have_prev2 = true; }
      uint32_t w = ends_excl;
      for (uint32_t i = c0; i < c1; i++) {
        bool head = !have_prev2 || lk[PIDX(i)] != prev_key2;
        if (head) running = lv[PIDX(i)];
        else running += lv[PIDX(i)];
        bool is_end = (start + i == n - 1) ||
                      ((i + 1 < tile_n) ? lk[PIDX(i + 1)] != lk[PIDX(i)]
                                        : lk[PIDX(tile_n)] != lk[PIDX(i)]);
        if (is_end) {
          keys_scr[start + w] = lk[PIDX(i)];
          sums_scr[start + w] = running;
          w++;
        }
        prev_key2 = lk[PIDX(i)];
        have_prev2 = true;
      }
      /* last thread with data records the tile trailing state */
      if (c1 == tile_n) {
        bool last_is_end = (start + tile_n - 1 == n - 1) ||
                           (lk[PIDX(tile_n)] != lk[PIDX(tile_n - 1)]);
        tile_T[t] = last_is_end ? 0.0 : running;
        /* any_end for the tile = total ends > 0 */
        tile_anyend[t] = (t_ends[NTHREADS - 1] > 0) ? 1u : 0u;
        tile_nruns[t] = t_ends[NTHREADS - 1];
      }
    }
    __syncthreads();
  }
}

/* stitch A: per-chunk segmented scan over tiles of (T, any_end) -> the
 * WITHIN-CHUNK incoming carry per tile, plus per-chunk aggregates; also
 * within-chunk exclusive run-count offsets.  Chunk = STITCH_CHUNK tiles,
 * one workgroup per chunk.  tile_C gets the within-chunk value; whether
 * the chain reaches the chunk start is recorded in chain_open[t].
 * (Replaces the single-workgroup stitch — 0.19 ms at 24k tiles, 2x at
 * 48k — with a 3-kernel decoupled scan.) */
__global__ __launch_bounds__(NTHREADS) void k_stitchA(
    const double *__restrict__ tile_T, const uint32_t *__restrict__ tile_anyend,
    const uint32_t *__restrict__ tile_nruns, uint64_t ntiles,
    double *__restrict__ tile_C, uint8_t *__restrict__ chain_open,
    uint64_t *__restrict__ tile_outoff,
    double *__restrict__ chunk_T, uint32_t *__restrict__ chunk_F,
    uint64_t *__restrict__ chunk_N) {
  __shared__ double sT[NTHREADS];
  __shared__ uint32_t sF[NTHREADS];
  __shared__ uint64_t sN[NTHREADS];
  uint64_t c0 = (uint64_t)blockIdx.x * STITCH_CHUNK;
  uint64_t c1 = min(c0 + STITCH_CHUNK, ntiles);
  double carry = 0.0;      /* within-chunk running open sum   */
  uint32_t closed = 0;     /* any_end seen within chunk       */
  uint64_t off_run = 0;
  for (uint64_t c = c0; c < c1; c += NTHREADS) {
    uint64_t t = c + threadIdx.x;
    bool in = t < c1;
    double myT = in ? tile_T[t] : 0.0;
    uint32_t myF = in ? tile_anyend[t] : 1u;
    uint64_t myN = in ? (uint64_t)tile_nruns[t] : 0;
    sT[threadIdx.x] = myT;
    sF[threadIdx.x] = myF;
    sN[threadIdx.x] = myN;
    __syncthreads();
    for (int off = 1; off < NTHREADS; off <<= 1) {
      double s_prev = 0.0;
      uint32_t f_prev = 0;
      uint64_t n_prev = 0;
      if (threadIdx.x >= (unsigned)off) {
        s_prev = sT[threadIdx.x - off];
        f_prev = sF[threadIdx.x - off];
        n_prev = sN[threadIdx.x - off];
      }
      __syncthreads();
      if (threadIdx.x >= (unsigned)off) {
        if (!sF[threadIdx.x]) {
          sT[threadIdx.x] += s_prev;
          sF[threadIdx.x] = f_prev;
        }
        sN[threadIdx.x] += n_prev;
      }
      __syncthreads();
    }
    if (in) {
      double excl_T = threadIdx.x ? sT[threadIdx.x - 1] : 0.0;
      uint32_t excl_F = threadIdx.x ? sF[threadIdx.x - 1] : 0;
      tile_C[t] = excl_T + (excl_F ? 0.0 : carry);
      chain_open[t] = (uint8_t)!(excl_F | closed);
      tile_outoff[t] = off_run + (sN[threadIdx.x] - myN);
    }
    __syncthreads();
    double blk_T = sT[NTHREADS - 1];
    uint32_t blk_F = sF[NTHREADS - 1];
    uint64_t blk_N = sN[NTHREADS - 1];
    __syncthreads();
    carry = blk_F ? blk_T : (blk_T + carry);
    closed |= blk_F;
    off_run += blk_N;
  }
  if (threadIdx.x == 0) {
    chunk_T[blockIdx.x] = carry;
    chunk_F[blockIdx.x] = closed;
    chunk_N[blockIdx.x] = off_run;
  }
}

/* stitch B (single WG): segmented exclusive scan over chunk aggregates;
 * chunk_T/F become the carry INTO each chunk; chunk_N becomes the run
 * offset of each chunk; total runs -> d_total. */
__global__ __launch_bounds__(NTHREADS) void k_stitchB(
    double *__restrict__ chunk_T, uint32_t *__restrict__ chunk_F,
    uint64_t *__restrict__ chunk_N, uint64_t nchunks,
    uint64_t *__restrict__ d_total) {
  if (threadIdx.x == 0) {
    double carry = 0.0;
    uint64_t off = 0;
    for (uint64_t s = 0; s < nchunks; s++) {
      double T = chunk_T[s];
      uint32_t F = chunk_F[s];
      uint64_t N = chunk_N[s];
      chunk_T[s] = carry;
      chunk_N[s] = off;
      carry = F ? T : (T + carry);
      off += N;
    }
    *d_total = off;
  }
}

/* stitch C: apply chunk carries */
__global__ __launch_bounds__(NTHREADS) void k_stitchC(
    double *__restrict__ tile_C, const uint8_t *__restrict__ chain_open,
    uint64_t *__restrict__ tile_outoff, uint64_t ntiles,
    const double *__restrict__ chunk_T, const uint64_t *__restrict__ chunk_N) {
  uint64_t c0 = (uint64_t)blockIdx.x * STITCH_CHUNK;
  uint64_t c1 = min(c0 + STITCH_CHUNK, ntiles);
  double cin = chunk_T[blockIdx.x];
  uint64_t base = chunk_N[blockIdx.x];
  for (uint64_t t = c0 + threadIdx.x; t < c1; t += NTHREADS) {
    if (chain_open[t]) tile_C[t] += cin;
    tile_outoff[t] += base;
  }
}

/* compact: copy each tile's runs from scratch to the final arrays; the
 * FIRST run of a tile absorbs the stitched carry C_t (C_t == 0.0 whenever
 * the tile's leading segment is empty — see DESIGN.md). */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_compact_runs(
    const K *__restrict__ keys_scr, const double *__restrict__ sums_scr,
    const uint32_t *__restrict__ tile_nruns,
    const uint64_t *__restrict__ tile_outoff,
    const double *__restrict__ tile_C, uint64_t ntiles,
    K *__restrict__ keys_out, double *__restrict__ sums_out) {
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    uint32_t nr = tile_nruns[t];
    uint64_t base = tile_outoff[t];
    uint64_t src = t * (uint64_t)RTILE;
    for (uint32_t j = threadIdx.x; j < nr; j += NTHREADS) {
      double s = sums_scr[src + j];
      if (j == 0) s += tile_C[t];
      keys_out[base + j] = keys_scr[src + j];
      sums_out[base + j] = s;
    }
  }
}

/* ------------------------------------------------------------------ */
/* kernels: run extraction (groupByKey / join structure)               */
/* ------------------------------------------------------------------ */

/* per tile: compact (key, head_global_index) for every run HEAD */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_head_tiles(
    const K *__restrict__ keys, uint64_t n,
    K *__restrict__ keys_scr, uint64_t *__restrict__ idx_scr,
    uint32_t *__restrict__ tile_nheads, uint64_t ntiles) {
  __shared__ uint32_t cnt[NTHREADS];
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    const uint64_t start = t * (uint64_t)RTILE;
    const uint64_t end = min(start + RTILE, n);
    /* count heads per thread (strided chunks of ELEMS_PER_THREAD) */
    const uint64_t c0 = start + threadIdx.x * EPT_R;
    const uint64_t c1 = min(c0 + EPT_R, end);
    uint32_t nh = 0;
    for (uint64_t i = c0; i < c1 && i < end; i++) {
      bool head = (i == 0) || keys[i] != keys[i - 1];
      if (head) nh++;
    }
    cnt[threadIdx.x] = nh;
    __syncthreads();
    for (int off = 1; off < NTHREADS; off <<= 1) {
      uint32_t prev = (threadIdx.x >= (unsigned)off) ? cnt[threadIdx.x - off] : 0;
      __syncthreads();
      cnt[threadIdx.x] += prev;
      __syncthreads();
    }
    uint32_t w = cnt[threadIdx.x] - nh;
    for (uint64_t i = c0; i < c1 && i < end; i++) {
      bool head = (i == 0) || keys[i] != keys[i - 1];
      if (head) {
        keys_scr[start + w] = keys[i];
        idx_scr[start + w] = i;
        w++;
      }
    }
    if (threadIdx.x == NTHREADS - 1) tile_nheads[t] = cnt[threadIdx.x];
    __syncthreads();
  }
}

/* compact heads -> (unique_keys, starts) using scanned tile offsets */
/* heads compact ALWAYS emits int64 unique keys (grp_keys) — the join /
 * group machinery downstream is key-width independent */
template <typename K>
__global__ __launch_bounds__(NTHREADS) void k_compact_heads(
    const K *__restrict__ keys_scr, const uint64_t *__restrict__ idx_scr,
    const uint32_t *__restrict__ tile_nheads,
    const uint64_t *__restrict__ tile_outoff, uint64_t ntiles,
    int64_t *__restrict__ uk, uint64_t *__restrict__ starts) {
  for (uint64_t t = blockIdx.x; t < ntiles; t += gridDim.x) {
    uint32_t nh = tile_nheads[t];
    uint64_t base = tile_outoff[t];
    uint64_t src = t * (uint64_t)RTILE;
    for (uint32_t j = threadIdx.x; j < nh; j += NTHREADS) {
      uk[base + j] = (int64_t)KT<K>::raw(keys_scr[src + j]);
      starts[base + j] = idx_scr[src + j];
    }
  }
}

__global__ __launch_bounds__(NTHREADS) void k_run_counts(
    const uint64_t *__restrict__ starts, uint64_t g, uint64_t n,
    uint64_t *__restrict__ counts) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x; i < g;
       i += stride)
    counts[i] = ((i + 1 < g) ? starts[i + 1] : n) - starts[i];
}

/* plain exclusive-scan of tile_nheads -> tile_outoff (+ total) — single
 * workgroup, chunked (ntiles <= 1M => <=4096 chunks) */
__global__ __launch_bounds__(NTHREADS) void k_scan_u32_single(
    const uint32_t *__restrict__ in, uint64_t m,
    uint64_t *__restrict__ out, uint64_t *__restrict__ d_total) {
  __shared__ uint64_t sh[NTHREADS];
  uint64_t run = 0;
  for (uint64_t c = 0; c < m; c += NTHREADS) {
    uint64_t i = c + threadIdx.x;
    uint64_t v = (i < m) ? (uint64_t)in[i] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < NTHREADS; off <<= 1) {
      uint64_t prev = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += prev;
      __syncthreads();
    }
    if (i < m) out[i] = run + sh[threadIdx.x] - v;
    uint64_t tot = sh[NTHREADS - 1];
    __syncthreads();
    run += tot;
  }
  if (threadIdx.x == 0 && d_total) *d_total = run;
}

/* ------------------------------------------------------------------ */
/* kernels: generic u64 exclusive scan (3-kernel, for large arrays)    */
/* ------------------------------------------------------------------ */
__global__ __launch_bounds__(NTHREADS) void k_scanA(
    uint64_t *__restrict__ data, uint64_t m, uint64_t *__restrict__ part) {
  __shared__ uint64_t sh[NTHREADS];
  uint64_t c0 = (uint64_t)blockIdx.x * SCAN_CHUNK;
  uint64_t c1 = min(c0 + SCAN_CHUNK, m);
  uint64_t run = 0;
  for (uint64_t c = c0; c < c1; c += NTHREADS) {
    uint64_t i = c + threadIdx.x;
    uint64_t v = (i < c1) ? data[i] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < NTHREADS; off <<= 1) {
      uint64_t prev = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += prev;
      __syncthreads();
    }
    if (i < c1) data[i] = run + sh[threadIdx.x] - v; /* exclusive in place */
    uint64_t tot = sh[NTHREADS - 1];
    __syncthreads();
    run += tot;
  }
  if (threadIdx.x == 0) part[blockIdx.x] = run;
}

__global__ __launch_bounds__(NTHREADS) void k_scanB(
    uint64_t *__restrict__ part, uint64_t nparts, uint64_t *__restrict__ d_total) {
  /* same single-WG chunked exclusive scan, in place */
  __shared__ uint64_t sh[NTHREADS];
  uint64_t run = 0;
  for (uint64_t c = 0; c < nparts; c += NTHREADS) {
    uint64_t i = c + threadIdx.x;
    uint64_t v = (i < nparts) ? part[i] : 0;
    sh[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < NTHREADS; off <<= 1) {
      uint64_t prev = (threadIdx.x >= (unsigned)off) ? sh[threadIdx.x - off] : 0;
      __syncthreads();
      sh[threadIdx.x] += prev;
      __syncthreads();
    }
    if (i < nparts) part[i] = run + sh[threadIdx.x] - v;
    uint64_t tot = sh[NTHREADS - 1];
    __syncthreads();
    run += tot;
  }
  if (threadIdx.x == 0 && d_total) *d_total = run;
}

__global__ __launch_bounds__(NTHREADS) void k_scanC(
    uint64_t *__restrict__ data, uint64_t m, const uint64_t *__restrict__ part) {
  uint64_t c0 = (uint64_t)blockIdx.x * SCAN_CHUNK;
  uint64_t c1 = min(c0 + SCAN_CHUNK, m);
  uint64_t base = part[blockIdx.x];
  for (uint64_t i = c0 + threadIdx.x; i < c1; i += NTHREADS) data[i] += base;
}

/* ------------------------------------------------------------------ */
/* kernels: join                                                       */
/* ------------------------------------------------------------------ */

/* per A-run: find the key among B's unique keys; product size.
 * Both unique-key arrays are sorted, so a wave handling 64 CONSECUTIVE
 * A-runs first brackets the matching B-range with two deep binary
 * searches (lanes 0/1), then every lane searches the narrow bracket
 * (L2-hot) — ~10x fewer dependent deep probes than naive per-lane
 * binary search at ga ~ 10^7. */
__global__ __launch_bounds__(NTHREADS) void k_join_count(
    const int64_t *__restrict__ uka, const uint64_t *__restrict__ cnta,
    uint64_t ga, const int64_t *__restrict__ ukb,
    const uint64_t *__restrict__ cntb, uint64_t gb,
    uint64_t *__restrict__ prod, uint64_t *__restrict__ bidx) {
  const int lane = threadIdx.x % 64;
  const uint64_t nwaves = (uint64_t)gridDim.x * (NTHREADS / 64);
  const uint64_t wid = (uint64_t)blockIdx.x * (NTHREADS / 64) + threadIdx.x / 64;
  for (uint64_t r0 = wid * 64; r0 < ga; r0 += nwaves * 64) {
    uint64_t r = r0 + lane;
    uint64_t rlast = min(r0 + 63, ga - 1);
    /* bracket [blo, bhi) of B covering this batch's key range */
    uint64_t blo = 0, bhi = gb;
    if (lane == 0) {
      int64_t k = uka[r0];
      uint64_t lo = 0, hi = gb;
      while (lo < hi) { uint64_t m = (lo + hi) >> 1;
                        if (ukb[m] < k) lo = m + 1; else hi = m; }
      blo = lo;
    }
    if (lane == 1) {
      int64_t k = uka[rlast];
      uint64_t lo = 0, hi = gb;
      while (lo < hi) { uint64_t m = (lo + hi) >> 1;
                        if (ukb[m] <= k) lo = m + 1; else hi = m; }
      bhi = lo;
    }
    blo = (uint64_t)__shfl((long long)blo, 0, 64);
    bhi = (uint64_t)__shfl((long long)bhi, 1, 64);
    if (r < ga) {
      int64_t k = uka[r];
      uint64_t lo = blo, hi = bhi;
      while (lo < hi) { uint64_t m = (lo + hi) >> 1;
                        if (ukb[m] < k) lo = m + 1; else hi = m; }
      if (lo < gb && ukb[lo] == k) {
        prod[r] = cnta[r] * cntb[lo];
        bidx[r] = lo;
      } else {
        prod[r] = 0;
        bidx[r] = (uint64_t)-1;
      }
    }
  }
}

/* merge co-walk variant of the match step (DPRK_JOIN_MP experiment):
 * one thread owns WCHUNK consecutive A-runs — ONE binary search finds
 * its B start, then both sorted unique lists advance together
 * sequentially (L2-hot streams), ~0.4+2 memory ops per run vs ~8
 * probes for the wave-bracketed per-lane searches. */
#define JOIN_WCHUNK 64
__global__ __launch_bounds__(NTHREADS) void k_join_count_mp(
    const int64_t *__restrict__ uka, const uint64_t *__restrict__ cnta,
    uint64_t ga, const int64_t *__restrict__ ukb,
    const uint64_t *__restrict__ cntb, uint64_t gb,
    uint64_t *__restrict__ prod, uint64_t *__restrict__ bidx) {
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  uint64_t nchunks = (ga + JOIN_WCHUNK - 1) / JOIN_WCHUNK;
  for (uint64_t c = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
       c < nchunks; c += stride) {
    uint64_t r0 = c * JOIN_WCHUNK;
    uint64_t r1 = min(r0 + JOIN_WCHUNK, ga);
    /* lower_bound(B, uka[r0]) */
    int64_t k0 = uka[r0];
    uint64_t lo = 0, hi = gb;
    while (lo < hi) {
      uint64_t m = (lo + hi) >> 1;
      if (ukb[m] < k0) lo = m + 1; else hi = m;
    }
    uint64_t b = lo;
    for (uint64_t r = r0; r < r1; r++) {
      int64_t k = uka[r];
      while (b < gb && ukb[b] < k) b++;
      if (b < gb && ukb[b] == k) {
        prod[r] = cnta[r] * cntb[b];
        bidx[r] = b;
      } else {
        prod[r] = 0;
        bidx[r] = (uint64_t)-1;
      }
    }
  }
}

/* emit: per-key cross products (dpark join semantics — SURVEY.md §3.3).
 * One GLANES-lane sub-wave group per A-run (products are small, so a
 * full wave per run left most lanes idle at low multiplicity — measured
 * r01; GLANES tunable: wider groups write fuller lines per iteration).
 * Consecutive groups emit consecutive runs, whose output ranges are
 * adjacent => coalesced-ish stores. */
template <int GLANES>
__global__ __launch_bounds__(NTHREADS) void k_join_emit(
    const int64_t *__restrict__ uka, const uint64_t *__restrict__ starta,
    uint64_t ga,
    const uint64_t *__restrict__ startb, const uint64_t *__restrict__ cntb,
    const uint64_t *__restrict__ prod_off, const uint64_t *__restrict__ prod,
    const uint64_t *__restrict__ bidx, const double *__restrict__ va,
    const double *__restrict__ vb, int64_t *__restrict__ out_k,
    double *__restrict__ out_a, double *__restrict__ out_b) {
  const uint32_t laneg = threadIdx.x & (GLANES - 1);
  const uint64_t ngroups = (uint64_t)gridDim.x * (NTHREADS / GLANES);
  const uint64_t gid =
      (uint64_t)blockIdx.x * (NTHREADS / GLANES) + threadIdx.x / GLANES;
  for (uint64_t r = gid; r < ga; r += ngroups) {
    uint64_t p = prod[r];
    if (p == 0) continue;
    int64_t k = uka[r];
    uint64_t sa = starta[r];
    uint64_t b = bidx[r];
    uint64_t sb = startb[b];
    uint32_t cb = (uint32_t)cntb[b];
    uint64_t base = prod_off[r];
    /* index math in u32 (u64 div is a ~50-100 cycle libcall, paid twice
     * per output element); cb==1 needs no division at all */
    if (cb == 1) {
      double vbv = vb[sb];
      for (uint64_t j = laneg; j < p; j += GLANES) {
        out_k[base + j] = k;
        out_a[base + j] = va[sa + j];
        out_b[base + j] = vbv;
      }
    } else if (p <= 0xFFFFFFFFull) {
      for (uint32_t j = laneg; j < (uint32_t)p; j += GLANES) {
        out_k[base + j] = k;
        out_a[base + j] = va[sa + j / cb];
        out_b[base + j] = vb[sb + j % cb];
      }
    } else {
      for (uint64_t j = laneg; j < p; j += GLANES) {
        out_k[base + j] = k;
        out_a[base + j] = va[sa + j / cb];
        out_b[base + j] = vb[sb + j % cb];
      }
    }
  }
}

/* ------------------------------------------------------------------ */
/* host-side helpers                                                   */
/* ------------------------------------------------------------------ */
static uint64_t ceil_div(uint64_t a, uint64_t b) { return (a + b - 1) / b; }

/* memory-bound grid size: enough workgroups to fill 256 CUs at the
 * kernel's blocks/CU, grid-stride beyond (Guideline 11) */
static int stream_grid(uint64_t work_items) {
  uint64_t g = ceil_div(work_items, NTHREADS);
  if (g > 2048) g = 2048;
  if (g == 0) g = 1;
  return (int)g;
}

static int free_all(dprk_ctx *c);
static int materialize_lazy(dprk_ctx *c);

/* workgroups for tile-mapped kernels: min(ntiles, cap); cap tunable via
 * DPRK_GRID_CAP (default 2048 — measured flat 512..4096, see DESIGN) */
static int run_scan_u64(dprk_ctx *c, uint64_t *data, uint64_t m,
                        uint64_t *total_out);

static int tile_grid(uint64_t ntiles) {
  static int cap = 0;
  if (!cap) {
    const char *e = getenv("DPRK_GRID_CAP");
    cap = e ? atoi(e) : 2048;
    if (cap < 64 || cap > 65536) cap = 2048;
  }
  return (int)min(ntiles, (uint64_t)cap);
}

/* lookback L3-residency gate in bytes (granule array must stay L3-hot;
 * DPRK_LB_GATE_MB tunes).  Default 300 MB, measured r02 on the 1B
 * shapes WITH the pipelined scatters: pairs 28.4 -> 29.5 G pairs/s,
 * keys-only sort 16.0 -> 19.6 G keys/s (the earlier 200 MB keys-only
 * preference belonged to the pre-pipeline kernel).  A pure function of
 * n => input-deterministic path choice. */
static uint64_t lb_gate_bytes(int with_vals) {
  (void)with_vals;
  static long gate_mb = -1;
  if (gate_mb < 0) {
    const char *e = getenv("DPRK_LB_GATE_MB");
    gate_mb = e ? atol(e) : 300;
  }
  return (uint64_t)gate_mb << 20;
}

extern "C" int dprk_init(dprk_ctx **out, int device_id) {
  *out = nullptr;
  HIP_CHECK(hipSetDevice(device_id));
  dprk_ctx *c = new dprk_ctx();
  c->device = device_id;
  memset(&c->stats, 0, sizeof(c->stats));
  if (hipStreamCreateWithFlags(&c->stream, hipStreamNonBlocking) !=
      hipSuccess) {
    delete c;
    set_err("hipStreamCreate failed");
    return 1;
  }
  *out = c;
  return 0;
}

extern "C" void dprk_free(dprk_ctx *c) {
  if (!c) return;
  (void)hipSetDevice(c->device);
  free_all(c);
  for (int i = 0; i < c->n_evts_created; i++)
    (void)hipEventDestroy(c->evts[i]);
  if (c->comm) ncclCommDestroy(c->comm);
  if (c->stream) (void)hipStreamDestroy(c->stream);
  delete c;
}

static int free_group_join(dprk_ctx *c) {
  if (c->grp_keys) HIP_CHECK(hipFree(c->grp_keys));
  if (c->grp_starts) HIP_CHECK(hipFree(c->grp_starts));
  if (c->grp_counts) HIP_CHECK(hipFree(c->grp_counts));
  c->grp_keys = nullptr; c->grp_starts = nullptr; c->grp_counts = nullptr;
  c->n_groups = 0; c->grp_cap = 0;
  if (c->join_k) HIP_CHECK(hipFree(c->join_k));
  if (c->join_a) HIP_CHECK(hipFree(c->join_a));
  if (c->join_b) HIP_CHECK(hipFree(c->join_b));
  c->join_k = nullptr; c->join_a = nullptr; c->join_b = nullptr;
  c->join_n = 0; c->join_cap = 0;
  if (c->run_prod) HIP_CHECK(hipFree(c->run_prod));
  if (c->run_boff) HIP_CHECK(hipFree(c->run_boff));
  if (c->d_bidx) HIP_CHECK(hipFree(c->d_bidx));
  c->run_prod = nullptr; c->run_boff = nullptr; c->d_bidx = nullptr;
  c->run_prod_cap = 0;
  return 0;
}

/* capacity-cached group-array (re)allocation: repeated ops (bench steps)
 * must not pay GB-scale hipMalloc/hipFree every call */
static int ensure_grp(dprk_ctx *c, uint64_t g) {
  if (g <= c->grp_cap && c->grp_keys) return 0;
  if (c->grp_keys) HIP_CHECK(hipFree(c->grp_keys));
  if (c->grp_starts) HIP_CHECK(hipFree(c->grp_starts));
  if (c->grp_counts) HIP_CHECK(hipFree(c->grp_counts));
  uint64_t cap = g + g / 8 + 1;
  HIP_CHECK(hipMalloc(&c->grp_keys, cap * 8));
  HIP_CHECK(hipMalloc(&c->grp_starts, cap * 8));
  HIP_CHECK(hipMalloc(&c->grp_counts, cap * 8));
  c->grp_cap = cap;
  return 0;
}

static int free_fused(dprk_ctx *c) {
  if (c->f_cnt) HIP_CHECK(hipFree(c->f_cnt));
  if (c->f_lex) HIP_CHECK(hipFree(c->f_lex));
  if (c->f_goff) HIP_CHECK(hipFree(c->f_goff));
  c->f_cnt = nullptr; c->f_lex = nullptr; c->f_goff = nullptr;
  c->fused_cap_rec = 0;
  return 0;
}

static int ensure_fused(dprk_ctx *c, uint64_t nrec) {
  if (nrec <= c->fused_cap_rec && c->f_cnt) return 0;
  free_fused(c);
  HIP_CHECK(hipMalloc(&c->f_cnt, nrec * 4));
  HIP_CHECK(hipMalloc(&c->f_lex, nrec * 4));
  HIP_CHECK(hipMalloc(&c->f_goff, nrec * 4));
  c->fused_cap_rec = nrec;
  return 0;
}

static int free_big(dprk_ctx *c) {
  if (c->big_k) HIP_CHECK(hipFree(c->big_k));
  if (c->big_v) HIP_CHECK(hipFree(c->big_v));
  c->big_k = nullptr;
  c->big_v = nullptr;
  c->big_cap_total = 0;
  c->big_cap_b = 0;
  c->big_nbuckets = 0;
  c->big_packed = -1;
  c->big_out_n = 0;
  c->big_valid = 0;
  memset(c->big_fill, 0, sizeof(c->big_fill));
  return 0;
}

static int free_all(dprk_ctx *c) {
  if (free_fused(c)) return 1;
  if (free_big(c)) return 1;
  for (int i = 0; i < 2; i++) {
    if (c->keys[i]) HIP_CHECK(hipFree(c->keys[i]));
    if (c->vals[i]) HIP_CHECK(hipFree(c->vals[i]));
    c->keys[i] = nullptr; c->vals[i] = nullptr;
  }
  if (c->d_lb) HIP_CHECK(hipFree(c->d_lb));
  if (c->d_fail) HIP_CHECK(hipFree(c->d_fail));
  if (c->d_ghist) HIP_CHECK(hipFree(c->d_ghist));
  if (c->d_dbase) HIP_CHECK(hipFree(c->d_dbase));
  c->d_lb = nullptr; c->d_fail = nullptr;
  c->d_ghist = nullptr; c->d_dbase = nullptr; c->lb_cap_tiles = 0;
  if (c->tile_hist) HIP_CHECK(hipFree(c->tile_hist));
  if (c->tile_off) HIP_CHECK(hipFree(c->tile_off));
  if (c->seg_sums) HIP_CHECK(hipFree(c->seg_sums));
  if (c->row_tot) HIP_CHECK(hipFree(c->row_tot));
  if (c->d_orand) HIP_CHECK(hipFree(c->d_orand));
  if (c->alias_table) HIP_CHECK(hipFree(c->alias_table));
  if (c->tile_T) HIP_CHECK(hipFree(c->tile_T));
  if (c->tile_anyend) HIP_CHECK(hipFree(c->tile_anyend));
  if (c->tile_nruns) HIP_CHECK(hipFree(c->tile_nruns));
  if (c->tile_outoff) HIP_CHECK(hipFree(c->tile_outoff));
  if (c->tile_C) HIP_CHECK(hipFree(c->tile_C));
  if (c->chain_open) HIP_CHECK(hipFree(c->chain_open));
  if (c->chunk_T) HIP_CHECK(hipFree(c->chunk_T));
  if (c->chunk_F) HIP_CHECK(hipFree(c->chunk_F));
  if (c->chunk_N) HIP_CHECK(hipFree(c->chunk_N));
  if (c->d_scalar) HIP_CHECK(hipFree(c->d_scalar));
  if (c->scan_part) HIP_CHECK(hipFree(c->scan_part));
  c->tile_hist = nullptr; c->tile_off = nullptr; c->seg_sums = nullptr;
  c->row_tot = nullptr; c->d_orand = nullptr; c->tile_T = nullptr;
  c->alias_table = nullptr; c->alias_universe = 0;
  c->tile_anyend = nullptr; c->tile_nruns = nullptr; c->tile_outoff = nullptr;
  c->tile_C = nullptr; c->d_scalar = nullptr; c->scan_part = nullptr;
  c->chain_open = nullptr; c->chunk_T = nullptr; c->chunk_F = nullptr;
  c->chunk_N = nullptr;
  c->scan_part_cap = 0;
  c->cap = 0; c->max_tiles = 0; c->n = 0;
  return free_group_join(c);
}

extern "C" int dprk_reserve(dprk_ctx *c, uint64_t max_n, int with_vals) {
  if (!c) { set_err("null ctx"); return 1; }
  if (max_n >= (1ull << 32)) {
    set_err("dprk_reserve: max_n %llu >= 2^32 unsupported (u32 radix "
            "offsets; out-of-core chunking is a next-row item, SURVEY.md "
            "§8(f))", (unsigned long long)max_n);
    return 1;
  }
  HIP_CHECK(hipSetDevice(c->device));
  if (max_n <= c->cap && (!with_vals || c->with_vals)) {
    c->with_vals = with_vals;
    return 0;
  }
  free_all(c);
  if (max_n == 0) max_n = 1;
  uint64_t ntiles = ceil_div(max_n, TILE);
  uint64_t ntiles_r = ceil_div(max_n, RTILE); /* reduce/head tiling */
  HIP_CHECK(hipMalloc(&c->keys[0], max_n * 8));
  HIP_CHECK(hipMalloc(&c->keys[1], max_n * 8));
  if (with_vals) {
    HIP_CHECK(hipMalloc(&c->vals[0], max_n * 8));
    HIP_CHECK(hipMalloc(&c->vals[1], max_n * 8));
  }
  HIP_CHECK(hipMalloc(&c->tile_hist, (uint64_t)RADIX * ntiles * 4));
  HIP_CHECK(hipMalloc(&c->tile_off, (uint64_t)RADIX * ntiles * 4));
  HIP_CHECK(hipMalloc(&c->seg_sums, (uint64_t)RADIX * RADIX * 4));
  HIP_CHECK(hipMalloc(&c->row_tot, RADIX * 8));
  HIP_CHECK(hipMalloc(&c->d_orand, 2 * 8));
  HIP_CHECK(hipMalloc(&c->tile_T, ntiles_r * 8));
  HIP_CHECK(hipMalloc(&c->tile_anyend, ntiles_r * 4));
  HIP_CHECK(hipMalloc(&c->tile_nruns, ntiles_r * 4));
  HIP_CHECK(hipMalloc(&c->tile_outoff, ntiles_r * 8));
  HIP_CHECK(hipMalloc(&c->tile_C, ntiles_r * 8));
  uint64_t nchunks_r = ceil_div(ntiles_r, (uint64_t)STITCH_CHUNK);
  HIP_CHECK(hipMalloc(&c->chain_open, ntiles_r));
  HIP_CHECK(hipMalloc(&c->chunk_T, nchunks_r * 8));
  HIP_CHECK(hipMalloc(&c->chunk_F, nchunks_r * 4));
  HIP_CHECK(hipMalloc(&c->chunk_N, nchunks_r * 8));
  HIP_CHECK(hipMalloc(&c->d_scalar, 4 * 8));
  c->cap = max_n;
  c->max_tiles = ntiles;
  c->with_vals = with_vals;
  c->n = 0;
  c->cur = 0;
  c->sorted = 0;
  c->packed = 0;
  c->lazy_gen = 0;
  return 0;
}

extern "C" int dprk_count(dprk_ctx *c, uint64_t *n_out) {
  if (!c) { set_err("null ctx"); return 1; }
  *n_out = c->n;
  return 0;
}

extern "C" int dprk_set_alias_table(dprk_ctx *c, const uint64_t *table,
                                    uint64_t universe) {
  if (!c) { set_err("null ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->alias_table) { HIP_CHECK(hipFree(c->alias_table));
                        c->alias_table = nullptr; }
  HIP_CHECK(hipMalloc(&c->alias_table, universe * 8));
  HIP_CHECK(hipMemcpyAsync(c->alias_table, table, universe * 8,
                           hipMemcpyHostToDevice, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  c->alias_universe = universe;
  return 0;
}

template <typename K>
static int gen_pairs_t(dprk_ctx *c, uint64_t seed_hi, uint64_t n, int dist,
                       uint64_t key_universe, uint64_t keymask) {
  if (dist == 1) {
    TIMED(c, K_GEN,
          hipLaunchKernelGGL(k_gen_alias<K>, dim3(stream_grid(n)),
                             dim3(NTHREADS),
                             0, c->stream, (K *)c->keys[c->cur],
                             c->with_vals ? c->vals[c->cur] : nullptr, n,
                             seed_hi, c->alias_table, key_universe));
  } else {
    /* fused gen+hist: per-tile hist for the first sort pass (base path)
     * AND global per-pass histograms (lookback path) come out of the
     * generation pass itself — sorting gen-sourced data then reads the
     * keys zero extra times before the first scatter.  Active passes
     * for masked keys = the nonzero mask bytes (biased bit 63 is
     * constant for universes < 2^63). */
    int shift0 = 0;
    while (keymask != ~0ull && ((keymask >> shift0) & 0xFF) == 0 &&
           shift0 < 56)
      shift0 += 8;
    uint64_t shifts = 0;
    int nact = 0;
    for (int p = 0; p < 8; p++) {
      uint32_t mb = (uint32_t)(keymask >> (8 * p)) & 0xFF;
      if (keymask == ~0ull || mb)
        shifts |= ((uint64_t)(8 * p)) << (8 * (nact++));
    }
    /* the global hists feed only the lookback path; when n exceeds its
     * L3-residency gate (see dprk_sort_pairs) skip the extra LDS-atomic
     * work here */
    if (ceil_div(n, (uint64_t)TILE) * RADIX * 8 > lb_gate_bytes(c->with_vals))
      nact = 0;
    if (!c->d_ghist) HIP_CHECK(hipMalloc(&c->d_ghist, 8 * RADIX * 4));
    HIP_CHECK(hipMemsetAsync(c->d_ghist, 0, 8 * RADIX * 4, c->stream));
    uint64_t oa_init[2] = {0, ~0ull};
    HIP_CHECK(hipMemcpyAsync(c->d_orand, oa_init, 16, hipMemcpyHostToDevice,
                             c->stream));
    uint64_t ntiles = ceil_div(n, (uint64_t)TILE);
    TIMED(c, K_GEN,
          hipLaunchKernelGGL(k_gen_hist<K>,
                             dim3((uint32_t)min(ntiles, (uint64_t)2048)),
                             dim3(NTHREADS), 0, c->stream,
                             (K *)c->keys[c->cur],
                             c->with_vals ? c->vals[c->cur] : nullptr, n,
                             seed_hi, keymask, c->tile_hist, ntiles,
                             shift0, shifts, nact, c->d_ghist, c->d_orand));
    uint64_t oa[2];
    HIP_CHECK(hipMemcpyAsync(oa, c->d_orand, 16, hipMemcpyDeviceToHost,
                             c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    c->hist_shift_valid = shift0;
    c->ghist_shifts = shifts;
    c->ghist_nact = nact;
    c->ghist_valid = 1;
    /* EXACT biased-key bounds from generation: digit-skip needs no
     * separate k_minmax_or read (matters for full-universe keys, where
     * the conservative mask bound is useless) */
    c->orand_valid = 1;
    c->known_or = oa[0];
    c->known_and = oa[1];
  }
  c->n = n;
  c->sorted = 0;
  /* keys = splitmix64 & keymask, so biased keys lie in
   * [SIGNFLIP, SIGNFLIP | keymask]: conservative OR/AND bound — may only
   * RUN an avoidable pass, never skip a needed one */
  if (dist == 0) {
    /* exact values were read back from k_gen_hist above */
  } else if (dist == 1 && (key_universe & (key_universe - 1)) == 0) {
    c->orand_valid = 1;   /* alias keys are < universe (power of two) */
    if (c->packed) {      /* packed bias = the key value itself */
      c->known_or = key_universe - 1;
      c->known_and = 0;
    } else {
      c->known_or = SIGNFLIP | (key_universe - 1);
      c->known_and = SIGNFLIP;
    }
  } else {
    c->orand_valid = 0;
  }
  return drain_timing(c);
}

/* start: index offset into the synthetic stream (x_i = (seed<<32) +
 * start + i) — lets out-of-core chunks continue ONE stream; start=0
 * reproduces the original spec exactly. */
extern "C" int dprk_gen_pairs_at(dprk_ctx *c, uint64_t seed, uint64_t start,
                                 uint64_t n, int dist,
                                 uint64_t key_universe) {
  if (!c) { set_err("null ctx"); return 1; }
  if (dist != 0 && dist != 1) { set_err("gen: unknown dist %d", dist);
                                return 1; }
  if (dist == 1 && (!c->alias_table || c->alias_universe != key_universe)) {
    set_err("gen dist=1: call dprk_set_alias_table(universe=%llu) first",
            (unsigned long long)key_universe);
    return 1;
  }
  if (n > c->cap) { set_err("gen: n > reserved cap"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  uint64_t keymask =
      (key_universe == 0) ? ~0ull : (key_universe - 1); /* power of two */
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  /* packed-u32 fast path: generated keys are in [0, universe) — pack
   * whenever the universe fits 32 bits (uniform policy, a pure function
   * of the gen arguments => bit-repeatable path selection).
   * DPRK_NO_PACK=1 disables (A/B harness). */
  c->packed = (key_universe != 0 && key_universe <= (1ull << 32) &&
               !getenv("DPRK_NO_PACK"));
  uint64_t seed_hi = (seed << 32) + start;
  /* gen->sort fusion (dist 0): only RECORD the stream — if the next op
   * is a sort/reduce, the first pass GENERATES pairs straight into
   * partitioned layout (saves the gen write + pass-1 read, 24 B/pair);
   * any other op materializes first (materialize_lazy).  Results are
   * deterministic per CALL SEQUENCE; fused vs materialized reductions
   * may differ in the last ulp (different element permutations, same
   * <=1e-6 oracle contract — like every other path pair here). */
  if (dist == 0 && n > 0 && !getenv("DPRK_NO_FUSED_GEN")) {
    c->lazy_gen = 1;
    c->lazy_seed_hi = seed_hi;
    c->lazy_keymask = keymask;
    c->lazy_universe = key_universe;
    c->n = n;
    c->sorted = 0;
    c->orand_valid = 0;
    return 0;
  }
  c->lazy_gen = 0;
  return c->packed
             ? gen_pairs_t<uint32_t>(c, seed_hi, n, dist, key_universe,
                                     keymask)
             : gen_pairs_t<int64_t>(c, seed_hi, n, dist, key_universe,
                                    keymask);
}

/* materialize a lazily-recorded uniform generation (the normal fused
 * k_gen_hist path); no-op otherwise */
static int materialize_lazy(dprk_ctx *c) {
  if (!c->lazy_gen) return 0;
  c->lazy_gen = 0;
  HIP_CHECK(hipSetDevice(c->device));
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  return c->packed
             ? gen_pairs_t<uint32_t>(c, c->lazy_seed_hi, c->n, 0,
                                     c->lazy_universe, c->lazy_keymask)
             : gen_pairs_t<int64_t>(c, c->lazy_seed_hi, c->n, 0,
                                    c->lazy_universe, c->lazy_keymask);
}

extern "C" int dprk_gen_pairs(dprk_ctx *c, uint64_t seed, uint64_t n,
                              int dist, uint64_t key_universe) {
  return dprk_gen_pairs_at(c, seed, 0, n, dist, key_universe);
}

/* put-side pack gate: run k_minmax_or on the fresh int64 keys (the sort
 * would need it anyway — this just moves it to put time), then pack to
 * u32 if every key fits [0, 2^32).  Input-deterministic. */
static int maybe_pack_after_put(dprk_ctx *c, uint64_t n) {
  uint64_t init[2] = {0, ~0ull};
  HIP_CHECK(hipMemcpyAsync(c->d_orand, init, 16, hipMemcpyHostToDevice,
                           c->stream));
  TIMED(c, K_HIST,
        hipLaunchKernelGGL(k_minmax_or<int64_t>, dim3(stream_grid(n)),
                           dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                           n, c->d_orand));
  uint64_t oa[2];
  HIP_CHECK(hipMemcpyAsync(oa, c->d_orand, 16, hipMemcpyDeviceToHost,
                           c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  c->stats.bytes_total += n * 8;
  c->orand_valid = 1;
  c->known_or = oa[0];
  c->known_and = oa[1];
  /* all keys in [0, 2^32)  <=>  biased OR's high 32 bits == 0x80000000
   * (bit 63 set: non-negative; bits 32..62 clear: < 2^32) */
  if ((oa[0] >> 32) == 0x80000000ull && !getenv("DPRK_NO_PACK")) {
    TIMED(c, K_OTHER,
          hipLaunchKernelGGL(k_pack_keys, dim3(stream_grid(n)),
                             dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                             (uint32_t *)c->keys[c->cur ^ 1], n));
    /* pointer-swap keys only: vals stay in place, cur unchanged */
    int64_t *tmp = c->keys[c->cur];
    c->keys[c->cur] = c->keys[c->cur ^ 1];
    c->keys[c->cur ^ 1] = tmp;
    c->packed = 1;
    /* re-express the known bounds in the packed bias domain */
    c->known_or = oa[0] & 0xFFFFFFFFull;
    c->known_and = oa[1] & 0xFFFFFFFFull;
    c->stats.bytes_total += n * 12;
  } else {
    c->packed = 0;
  }
  return drain_timing(c);
}

extern "C" int dprk_put_pairs(dprk_ctx *c, const int64_t *keys,
                              const double *vals, uint64_t n) {
  if (!c) { set_err("null ctx"); return 1; }
  if (n > c->cap) { set_err("put: n > reserved cap"); return 1; }
  c->lazy_gen = 0;   /* overwritten */
  if (vals && !c->with_vals) { set_err("put: ctx reserved keys-only"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  HIP_CHECK(hipMemcpyAsync(c->keys[c->cur], keys, n * 8,
                           hipMemcpyHostToDevice, c->stream));
  if (vals && c->with_vals)
    HIP_CHECK(hipMemcpyAsync(c->vals[c->cur], vals, n * 8,
                             hipMemcpyHostToDevice, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  c->n = n;
  c->sorted = 0;
  c->orand_valid = 0;
  c->packed = 0;
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  if (n > 0 && maybe_pack_after_put(c, n)) return 1;
  return 0;
}

extern "C" int dprk_get_pairs(dprk_ctx *c, int64_t *keys, double *vals,
                              uint64_t cap, uint64_t *n_out) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  *n_out = c->n;
  if (c->n > cap) { set_err("get: cap %llu < n %llu",
                            (unsigned long long)cap,
                            (unsigned long long)c->n); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (keys) {
    if (c->packed) {
      /* unpack into the spare ping-pong buffer (scratch), copy from it;
       * the resident dataset stays packed */
      TIMED(c, K_OTHER,
            hipLaunchKernelGGL(k_unpack_keys, dim3(stream_grid(c->n)),
                               dim3(NTHREADS), 0, c->stream,
                               (const uint32_t *)c->keys[c->cur],
                               c->keys[c->cur ^ 1], c->n));
      HIP_CHECK(hipMemcpyAsync(keys, c->keys[c->cur ^ 1], c->n * 8,
                               hipMemcpyDeviceToHost, c->stream));
    } else {
      HIP_CHECK(hipMemcpyAsync(keys, c->keys[c->cur], c->n * 8,
                               hipMemcpyDeviceToHost, c->stream));
    }
  }
  if (vals && c->with_vals)
    HIP_CHECK(hipMemcpyAsync(vals, c->vals[c->cur], c->n * 8,
                             hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  return drain_timing(c);
}

/* string-key path: hash a packed bytes buffer
 * of n_words words (offs: n_words+1 byte offsets) into the ctx's keys
 * via k_hash_bytes; vals uploaded alongside (NULL => all 1.0 is NOT
 * assumed — caller supplies vals or reserves keys-only).  The dataset
 * then flows through the normal sort/reduce/group pipeline keyed by the
 * 64-bit hash; the HOST resolves hash -> word via exemplar rows (see
 * dpark_amd/strings.py).  Collisions: P ~ K^2/2^65 for K distinct words
 * (~1e-9 at K=10M); DPRK_VERIFY_STRING_HASH=1 host-verifies exactly. */
extern "C" int dprk_put_bytes_keys(dprk_ctx *c, const uint8_t *buf,
                                   uint64_t nbytes, const uint64_t *offs,
                                   uint64_t n_words, const double *vals) {
  if (!c) { set_err("null ctx"); return 1; }
  if (n_words > c->cap) { set_err("put_bytes: n > reserved cap"); return 1; }
  c->lazy_gen = 0;   /* overwritten */
  if (vals && !c->with_vals) { set_err("put_bytes: ctx reserved keys-only");
                               return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  uint8_t *d_buf = nullptr;
  uint64_t *d_offs = nullptr;
  HIP_CHECK(hipMalloc(&d_buf, nbytes ? nbytes : 1));
  HIP_CHECK(hipMalloc(&d_offs, (n_words + 1) * 8));
  HIP_CHECK(hipMemcpyAsync(d_buf, buf, nbytes, hipMemcpyHostToDevice,
                           c->stream));
  HIP_CHECK(hipMemcpyAsync(d_offs, offs, (n_words + 1) * 8,
                           hipMemcpyHostToDevice, c->stream));
  TIMED(c, K_HIST,
        hipLaunchKernelGGL(k_hash_bytes, dim3(stream_grid(n_words)),
                           dim3(NTHREADS), 0, c->stream, d_buf, d_offs,
                           n_words, c->keys[c->cur]));
  if (vals && c->with_vals)
    HIP_CHECK(hipMemcpyAsync(c->vals[c->cur], vals, n_words * 8,
                             hipMemcpyHostToDevice, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  HIP_CHECK(hipFree(d_buf));
  HIP_CHECK(hipFree(d_offs));
  c->stats.bytes_total += nbytes + n_words * 16;
  c->n = n_words;
  c->sorted = 0;
  c->packed = 0;      /* hashes span the full 64-bit space */
  c->orand_valid = 0;
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  return drain_timing(c);
}

/* packed-state introspection + forcing (multi-rank exchange agreement:
 * every rank must ship the same key width — the host layer all-reduces
 * the packed flags and calls set_packed to align disagreeing ranks) */
extern "C" int dprk_is_packed(dprk_ctx *c, int *out) {
  if (!c || !out) { set_err("null arg"); return 1; }
  *out = c->packed;
  return 0;
}

extern "C" int dprk_set_packed(dprk_ctx *c, int want) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  HIP_CHECK(hipSetDevice(c->device));
  if (want == c->packed) return 0;
  if (c->n == 0) { c->packed = want ? 1 : 0; return 0; }
  if (!want) {
    /* unpack in place: u32 -> int64 into the spare buffer, pointer-swap */
    TIMED(c, K_OTHER,
          hipLaunchKernelGGL(k_unpack_keys, dim3(stream_grid(c->n)),
                             dim3(NTHREADS), 0, c->stream,
                             (const uint32_t *)c->keys[c->cur],
                             c->keys[c->cur ^ 1], c->n));
    int64_t *tmp = c->keys[c->cur];
    c->keys[c->cur] = c->keys[c->cur ^ 1];
    c->keys[c->cur ^ 1] = tmp;
    c->packed = 0;
    if (c->orand_valid) {   /* packed bias -> int64 bias domain */
      c->known_or |= SIGNFLIP;
      c->known_and |= SIGNFLIP;
    }
    c->stats.bytes_total += c->n * 12;
    return drain_timing(c);
  }
  /* want packed: only valid if every key fits [0, 2^32) */
  if (!c->orand_valid) {
    uint64_t init[2] = {0, ~0ull};
    HIP_CHECK(hipMemcpyAsync(c->d_orand, init, 16, hipMemcpyHostToDevice,
                             c->stream));
    TIMED(c, K_HIST,
          hipLaunchKernelGGL(k_minmax_or<int64_t>, dim3(stream_grid(c->n)),
                             dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                             c->n, c->d_orand));
    uint64_t oa[2];
    HIP_CHECK(hipMemcpyAsync(oa, c->d_orand, 16, hipMemcpyDeviceToHost,
                             c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    c->orand_valid = 1;
    c->known_or = oa[0];
    c->known_and = oa[1];
  }
  if ((c->known_or >> 32) != 0x80000000ull) {
    set_err("set_packed: keys do not fit [0, 2^32)");
    return 1;
  }
  TIMED(c, K_OTHER,
        hipLaunchKernelGGL(k_pack_keys, dim3(stream_grid(c->n)),
                           dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                           (uint32_t *)c->keys[c->cur ^ 1], c->n));
  int64_t *tmp = c->keys[c->cur];
  c->keys[c->cur] = c->keys[c->cur ^ 1];
  c->keys[c->cur ^ 1] = tmp;
  c->packed = 1;
  c->known_or &= 0xFFFFFFFFull;
  c->known_and &= 0xFFFFFFFFull;
  c->stats.bytes_total += c->n * 12;
  return drain_timing(c);
}

/* deterministic f64 sum of the current values (conservation checks at
 * out-of-core sizes): per-block fixed-tree partials + host combine in
 * block order. */
extern "C" int dprk_val_sum(dprk_ctx *c, double *out) {
  if (!c || !out) { set_err("null arg"); return 1; }
  if (materialize_lazy(c)) return 1;
  if (!c->with_vals) { set_err("val_sum: keys-only ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  *out = 0.0;
  if (c->n == 0) return 0;
  int grid = stream_grid(c->n);
  double *d_part = nullptr;
  HIP_CHECK(hipMalloc(&d_part, (uint64_t)grid * 8));
  TIMED(c, K_OTHER,
        hipLaunchKernelGGL(k_val_sum, dim3(grid), dim3(NTHREADS), 0,
                           c->stream, c->vals[c->cur], c->n, d_part));
  double *h = (double *)malloc((uint64_t)grid * 8);
  HIP_CHECK(hipMemcpyAsync(h, d_part, (uint64_t)grid * 8,
                           hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  double s = 0.0;
  for (int i = 0; i < grid; i++) s += h[i];
  free(h);
  HIP_CHECK(hipFree(d_part));
  *out = s;
  return drain_timing(c);
}

/* ------------------------------------------------------------------ */
/* out-of-core "big" reduceByKey (n >= 2^32).  */
/* Chunks (each < 2^32, ingested via gen_pairs_at / put_pairs) are     */
/* hash-partitioned (the same splitmix64 top-digit HashPartitioner as  */
/* the multi-rank path) into nbuckets device ARENAS; every key lands   */
/* wholly in one bucket, so each bucket then runs the normal in-core   */
/* sort + segmented reduce, writing its (unique key, sum) output       */
/* compacted forward in the arena.  Output order: bucket-major (hash   */
/* digit), key-sorted within a bucket — documented, not globally key-  */
/* sorted (reduceByKey's contract is a map).         */
/* dpark's external sorted-run Merger is the semantic precedent [U].   */
/* ------------------------------------------------------------------ */

extern "C" int dprk_big_begin(dprk_ctx *c, uint64_t total_cap,
                              int nbuckets) {
  if (!c) { set_err("null ctx"); return 1; }
  if (nbuckets < 2 || nbuckets > 256 || (nbuckets & (nbuckets - 1))) {
    set_err("big_begin: nbuckets must be a power of two in [2,256]");
    return 1;
  }
  if (!c->with_vals) { set_err("big_begin: reserve with_vals first");
                       return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  /* re-begin with identical geometry REUSES the arenas (a ~100 GB
   * hipMalloc costs seconds; measured 4.1 s of a 4.5 s 8B-pair job
   * before this) — only the fill state resets */
  if (c->big_k && c->big_cap_total == total_cap &&
      c->big_nbuckets == nbuckets) {
    memset(c->big_fill, 0, sizeof(c->big_fill));
    c->big_out_n = 0;
    c->big_or = 0;
    c->big_and = ~0ull;
    c->big_valid = 1;
    return 0;
  }
  if (free_big(c)) return 1;
  /* arenas are allocated lazily at the first chunk (key width unknown
   * until then); per-bucket slack covers hash-uniform fluctuation —
   * heavy skew that overflows a bucket errors clearly (hot-key
   * pre-reduction is the documented mitigation) */
  c->big_cap_total = total_cap;
  c->big_cap_b = total_cap / nbuckets + total_cap / (nbuckets * 8) + 4096;
  c->big_nbuckets = nbuckets;
  c->big_packed = -1;
  c->big_or = 0;
  c->big_and = ~0ull;
  c->big_valid = 1;
  return 0;
}

extern "C" int dprk_big_add_chunk(dprk_ctx *c) {
  if (!c || !c->big_valid) { set_err("big_add_chunk: call big_begin");
                             return 1; }
  if (c->n == 0) return 0;
  HIP_CHECK(hipSetDevice(c->device));
  if (c->big_packed < 0) {
    c->big_packed = c->packed;
    uint64_t ksz = c->big_packed ? 4 : 8;
    uint64_t nb = (uint64_t)c->big_nbuckets;
    if (hipMalloc(&c->big_k, nb * c->big_cap_b * ksz) != hipSuccess) {
      set_err("big: keys arena alloc failed (%llu B)",
              (unsigned long long)(nb * c->big_cap_b * ksz));
      return 1;
    }
    if (hipMalloc(&c->big_v, nb * c->big_cap_b * 8) != hipSuccess) {
      (void)hipFree(c->big_k);
      c->big_k = nullptr;
      set_err("big: vals arena alloc failed");
      return 1;
    }
  }
  if (c->packed != c->big_packed) {
    /* align the chunk's key width to the aggregate's (first chunk wins;
     * set_packed(1) fails loudly if the chunk's keys do not fit) */
    if (dprk_set_packed(c, c->big_packed)) return 1;
  }
  if (c->orand_valid) {
    c->big_or |= c->known_or;
    c->big_and &= c->known_and;
  } else {
    c->big_or = ~0ull;
    c->big_and = 0;
  }
  uint64_t counts[256];
  if (dprk_partition(c, c->big_nbuckets, counts)) return 1;
  uint64_t ksz = c->big_packed ? 4 : 8;
  uint64_t off = 0;
  for (int b = 0; b < c->big_nbuckets; b++) {
    if (c->big_fill[b] + counts[b] > c->big_cap_b) {
      set_err("big: bucket %d overflows its arena (%llu + %llu > %llu) — "
              "skew beyond hash balancing; reduce chunk size or pre-reduce "
              "chunks", b, (unsigned long long)c->big_fill[b],
              (unsigned long long)counts[b],
              (unsigned long long)c->big_cap_b);
      return 1;
    }
    if (counts[b]) {
      char *dk = (char *)c->big_k + (b * c->big_cap_b + c->big_fill[b]) * ksz;
      HIP_CHECK(hipMemcpyAsync(dk, (char *)c->keys[c->cur] + off * ksz,
                               counts[b] * ksz, hipMemcpyDeviceToDevice,
                               c->stream));
      HIP_CHECK(hipMemcpyAsync(c->big_v + b * c->big_cap_b + c->big_fill[b],
                               c->vals[c->cur] + off, counts[b] * 8,
                               hipMemcpyDeviceToDevice, c->stream));
      c->big_fill[b] += counts[b];
    }
    off += counts[b];
  }
  HIP_CHECK(hipStreamSynchronize(c->stream));
  c->stats.bytes_total += c->n * (ksz + 8) * 2;
  c->n = 0;   /* chunk consumed */
  return drain_timing(c);
}

extern "C" int dprk_big_reduce(dprk_ctx *c, uint64_t *n_out) {
  if (!c || !c->big_valid) { set_err("big_reduce: call big_begin");
                             return 1; }
  if (c->big_packed < 0) { c->big_out_n = 0;
                           if (n_out) *n_out = 0; return 0; }
  HIP_CHECK(hipSetDevice(c->device));
  uint64_t ksz = c->big_packed ? 4 : 8;
  uint64_t out_off = 0;
  for (int b = 0; b < c->big_nbuckets; b++) {
    uint64_t n_b = c->big_fill[b];
    if (!n_b) continue;
    if (n_b > c->cap) {
      set_err("big_reduce: bucket %d (%llu pairs) exceeds the chunk "
              "capacity %llu", b, (unsigned long long)n_b,
              (unsigned long long)c->cap);
      return 1;
    }
    char *sk = (char *)c->big_k + b * c->big_cap_b * ksz;
    HIP_CHECK(hipMemcpyAsync(c->keys[c->cur], sk, n_b * ksz,
                             hipMemcpyDeviceToDevice, c->stream));
    HIP_CHECK(hipMemcpyAsync(c->vals[c->cur], c->big_v + b * c->big_cap_b,
                             n_b * 8, hipMemcpyDeviceToDevice, c->stream));
    c->n = n_b;
    c->sorted = 0;
    c->packed = c->big_packed;
    c->orand_valid = 1;             /* conservative union over chunks */
    c->known_or = c->big_or;
    c->known_and = c->big_and;
    c->hist_shift_valid = -1;
    c->ghist_valid = 0;
    uint64_t nu = 0;
    if (dprk_reduce_by_key_sum(c, &nu)) return 1;
    /* compact the bucket's output forward in the arenas (out_off never
     * passes the consumed input region: output <= input per bucket) */
    HIP_CHECK(hipMemcpyAsync((char *)c->big_k + out_off * ksz,
                             c->keys[c->cur], nu * ksz,
                             hipMemcpyDeviceToDevice, c->stream));
    HIP_CHECK(hipMemcpyAsync(c->big_v + out_off, c->vals[c->cur], nu * 8,
                             hipMemcpyDeviceToDevice, c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    c->stats.bytes_total += (n_b + nu) * (ksz + 8);
    out_off += nu;
  }
  c->big_out_n = out_off;
  c->n = 0;   /* the ctx dataset is scratch now */
  if (n_out) *n_out = out_off;
  return 0;
}

extern "C" int dprk_big_get(dprk_ctx *c, int64_t *keys, double *vals,
                            uint64_t cap, uint64_t *n_out) {
  if (!c || !c->big_valid) { set_err("big_get: call big_begin"); return 1; }
  *n_out = c->big_out_n;
  if (c->big_out_n > cap) { set_err("big_get: cap too small"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->big_out_n == 0) return 0;
  if (vals)
    HIP_CHECK(hipMemcpyAsync(vals, c->big_v, c->big_out_n * 8,
                             hipMemcpyDeviceToHost, c->stream));
  if (keys) {
    if (c->big_packed) {
      /* unpack through the chunk ctx's spare buffer, piecewise */
      uint64_t done = 0;
      while (done < c->big_out_n) {
        uint64_t take = min(c->big_out_n - done, c->cap);
        hipLaunchKernelGGL(k_unpack_keys, dim3(stream_grid(take)),
                           dim3(NTHREADS), 0, c->stream,
                           (const uint32_t *)c->big_k + done,
                           c->keys[c->cur ^ 1], take);
        HIP_CHECK(hipGetLastError());
        HIP_CHECK(hipMemcpyAsync(keys + done, c->keys[c->cur ^ 1], take * 8,
                                 hipMemcpyDeviceToHost, c->stream));
        HIP_CHECK(hipStreamSynchronize(c->stream));
        done += take;
      }
    } else {
      HIP_CHECK(hipMemcpyAsync(keys, c->big_k, c->big_out_n * 8,
                               hipMemcpyDeviceToHost, c->stream));
    }
  }
  HIP_CHECK(hipStreamSynchronize(c->stream));
  return 0;
}

/* big-output val sum without D2H (conservation check) */
extern "C" int dprk_big_val_sum(dprk_ctx *c, double *out) {
  if (!c || !c->big_valid || !out) { set_err("big_val_sum: bad args");
                                     return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  *out = 0.0;
  if (c->big_out_n == 0) return 0;
  int grid = stream_grid(c->big_out_n);
  double *d_part = nullptr;
  HIP_CHECK(hipMalloc(&d_part, (uint64_t)grid * 8));
  hipLaunchKernelGGL(k_val_sum, dim3(grid), dim3(NTHREADS), 0, c->stream,
                     c->big_v, c->big_out_n, d_part);
  HIP_CHECK(hipGetLastError());
  double *h = (double *)malloc((uint64_t)grid * 8);
  HIP_CHECK(hipMemcpyAsync(h, d_part, (uint64_t)grid * 8,
                           hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  double s = 0.0;
  for (int i = 0; i < grid; i++) s += h[i];
  free(h);
  HIP_CHECK(hipFree(d_part));
  *out = s;
  return 0;
}

extern "C" int dprk_big_free(dprk_ctx *c) {
  if (!c) { set_err("null ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  return free_big(c);
}

/* ---- one radix pass: tile_hist -> 3-level scan -> scatter ---------- */
template <typename K>
static int radix_pass_t(dprk_ctx *c, int mode, int shift, uint32_t mask,
                        int want_row_tot, const int64_t *d_split = nullptr,
                        int nsplit = 0) {
  uint64_t n = c->n;
  uint64_t ntiles = ceil_div(n, TILE);
  uint64_t seg_tiles = ceil_div(ntiles, RADIX); /* nseg <= 256 */
  uint64_t nseg = ceil_div(ntiles, seg_tiles);
  int src = c->cur, dst = c->cur ^ 1;
  int thgrid = tile_grid(ntiles);
  int hist_ran = !(mode == 0 && c->hist_shift_valid == shift);
  if (hist_ran) {
    TIMED(c, K_TILEHIST,
          hipLaunchKernelGGL(k_tile_hist<K>, dim3(thgrid), dim3(NTHREADS), 0,
                             c->stream, (const K *)c->keys[src], n,
                             c->tile_hist, ntiles, mode, shift, mask,
                             d_split, nsplit));
  }
  c->hist_shift_valid = -1; /* consumed; scatter reorders the data */
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_seg_sums, dim3((uint32_t)nseg), dim3(NTHREADS),
                           0, c->stream, c->tile_hist, ntiles, seg_tiles,
                           c->seg_sums));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_scan_mid, dim3(1), dim3(NTHREADS), 0, c->stream,
                           c->seg_sums, nseg,
                           want_row_tot ? c->row_tot : nullptr));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_tile_off, dim3((uint32_t)nseg), dim3(NTHREADS),
                           0, c->stream, c->tile_hist, ntiles, seg_tiles,
                           c->seg_sums, c->tile_off));
  static int rank_atomic = -1;
  if (rank_atomic < 0) rank_atomic = getenv("DPRK_RANK_BALLOT") ? 0 : 1;  /* atomic rank default (r02 A/B) */
  if (c->with_vals) {
    if (rank_atomic) {
      TIMED(c, K_SCATTER,
            hipLaunchKernelGGL((k_scatter<K, true, true>), dim3(thgrid),
                               dim3(NTHREADS_SC),
                               0, c->stream, (const K *)c->keys[src],
                               c->vals[src],
                               (K *)c->keys[dst], c->vals[dst], n,
                               c->tile_hist,
                               c->tile_off, ntiles, mode, shift, mask,
                               d_split, nsplit));
    } else {
      TIMED(c, K_SCATTER,
            hipLaunchKernelGGL((k_scatter<K, true>), dim3(thgrid),
                               dim3(NTHREADS_SC),
                               0, c->stream, (const K *)c->keys[src],
                               c->vals[src],
                               (K *)c->keys[dst], c->vals[dst], n,
                               c->tile_hist,
                               c->tile_off, ntiles, mode, shift, mask,
                               d_split, nsplit));
    }
  } else {
    if (rank_atomic) {
      TIMED(c, K_SCATTER,
            hipLaunchKernelGGL((k_scatter<K, false, true>), dim3(thgrid),
                               dim3(NTHREADS_SC),
                               0, c->stream, (const K *)c->keys[src], nullptr,
                               (K *)c->keys[dst], nullptr, n, c->tile_hist,
                               c->tile_off, ntiles, mode, shift, mask,
                               d_split, nsplit));
    } else {
      TIMED(c, K_SCATTER,
            hipLaunchKernelGGL((k_scatter<K, false>), dim3(thgrid),
                               dim3(NTHREADS_SC),
                               0, c->stream, (const K *)c->keys[src], nullptr,
                               (K *)c->keys[dst], nullptr, n, c->tile_hist,
                               c->tile_off, ntiles, mode, shift, mask,
                               d_split, nsplit));
    }
  }
  uint64_t bpp = 2 * (sizeof(K) + (c->with_vals ? 8 : 0));
  c->stats.bytes_scatter += n * bpp;
  c->stats.bytes_total += n * (bpp + (hist_ran ? sizeof(K) : 0));
  c->cur = dst;
  return 0;
}

static int radix_pass(dprk_ctx *c, int mode, int shift, uint32_t mask,
                      int want_row_tot, const int64_t *d_split = nullptr,
                      int nsplit = 0) {
  return c->packed
             ? radix_pass_t<uint32_t>(c, mode, shift, mask, want_row_tot,
                                      d_split, nsplit)
             : radix_pass_t<int64_t>(c, mode, shift, mask, want_row_tot,
                                     d_split, nsplit);
}

extern "C" int dprk_sample_keys(dprk_ctx *c, uint64_t n_samples,
                                int64_t *out_host) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  HIP_CHECK(hipSetDevice(c->device));
  if (c->n == 0 || n_samples == 0) return 0;
  if (n_samples > c->n) n_samples = c->n;
  uint64_t stride_elems = c->n / n_samples;
  int64_t *d_out = nullptr;
  HIP_CHECK(hipMalloc(&d_out, n_samples * 8));
  if (c->packed) {
    TIMED(c, K_OTHER,
          hipLaunchKernelGGL(k_sample<uint32_t>, dim3(stream_grid(n_samples)),
                             dim3(NTHREADS), 0, c->stream,
                             (const uint32_t *)c->keys[c->cur],
                             c->n, stride_elems, d_out, n_samples));
  } else {
    TIMED(c, K_OTHER,
          hipLaunchKernelGGL(k_sample<int64_t>, dim3(stream_grid(n_samples)),
                             dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                             c->n, stride_elems, d_out, n_samples));
  }
  HIP_CHECK(hipMemcpyAsync(out_host, d_out, n_samples * 8,
                           hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  HIP_CHECK(hipFree(d_out));
  return drain_timing(c);
}

/* range partition by sorted splitters (nranks-1 of them): bucket r =
 * #{j : key >= splitters[j]} — the engine RangePartitioner
 * (dpark/dependency.py :: RangePartitioner [U]); order-preserving across
 * ranks, used by distributed sortByKey. */
extern "C" int dprk_partition_range(dprk_ctx *c, const int64_t *splitters,
                                    int nbuckets, uint64_t *bucket_counts) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  if (nbuckets < 1 || nbuckets > 256) {
    set_err("partition_range: nbuckets must be in [1,256]");
    return 1;
  }
  HIP_CHECK(hipSetDevice(c->device));
  if (nbuckets == 1 || c->n == 0) {
    if (bucket_counts) {
      for (int b = 0; b < nbuckets; b++) bucket_counts[b] = 0;
      if (nbuckets >= 1) bucket_counts[0] = c->n;
    }
    return 0;
  }
  int64_t *d_split = nullptr;
  HIP_CHECK(hipMalloc(&d_split, (nbuckets - 1) * 8));
  HIP_CHECK(hipMemcpyAsync(d_split, splitters, (nbuckets - 1) * 8,
                           hipMemcpyHostToDevice, c->stream));
  int rc = radix_pass(c, /*mode=*/2, 0, 0, 1, d_split, nbuckets - 1);
  c->sorted = 0;
  c->orand_valid = 0;
  if (!rc && bucket_counts) {
    uint64_t rt[RADIX];
    HIP_CHECK(hipMemcpyAsync(rt, c->row_tot, RADIX * 8,
                             hipMemcpyDeviceToHost, c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    for (int b = 0; b < nbuckets; b++) bucket_counts[b] = rt[b];
  }
  int rc2 = drain_timing(c);
  HIP_CHECK(hipFree(d_split));
  return rc ? rc : rc2;
}

static int ensure_lookback(dprk_ctx *c, uint64_t ntiles) {
  if (ntiles <= c->lb_cap_tiles && c->d_lb) return 0;
  if (c->d_lb) HIP_CHECK(hipFree(c->d_lb));
  HIP_CHECK(hipMalloc(&c->d_lb, ntiles * RADIX * 8));
  if (!c->d_fail) HIP_CHECK(hipMalloc(&c->d_fail, 4));
  if (!c->d_ghist) HIP_CHECK(hipMalloc(&c->d_ghist, 8 * RADIX * 4));
  if (!c->d_dbase) HIP_CHECK(hipMalloc(&c->d_dbase, RADIX * 4));
  c->lb_cap_tiles = ntiles;
  return 0;
}

/* lookback sort (DPRK_LOOKBACK): one upfront read builds every active
 * pass's global digit histogram; each pass is then a single
 * k_scatter_lb launch (per-tile hist + decoupled-lookback offsets).
 * Falls back per pass to the 3-kernel path on bounded-spin failure. */
template <typename K>
static int sort_pairs_lookback(dprk_ctx *c, const int *passes, int np) {
  uint64_t n = c->n;
  uint64_t ntiles = ceil_div(n, TILE);
  if (ensure_lookback(c, ceil_div(c->cap, (uint64_t)TILE))) return 1;
  /* global histograms for all active passes (8 KiB readback) */
  uint64_t shifts = 0;
  for (int i = 0; i < np; i++)
    shifts |= ((uint64_t)(8 * passes[i])) << (8 * i);
  int gh_map[8];
  if (c->ghist_valid) {
    /* the generator already built global hists; its pass list covers a
     * superset (all nonzero mask bytes == the active passes).  Map each
     * active pass to its slot. */
    for (int i = 0; i < np; i++) {
      gh_map[i] = -1;
      for (int j = 0; j < c->ghist_nact; j++)
        if (((c->ghist_shifts >> (8 * j)) & 0xFF) ==
            (uint64_t)(8 * passes[i]))
          gh_map[i] = j;
    }
    bool all = true;
    for (int i = 0; i < np; i++) all = all && gh_map[i] >= 0;
    if (!all) c->ghist_valid = 0;
  }
  if (!c->ghist_valid) {
    HIP_CHECK(hipMemsetAsync(c->d_ghist, 0, 8 * RADIX * 4, c->stream));
    TIMED(c, K_HIST,
          hipLaunchKernelGGL(k_hist_all<K>, dim3(stream_grid(n / 2 + 1)),
                             dim3(NTHREADS), 0, c->stream,
                             (const K *)c->keys[c->cur],
                             n, shifts, np, c->d_ghist));
    c->stats.bytes_total += n * sizeof(K);
    for (int i = 0; i < np; i++) gh_map[i] = i;
  }
  uint32_t gh[8 * RADIX];
  HIP_CHECK(hipMemcpyAsync(gh, c->d_ghist, 8 * RADIX * 4,
                           hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  for (int i = 0; i < np; i++) {
    /* exclusive digit bases for this pass */
    uint32_t base[RADIX], run = 0;
    for (int d = 0; d < RADIX; d++) {
      base[d] = run;
      run += gh[gh_map[i] * RADIX + d];
    }
    HIP_CHECK(hipMemcpyAsync(c->d_dbase, base, RADIX * 4,
                             hipMemcpyHostToDevice, c->stream));
    HIP_CHECK(hipMemsetAsync(c->d_lb, 0, ntiles * RADIX * 8, c->stream));
    HIP_CHECK(hipMemsetAsync(c->d_fail, 0, 4, c->stream));
    int src = c->cur, dst = c->cur ^ 1;
    /* grid = guaranteed-resident block count (static issue-ordered
     * assignment needs every block resident; the occupancy API can
     * over-report by one for SGPR-heavy kernels — clamp to 2/CU, which
     * the 74 KiB LDS independently bounds) */
    static int occ_v = 0, occ_k = 0;   /* per-K instantiation cache */
    if (!occ_v) {
      HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &occ_v, (const void *)(k_scatter_lb<K, true>), NTHREADS_SC, 0));
      HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
          &occ_k, (const void *)(k_scatter_lb<K, false>), NTHREADS_SC, 0));
      hipDeviceProp_t props;
      HIP_CHECK(hipGetDeviceProperties(&props, c->device));
      occ_v = min(2, max(1, occ_v)) * props.multiProcessorCount;
      occ_k = min(2, max(1, occ_k)) * props.multiProcessorCount;
    }
    int grid = (int)min(ntiles, (uint64_t)(c->with_vals ? occ_v : occ_k));
    static int rank_atomic = -1;
    if (rank_atomic < 0) rank_atomic = getenv("DPRK_RANK_BALLOT") ? 0 : 1;  /* atomic rank default (r02 A/B) */
    static int pipe = -1;
    if (pipe < 0) pipe = getenv("DPRK_NO_PIPE") ? 0 : 1;  /* default on:
        r02 same-box A/B: 0.868 -> 0.715 ms/launch at 100M (+16% value),
        +4.7% at 1B, +1.3% zipf; parity suite green */
    static int pipe2 = -1;
    if (pipe2 < 0) pipe2 = getenv("DPRK_PIPE2") ? 1 : 0;
    if (c->lazy_gen && i == 0) {
      /* FUSED gen + first pass: pairs are generated straight into
         pass-0-partitioned layout (write-only: saves the gen write +
         pass read, 24 B/pair).  Eligibility (pipe form exists) was
         checked before the prepass in dprk_sort_pairs. */
      bool launched = false;
      static int occ_gv = 0, occ_gk = 0;
      if constexpr (sizeof(K) == 4) {
        if (!occ_gv) {
          HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
              &occ_gv, (const void *)(k_scatter_lb_pipe<K, true, true>),
              NTHREADS_SC, 0));
          HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
              &occ_gk, (const void *)(k_scatter_lb_pipe<K, false, true>),
              NTHREADS_SC, 0));
          hipDeviceProp_t props;
          HIP_CHECK(hipGetDeviceProperties(&props, c->device));
          occ_gv = min(2, max(1, occ_gv)) * props.multiProcessorCount;
          occ_gk = min(2, max(1, occ_gk)) * props.multiProcessorCount;
        }
        int ggrid = (int)min(ntiles, (uint64_t)(c->with_vals ? occ_gv
                                                             : occ_gk));
        if (c->with_vals) {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe<K, true, true>),
                                   dim3(ggrid), dim3(NTHREADS_SC), 0,
                                   c->stream, nullptr, nullptr,
                                   (K *)c->keys[dst], c->vals[dst], n,
                                   ntiles, 8 * passes[i], c->d_dbase,
                                   c->d_lb, c->d_fail, c->lazy_seed_hi,
                                   c->lazy_keymask));
        } else {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe<K, false, true>),
                                   dim3(ggrid), dim3(NTHREADS_SC), 0,
                                   c->stream, nullptr, nullptr,
                                   (K *)c->keys[dst], nullptr, n,
                                   ntiles, 8 * passes[i], c->d_dbase,
                                   c->d_lb, c->d_fail, c->lazy_seed_hi,
                                   c->lazy_keymask));
        }
        launched = true;
      } else {
        if (!c->with_vals) {
          if (!occ_gk) {
            HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
                &occ_gk, (const void *)(k_scatter_lb_pipe<K, false, true>),
                NTHREADS_SC, 0));
            hipDeviceProp_t props;
            HIP_CHECK(hipGetDeviceProperties(&props, c->device));
            occ_gk = min(2, max(1, occ_gk)) * props.multiProcessorCount;
          }
          int ggrid = (int)min(ntiles, (uint64_t)occ_gk);
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe<K, false, true>),
                                   dim3(ggrid), dim3(NTHREADS_SC), 0,
                                   c->stream, nullptr, nullptr,
                                   (K *)c->keys[dst], nullptr, n,
                                   ntiles, 8 * passes[i], c->d_dbase,
                                   c->d_lb, c->d_fail, c->lazy_seed_hi,
                                   c->lazy_keymask));
          launched = true;
        }
      }
      if (launched) {
        uint32_t failg = 0;
        HIP_CHECK(hipMemcpyAsync(&failg, c->d_fail, 4,
                                 hipMemcpyDeviceToHost, c->stream));
        HIP_CHECK(hipStreamSynchronize(c->stream));
        if (failg) {
          if (materialize_lazy(c)) return 1;
          if (radix_pass_t<K>(c, 0, 8 * passes[i], 0xFF, 0)) return 1;
        } else {
          c->lazy_gen = 0;
          uint64_t bpp = sizeof(K) + (c->with_vals ? 8 : 0); /* write only */
          c->stats.bytes_scatter += n * bpp;
          c->stats.bytes_total += n * bpp;
          c->cur = dst;
        }
        c->hist_shift_valid = -1;
        c->stats.sort_passes++;
        continue;
      }
      if (materialize_lazy(c)) return 1;   /* safety net */
    }
    if constexpr (sizeof(K) == 4) {
      if (pipe) {
      /* software-pipelined variant (u32 only); DPRK_PIPE2 selects the
       * two-deep key prefetch form */
      static int occ_pv = 0, occ_pk = 0;
      if (!occ_pv) {
        HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &occ_pv, pipe2 ? (const void *)(k_scatter_lb_pipe2<K, true>)
                           : (const void *)(k_scatter_lb_pipe<K, true>),
            NTHREADS_SC, 0));
        HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &occ_pk, pipe2 ? (const void *)(k_scatter_lb_pipe2<K, false>)
                           : (const void *)(k_scatter_lb_pipe<K, false>),
            NTHREADS_SC, 0));
        hipDeviceProp_t props;
        HIP_CHECK(hipGetDeviceProperties(&props, c->device));
        occ_pv = min(2, max(1, occ_pv)) * props.multiProcessorCount;
        occ_pk = min(2, max(1, occ_pk)) * props.multiProcessorCount;
      }
      int pgrid = (int)min(ntiles, (uint64_t)(c->with_vals ? occ_pv
                                                           : occ_pk));
      if (c->with_vals) {
        if (pipe2) {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe2<K, true>),
                                   dim3(pgrid),
                                   dim3(NTHREADS_SC), 0, c->stream,
                                   (const K *)c->keys[src], c->vals[src],
                                   (K *)c->keys[dst], c->vals[dst], n,
                                   ntiles,
                                   8 * passes[i], c->d_dbase, c->d_lb,
                                   c->d_fail));
        } else {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe<K, true>), dim3(pgrid),
                                   dim3(NTHREADS_SC), 0, c->stream,
                                   (const K *)c->keys[src], c->vals[src],
                                   (K *)c->keys[dst], c->vals[dst], n,
                                   ntiles,
                                   8 * passes[i], c->d_dbase, c->d_lb,
                                   c->d_fail));
        }
      } else {
        if (pipe2) {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe2<K, false>),
                                   dim3(pgrid),
                                   dim3(NTHREADS_SC), 0, c->stream,
                                   (const K *)c->keys[src], nullptr,
                                   (K *)c->keys[dst], nullptr, n, ntiles,
                                   8 * passes[i], c->d_dbase, c->d_lb,
                                   c->d_fail));
        } else {
          TIMED(c, K_SCATTER,
                hipLaunchKernelGGL((k_scatter_lb_pipe<K, false>),
                                   dim3(pgrid),
                                   dim3(NTHREADS_SC), 0, c->stream,
                                   (const K *)c->keys[src], nullptr,
                                   (K *)c->keys[dst], nullptr, n, ntiles,
                                   8 * passes[i], c->d_dbase, c->d_lb,
                                   c->d_fail));
        }
      }
      uint32_t failp = 0;
      HIP_CHECK(hipMemcpyAsync(&failp, c->d_fail, 4, hipMemcpyDeviceToHost,
                               c->stream));
      HIP_CHECK(hipStreamSynchronize(c->stream));
      if (failp) {
        if (radix_pass_t<K>(c, 0, 8 * passes[i], 0xFF, 0)) return 1;
      } else {
        uint64_t bpp = 2 * (sizeof(K) + (c->with_vals ? 8 : 0));
        c->stats.bytes_scatter += n * bpp;
        c->stats.bytes_total += n * bpp;
        c->cur = dst;
      }
      c->hist_shift_valid = -1;
      c->stats.sort_passes++;
      continue;
      }
    } else if (!c->with_vals) {
      /* int64 KEYS-ONLY also fits the pipelined kernel (64 KiB staging
       * + 32 KiB double hist ~ 98 KiB LDS); with-values int64 does not
       * (160+ KiB) and takes the non-pipelined path below */
      if (pipe) {
        static int occ_p8 = 0;
        if (!occ_p8) {
          HIP_CHECK(hipOccupancyMaxActiveBlocksPerMultiprocessor(
              &occ_p8, (const void *)(k_scatter_lb_pipe<K, false>),
              NTHREADS_SC, 0));
          hipDeviceProp_t props;
          HIP_CHECK(hipGetDeviceProperties(&props, c->device));
          occ_p8 = min(2, max(1, occ_p8)) * props.multiProcessorCount;
        }
        int pgrid = (int)min(ntiles, (uint64_t)occ_p8);
        TIMED(c, K_SCATTER,
              hipLaunchKernelGGL((k_scatter_lb_pipe<K, false>), dim3(pgrid),
                                 dim3(NTHREADS_SC), 0, c->stream,
                                 (const K *)c->keys[src], nullptr,
                                 (K *)c->keys[dst], nullptr, n, ntiles,
                                 8 * passes[i], c->d_dbase, c->d_lb,
                                 c->d_fail));
        uint32_t failp = 0;
        HIP_CHECK(hipMemcpyAsync(&failp, c->d_fail, 4,
                                 hipMemcpyDeviceToHost, c->stream));
        HIP_CHECK(hipStreamSynchronize(c->stream));
        if (failp) {
          if (radix_pass_t<K>(c, 0, 8 * passes[i], 0xFF, 0)) return 1;
        } else {
          c->stats.bytes_scatter += n * 2 * sizeof(K);
          c->stats.bytes_total += n * 2 * sizeof(K);
          c->cur = dst;
        }
        c->hist_shift_valid = -1;
        c->stats.sort_passes++;
        continue;
      }
    }
    if (c->with_vals) {
      if (rank_atomic) {
        TIMED(c, K_SCATTER,
              hipLaunchKernelGGL((k_scatter_lb<K, true, true>), dim3(grid),
                                 dim3(NTHREADS_SC), 0, c->stream,
                                 (const K *)c->keys[src], c->vals[src],
                                 (K *)c->keys[dst],
                                 c->vals[dst], n, ntiles, 8 * passes[i],
                                 c->d_dbase, c->d_lb, c->d_fail));
      } else {
        TIMED(c, K_SCATTER,
              hipLaunchKernelGGL((k_scatter_lb<K, true>), dim3(grid),
                                 dim3(NTHREADS_SC), 0, c->stream,
                                 (const K *)c->keys[src], c->vals[src],
                                 (K *)c->keys[dst],
                                 c->vals[dst], n, ntiles, 8 * passes[i],
                                 c->d_dbase, c->d_lb, c->d_fail));
      }
    } else {
      if (rank_atomic) {
        TIMED(c, K_SCATTER,
              hipLaunchKernelGGL((k_scatter_lb<K, false, true>), dim3(grid),
                                 dim3(NTHREADS_SC), 0, c->stream,
                                 (const K *)c->keys[src], nullptr,
                                 (K *)c->keys[dst], nullptr,
                                 n, ntiles, 8 * passes[i], c->d_dbase,
                                 c->d_lb, c->d_fail));
      } else {
        TIMED(c, K_SCATTER,
              hipLaunchKernelGGL((k_scatter_lb<K, false>), dim3(grid),
                                 dim3(NTHREADS_SC), 0, c->stream,
                                 (const K *)c->keys[src], nullptr,
                                 (K *)c->keys[dst], nullptr,
                                 n, ntiles, 8 * passes[i], c->d_dbase,
                                 c->d_lb, c->d_fail));
      }
    }
    uint32_t fail = 0;
    HIP_CHECK(hipMemcpyAsync(&fail, c->d_fail, 4, hipMemcpyDeviceToHost,
                             c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    if (fail) {
      /* bounded-spin bailout: src buffers are intact — redo this pass on
       * the decoupled 3-kernel path (never observed; safety net) */
      if (radix_pass_t<K>(c, 0, 8 * passes[i], 0xFF, 0)) return 1;
    } else {
      uint64_t bpp = 2 * (sizeof(K) + (c->with_vals ? 8 : 0));
      c->stats.bytes_scatter += n * bpp;
      c->stats.bytes_total += n * bpp;
      c->cur = dst;
    }
    c->hist_shift_valid = -1;
    c->stats.sort_passes++;
  }
  if (drain_timing(c)) return 1;
  c->sorted = 1;
  return 0;
}

extern "C" int dprk_sort_pairs(dprk_ctx *c) {
  if (!c) { set_err("null ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->n == 0) { c->sorted = 1; c->lazy_gen = 0; return 0; }
  if (c->sorted) return 0;
  if (c->lazy_gen) {
    /* gen->sort fusion: the fused first pass needs the lookback path
       and a pipelined kernel form; otherwise materialize and go the
       normal way */
    uint64_t lbt = ceil_div(c->n, (uint64_t)TILE);
    int pipe_ok = !getenv("DPRK_NO_PIPE") &&
                  (c->packed || !c->with_vals);
    if (getenv("DPRK_NO_LOOKBACK") || !pipe_ok ||
        lbt * RADIX * 8 > lb_gate_bytes(c->with_vals)) {
      if (materialize_lazy(c)) return 1;
    } else {
      /* hist-only PRNG prepass: every candidate pass's global digit
         histogram + the exact biased OR/AND — zero data traffic */
      uint64_t km = c->lazy_keymask;
      uint64_t shifts = 0;
      int nact = 0;
      for (int p = 0; p < 8; p++) {
        uint32_t mb = (uint32_t)(km >> (8 * p)) & 0xFF;
        if (km == ~0ull || mb)
          shifts |= ((uint64_t)(8 * p)) << (8 * (nact++));
      }
      if (!c->d_ghist) HIP_CHECK(hipMalloc(&c->d_ghist, 8 * RADIX * 4));
      HIP_CHECK(hipMemsetAsync(c->d_ghist, 0, 8 * RADIX * 4, c->stream));
      uint64_t oa_init[2] = {0, ~0ull};
      HIP_CHECK(hipMemcpyAsync(c->d_orand, oa_init, 16,
                               hipMemcpyHostToDevice, c->stream));
      if (c->packed) {
        TIMED(c, K_GEN,
              hipLaunchKernelGGL(k_gen_hist_only<uint32_t>,
                                 dim3(stream_grid(c->n)), dim3(NTHREADS), 0,
                                 c->stream, c->n, c->lazy_seed_hi, km,
                                 shifts, nact, c->d_ghist, c->d_orand));
      } else {
        TIMED(c, K_GEN,
              hipLaunchKernelGGL(k_gen_hist_only<int64_t>,
                                 dim3(stream_grid(c->n)), dim3(NTHREADS), 0,
                                 c->stream, c->n, c->lazy_seed_hi, km,
                                 shifts, nact, c->d_ghist, c->d_orand));
      }
      uint64_t oa[2];
      HIP_CHECK(hipMemcpyAsync(oa, c->d_orand, 16, hipMemcpyDeviceToHost,
                               c->stream));
      HIP_CHECK(hipStreamSynchronize(c->stream));
      c->ghist_shifts = shifts;
      c->ghist_nact = nact;
      c->ghist_valid = 1;
      c->orand_valid = 1;
      c->known_or = oa[0];
      c->known_and = oa[1];
    }
  }
  /* digit-skip detection: OR/AND of biased keys (or the conservative
   * generation-time bound, saving one full read of the keys) */
  uint64_t orand[2];
  if (c->orand_valid) {
    orand[0] = c->known_or;
    orand[1] = c->known_and;
  } else {
    uint64_t init[2] = {0, ~0ull};
    HIP_CHECK(hipMemcpyAsync(c->d_orand, init, 16, hipMemcpyHostToDevice,
                             c->stream));
    if (c->packed) {
      TIMED(c, K_HIST,
            hipLaunchKernelGGL(k_minmax_or<uint32_t>,
                               dim3(stream_grid(c->n)),
                               dim3(NTHREADS), 0, c->stream,
                               (const uint32_t *)c->keys[c->cur],
                               c->n, c->d_orand));
    } else {
      TIMED(c, K_HIST,
            hipLaunchKernelGGL(k_minmax_or<int64_t>,
                               dim3(stream_grid(c->n)),
                               dim3(NTHREADS), 0, c->stream, c->keys[c->cur],
                               c->n, c->d_orand));
    }
    HIP_CHECK(hipMemcpyAsync(orand, c->d_orand, 16, hipMemcpyDeviceToHost,
                             c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    c->stats.bytes_total += c->n * (c->packed ? 4 : 8);
  }
  c->stats.sort_passes = 0;
  int passes[8], np = 0;
  for (int p = 0; p < 8; p++) {
    uint32_t ob = (uint32_t)(orand[0] >> (8 * p)) & 0xFF;
    uint32_t ab = (uint32_t)(orand[1] >> (8 * p)) & 0xFF;
    if (ob != ab) passes[np++] = p;
  }
  /* default path: decoupled-lookback scatter (bit-identical output to
   * the 3-kernel path — same permutation; interleaved A/B: 5.10 vs 5.33
   * ms/step on config 2).  DPRK_NO_LOOKBACK restores the old path. */
  /* the granule array must stay L3-resident (256 MiB): past it every
   * lookback hop is an HBM-random load and the walk cost explodes
   * (measured: 1B keys, 500 MB granules -> +4.3 ms/pass).  The gate is
   * a pure function of n => path choice stays input-deterministic. */
  if (c->lazy_gen && np == 0) {
    /* all keys identical: nothing to sort — materialize and finish */
    if (materialize_lazy(c)) return 1;
  }
  uint64_t lb_tiles = ceil_div(c->n, (uint64_t)TILE);
  if (np && !getenv("DPRK_NO_LOOKBACK") &&
      lb_tiles * RADIX * 8 <= lb_gate_bytes(c->with_vals))
    return c->packed ? sort_pairs_lookback<uint32_t>(c, passes, np)
                     : sort_pairs_lookback<int64_t>(c, passes, np);
  if (c->lazy_gen && materialize_lazy(c)) return 1;
  for (int i = 0; i < np; i++) {
    if (radix_pass(c, /*mode=*/0, /*shift=*/8 * passes[i], /*mask=*/0xFF, 0))
      return 1;
    c->stats.sort_passes++;
  }
  c->sorted = 1;
  return drain_timing(c);
}

extern "C" int dprk_partition(dprk_ctx *c, int nbuckets,
                              uint64_t *bucket_counts) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  if (nbuckets < 1 || nbuckets > 256 || (nbuckets & (nbuckets - 1))) {
    set_err("partition: nbuckets must be a power of two in [1,256]");
    return 1;
  }
  HIP_CHECK(hipSetDevice(c->device));
  if (nbuckets == 1) {
    if (bucket_counts) bucket_counts[0] = c->n;
    return 0;
  }
  int bits = 0;
  while ((1 << bits) < nbuckets) bits++;
  if (c->n > 0) {
    if (radix_pass(c, /*mode=*/1, /*shift=*/64 - bits,
                   /*mask=*/(uint32_t)(nbuckets - 1), 1))
      return 1;
    c->sorted = 0;
  } else {
    HIP_CHECK(hipMemsetAsync(c->row_tot, 0, RADIX * 8, c->stream));
  }
  if (bucket_counts) {
    uint64_t rt[RADIX];
    HIP_CHECK(hipMemcpyAsync(rt, c->row_tot, RADIX * 8,
                             hipMemcpyDeviceToHost, c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
    for (int b = 0; b < nbuckets; b++) bucket_counts[b] = rt[b];
    if (c->n == 0)
      for (int b = 0; b < nbuckets; b++) bucket_counts[b] = 0;
  }
  return drain_timing(c);
}

/* generic u64 exclusive scan (in place) + total readback */
static int run_scan_u64(dprk_ctx *c, uint64_t *data, uint64_t m,
                        uint64_t *total_out) {
  uint64_t nchunks = ceil_div(m, (uint64_t)SCAN_CHUNK);
  if (c->scan_part_cap < nchunks) {
    if (c->scan_part) HIP_CHECK(hipFree(c->scan_part));
    HIP_CHECK(hipMalloc(&c->scan_part, nchunks * 8));
    c->scan_part_cap = nchunks;
  }
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_scanA, dim3((uint32_t)nchunks), dim3(NTHREADS),
                           0, c->stream, data, m, c->scan_part));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_scanB, dim3(1), dim3(NTHREADS), 0, c->stream,
                           c->scan_part, nchunks, c->d_scalar));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_scanC, dim3((uint32_t)nchunks), dim3(NTHREADS),
                           0, c->stream, data, m, c->scan_part));
  if (total_out) {
    HIP_CHECK(hipMemcpyAsync(total_out, c->d_scalar, 8,
                             hipMemcpyDeviceToHost, c->stream));
    HIP_CHECK(hipStreamSynchronize(c->stream));
  }
  return 0;
}

template <typename K>
static int segmented_reduce_sorted(dprk_ctx *c, uint64_t *n_out);

/* the fused last-pass reduce path (DPRK_FUSED_REDUCE): runs all but the
 * last radix pass normally, then k_scatter_reduce (fragment emission) +
 * d-major reorder + the base reduce over the small fragment stream.
 * shift = the last pass's digit shift. */
template <typename K>
static int fused_reduce_last_pass(dprk_ctx *c, int shift, uint64_t *n_out) {
  uint64_t n = c->n;
  uint64_t ntiles = ceil_div(n, TILE);
  uint64_t nrec = ntiles * RADIX;
  uint64_t seg_tiles = ceil_div(ntiles, (uint64_t)RADIX);
  uint64_t nseg = ceil_div(ntiles, seg_tiles);
  int src = c->cur, dst = c->cur ^ 1;
  int grid = tile_grid(ntiles);
  if (ensure_fused(c, ceil_div(c->cap, (uint64_t)TILE) * RADIX)) return 1;
  if (c->hist_shift_valid != shift) {
    TIMED(c, K_TILEHIST,
          hipLaunchKernelGGL(k_tile_hist<K>, dim3(grid), dim3(NTHREADS), 0,
                             c->stream, (const K *)c->keys[src], n,
                             c->tile_hist,
                             ntiles, 0, shift, 0xFF, nullptr, 0));
  }
  c->hist_shift_valid = -1;
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_scatter_reduce<K>, dim3(grid),
                           dim3(NTHREADS_SC), 0, c->stream,
                           (const K *)c->keys[src],
                           c->vals[src], (K *)c->keys[dst], c->vals[dst], n,
                           c->tile_hist, ntiles, shift, c->f_cnt, c->f_lex));
  /* d-major offsets of the fragment stream: the same 3-level scan used
   * for radix passes, run over the fragment-count matrix */
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_seg_sums, dim3((uint32_t)nseg), dim3(NTHREADS),
                           0, c->stream, c->f_cnt, ntiles, seg_tiles,
                           c->seg_sums));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_scan_mid, dim3(1), dim3(NTHREADS), 0, c->stream,
                           c->seg_sums, nseg, c->row_tot));
  TIMED(c, K_SCAN,
        hipLaunchKernelGGL(k_tile_off, dim3((uint32_t)nseg), dim3(NTHREADS),
                           0, c->stream, c->f_cnt, ntiles, seg_tiles,
                           c->seg_sums, c->f_goff));
  uint64_t row_tot[RADIX], n2 = 0;
  HIP_CHECK(hipMemcpyAsync(row_tot, c->row_tot, RADIX * 8,
                           hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  for (int d = 0; d < RADIX; d++) n2 += row_tot[d];
  /* gather tile-local fragments into the d-major stream (into the freed
   * pass-input buffers) */
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_fused_gather<K>, dim3(stream_grid(nrec)),
                           dim3(NTHREADS), 0, c->stream, c->f_cnt, c->f_lex,
                           c->f_goff, (const K *)c->keys[dst], c->vals[dst],
                           ntiles,
                           nrec, (K *)c->keys[src], c->vals[src]));
  c->stats.bytes_total += n * (sizeof(K) + 16) + n2 * 32 + nrec * 16;
  c->stats.sort_passes++; /* the fused pass counts as a pass */
  c->n = n2;
  c->cur = src;
  c->sorted = 1;   /* the fragment stream is key-sorted (d-major order) */
  if (n_out) *n_out = n2;
  /* the base reduce merges cross-tile fragments exactly */
  return segmented_reduce_sorted<K>(c, n_out);
}

/* the base segmented reduce over the (sorted) current dataset */
template <typename K>
static int segmented_reduce_sorted(dprk_ctx *c, uint64_t *n_out) {
  uint64_t n = c->n;
  uint64_t ntiles = ceil_div(n, RTILE);
  int src = c->cur, scr = c->cur ^ 1;
  int grid = tile_grid(ntiles);
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_reduce_tiles<K>, dim3(grid), dim3(NTHREADS), 0,
                           c->stream, (const K *)c->keys[src], c->vals[src], n,
                           (K *)c->keys[scr], c->vals[scr], c->tile_T,
                           c->tile_anyend, c->tile_nruns, ntiles));
  uint64_t nchunks = ceil_div(ntiles, (uint64_t)STITCH_CHUNK);
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_stitchA, dim3((uint32_t)nchunks), dim3(NTHREADS),
                           0, c->stream, c->tile_T, c->tile_anyend,
                           c->tile_nruns, ntiles, c->tile_C, c->chain_open,
                           c->tile_outoff, c->chunk_T, c->chunk_F,
                           c->chunk_N));
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_stitchB, dim3(1), dim3(NTHREADS), 0, c->stream,
                           c->chunk_T, c->chunk_F, c->chunk_N, nchunks,
                           c->d_scalar));
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_stitchC, dim3((uint32_t)nchunks), dim3(NTHREADS),
                           0, c->stream, c->tile_C, c->chain_open,
                           c->tile_outoff, ntiles, c->chunk_T, c->chunk_N));
  uint64_t total = 0;
  HIP_CHECK(hipMemcpyAsync(&total, c->d_scalar, 8, hipMemcpyDeviceToHost,
                           c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  TIMED(c, K_REDUCE,
        hipLaunchKernelGGL(k_compact_runs<K>, dim3(grid), dim3(NTHREADS), 0,
                           c->stream, (const K *)c->keys[scr], c->vals[scr],
                           c->tile_nruns, c->tile_outoff, c->tile_C, ntiles,
                           (K *)c->keys[src], c->vals[src]));
  c->stats.bytes_total += n * (sizeof(K) + 8) + total * (2 * sizeof(K) + 16);
  c->n = total;
  c->cur = src;
  c->sorted = 1;
  c->ghist_valid = 0;   /* dataset replaced */
  if (n_out) *n_out = total;
  return drain_timing(c);
}

extern "C" int dprk_reduce_by_key_sum(dprk_ctx *c, uint64_t *n_out) {
  if (!c) { set_err("null ctx"); return 1; }
  if (!c->with_vals) { set_err("reduce: ctx reserved keys-only"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->n == 0) { if (n_out) *n_out = 0; return 0; }
  /* Fused last pass (DPRK_FUSED_REDUCE=1, explicit opt-in only): pays
   * off on long-run (skewed) data — zipf config 5 measured 63.7 vs 65.7
   * ms/step — and loses ~9% on uniform data (the tiny-fragment d-major
   * transpose).  NOT auto-selected: history-dependent path choice would
   * make same-input results last-ulp dependent on ctx history, breaking
   * the run-to-run bit-identical contract the tests assert.  Both paths
   * are individually deterministic and oracle-exact. */
  if (!c->sorted && getenv("DPRK_FUSED_REDUCE")) {
    /* fused path: all but the last pass as normal sort passes, the last
     * fused with the segmented reduce (needs >=1 active pass).  This
     * opt-in path has no gen-fusion form — materialize lazy gen. */
    if (materialize_lazy(c)) return 1;
    uint64_t orand[2];
    if (c->orand_valid) {
      orand[0] = c->known_or;
      orand[1] = c->known_and;
    } else {
      uint64_t init[2] = {0, ~0ull};
      HIP_CHECK(hipMemcpyAsync(c->d_orand, init, 16, hipMemcpyHostToDevice,
                               c->stream));
      if (c->packed) {
        TIMED(c, K_HIST,
              hipLaunchKernelGGL(k_minmax_or<uint32_t>,
                                 dim3(stream_grid(c->n)),
                                 dim3(NTHREADS), 0, c->stream,
                                 (const uint32_t *)c->keys[c->cur],
                                 c->n, c->d_orand));
      } else {
        TIMED(c, K_HIST,
              hipLaunchKernelGGL(k_minmax_or<int64_t>,
                                 dim3(stream_grid(c->n)),
                                 dim3(NTHREADS), 0, c->stream,
                                 c->keys[c->cur],
                                 c->n, c->d_orand));
      }
      HIP_CHECK(hipMemcpyAsync(orand, c->d_orand, 16, hipMemcpyDeviceToHost,
                               c->stream));
      HIP_CHECK(hipStreamSynchronize(c->stream));
      c->stats.bytes_total += c->n * (c->packed ? 4 : 8);
      c->orand_valid = 1;
      c->known_or = orand[0];
      c->known_and = orand[1];
    }
    int passes[8], np = 0;
    for (int p = 0; p < 8; p++) {
      uint32_t ob = (uint32_t)(orand[0] >> (8 * p)) & 0xFF;
      uint32_t ab = (uint32_t)(orand[1] >> (8 * p)) & 0xFF;
      if (ob != ab) passes[np++] = p;
    }
    if (np >= 1) {
      c->stats.sort_passes = 0;
      for (int i = 0; i + 1 < np; i++) {
        if (radix_pass(c, 0, 8 * passes[i], 0xFF, 0)) return 1;
        c->stats.sort_passes++;
      }
      return c->packed
                 ? fused_reduce_last_pass<uint32_t>(c, 8 * passes[np - 1],
                                                    n_out)
                 : fused_reduce_last_pass<int64_t>(c, 8 * passes[np - 1],
                                                   n_out);
    }
    /* np == 0: all keys equal — trivial; fall through to the base path */
  }
  if (!c->sorted && dprk_sort_pairs(c)) return 1;
  return c->packed ? segmented_reduce_sorted<uint32_t>(c, n_out)
                   : segmented_reduce_sorted<int64_t>(c, n_out);
}

/* run extraction into ctx group arrays (assumes sorted) */
template <typename K>
static int extract_runs_t(dprk_ctx *c) {
  uint64_t n = c->n;
  uint64_t ntiles = ceil_div(n, RTILE);
  int src = c->cur, scr = c->cur ^ 1;
  int grid = tile_grid(ntiles);
  /* idx scratch reuses the alt VALS buffer as u64 (same 8B stride);
   * keys scratch uses the alt keys buffer */
  if (!c->with_vals) { set_err("groups need vals buffers reserved"); return 1; }
  TIMED(c, K_RUNEXT,
        hipLaunchKernelGGL(k_head_tiles<K>, dim3(grid), dim3(NTHREADS), 0,
                           c->stream, (const K *)c->keys[src], n,
                           (K *)c->keys[scr],
                           (uint64_t *)c->vals[scr], c->tile_nruns, ntiles));
  TIMED(c, K_RUNEXT,
        hipLaunchKernelGGL(k_scan_u32_single, dim3(1), dim3(NTHREADS), 0,
                           c->stream, c->tile_nruns, ntiles, c->tile_outoff,
                           c->d_scalar));
  uint64_t g = 0;
  HIP_CHECK(hipMemcpyAsync(&g, c->d_scalar, 8, hipMemcpyDeviceToHost,
                           c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  if (ensure_grp(c, g ? g : 1)) return 1;
  TIMED(c, K_RUNEXT,
        hipLaunchKernelGGL(k_compact_heads<K>, dim3(grid), dim3(NTHREADS), 0,
                           c->stream, (const K *)c->keys[scr],
                           (uint64_t *)c->vals[scr],
                           c->tile_nruns, c->tile_outoff, ntiles,
                           c->grp_keys, c->grp_starts));
  TIMED(c, K_RUNEXT,
        hipLaunchKernelGGL(k_run_counts, dim3(stream_grid(g)), dim3(NTHREADS),
                           0, c->stream, c->grp_starts, g, n, c->grp_counts));
  c->stats.bytes_total += n * sizeof(K) + g * 24;
  c->n_groups = g;
  return drain_timing(c);
}

static int extract_runs(dprk_ctx *c) {
  return c->packed ? extract_runs_t<uint32_t>(c) : extract_runs_t<int64_t>(c);
}

extern "C" int dprk_group_by_key(dprk_ctx *c, uint64_t *n_groups_out) {
  if (!c) { set_err("null ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->n == 0) { free_group_join(c); c->n_groups = 0;
                   if (n_groups_out) *n_groups_out = 0; return 0; }
  if (!c->sorted && dprk_sort_pairs(c)) return 1;
  if (extract_runs(c)) return 1;
  if (n_groups_out) *n_groups_out = c->n_groups;
  return 0;
}

extern "C" int dprk_get_groups(dprk_ctx *c, int64_t *unique_keys,
                               uint64_t *starts, uint64_t *counts,
                               uint64_t cap, uint64_t *n_out) {
  if (!c) { set_err("null ctx"); return 1; }
  *n_out = c->n_groups;
  if (c->n_groups > cap) { set_err("get_groups: cap too small"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->n_groups == 0) return 0;
  if (unique_keys)
    HIP_CHECK(hipMemcpyAsync(unique_keys, c->grp_keys, c->n_groups * 8,
                             hipMemcpyDeviceToHost, c->stream));
  if (starts)
    HIP_CHECK(hipMemcpyAsync(starts, c->grp_starts, c->n_groups * 8,
                             hipMemcpyDeviceToHost, c->stream));
  if (counts)
    HIP_CHECK(hipMemcpyAsync(counts, c->grp_counts, c->n_groups * 8,
                             hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  return 0;
}

extern "C" int dprk_join(dprk_ctx *a, dprk_ctx *b, uint64_t *n_out) {
  if (!a || !b) { set_err("null ctx"); return 1; }
  if (a->device != b->device) { set_err("join: ctxs on different devices");
                                return 1; }
  HIP_CHECK(hipSetDevice(a->device));
  if (a->n == 0 || b->n == 0) {
    free_group_join(a);
    a->join_n = 0;
    if (n_out) *n_out = 0;
    return 0;
  }
  /* sort + run-extract both sides (group layout lives per ctx) */
  if (!a->sorted && dprk_sort_pairs(a)) return 1;
  if (!b->sorted && dprk_sort_pairs(b)) return 1;
  if (extract_runs(a)) return 1;
  if (extract_runs(b)) return 1;
  uint64_t ga = a->n_groups, gb = b->n_groups;
  /* per-A-run product sizes + B match indices (capacity-cached) */
  if (ga > a->run_prod_cap || !a->run_prod) {
    if (a->run_prod) HIP_CHECK(hipFree(a->run_prod));
    if (a->run_boff) HIP_CHECK(hipFree(a->run_boff));
    if (a->d_bidx) HIP_CHECK(hipFree(a->d_bidx));
    uint64_t cap = ga + ga / 8 + 1;
    HIP_CHECK(hipMalloc(&a->run_prod, cap * 8));
    HIP_CHECK(hipMalloc(&a->run_boff, cap * 8));
    HIP_CHECK(hipMalloc(&a->d_bidx, cap * 8));
    a->run_prod_cap = cap;
  }
  uint64_t *d_bidx = a->d_bidx;
  static int join_mp = -1;
  if (join_mp < 0) join_mp = getenv("DPRK_JOIN_MP") ? 1 : 0;
  if (join_mp) {
    TIMED(a, K_JOIN,
          hipLaunchKernelGGL(k_join_count_mp,
                             dim3(stream_grid(ga / JOIN_WCHUNK + 1)),
                             dim3(NTHREADS),
                             0, a->stream, a->grp_keys, a->grp_counts, ga,
                             b->grp_keys, b->grp_counts, gb, a->run_prod,
                             d_bidx));
  } else {
    TIMED(a, K_JOIN,
          hipLaunchKernelGGL(k_join_count, dim3(stream_grid(ga)),
                             dim3(NTHREADS),
                             0, a->stream, a->grp_keys, a->grp_counts, ga,
                             b->grp_keys, b->grp_counts, gb, a->run_prod,
                             d_bidx));
  }
  /* exclusive scan of products -> output offsets (run_boff := scan copy) */
  HIP_CHECK(hipMemcpyAsync(a->run_boff, a->run_prod, ga * 8,
                           hipMemcpyDeviceToDevice, a->stream));
  uint64_t total = 0;
  if (run_scan_u64(a, a->run_boff, ga, &total)) return 1;
  /* output buffers, capacity-cached across calls */
  if (total > a->join_cap || !a->join_k) {
    if (a->join_k) HIP_CHECK(hipFree(a->join_k));
    if (a->join_a) HIP_CHECK(hipFree(a->join_a));
    if (a->join_b) HIP_CHECK(hipFree(a->join_b));
    uint64_t cap = total + total / 16 + 1;
    HIP_CHECK(hipMalloc(&a->join_k, cap * 8));
    HIP_CHECK(hipMalloc(&a->join_a, cap * 8));
    HIP_CHECK(hipMalloc(&a->join_b, cap * 8));
    a->join_cap = cap;
  }
  {
    static int glanes = -1;
    if (glanes < 0) {
      const char *e = getenv("DPRK_JOIN_GLANES");
      glanes = e ? atoi(e) : 16;
      if (glanes != 16 && glanes != 32 && glanes != 64) glanes = 16;
    }
    if (glanes == 64) {
      TIMED(a, K_JOIN,
            hipLaunchKernelGGL(k_join_emit<64>, dim3(2048), dim3(NTHREADS),
                               0,
                               a->stream, a->grp_keys, a->grp_starts,
                               ga, b->grp_starts, b->grp_counts,
                               a->run_boff, a->run_prod, d_bidx,
                               a->vals[a->cur], b->vals[b->cur], a->join_k,
                               a->join_a, a->join_b));
    } else if (glanes == 32) {
      TIMED(a, K_JOIN,
            hipLaunchKernelGGL(k_join_emit<32>, dim3(2048), dim3(NTHREADS),
                               0,
                               a->stream, a->grp_keys, a->grp_starts,
                               ga, b->grp_starts, b->grp_counts,
                               a->run_boff, a->run_prod, d_bidx,
                               a->vals[a->cur], b->vals[b->cur], a->join_k,
                               a->join_a, a->join_b));
    } else {
      TIMED(a, K_JOIN,
            hipLaunchKernelGGL(k_join_emit<16>, dim3(2048), dim3(NTHREADS),
                               0,
                               a->stream, a->grp_keys, a->grp_starts,
                               ga, b->grp_starts, b->grp_counts,
                               a->run_boff, a->run_prod, d_bidx,
                               a->vals[a->cur], b->vals[b->cur], a->join_k,
                               a->join_a, a->join_b));
    }
  }
  a->stats.bytes_total += total * 24 + ga * 32 + gb * 16;
  int rc = drain_timing(a);
  (void)d_bidx;
  a->join_n = total;
  if (n_out) *n_out = total;
  return rc;
}

extern "C" int dprk_get_join(dprk_ctx *c, int64_t *keys, double *va,
                             double *vb, uint64_t cap, uint64_t *n_out) {
  if (!c) { set_err("null ctx"); return 1; }
  *n_out = c->join_n;
  if (c->join_n > cap) { set_err("get_join: cap too small"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  if (c->join_n == 0) return 0;
  if (keys)
    HIP_CHECK(hipMemcpyAsync(keys, c->join_k, c->join_n * 8,
                             hipMemcpyDeviceToHost, c->stream));
  if (va)
    HIP_CHECK(hipMemcpyAsync(va, c->join_a, c->join_n * 8,
                             hipMemcpyDeviceToHost, c->stream));
  if (vb)
    HIP_CHECK(hipMemcpyAsync(vb, c->join_b, c->join_n * 8,
                             hipMemcpyDeviceToHost, c->stream));
  HIP_CHECK(hipStreamSynchronize(c->stream));
  return 0;
}

/* ---- multi-GPU exchange ------------------------------------------- */
extern "C" int dprk_comm_unique_id(char uid[128]) {
  static_assert(sizeof(ncclUniqueId) == 128, "ncclUniqueId size");
  NCCL_CHECK(ncclGetUniqueId((ncclUniqueId *)uid));
  return 0;
}

extern "C" int dprk_comm_init(dprk_ctx *c, int nranks, int rank,
                              const char uid[128]) {
  if (!c) { set_err("null ctx"); return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  ncclUniqueId id;
  memcpy(&id, uid, sizeof(id));
  NCCL_CHECK(ncclCommInitRank(&c->comm, nranks, id, rank));
  c->nranks = nranks;
  c->rank = rank;
  return 0;
}

/* One grouped send/recv all-to-allv of the rank-partitioned dataset —
 * the ShuffleFetcher replacement (pairwise-direct over the
 * 7 xGMI p2p links; never a ring for this exchange). */
extern "C" int dprk_exchange(dprk_ctx *c, const uint64_t *send_counts,
                             const uint64_t *recv_counts, int nranks) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  /* N=1 is a no-op — unless DPRK_FORCE_SELF_EXCHANGE is set, which runs
   * the real grouped ncclSend/ncclRecv self path so the RCCL exchange is
   * exercised on the single-GPU test box. */
  if (nranks == 1 && !getenv("DPRK_FORCE_SELF_EXCHANGE")) return 0;
  if (!c->comm) { set_err("exchange: comm not initialized"); return 1; }
  if (nranks != c->nranks) { set_err("exchange: nranks mismatch"); return 1; }
  if (nranks < 1 || nranks > 256) {
    set_err("exchange: nranks %d out of range [1,256]", nranks);
    return 1;
  }
  HIP_CHECK(hipSetDevice(c->device));
  uint64_t send_off[256], recv_off[256], total_recv = 0, total_send = 0;
  for (int r = 0; r < nranks; r++) {
    send_off[r] = total_send;
    recv_off[r] = total_recv;
    total_send += send_counts[r];
    total_recv += recv_counts[r];
  }
  if (total_send != c->n) { set_err("exchange: send counts != n"); return 1; }
  if (total_recv > c->cap) { set_err("exchange: recv %llu > cap",
                                     (unsigned long long)total_recv);
                             return 1; }
  int src = c->cur, dst = c->cur ^ 1;
  /* packed keys travel as u32 over xGMI (half the key bytes); every rank
   * must agree on the width — the host layer aligns packed state before
   * calling (Engine.sync_packed / bench exchange paths) */
  ncclDataType_t kdt = c->packed ? ncclUint32 : ncclInt64;
  size_t ksz = c->packed ? 4 : 8;
  int began = timed_begin(c, K_EXCHANGE);
  NCCL_CHECK(ncclGroupStart());
  for (int r = 0; r < nranks; r++) {
    if (send_counts[r])
      NCCL_CHECK(ncclSend((char *)c->keys[src] + send_off[r] * ksz,
                          send_counts[r], kdt, r, c->comm, c->stream));
    if (recv_counts[r])
      NCCL_CHECK(ncclRecv((char *)c->keys[dst] + recv_off[r] * ksz,
                          recv_counts[r], kdt, r, c->comm, c->stream));
    if (c->with_vals) {
      if (send_counts[r])
        NCCL_CHECK(ncclSend(c->vals[src] + send_off[r], send_counts[r],
                            ncclDouble, r, c->comm, c->stream));
      if (recv_counts[r])
        NCCL_CHECK(ncclRecv(c->vals[dst] + recv_off[r], recv_counts[r],
                            ncclDouble, r, c->comm, c->stream));
    }
  }
  NCCL_CHECK(ncclGroupEnd());
  timed_end(c, began);
  c->stats.bytes_total +=
      (total_send + total_recv) * (ksz + (c->with_vals ? 8 : 0));
  c->cur = dst;
  c->n = total_recv;
  c->sorted = 0;
  c->orand_valid = 0;
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  return drain_timing(c);
}

/* Bucket-granular all-to-allv: the dataset laid out by
 * dprk_partition(ctx, nbuckets) is exchanged under an arbitrary
 * bucket->rank assignment — one grouped ncclSend/Recv per nonempty
 * bucket, both sides iterating buckets ascending so point-to-point
 * pairs match.  With the count matrix all-gathered on the host, a
 * deterministic greedy assignment balances skewed buckets across ranks
 * (SURVEY.md §8(e) hot buckets); each bucket stays whole, so reduceByKey
 * needs no second combine.  Residual limit: a single bucket larger than
 * a rank's capacity still fails (clearly) — splitting ONE key's partial
 * sums across ranks is future work.
 * recv_matrix[s*nbuckets + b] = elements source rank s sends this rank
 * from bucket b (zero for buckets not assigned here). */
extern "C" int dprk_exchange_buckets(dprk_ctx *c,
                                     const uint64_t *bucket_counts,
                                     const int *bucket_rank, int nbuckets,
                                     const uint64_t *recv_matrix,
                                     int nranks) {
  if (!c) { set_err("null ctx"); return 1; }
  if (materialize_lazy(c)) return 1;
  if (!c->comm) { set_err("exchange_buckets: comm not initialized");
                  return 1; }
  if (nranks != c->nranks) { set_err("exchange_buckets: nranks mismatch");
                             return 1; }
  HIP_CHECK(hipSetDevice(c->device));
  uint64_t total_send = 0, total_recv = 0;
  for (int b = 0; b < nbuckets; b++) total_send += bucket_counts[b];
  for (int s = 0; s < nranks; s++)
    for (int b = 0; b < nbuckets; b++)
      total_recv += recv_matrix[(uint64_t)s * nbuckets + b];
  if (total_send != c->n) { set_err("exchange_buckets: counts != n");
                            return 1; }
  if (total_recv > c->cap) { set_err("exchange_buckets: recv %llu > cap %llu"
                                     " (skew beyond balancing: a single "
                                     "key exceeds a rank's share)",
                                     (unsigned long long)total_recv,
                                     (unsigned long long)c->cap);
                             return 1; }
  int src = c->cur, dst = c->cur ^ 1;
  ncclDataType_t kdt = c->packed ? ncclUint32 : ncclInt64;
  size_t ksz = c->packed ? 4 : 8;
  int began = timed_begin(c, K_EXCHANGE);
  NCCL_CHECK(ncclGroupStart());
  uint64_t soff = 0;
  for (int b = 0; b < nbuckets; b++) {
    uint64_t cnt = bucket_counts[b];
    if (cnt) {
      int peer = bucket_rank[b];
      NCCL_CHECK(ncclSend((char *)c->keys[src] + soff * ksz, cnt, kdt, peer,
                          c->comm, c->stream));
      if (c->with_vals)
        NCCL_CHECK(ncclSend(c->vals[src] + soff, cnt, ncclDouble, peer,
                            c->comm, c->stream));
    }
    soff += cnt;
  }
  /* recvs: src-major, ascending bucket — matches each sender's ascending
   * bucket order of sends to us */
  uint64_t roff = 0;
  for (int s = 0; s < nranks; s++) {
    for (int b = 0; b < nbuckets; b++) {
      uint64_t cnt = recv_matrix[(uint64_t)s * nbuckets + b];
      if (cnt) {
        NCCL_CHECK(ncclRecv((char *)c->keys[dst] + roff * ksz, cnt, kdt, s,
                            c->comm, c->stream));
        if (c->with_vals)
          NCCL_CHECK(ncclRecv(c->vals[dst] + roff, cnt, ncclDouble, s,
                              c->comm, c->stream));
        roff += cnt;
      }
    }
  }
  NCCL_CHECK(ncclGroupEnd());
  timed_end(c, began);
  c->stats.bytes_total +=
      (total_send + total_recv) * (ksz + (c->with_vals ? 8 : 0));
  c->cur = dst;
  c->n = total_recv;
  c->sorted = 0;
  c->orand_valid = 0;
  c->hist_shift_valid = -1;
  c->ghist_valid = 0;
  return drain_timing(c);
}

/* ---- stats --------------------------------------------------------- */
extern "C" int dprk_stats(dprk_ctx *c, dprk_stats_t *out) {
  if (!c || !out) { set_err("null arg"); return 1; }
  *out = c->stats;
  return 0;
}

extern "C" int dprk_stats_reset(dprk_ctx *c) {
  if (!c) { set_err("null ctx"); return 1; }
  memset(&c->stats, 0, sizeof(c->stats));
  return 0;
}
