// kernels_agg3.hip — v3 partition scatter for the two-phase aggregation:
// LDS-staged packet flushes into 64B-aligned per-(block,bucket) ranges.
//
// Round-1/2 PMC evidence (profiles/r01_final_pmc.md, r2chain stats): the
// per-record scatter wrote partial cache lines across ~512K open regions and
// paid ~2x write-back amplification — 30 ms of the 41 ms step at 1.35 TB/s
// effective. Here every (block,bucket) range starts 64B-aligned (the hist
// scan pads sizes to whole lines) and records are staged in LDS until a full
// 8-record packet (8 x 24 B = 192 B = 3 exact lines) can be written, so the
// L2 never holds a partially written line between flushes. Layout per block:
//   staging[512 buckets][12 records][24 B]  = 147 KB LDS
//   cnt/fl counters                          = 4 KB
// (one 1024-thread workgroup per CU; the scatter is bandwidth-bound, not
// latency-bound, so 16 waves suffice — measured round 2.)
//
// The bucket aggregation kernel widens its LDS hash window to 4096 slots
// (512 buckets x ~2000 distinct keys/bucket at the 1M-group north star needs
// >2048) at 24 B/slot SoA = 96 KB -> one 1024-thread workgroup per CU.
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "kernels.h"

namespace auron {

namespace {
inline void check_launch3(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("kernel launch failed: ") + name +
                             ": " + hipGetErrorString(e));
}
constexpr int64_t KEY_EMPTY3 = INT64_MIN;

__device__ __forceinline__ uint64_t mix64_3(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__device__ __forceinline__ bool bit_get3(const uint8_t* bm, int64_t i) {
  return (bm[i >> 3] >> (i & 7)) & 1;
}

__device__ __forceinline__ void sum_accum3(double* acc, double v,
                                           bool is_int) {
  if (is_int) {
    uint64_t b;
    memcpy(&b, &v, 8);
    atomicAdd(reinterpret_cast<unsigned long long*>(acc),
              (unsigned long long)b);
  } else {
    unsafeAtomicAdd(acc, v);
  }
}
}  // namespace

// ---- byte-line offsets: pad each (block,bucket) range to whole 64B lines --
// line_sizes[i] = ceil(counts[i] * 24 / 64): the exclusive scan of THIS
// matrix (in 64-byte line units, so 6 GB chunks still fit u32) gives every
// range a line-aligned start.
__global__ void k_agg3_line_sizes(const uint32_t* __restrict__ counts,
                                  int64_t n, uint32_t rec,
                                  uint32_t* __restrict__ sizes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    sizes[i] = (counts[i] * rec + 63u) >> 6;
}

void launch_agg3_line_sizes(const uint32_t* counts, int64_t n,
                            uint32_t* sizes, int rec, hipStream_t s) {
  hipLaunchKernelGGL(k_agg3_line_sizes, dim3(256), dim3(256), 0, s, counts, n,
                     (uint32_t)rec, sizes);
  check_launch3("k_agg3_line_sizes");
}

// ---- v3 scatter ------------------------------------------------------------
static constexpr int A3_CAP = 12;      // staged records per bucket (24B)
static constexpr int A3_CAP16 = 15;    // 16B records: more slots in less
                                       // LDS (512*15*16 = 123KB). HARD CAP:
                                       // pos must stay < 16 so a bucket
                                       // crosses the 8-record packet
                                       // boundary at most ONCE per tile —
                                       // two queue entries for one bucket
                                       // would race two flush subgroups
                                       // (observed as corrupted group keys
                                       // at CAP 18)
static constexpr int A3_QUANT = 8;     // records per flushed packet (192 B)
// staging slot stride: 24 B packed (a 28 B bank-spread pad was measured
// SLOWER — 7.71 vs 6.58 ms/chunk — the kernel is flush-latency-bound, not
// LDS-bank-bound)
static constexpr int A3_SLOT = 24;
// a thread that cannot stage its row after a few tiles (hot bucket under
// key skew: drain is bounded by CAP per tile) bypasses to the leftover
// list; the single-phase leftover kernel handles it — correct and bounded
// for ANY distribution, perf-relevant only under adversarial skew
static constexpr int A3_RETRY_BYPASS = 4;

// A3_RPT rows per thread per tile; a dirty QUEUE records buckets that
// crossed a packet boundary so the flush phase touches only those (the v3.0
// per-tile scan of every bucket made the kernel instruction-bound: PMC
// showed 20x the dynamic instructions of v2 and 77% parked waves while HBM
// writes ran at 1.0x algorithmic — gpurun_out/r2_pmc_scatter.json).
static constexpr int A3_RPT = 2;  // default; AURON_AGG2_RPT=4 selects the
                                  // wider tile instantiation

// NORMAL-key min/max probe (order-mapped u64): lets the engine choose the
// packed-16B record layout BEFORE the first chunk's scatter (the probe is
// one 8 B/row read, ~0.35 ms per 256M rows — the 24B chunk it replaces
// costs ~2.8 ms more than a packed one).
__global__ void k_keys_minmax(const int64_t* __restrict__ keys,
                              const uint8_t* __restrict__ key_valid,
                              int64_t n,
                              unsigned long long* __restrict__ kminmax) {
  unsigned long long kmin = ~0ull, kmax = 0ull;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (key_valid && !bit_get3(key_valid, i)) continue;
    int64_t k = keys[i];
    if (k == KEY_EMPTY3) continue;
    unsigned long long m = (unsigned long long)k ^ 0x8000000000000000ull;
    kmin = m < kmin ? m : kmin;
    kmax = m > kmax ? m : kmax;
  }
  if (kmin != ~0ull) {
    atomicMin(&kminmax[0], kmin);
    atomicMax(&kminmax[1], kmax);
  }
}

void launch_keys_minmax(const int64_t* keys, const uint8_t* key_valid,
                        int64_t n, unsigned long long* kminmax,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_keys_minmax, dim3(512), dim3(256), 0, s, keys,
                     key_valid, n, kminmax);
  check_launch3("k_keys_minmax");
}

// REC = 24: [i64 key][f64 val][u32 rowv] (pad to 24). REC = 16: the key is
// stored as a u32 OFFSET from key_base packed with rowv into one u64 —
// [u64 koff|rowv<<32][f64 val] — 33% less scatter/bucket HBM traffic. Rows
// whose key falls outside [key_base, key_base+2^32) take the leftover
// bypass (counted in the correction matrix), so the packed path is exact
// for ANY input; the engine only selects it after observing chunk-1's key
// range (hist min/max).
template <int A3_RPT, int REC>
__global__ void __launch_bounds__(1024) k_agg3_scatter(
    const int64_t* __restrict__ keys, const uint8_t* __restrict__ key_valid,
    const double* __restrict__ vals, const uint8_t* __restrict__ val_valid,
    int64_t n, int nbuck_log2, int grid_log2, int64_t key_base,
    const uint32_t* __restrict__ line_scan, uint8_t* __restrict__ out,
    PartRow* __restrict__ leftover, unsigned long long* __restrict__ lo_n,
    uint32_t* __restrict__ bypass_matrix, uint32_t* __restrict__ err_flag) {
  const uint32_t nbuck = 1u << nbuck_log2;
  extern __shared__ uint8_t lds[];
  // layout: records [nbuck][A3_CAP][24] | cnt[nbuck] | fl[nbuck] |
  //         base_line[nbuck] | dirty queue [nbuck] u16-as-u32 | qn
  uint8_t* stage = lds;
  constexpr int CAP = REC == 16 ? A3_CAP16 : A3_CAP;
  uint32_t* cnt = (uint32_t*)(lds + (size_t)nbuck * CAP * REC);
  uint32_t* fl = cnt + nbuck;
  uint32_t* base_line = fl + nbuck;
  uint16_t* queue = (uint16_t*)(base_line + nbuck);
  uint32_t* qn = (uint32_t*)(queue + nbuck + 64);
  uint32_t* byp = qn + 1;  // rows bypassed to the leftover list, per bucket

  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x) {
    cnt[b] = 0;
    fl[b] = 0;
    byp[b] = 0;
    base_line[b] = line_scan[((size_t)b << grid_log2) | blockIdx.x];
  }
  if (threadIdx.x == 0) *qn = 0;
  __syncthreads();

  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int nwave = (int)(blockDim.x >> 6);
  const int wave = (int)(threadIdx.x >> 6);
  const int lane = (int)(threadIdx.x & 63);
  int64_t my[A3_RPT];
  int tries[A3_RPT] = {};
  int64_t k_cur[A3_RPT];
  double v_cur[A3_RPT];
#pragma unroll
  for (int r = 0; r < A3_RPT; r++) {
    my[r] = (int64_t)blockIdx.x * blockDim.x + threadIdx.x +
            (int64_t)r * stride;
    if (my[r] < n) {  // preload; refreshed right after each advance so the
      k_cur[r] = keys[my[r]];     // next tile's loads hide behind the flush
      v_cur[r] = vals[my[r]];
    }
  }
  const int64_t step = (int64_t)A3_RPT * stride;
  // hard tile bound: rows/thread plus a generous retry allowance — a full
  // staging bucket retries a row for a few tiles at most; exceeding this
  // means a logic bug, and the kernel must FAIL, never hang the box
  const int64_t max_tiles = 8 * ((n + step - 1) / step) + 256;
  int64_t tile = 0;

  for (;;) {
    if (++tile > max_tiles) {
      atomicOr(err_flag, 32u);
      break;
    }
    int mine = 0;
#pragma unroll
    for (int r = 0; r < A3_RPT; r++) {
      if (my[r] >= n) continue;
      mine = 1;
      int64_t k = k_cur[r];
      bool advanced = false;
      bool knull = key_valid && !bit_get3(key_valid, my[r]);
      if (knull || k == KEY_EMPTY3) {
        my[r] += step;  // specials handled by k_agg2_specials
        mine |= (my[r] < n);
        advanced = true;
      } else {
        uint32_t b = (uint32_t)(mix64_3((uint64_t)k) >> (64 - nbuck_log2));
        bool range_ok = true;
        uint64_t koff = 0;
        if (REC == 16) {
          koff = (uint64_t)(k - key_base);
          range_ok = koff <= 0xFFFFFFFFull;
        }
        if (!range_ok) {
          // outside the packed key window: leftover bypass (exactness for
          // any input; the hist counted this row, so correct the matrix)
          bool vvalid = !val_valid || bit_get3(val_valid, my[r]);
          unsigned long long p = atomicAdd(lo_n, 1ull);
          leftover[p] = PartRow{
              k, v_cur[r], (uint32_t)my[r] | (vvalid ? 0x80000000u : 0u), 0};
          atomicAdd(&byp[b], 1u);
          my[r] += step;
          tries[r] = 0;
          advanced = true;
          if (advanced && my[r] < n) {
            k_cur[r] = keys[my[r]];
            v_cur[r] = vals[my[r]];
          }
          continue;
        }
        uint32_t pos = atomicAdd(&cnt[b], 1u);
        if (pos < CAP) {
          bool vvalid = !val_valid || bit_get3(val_valid, my[r]);
          uint8_t* rec = stage + ((size_t)b * CAP + pos) * REC;
          uint32_t rowv = (uint32_t)my[r] | (vvalid ? 0x80000000u : 0u);
          if (REC == 16) {
            *(uint64_t*)rec = koff | ((uint64_t)rowv << 32);
            *(double*)(rec + 8) = v_cur[r];
          } else {
            *(int64_t*)rec = k;
            *(double*)(rec + 8) = v_cur[r];
            *(uint32_t*)(rec + 16) = rowv;
          }
          if ((pos & (A3_QUANT - 1)) == A3_QUANT - 1) {  // crossed a packet
            uint32_t qi = atomicAdd(qn, 1u);
            if (qi < nbuck + 64) queue[qi] = (uint16_t)b;  // <=1 push per
          }                                                // bucket per tile
          my[r] += step;
          tries[r] = 0;
          advanced = true;
        } else {
          atomicSub(&cnt[b], 1u);
          if (++tries[r] >= A3_RETRY_BYPASS) {
            // hot bucket (skew): route the row to the leftover list; the
            // histogram already counted it, so record the correction the
            // bucket kernel must subtract from this (block,bucket) range
            bool vvalid = !val_valid || bit_get3(val_valid, my[r]);
            unsigned long long p = atomicAdd(lo_n, 1ull);
            leftover[p] = PartRow{
                k, v_cur[r],
                (uint32_t)my[r] | (vvalid ? 0x80000000u : 0u), 0};
            atomicAdd(&byp[b], 1u);
            my[r] += step;
            tries[r] = 0;
            advanced = true;
          }  // else: retry next tile (k_cur/v_cur stay valid)
        }
      }
      if (advanced && my[r] < n) {
        k_cur[r] = keys[my[r]];
        v_cur[r] = vals[my[r]];
      }
    }
    // block-wide termination check doubles as the pre-flush barrier
    int live = __syncthreads_count(mine);
    uint32_t nq = *qn;
    if (nq > nbuck + 64) nq = nbuck + 64;
    if (live == 0) {
      // drain: flush every bucket's remainder as (possibly partial) packets
      for (uint32_t b = wave; b < nbuck; b += nwave) {
        uint32_t c = cnt[b];
        if (!c) continue;
        uint8_t* dst = out + ((size_t)base_line[b] << 6) + (size_t)fl[b] * REC;
        const uint8_t* src = stage + (size_t)b * CAP * REC;
        for (uint32_t d = lane; d < c * (REC / 4); d += 64)
          ((uint32_t*)dst)[d] = ((const uint32_t*)src)[d];
      }
      __syncthreads();
      break;
    }
    // flush only queued (full-packet) buckets. Each entry is handled by an
    // 8-LANE subgroup (8 independent entries in flight per wave) — a
    // whole-wave-per-entry loop was a serial ~10-LDS-op latency chain per
    // entry and dominated the kernel.
    {
      const int sg = lane >> 3;        // subgroup 0..7 within the wave
      const int sl = lane & 7;         // lane within subgroup
      for (uint32_t i = wave * 8 + sg; i < nq; i += nwave * 8) {
        uint32_t b = queue[i];
        uint32_t c = cnt[b];
        uint32_t nfl = c & ~(uint32_t)(A3_QUANT - 1);
        if (!nfl) continue;
        uint8_t* dst = out + ((size_t)base_line[b] << 6) + (size_t)fl[b] * REC;
        uint8_t* src = stage + (size_t)b * CAP * REC;
        for (uint32_t d = sl; d < nfl * (REC / 4); d += 8)
          ((uint32_t*)dst)[d] = ((const uint32_t*)src)[d];
        uint32_t rem = c - nfl;
        for (uint32_t d = sl; d < rem * (REC / 4); d += 8)
          ((uint32_t*)src)[d] =
              ((const uint32_t*)(src + (size_t)nfl * REC))[d];
        if (sl == 0) {
          fl[b] += nfl;
          cnt[b] = rem;
        }
      }
    }
    // queue ENTRIES below nq are no longer needed; resetting the counter
    // concurrently with other waves' reads of those entries is benign
    if (threadIdx.x == 0) *qn = 0;
    __syncthreads();
  }
  // every (bucket, block) entry is written by its own block — on EVERY exit
  // path (incl. the tile-bound failure), so the bucket kernel never reads a
  // stale matrix
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x)
    bypass_matrix[((size_t)b << grid_log2) | blockIdx.x] = byp[b];
}

void launch_agg3_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* line_scan, uint8_t* out,
                         PartRow* leftover, unsigned long long* lo_n,
                         uint32_t* bypass_matrix, uint32_t* err_flag,
                         int rec, int64_t key_base, hipStream_t s) {
  const uint32_t nbuck = 1u << nbuck_log2;
  const int cap = rec == 16 ? A3_CAP16 : A3_CAP;
  size_t lds = (size_t)nbuck * cap * (size_t)rec + (size_t)nbuck * 16 +
               ((size_t)nbuck + 64) * 2 + 8;  // + queue + counters + bypass
  if (lds > 160 * 1024)
    throw std::runtime_error("agg3 scatter LDS over 160KB");
  static int rpt = [] {
    const char* e = getenv("AURON_AGG2_RPT");
    int v = e ? atoi(e) : 2;
    return (v >= 2 && v <= 4) ? v : 2;
  }();
  const void* fn;
  if (rec == 16)
    fn = rpt == 4   ? (const void*)k_agg3_scatter<4, 16>
         : rpt == 3 ? (const void*)k_agg3_scatter<3, 16>
                    : (const void*)k_agg3_scatter<2, 16>;
  else
    fn = rpt == 4   ? (const void*)k_agg3_scatter<4, 24>
         : rpt == 3 ? (const void*)k_agg3_scatter<3, 24>
                    : (const void*)k_agg3_scatter<2, 24>;
  hipError_t e = hipFuncSetAttribute(
      fn, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
  if (e != hipSuccess)
    throw std::runtime_error("agg3 scatter LDS attribute failed");
#define A3_LAUNCH(R, P)                                                     \
  hipLaunchKernelGGL((k_agg3_scatter<R, P>), dim3(1 << grid_log2),          \
                     dim3(1024), lds, s, keys, key_valid, vals, val_valid,  \
                     n, nbuck_log2, grid_log2, key_base, line_scan, out,    \
                     leftover, lo_n, bypass_matrix, err_flag)
  if (rec == 16) {
    if (rpt == 4) A3_LAUNCH(4, 16);
    else if (rpt == 3) A3_LAUNCH(3, 16);
    else A3_LAUNCH(2, 16);
  } else {
    if (rpt == 4) A3_LAUNCH(4, 24);
    else if (rpt == 3) A3_LAUNCH(3, 24);
    else A3_LAUNCH(2, 24);
  }
#undef A3_LAUNCH
  check_launch3("k_agg3_scatter");
}

// ---- v4 scatter: barrier-free producer/flusher rings -----------------------
// The v3 tile loop parks waves 73% of the time on its per-tile barriers
// (SQ_WAIT_ANY ~7x ACTIVE, gpurun_out sq pass). v4 removes every barrier
// from the hot loop: WORKER waves (AURON_AGG2_V4_WW, default 8 of 16)
// append records into per-bucket 8-slot LDS rings; the remaining FLUSHER
// waves (8-lane subgroups, 8 buckets in flight each)
// continuously drain 4-record 96 B quanta into the same 64B-aligned
// per-(block,bucket) ranges. Commit protocol: each ring slot carries a
// sequence TAG (= the absolute ring position) written AFTER the record by
// the same lane — LDS executes one wave's DS ops in order — so the flusher
// flushes exactly the contiguous committed prefix, tolerating out-of-order
// commits across workers. Hot buckets bypass to the leftover list BEFORE
// reserving (the reservation sequence must stay dense), with the same
// per-(block,bucket) correction matrix as v3.
static constexpr int A4_RING = 8;   // ring slots per bucket (pow2)
static constexpr int A4_QUANT = 8;  // records per flush quantum (192 B —
                                    // one full ring; halves flusher visits
                                    // per record vs the 96 B quantum)

#define A4_LD_RLX(p) \
  __hip_atomic_load((p), __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP)
#define A4_LD_ACQ(p) \
  __hip_atomic_load((p), __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_WORKGROUP)
#define A4_ST_REL(p, v) \
  __hip_atomic_store((p), (v), __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_WORKGROUP)
// LDS-only release: wait the wave's outstanding DS ops, then a relaxed
// store. A generic workgroup-scope RELEASE also emits s_waitcnt vmcnt(0),
// which stalls ~800 cycles for in-flight HBM stores/prefetches that are
// irrelevant to LDS ring reuse — measured as the v4 flusher bottleneck.
#define A4_ST_REL_LDS(p, v)                        \
  do {                                             \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory"); \
    __hip_atomic_store((p), (v), __ATOMIC_RELAXED, \
                       __HIP_MEMORY_SCOPE_WORKGROUP); \
  } while (0)

__global__ void __launch_bounds__(1024) k_agg4_scatter(
    const int64_t* __restrict__ keys, const uint8_t* __restrict__ key_valid,
    const double* __restrict__ vals, const uint8_t* __restrict__ val_valid,
    int64_t n, int nbuck_log2, int grid_log2, int worker_waves,
    const uint32_t* __restrict__ line_scan, uint8_t* __restrict__ out,
    PartRow* __restrict__ leftover, unsigned long long* __restrict__ lo_n,
    uint32_t* __restrict__ bypass_matrix, uint32_t* __restrict__ err_flag) {
  const uint32_t nbuck = 1u << nbuck_log2;
  extern __shared__ uint8_t lds[];
  // layout: ring[nbuck][A4_RING][24] | tag[nbuck][A4_RING] u32 |
  //         cnt[nbuck] | fl[nbuck] | base_line[nbuck] | byp[nbuck] | done
  uint8_t* ring = lds;
  uint32_t* tag = (uint32_t*)(lds + (size_t)nbuck * A4_RING * 24);
  uint32_t* cnt = tag + (size_t)nbuck * A4_RING;
  uint32_t* fl = cnt + nbuck;
  uint32_t* base_line = fl + nbuck;
  uint32_t* byp = base_line + nbuck;
  uint32_t* done = byp + nbuck;

  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x) {
    cnt[b] = 0;
    fl[b] = 0;
    byp[b] = 0;
    base_line[b] = line_scan[((size_t)b << grid_log2) | blockIdx.x];
  }
  for (uint32_t i = threadIdx.x; i < nbuck * A4_RING; i += blockDim.x)
    tag[i] = 0xFFFFFFFFu;  // != any pos; never reset (successive uses of a
                           // slot carry distinct pos values: pos = slot mod 8)
  if (threadIdx.x == 0) *done = 0;
  __syncthreads();

  const int wave = (int)(threadIdx.x >> 6);
  const int lane = (int)(threadIdx.x & 63);
  const int nwave = (int)(blockDim.x >> 6);

  if (wave < worker_waves) {
    // ---- worker: append rows; no barriers. VIRTUAL-LANE row mapping: rows
    // are credited to blocks by the 1024-thread histogram stride, so worker
    // thread t covers virtual lanes {t, t+NW, ...} < 1024 of each tile —
    // the hist keeps its full 1024-thread launch (384-thread hist measured
    // 2.2x slower), and each worker carries 2-3 rows per tile of ILP.
    //
    // A lane that reserved a ring position it cannot yet write (backlog)
    // must NOT spin in an inner loop: divergent serialization would park
    // its sibling lanes, including ones holding the very positions the
    // flusher needs next — an intra-wave deadlock (measured: spin-bound
    // trips under skew/bursts). Reservations it cannot complete go into a
    // small per-lane pending FIFO retried once per pass (leftover-bypass
    // when full), so every lane makes bounded attempts and siblings always
    // progress.
    const int NW = worker_waves * 64;
    const int tid = (int)threadIdx.x;
    const int64_t tile_stride = (int64_t)gridDim.x << 10;
    constexpr int MAXSUB = 3;   // ceil(1024 / (6 waves * 64))
    constexpr int PCAP = 4;     // pending FIFO slots (pow2)
    int64_t rk[MAXSUB];
    double rv[MAXSUB];
    uint32_t rvalid[MAXSUB];    // bit0: in range+normal key; bit1: val valid
    int64_t rrow[MAXSUB];
    uint32_t p_b[PCAP], p_pos[PCAP], p_rowv[PCAP];
    int64_t p_k[PCAP];
    double p_v[PCAP];
    uint32_t p_hd = 0, p_tl = 0;
    int p_spin = 0;
    int64_t tb = (int64_t)blockIdx.x << 10;
    int nr = 0;
    auto load_tile = [&](int64_t base) {
      nr = 0;
      if (base >= n) return;
#pragma unroll
      for (int s = 0; s < MAXSUB; s++) {
        int vt = tid + s * NW;
        int64_t i2 = base + vt;
        if (vt < 1024 && i2 < n) {
          rk[nr] = keys[i2];
          rv[nr] = vals[i2];
          bool knull = key_valid && !bit_get3(key_valid, i2);
          bool vvalid = !val_valid || bit_get3(val_valid, i2);
          rvalid[nr] = (!knull && rk[nr] != KEY_EMPTY3 ? 1u : 0u) |
                       (vvalid ? 2u : 0u);
          rrow[nr] = i2;
          nr++;
        }
      }
    };
    auto append = [&](uint32_t b, uint32_t pos, int64_t k, double v,
                      uint32_t rowv) {
      uint32_t slot = pos & (A4_RING - 1);
      uint8_t* rec = ring + ((size_t)b * A4_RING + slot) * 24;
      *(int64_t*)rec = k;
      *(double*)(rec + 8) = v;
      *(uint32_t*)(rec + 16) = rowv;
      // commit: LDS-only release — the record stores are DS ops,
      // lgkmcnt(0) fences all 64 lanes of the wave
      A4_ST_REL_LDS(&tag[b * A4_RING + slot], pos);
    };
    load_tile(tb);
    while (nr > 0 || p_hd != p_tl) {
      bool progress = false;
      if (p_hd != p_tl) {  // retry the oldest pending reservation
        uint32_t q = p_tl & (PCAP - 1);
        if (p_pos[q] - A4_LD_ACQ(&fl[p_b[q]]) < A4_RING) {
          append(p_b[q], p_pos[q], p_k[q], p_v[q], p_rowv[q]);
          p_tl++;
          p_spin = 0;
          progress = true;
        } else if (++p_spin > (1 << 22)) {
          atomicOr(err_flag, 64u);  // fails the chunk, never hangs the box
          p_tl = p_hd;
        }
      }
      // process the tile only when the FIFO can absorb every row going
      // pending — this keeps the reserved-but-unwritable inline spin below
      // unreachable (an inline spin can deadlock the wave: siblings parked
      // by divergence may hold the window positions the flusher needs)
      bool take = (p_hd - p_tl) + (uint32_t)nr <= PCAP;
      for (int r = 0; take && r < nr; r++) {
        if (!(rvalid[r] & 1u)) continue;  // special: handled elsewhere
        int64_t k = rk[r];
        uint32_t b = (uint32_t)(mix64_3((uint64_t)k) >> (64 - nbuck_log2));
        uint32_t rowv =
            (uint32_t)rrow[r] | ((rvalid[r] & 2u) ? 0x80000000u : 0u);
        // hot-bucket bypass BEFORE reserving (the reservation sequence
        // must stay dense — a reserved slot can never be abandoned)
        if (A4_LD_RLX(&cnt[b]) - A4_LD_RLX(&fl[b]) >=
            A4_RING + A4_RING / 2) {
          unsigned long long p = atomicAdd(lo_n, 1ull);
          leftover[p] = PartRow{k, rv[r], rowv, 0};
          atomicAdd(&byp[b], 1u);
          continue;
        }
        uint32_t pos = atomicAdd(&cnt[b], 1u);
        if (pos - A4_LD_ACQ(&fl[b]) < A4_RING) {
          append(b, pos, k, rv[r], rowv);
        } else {  // backlogged: park in the FIFO (room guaranteed above)
          uint32_t q = p_hd & (PCAP - 1);
          p_b[q] = b;
          p_pos[q] = pos;
          p_rowv[q] = rowv;
          p_k[q] = k;
          p_v[q] = rv[r];
          p_hd++;
        }
      }
      if (take) {
        if (nr) progress = true;
        tb += tile_stride;
        // prefetch the next tile AFTER the appends: issuing these loads
        // earlier would drag them into the appends' vmcnt waits
        load_tile(tb);
      }
      // back off only when the WHOLE wave is blocked on ring space
      if (__ballot(progress) == 0) __builtin_amdgcn_s_sleep(2);
    }
    // completion signal: RELEASE pairs with the flusher's ACQUIRE so every
    // tag committed by this wave is visible once done == worker_waves
    if (lane == 0)
      __hip_atomic_fetch_add(done, 1u, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_WORKGROUP);
  } else {
    // ---- flusher: 8-lane subgroups drain committed prefixes ----
    static_assert(A4_RING == 8, "ballot tag check assumes 8-slot rings");
    const int nfw = nwave - worker_waves;
    const int sg_global = (wave - worker_waves) * 8 + (lane >> 3);
    const int sg_w = lane >> 3;  // subgroup within this wave
    const int sl = lane & 7;
    const uint32_t sg_stride = (uint32_t)nfw * 8;
    bool draining = false;
    // software-pipelined sweep: the (fl, tag) reads for bucket b+stride are
    // issued while bucket b's copy is in flight, hiding the two dependent
    // LDS round-trips (~120 cycles) that otherwise bound every visit
    for (;;) {
      bool all_done = (A4_LD_ACQ(done) == (uint32_t)worker_waves);
      uint32_t moved = 0;
      uint32_t b = (uint32_t)sg_global;
      uint32_t f = fl[b];  // this sg is the only writer of fl[b]
      uint32_t t = A4_LD_ACQ(&tag[b * A4_RING +
                                  ((f + (uint32_t)sl) & (A4_RING - 1))]);
      while (b < nbuck) {
        uint32_t b2 = b + sg_stride;
        // ballot-match all 8 tags read in ONE wave instruction (lane sl
        // covered slot (f+sl)&7): committed-prefix length for this bucket
        uint64_t m = __ballot(t == f + (uint32_t)sl);
        uint32_t bits = (uint32_t)(m >> (sg_w * 8)) & 0xFFu;
        uint32_t ready = __builtin_ctz(~bits | 0x100u);
        uint32_t k = draining ? ready : ready & ~(uint32_t)(A4_QUANT - 1);
        // prefetch the NEXT bucket's fl+tag before this bucket's copy
        uint32_t f2 = 0, t2 = 0;
        if (b2 < nbuck) {
          f2 = fl[b2];
          t2 = A4_LD_ACQ(&tag[b2 * A4_RING +
                              ((f2 + (uint32_t)sl) & (A4_RING - 1))]);
        }
        if (k) {
          uint64_t* dst = (uint64_t*)(out + ((size_t)base_line[b] << 6) +
                                      (size_t)f * 24);
          const uint64_t* src =
              (const uint64_t*)(ring + (size_t)b * A4_RING * 24);
          for (uint32_t d = sl; d < k * 3; d += 8) {
            uint32_t r = d / 3;
            uint32_t slot = (f + r) & (A4_RING - 1);
            dst[d] = src[(size_t)slot * 3 + d % 3];
          }
          // LDS-only release: the slot reads must complete before workers
          // may overwrite them; in-flight HBM stores are irrelevant
          if (sl == 0) A4_ST_REL_LDS(&fl[b], f + k);
          moved += k;
        }
        b = b2;
        f = f2;
        t = t2;
      }
      if (draining && moved == 0) break;
      if (all_done) draining = true;  // final sweeps flush partial quanta
    }
  }
  // every (bucket, block) entry is written by its own block — on EVERY exit
  // path — so the bucket kernel never reads a stale correction matrix
  __syncthreads();
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x)
    bypass_matrix[((size_t)b << grid_log2) | blockIdx.x] = byp[b];
}

// The histogram MUST be launched with agg4_worker_waves()*64 threads per
// block when the v4 scatter consumes its counts: the per-(block,bucket)
// ranges are credited by the hist's row->block grid-stride mapping, and the
// v4 workers traverse rows with exactly that stride.
int agg4_worker_waves() {
  static int ww = [] {
    const char* e = getenv("AURON_AGG2_V4_WW");
    int v = e ? atoi(e) : 8;  // measured optimum: flusher-throughput-bound
    return (v >= 6 && v <= 15) ? v : 8;
  }();
  return ww;
}

void launch_agg4_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* line_scan, uint8_t* out,
                         PartRow* leftover, unsigned long long* lo_n,
                         uint32_t* bypass_matrix, uint32_t* err_flag,
                         hipStream_t s) {
  const uint32_t nbuck = 1u << nbuck_log2;
  size_t lds = (size_t)nbuck * A4_RING * 24 + (size_t)nbuck * A4_RING * 4 +
               (size_t)nbuck * 16 + 64;
  if (lds > 160 * 1024)
    throw std::runtime_error("agg4 scatter LDS over 160KB");
  int ww = agg4_worker_waves();
  hipError_t e = hipFuncSetAttribute(
      (const void*)k_agg4_scatter,
      hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
  if (e != hipSuccess)
    throw std::runtime_error("agg4 scatter LDS attribute failed");
  hipLaunchKernelGGL(k_agg4_scatter, dim3(1 << grid_log2), dim3(1024), lds, s,
                     keys, key_valid, vals, val_valid, n, nbuck_log2,
                     grid_log2, ww, line_scan, out, leftover, lo_n,
                     bypass_matrix, err_flag);
  check_launch3("k_agg4_scatter");
}

// ---- v3 bucket aggregation (4096-slot LDS window, 1024 threads) ------------
static constexpr int L3SLOTS = 4096;
static constexpr int L3PROBE = 128;

template <int REC>
__global__ void __launch_bounds__(1024) k_agg3_bucket(
    const uint8_t* __restrict__ part, const uint32_t* __restrict__ counts,
    const uint32_t* __restrict__ bypass,
    const uint32_t* __restrict__ line_scan, int grid_log2, int is_int,
    int nbuckets, int64_t key_base, StagedGroup* __restrict__ staged,
    unsigned long long* __restrict__ staged_n, int64_t staged_cap,
    PartRow* __restrict__ leftover, unsigned long long* __restrict__ lo_n,
    uint32_t* __restrict__ error_flag) {
  __shared__ int64_t ls_key[L3SLOTS];
  __shared__ double ls_sum[L3SLOTS];
  __shared__ uint32_t ls_cnt[L3SLOTS];
  __shared__ uint32_t ls_first[L3SLOTS];
  // per-range cumulative record counts for this bucket (grid of ranges)
  __shared__ uint32_t cum[1 << AGG2_GRID_LOG2_MAX];
  const int nrange = 1 << grid_log2;

  for (int b = blockIdx.x; b < nbuckets; b += gridDim.x) {
    for (int s = threadIdx.x; s < L3SLOTS; s += blockDim.x) {
      ls_key[s] = KEY_EMPTY3;
      ls_sum[s] = 0.0;
      ls_cnt[s] = 0;
      ls_first[s] = 0xFFFFFFFFu;
    }
    // serial scan of <=1024 range counts (one thread; ~us)
    if (threadIdx.x == 0) {
      uint32_t acc = 0;
      for (int k = 0; k < nrange; k++) {
        size_t e = ((size_t)b << grid_log2) | k;
        acc += counts[e] - bypass[e];  // bypassed rows were never written
        cum[k] = acc;
      }
    }
    __syncthreads();
    uint32_t total = cum[nrange - 1];
    // 2-way ILP: both record loads (random-ish global) and both first LDS
    // probe reads go out before either resolution — halves exposed latency
    // at the 16-wave occupancy this 100KB-LDS kernel gets
    auto locate = [&](uint32_t j) -> const uint8_t* {
      int lo = 0, hi = nrange - 1;
      while (lo < hi) {
        int mid = (lo + hi) >> 1;
        if (j < cum[mid]) hi = mid; else lo = mid + 1;
      }
      uint32_t before = lo ? cum[lo - 1] : 0;
      return part + ((size_t)line_scan[((size_t)b << grid_log2) | lo] << 6) +
             (size_t)(j - before) * REC;
    };
    // typed record parse: REC 16 packs (key - key_base) u32 with rowv
    auto load_rec = [&](const uint8_t* r, int64_t* k, double* v,
                        uint32_t* rv) {
      if (REC == 16) {
        uint64_t w = *(const uint64_t*)r;
        *k = key_base + (int64_t)(uint32_t)w;
        *rv = (uint32_t)(w >> 32);
      } else {
        *k = *(const int64_t*)r;
        *rv = *(const uint32_t*)(r + 16);
      }
      *v = *(const double*)(r + 8);
    };
    auto resolve = [&](int64_t k, double v, uint32_t rowv, uint32_t h) {
      uint32_t row = rowv & 0x7FFFFFFFu;
      bool vvalid = (rowv & 0x80000000u) != 0;
      int found = -1;
      for (int p = 0; p < L3PROBE; p++) {
        int64_t cur = ls_key[h];
        if (cur == k) {
          found = (int)h;
          break;
        }
        if (cur == KEY_EMPTY3) {
          long long prev = atomicCAS((unsigned long long*)&ls_key[h],
                                     (unsigned long long)KEY_EMPTY3,
                                     (unsigned long long)k);
          if (prev == (long long)KEY_EMPTY3 || prev == (long long)k) {
            found = (int)h;
            break;
          }
        }
        h = (h + 1) & (L3SLOTS - 1);
      }
      if (found >= 0) {
        atomicMin(&ls_first[found], row);
        if (vvalid) {
          sum_accum3(&ls_sum[found], v, is_int);
          atomicAdd(&ls_cnt[found], 1u);
        }
      } else {
        unsigned long long p = atomicAdd(lo_n, 1ull);
        leftover[p] = PartRow{k, v, rowv, 0};
      }
    };
    uint32_t j = threadIdx.x;
    for (; j + 3 * blockDim.x < total; j += 4 * blockDim.x) {
      const uint8_t* r0 = locate(j);
      const uint8_t* r1 = locate(j + blockDim.x);
      const uint8_t* r2 = locate(j + 2 * blockDim.x);
      const uint8_t* r3 = locate(j + 3 * blockDim.x);
      int64_t k0, k1, k2, k3;       // all four record loads in flight
      double v0, v1, v2, v3;
      uint32_t rv0, rv1, rv2, rv3;
      load_rec(r0, &k0, &v0, &rv0);
      load_rec(r1, &k1, &v1, &rv1);
      load_rec(r2, &k2, &v2, &rv2);
      load_rec(r3, &k3, &v3, &rv3);
      uint32_t h0 = (uint32_t)mix64_3((uint64_t)k0) & (L3SLOTS - 1);
      uint32_t h1 = (uint32_t)mix64_3((uint64_t)k1) & (L3SLOTS - 1);
      uint32_t h2 = (uint32_t)mix64_3((uint64_t)k2) & (L3SLOTS - 1);
      uint32_t h3 = (uint32_t)mix64_3((uint64_t)k3) & (L3SLOTS - 1);
      resolve(k0, v0, rv0, h0);
      resolve(k1, v1, rv1, h1);
      resolve(k2, v2, rv2, h2);
      resolve(k3, v3, rv3, h3);
    }
    for (; j < total; j += blockDim.x) {
      const uint8_t* r = locate(j);
      int64_t k;
      double v;
      uint32_t rv;
      load_rec(r, &k, &v, &rv);
      resolve(k, v, rv, (uint32_t)mix64_3((uint64_t)k) & (L3SLOTS - 1));
    }
    __syncthreads();
    for (int s = threadIdx.x; s < L3SLOTS; s += blockDim.x) {
      if (ls_key[s] == KEY_EMPTY3) continue;
      unsigned long long p = atomicAdd(staged_n, 1ull);
      if ((int64_t)p >= staged_cap) {
        atomicOr(error_flag, 2u);
        continue;
      }
      staged[p].key = ls_key[s];
      staged[p].sum = ls_sum[s];
      staged[p].cnt_first =
          ((unsigned long long)ls_cnt[s] << 32) | ls_first[s];
    }
    __syncthreads();
  }
}

void launch_agg3_bucket(const uint8_t* part, const uint32_t* counts,
                        const uint32_t* bypass,
                        const uint32_t* line_scan, int grid_log2, int is_int,
                        int nbuckets, StagedGroup* staged,
                        unsigned long long* staged_n, int64_t staged_cap,
                        PartRow* leftover, unsigned long long* lo_n,
                        uint32_t* error_flag, int rec, int64_t key_base,
                        hipStream_t s) {
  if (rec == 16)
    hipLaunchKernelGGL((k_agg3_bucket<16>), dim3(nbuckets), dim3(1024), 0, s,
                       part, counts, bypass, line_scan, grid_log2, is_int,
                       nbuckets, key_base, staged, staged_n, staged_cap,
                       leftover, lo_n, error_flag);
  else
    hipLaunchKernelGGL((k_agg3_bucket<24>), dim3(nbuckets), dim3(1024), 0, s,
                       part, counts, bypass, line_scan, grid_log2, is_int,
                       nbuckets, key_base, staged, staged_n, staged_cap,
                       leftover, lo_n, error_flag);
  check_launch3("k_agg3_bucket");
}

}  // namespace auron
