// kernels_agg2.hip — two-phase hash aggregation for high-cardinality GROUP BY
// (SURVEY.md §7 hard part (b)): radix-partition rows by key-hash into buckets
// whose groups fit in LDS, then aggregate each bucket entirely in LDS and
// merge the (bounded, counted) per-bucket group lists into the global slot
// table. Replaces the single-phase kernel's per-row random slot-line round
// trips (~183 B/row measured, profiles/r01_agg_pmc.md) with sequential
// streams: ~8 B/row histogram + 16 B/row read + 20 B/row partition write +
// 20 B/row bucket read; group-table traffic becomes per-GROUP, not per-row.
#include <hip/hip_runtime.h>

#include <rocprim/device/device_scan.hpp>

#include <stdexcept>
#include <string>

#include "kernels.h"

namespace auron {

namespace {
inline void check_launch2(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("kernel launch failed: ") + name +
                             ": " + hipGetErrorString(e));
}
constexpr int BLOCK = 256;
constexpr int64_t MAX_BLOCKS = 256 * 8;
inline int grid2(int64_t n) {
  int64_t b = (n + BLOCK - 1) / BLOCK;
  if (b > MAX_BLOCKS) b = MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}
constexpr int64_t KEY_EMPTY2 = INT64_MIN;
// typed accumulate (see kernels.hip sum_accum): i64 mode adds raw bits with
// wrapping integer atomics; value bytes ride through double registers
__device__ __forceinline__ void sum_accum2(double* acc, double v, bool is_int) {
  if (is_int) {
    uint64_t b;
    memcpy(&b, &v, 8);
    atomicAdd(reinterpret_cast<unsigned long long*>(acc),
              (unsigned long long)b);
  } else {
    unsafeAtomicAdd(acc, v);
  }
}



__device__ __forceinline__ uint64_t mix64_2(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__device__ __forceinline__ bool bit_get2(const uint8_t* bm, int64_t i) {
  return (bm[i >> 3] >> (i & 7)) & 1;
}

// bucket id = HIGH bits of the mix (the LDS probe uses the LOW bits, so the
// two are independent)
__device__ __forceinline__ uint32_t bucket_of(int64_t key, int nbuck_log2) {
  return (uint32_t)(mix64_2((uint64_t)key) >> (64 - nbuck_log2));
}

}  // namespace

// ---- phase P1: per-BLOCK per-bucket histogram (reads keys only) ------------
// counts_matrix is bucket-major: counts[(b << GRID_LOG2) | blockIdx]. An
// exclusive scan over the flat matrix then gives every block a private,
// contiguous output range per bucket — the scatter needs NO global atomics.
// Launched at 256 or 1024 threads (AURON_AGG2_BLOCK): 1024-thread blocks
// reach 16 waves per block, so a 512-block grid fills all 8192 wave slots
// of the chip where 256-thread blocks reached only 8 waves/CU.

__global__ __launch_bounds__(1024) void k_agg2_hist(
    const int64_t* __restrict__ keys,
                            const uint8_t* __restrict__ key_valid, int64_t n,
                            int nbuck_log2, int grid_log2,
                            uint32_t* __restrict__ counts_matrix,
                            uint32_t* __restrict__ special_rows,
                            unsigned long long* __restrict__ kminmax) {
  extern __shared__ uint32_t lds_hist[];
  const uint32_t nbuck = 1u << nbuck_log2;
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x) lds_hist[b] = 0;
  __syncthreads();
  uint32_t special = 0;
  // optional NORMAL-key range probe (order-mapped u64 so unsigned
  // atomicMin/Max implement signed i64 min/max): feeds the packed-16B
  // partition record fast path (kernels_agg3.hip)
  unsigned long long kmin = ~0ull, kmax = 0ull;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get2(key_valid, i);
    int64_t k = keys[i];
    if (knull || k == KEY_EMPTY2) {
      special++;
      continue;
    }
    if (kminmax) {
      unsigned long long m = (unsigned long long)k ^ 0x8000000000000000ull;
      kmin = m < kmin ? m : kmin;
      kmax = m > kmax ? m : kmax;
    }
    atomicAdd(&lds_hist[bucket_of(k, nbuck_log2)], 1u);
  }
  __syncthreads();
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x)
    counts_matrix[((size_t)b << grid_log2) | blockIdx.x] = lds_hist[b];
  if (special) atomicAdd(special_rows, special);
  if (kminmax && kmin != ~0ull) {
    atomicMin(&kminmax[0], kmin);
    atomicMax(&kminmax[1], kmax);
  }
}

// offsets[b] = scanned[(b << GRID_LOG2) | 0]; offsets[nbuck] = the scan
// total (scanned[nbuck << GRID_LOG2], the extra slot) — read on device so the
// host never synchronizes for it
__global__ void k_agg2_offsets(const uint32_t* __restrict__ scanned,
                               int nbuck_log2, int grid_log2,
                               uint32_t* __restrict__ offsets) {
  int nbuck = 1 << nbuck_log2;
  for (int b = (int)(blockIdx.x * blockDim.x + threadIdx.x); b <= nbuck;
       b += (int)(gridDim.x * blockDim.x))
    offsets[b] = scanned[(size_t)b << grid_log2];
}

// ---- phase P2: scatter into per-(block,bucket) reserved ranges -------------
// Must use the SAME grid geometry and row traversal as k_agg2_hist.
// rowv packs the chunk-local row (bit 0..30) + value-validity (bit 31).
__global__ __launch_bounds__(1024) void k_agg2_scatter(
    const int64_t* __restrict__ keys,
                               const uint8_t* __restrict__ key_valid,
                               const double* __restrict__ vals,
                               const uint8_t* __restrict__ val_valid, int64_t n,
                               int nbuck_log2, int grid_log2,
                               const uint32_t* __restrict__ scanned,
                               PartKV* __restrict__ out_kv,
                               uint32_t* __restrict__ out_rowv) {
  extern __shared__ uint32_t lds_cursor[];
  const uint32_t nbuck = 1u << nbuck_log2;
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x)
    lds_cursor[b] = scanned[((size_t)b << grid_log2) | blockIdx.x];
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get2(key_valid, i);
    int64_t k = keys[i];
    if (knull || k == KEY_EMPTY2) continue;  // specials handled separately
    uint32_t b = bucket_of(k, nbuck_log2);
    uint32_t pos = atomicAdd(&lds_cursor[b], 1u);  // block-local LDS cursor
    bool vvalid = !val_valid || bit_get2(val_valid, i);
    out_kv[pos] = PartKV{k, vals[i]};   // one aligned 16-byte store
    out_rowv[pos] = (uint32_t)i | (vvalid ? 0x80000000u : 0u);
  }
}

// specials (null key / i64::MIN key): straight to the global special slots
__global__ void k_agg2_specials(const AggTable t,
                                const int64_t* __restrict__ keys,
                                const uint8_t* __restrict__ key_valid,
                                const double* __restrict__ vals,
                                const uint8_t* __restrict__ val_valid,
                                int64_t n, uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get2(key_valid, i);
    int64_t k = keys[i];
    if (!(knull || k == KEY_EMPTY2)) continue;
    int which = knull ? 1 : 0;
    if (atomicCAS(&t.special_used[which], 0u, 1u) == 0u)
      atomicAdd(t.num_groups, 1ull);
    AggSlot* sl = &t.slots[t.cap + which];
    uint64_t row = row_offset + (uint64_t)i;
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
    bool vvalid = !val_valid || bit_get2(val_valid, i);
    if (vvalid) {
      sum_accum2(&sl->sum, vals[i], t.sum_int);
      atomicAdd(&sl->cnt, 1ull);
    }
  }
}

// ---- phase A: one workgroup aggregates one bucket in LDS -------------------
// LDS table: LSLOTS entries of {key i64, sum f64, cnt u32, first u32} in SoA.
// Keys that cannot be placed within the probe window append their raw rows to
// the leftover list (reprocessed by the single-phase path — rare).
static constexpr int LSLOTS = AGG2_LSLOTS;  // 48 KB LDS -> 3 blocks/CU
static constexpr int LPROBE = 64;

__global__ void __launch_bounds__(1024) k_agg2_bucket(
    const PartKV* __restrict__ part, const uint32_t* __restrict__ part_rowv,
    const uint32_t* __restrict__ offsets, int is_int,
    int nbuckets, StagedGroup* __restrict__ staged,
    unsigned long long* __restrict__ staged_n, int64_t staged_cap,
    PartRow* __restrict__ leftover, unsigned long long* __restrict__ lo_n,
    uint32_t* __restrict__ error_flag) {
  __shared__ int64_t ls_key[LSLOTS];
  __shared__ double ls_sum[LSLOTS];
  __shared__ uint32_t ls_cnt[LSLOTS];
  __shared__ uint32_t ls_first[LSLOTS];

  for (int b = blockIdx.x; b < nbuckets; b += gridDim.x) {
    for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x) {
      ls_key[s] = KEY_EMPTY2;
      ls_sum[s] = 0.0;
      ls_cnt[s] = 0;
      ls_first[s] = 0xFFFFFFFFu;
    }
    __syncthreads();
    uint32_t beg = offsets[b], end = offsets[b + 1];
    for (uint32_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
      int64_t k = part[i].key;
      double v = part[i].val;
      uint32_t rowv = part_rowv[i];
      uint32_t row = rowv & 0x7FFFFFFFu;
      bool vvalid = (rowv & 0x80000000u) != 0;
      // probe LDS (low bits of the same mix; bucket used the high bits)
      uint32_t h = (uint32_t)mix64_2((uint64_t)k) & (LSLOTS - 1);
      int found = -1;
      for (int p = 0; p < LPROBE; p++) {
        int64_t cur = ls_key[h];
        if (cur == k) {
          found = (int)h;
          break;
        }
        if (cur == KEY_EMPTY2) {
          long long prev = atomicCAS((unsigned long long*)&ls_key[h],
                                     (unsigned long long)KEY_EMPTY2,
                                     (unsigned long long)k);
          if (prev == (long long)KEY_EMPTY2 || prev == (long long)k) {
            found = (int)h;
            break;
          }
        }
        h = (h + 1) & (LSLOTS - 1);
      }
      if (found >= 0) {
        atomicMin(&ls_first[found], row);
        if (vvalid) {
          sum_accum2(&ls_sum[found], v, is_int);
          atomicAdd(&ls_cnt[found], 1u);
        }
      } else {
        // LDS window full: append raw row to the leftover list
        unsigned long long p = atomicAdd(lo_n, 1ull);
        leftover[p] = PartRow{k, v, rowv, 0};
      }
    }
    __syncthreads();
    // flush occupied LDS slots to the staged group list
    for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x) {
      if (ls_key[s] == KEY_EMPTY2) continue;
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

// merge staged groups into the global slot table (per-GROUP work)
__global__ void k_agg2_merge_groups(const AggTable t,
                                    const StagedGroup* __restrict__ staged,
                                    int64_t n, uint64_t row_offset) {
  const int64_t mask = t.cap - 1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t key = staged[i].key;
    uint64_t h = mix64_2((uint64_t)key);
    int64_t s = (int64_t)(h & (uint64_t)mask);
    int64_t a = -1;
    for (int64_t probe = 0; probe <= mask; probe++) {
      int64_t cur = t.slots[s].key;
      if (cur == key) {
        a = s;
        break;
      }
      if (cur == KEY_EMPTY2) {
        long long prev = atomicCAS((unsigned long long*)&t.slots[s].key,
                                   (unsigned long long)KEY_EMPTY2,
                                   (unsigned long long)key);
        if (prev == (long long)KEY_EMPTY2) {
          atomicAdd(t.num_groups, 1ull);
          a = s;
          break;
        }
        if (prev == (long long)key) {
          a = s;
          break;
        }
      }
      s = (s + 1) & mask;
    }
    if (a < 0) {
      atomicOr(t.error_flag, 1u);
      continue;
    }
    AggSlot* sl = &t.slots[a];
    unsigned long long cf = staged[i].cnt_first;
    uint64_t row = row_offset + (uint32_t)cf;
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
    uint32_t cnt = (uint32_t)(cf >> 32);
    if (cnt) {  // cnt==0: group seen only via null values — key+order only
      sum_accum2(&sl->sum, staged[i].sum, t.sum_int);
      atomicAdd(&sl->cnt, (unsigned long long)cnt);
    }
  }
}

// leftover raw rows (LDS window overflow — rare): single-phase accumulate
__global__ void k_agg2_leftovers(const AggTable t,
                                 const PartRow* __restrict__ rows, int64_t n,
                                 uint64_t row_offset) {
  const int64_t mask = t.cap - 1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t key = rows[i].key;
    uint64_t h = mix64_2((uint64_t)key);
    int64_t s = (int64_t)(h & (uint64_t)mask);
    int64_t a = -1;
    for (int64_t probe = 0; probe <= mask; probe++) {
      int64_t cur = t.slots[s].key;
      if (cur == key) {
        a = s;
        break;
      }
      if (cur == KEY_EMPTY2) {
        long long prev = atomicCAS((unsigned long long*)&t.slots[s].key,
                                   (unsigned long long)KEY_EMPTY2,
                                   (unsigned long long)key);
        if (prev == (long long)KEY_EMPTY2) {
          atomicAdd(t.num_groups, 1ull);
          a = s;
          break;
        }
        if (prev == (long long)key) {
          a = s;
          break;
        }
      }
      s = (s + 1) & mask;
    }
    if (a < 0) {
      atomicOr(t.error_flag, 1u);
      continue;
    }
    AggSlot* sl = &t.slots[a];
    uint32_t rv = rows[i].rowv;
    uint64_t row = row_offset + (rv & 0x7FFFFFFFu);
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
    if (rv & 0x80000000u) {
      sum_accum2(&sl->sum, rows[i].val, t.sum_int);
      atomicAdd(&sl->cnt, 1ull);
    }
  }
}

void launch_agg2_leftovers(const AggTable& t, const PartRow* rows, int64_t n,
                           uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_agg2_leftovers, dim3(grid2(n)), dim3(BLOCK), 0, s, t,
                     rows, n, row_offset);
  check_launch2("k_agg2_leftovers");
}

void launch_agg2_hist(const int64_t* keys, const uint8_t* key_valid, int64_t n,
                      int nbuck_log2, int grid_log2, uint32_t* counts_matrix,
                      uint32_t* special_rows, int block, hipStream_t s,
                      unsigned long long* kminmax) {
  size_t lds = (size_t)(1u << nbuck_log2) * 4;
  hipLaunchKernelGGL(k_agg2_hist, dim3(1 << grid_log2), dim3(block), lds, s,
                     keys, key_valid, n, nbuck_log2, grid_log2, counts_matrix,
                     special_rows, kminmax);
  check_launch2("k_agg2_hist");
}

void launch_agg2_offsets(const uint32_t* scanned, int nbuck_log2,
                         int grid_log2, uint32_t* offsets, hipStream_t s) {
  hipLaunchKernelGGL(k_agg2_offsets, dim3(8), dim3(BLOCK), 0, s, scanned,
                     nbuck_log2, grid_log2, offsets);
  check_launch2("k_agg2_offsets");
}

void scan_counts_matrix(const uint32_t* counts, uint32_t* scanned, int64_t n,
                        void* temp, size_t* temp_bytes, hipStream_t s) {
  hipError_t e = rocprim::exclusive_scan(temp, *temp_bytes, counts, scanned,
                                         0u, (size_t)n,
                                         rocprim::plus<uint32_t>(), s);
  if (e != hipSuccess) abort();
}

void launch_agg2_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* scanned,
                         PartKV* out_kv, uint32_t* out_rowv, int block,
                         hipStream_t s) {
  size_t lds = (size_t)(1u << nbuck_log2) * 4;
  hipLaunchKernelGGL(k_agg2_scatter, dim3(1 << grid_log2), dim3(block), lds, s,
                     keys, key_valid, vals, val_valid, n, nbuck_log2,
                     grid_log2, scanned, out_kv, out_rowv);
  check_launch2("k_agg2_scatter");
}

void launch_agg2_specials(const AggTable& t, const int64_t* keys,
                          const uint8_t* key_valid, const double* vals,
                          const uint8_t* val_valid, int64_t n,
                          uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_agg2_specials, dim3(grid2(n)), dim3(BLOCK), 0, s, t,
                     keys, key_valid, vals, val_valid, n, row_offset);
  check_launch2("k_agg2_specials");
}

void launch_agg2_bucket(const PartKV* part_kv, const uint32_t* part_rowv,
                        const uint32_t* offsets, int is_int,
                        int nbuckets, StagedGroup* staged,
                        unsigned long long* staged_n, int64_t staged_cap,
                        PartRow* leftover, unsigned long long* lo_n,
                        uint32_t* error_flag, int block, hipStream_t s) {
  int blocks = nbuckets < (int)MAX_BLOCKS ? nbuckets : (int)MAX_BLOCKS;
  hipLaunchKernelGGL(k_agg2_bucket, dim3(blocks), dim3(block), 0, s, part_kv,
                     part_rowv, offsets, is_int, nbuckets, staged, staged_n,
                     staged_cap, leftover, lo_n, error_flag);
  check_launch2("k_agg2_bucket");
}


// ---- 24B AoS variants (A/B comparison path, selected by the engine) --------
__global__ __launch_bounds__(1024) void k_agg2_scatter24(
    const int64_t* __restrict__ keys,
                                 const uint8_t* __restrict__ key_valid,
                                 const double* __restrict__ vals,
                                 const uint8_t* __restrict__ val_valid,
                                 int64_t n, int nbuck_log2, int grid_log2,
                                 const uint32_t* __restrict__ scanned,
                                 PartRow* __restrict__ out) {
  extern __shared__ uint32_t lds_cursor[];
  const uint32_t nbuck = 1u << nbuck_log2;
  for (uint32_t b = threadIdx.x; b < nbuck; b += blockDim.x)
    lds_cursor[b] = scanned[((size_t)b << grid_log2) | blockIdx.x];
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get2(key_valid, i);
    int64_t k = keys[i];
    if (knull || k == KEY_EMPTY2) continue;
    uint32_t b = bucket_of(k, nbuck_log2);
    uint32_t pos = atomicAdd(&lds_cursor[b], 1u);
    bool vvalid = !val_valid || bit_get2(val_valid, i);
    PartRow r;
    r.key = k;
    r.val = vals[i];
    r.rowv = (uint32_t)i | (vvalid ? 0x80000000u : 0u);
    r._pad = 0;
    out[pos] = r;
  }
}

void launch_agg2_scatter24(const int64_t* keys, const uint8_t* key_valid,
                           const double* vals, const uint8_t* val_valid,
                           int64_t n, int nbuck_log2, int grid_log2,
                           const uint32_t* scanned,
                           PartRow* out, int block, hipStream_t s) {
  size_t lds = (size_t)(1u << nbuck_log2) * 4;
  hipLaunchKernelGGL(k_agg2_scatter24, dim3(1 << grid_log2), dim3(block), lds,
                     s, keys, key_valid, vals, val_valid, n, nbuck_log2,
                     grid_log2, scanned, out);
  check_launch2("k_agg2_scatter24");
}

__global__ void __launch_bounds__(1024) k_agg2_bucket24(
    const PartRow* __restrict__ part, const uint32_t* __restrict__ offsets,
    int is_int, int nbuckets, StagedGroup* __restrict__ staged,
    unsigned long long* __restrict__ staged_n, int64_t staged_cap,
    PartRow* __restrict__ leftover, unsigned long long* __restrict__ lo_n,
    uint32_t* __restrict__ error_flag) {
  __shared__ int64_t ls_key[LSLOTS];
  __shared__ double ls_sum[LSLOTS];
  __shared__ uint32_t ls_cnt[LSLOTS];
  __shared__ uint32_t ls_first[LSLOTS];
  for (int b = blockIdx.x; b < nbuckets; b += gridDim.x) {
    for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x) {
      ls_key[s] = KEY_EMPTY2;
      ls_sum[s] = 0.0;
      ls_cnt[s] = 0;
      ls_first[s] = 0xFFFFFFFFu;
    }
    __syncthreads();
    uint32_t beg = offsets[b], end = offsets[b + 1];
    for (uint32_t i = beg + threadIdx.x; i < end; i += blockDim.x) {
      int64_t k = part[i].key;
      double v = part[i].val;
      uint32_t rowv = part[i].rowv;
      uint32_t row = rowv & 0x7FFFFFFFu;
      bool vvalid = (rowv & 0x80000000u) != 0;
      uint32_t h = (uint32_t)mix64_2((uint64_t)k) & (LSLOTS - 1);
      int found = -1;
      for (int p = 0; p < LPROBE; p++) {
        int64_t cur = ls_key[h];
        if (cur == k) {
          found = (int)h;
          break;
        }
        if (cur == KEY_EMPTY2) {
          long long prev = atomicCAS((unsigned long long*)&ls_key[h],
                                     (unsigned long long)KEY_EMPTY2,
                                     (unsigned long long)k);
          if (prev == (long long)KEY_EMPTY2 || prev == (long long)k) {
            found = (int)h;
            break;
          }
        }
        h = (h + 1) & (LSLOTS - 1);
      }
      if (found >= 0) {
        atomicMin(&ls_first[found], row);
        if (vvalid) {
          sum_accum2(&ls_sum[found], v, is_int);
          atomicAdd(&ls_cnt[found], 1u);
        }
      } else {
        unsigned long long p = atomicAdd(lo_n, 1ull);
        leftover[p] = part[i];
      }
    }
    __syncthreads();
    for (int s = threadIdx.x; s < LSLOTS; s += blockDim.x) {
      if (ls_key[s] == KEY_EMPTY2) continue;
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

void launch_agg2_bucket24(const PartRow* part, const uint32_t* offsets,
                          int is_int, int nbuckets, StagedGroup* staged,
                          unsigned long long* staged_n, int64_t staged_cap,
                          PartRow* leftover, unsigned long long* lo_n,
                          uint32_t* error_flag, int block, hipStream_t s) {
  int blocks = nbuckets < (int)MAX_BLOCKS ? nbuckets : (int)MAX_BLOCKS;
  hipLaunchKernelGGL(k_agg2_bucket24, dim3(blocks), dim3(block), 0, s, part,
                     offsets, is_int, nbuckets, staged, staged_n, staged_cap,
                     leftover, lo_n, error_flag);
  check_launch2("k_agg2_bucket24");
}

void launch_agg2_merge_groups(const AggTable& t, const StagedGroup* staged,
                              int64_t n, uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_agg2_merge_groups, dim3(grid2(n)), dim3(BLOCK), 0, s, t,
                     staged, n, row_offset);
  check_launch2("k_agg2_merge_groups");
}

}  // namespace auron
