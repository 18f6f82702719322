// kernels.hip — hand-written CDNA4 (gfx950) kernels for the Auron hot path.
//
// This path is hash/scatter work bound by HBM streaming + cache-side atomics —
// MFMA is irrelevant (BASELINE.json north_star). Design notes:
//  * wave64: block size 256 (4 waves), grid-stride loops capped at
//    256 CU × 8 blocks (guide §6 G11), so one launch fills all 8 XCDs.
//  * input reads are coalesced 8 B/lane streams; the hash table is sized so
//    the hot accumulator lines live in L2/L3 (1M groups ≈ 50 MB < 256 MB L3),
//    keeping HBM traffic near the 16 B/row algorithmic minimum.
//  * f64 scatter-add uses the native global_atomic_add_f64 (unsafeAtomicAdd;
//    compiled with -munsafe-fp-atomics); i64/u64 counts use integer atomics
//    (bit-exact, order-free). Float SUM order is therefore arrival-order-free;
//    parity bar per BASELINE.json: 1e-6 relative for SUM/AVG(f64), bit-exact
//    integers.
//  * record order: first-occurrence (atomicMin on a per-group arrival index,
//    deterministic) matching AggHashMap's dense-index insertion order
//    (agg_hash_map.rs:77-168).
#include <hip/hip_runtime.h>

#include <rocprim/device/device_radix_sort.hpp>
#include <rocprim/device/device_scan.hpp>

#include <stdexcept>
#include <string>

#include "kernels.h"
#include "dev_agg.h"

namespace auron {

// raise launch-configuration errors loudly (hipLaunchKernelGGL itself
// reports nothing)
static inline void check_launch(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("kernel launch failed: ") + name +
                             ": " + hipGetErrorString(e));
}

static constexpr int BLOCK = 256;
static constexpr int64_t MAX_BLOCKS = 256 * 8;

static inline int grid_for(int64_t n) {
  int64_t b = (n + BLOCK - 1) / BLOCK;
  if (b > MAX_BLOCKS) b = MAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}

__global__ void k_hash_init(int32_t* hashes, int32_t seed, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    hashes[i] = seed;
}

__global__ void k_hash_fold_i64(const int64_t* __restrict__ vals,
                                const uint8_t* __restrict__ valid, int64_t n,
                                int32_t* __restrict__ hashes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (!valid || bit_get_dev(valid, i))
      hashes[i] = murmur3_long(vals[i], hashes[i]);
  }
}

__global__ void k_pmod(const int32_t* __restrict__ hashes, int64_t n, int32_t P,
                       uint32_t* __restrict__ part_ids) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t r = hashes[i] % P;           // rem_euclid, shuffle/mod.rs:180-183
    part_ids[i] = (uint32_t)(r < 0 ? r + P : r);
  }
}

void launch_hash_init(int32_t* hashes, int32_t seed, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_hash_init, dim3(grid_for(n)), dim3(BLOCK), 0, s, hashes,
                     seed, n);
  check_launch("k_hash_init");
}
void launch_hash_fold_i64(const int64_t* vals, const uint8_t* valid, int64_t n,
                          int32_t* hashes, hipStream_t s) {
  hipLaunchKernelGGL(k_hash_fold_i64, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     vals, valid, n, hashes);
  check_launch("k_hash_fold_i64");
}
__global__ void k_hash_fold_i32(const int32_t* __restrict__ vals,
                                const uint8_t* __restrict__ valid, int64_t n,
                                int32_t* __restrict__ hashes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (!valid || bit_get_dev(valid, i)) {
      // 4-byte LE word: one mix + fmix(len=4) (mur.rs:19-30 aligned path)
      int32_t h1 = mur_mix_h1(hashes[i], mur_mix_k1(vals[i]));
      hashes[i] = mur_fmix(h1, 4);
    }
  }
}

__global__ void k_widen_i32_i64(const int32_t* __restrict__ in, int64_t n,
                                int64_t* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int64_t)in[i];
}

__global__ void k_narrow_i64_i32(const int64_t* __restrict__ in, int64_t n,
                                 int32_t* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int32_t)in[i];
}

__global__ void k_robin_ids(int64_t n, uint32_t start, uint32_t P,
                            uint32_t* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (uint32_t)((start + (uint64_t)i) % P);
}

void launch_hash_fold_i32(const int32_t* vals, const uint8_t* valid, int64_t n,
                          int32_t* hashes, hipStream_t s) {
  hipLaunchKernelGGL(k_hash_fold_i32, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     vals, valid, n, hashes);
  check_launch("k_hash_fold_i32");
}
void launch_widen_i32_i64(const int32_t* in, int64_t n, int64_t* out,
                          hipStream_t s) {
  hipLaunchKernelGGL(k_widen_i32_i64, dim3(grid_for(n)), dim3(BLOCK), 0, s, in,
                     n, out);
  check_launch("k_widen_i32_i64");
}
void launch_narrow_i64_i32(const int64_t* in, int64_t n, int32_t* out,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_narrow_i64_i32, dim3(grid_for(n)), dim3(BLOCK), 0, s, in,
                     n, out);
  check_launch("k_narrow_i64_i32");
}
void launch_robin_ids(int64_t n, uint32_t start, uint32_t P, uint32_t* out,
                      hipStream_t s) {
  hipLaunchKernelGGL(k_robin_ids, dim3(grid_for(n)), dim3(BLOCK), 0, s, n,
                     start, P, out);
  check_launch("k_robin_ids");
}

void launch_pmod(const int32_t* hashes, int64_t n, int32_t P, uint32_t* part_ids,
                 hipStream_t s) {
  hipLaunchKernelGGL(k_pmod, dim3(grid_for(n)), dim3(BLOCK), 0, s, hashes, n, P,
                     part_ids);
  check_launch("k_pmod");
}

// ---- hash aggregation ------------------------------------------------------
// Per-row slot resolution + accumulate (the slow path of the batched kernel
// below; also correct standalone).
__device__ __forceinline__ void agg_accum_row(const AggTable t, int64_t key,
                                              bool knull, double val,
                                              bool vvalid, uint64_t row) {
  int64_t a = agg_upsert_slot(t, key, knull);
  if (a < 0) return;  // table full: error_flag raised, host aborts
  AggSlot* sl = &t.slots[a];
  // skip the atomic when first_row is already <= row: a stale (L1) read can
  // only be HIGHER than the true value (first_row only decreases), so the
  // skip is always safe
  if (sl->first_row > row) atomicMin(&sl->first_row, row);
  if (vvalid) {
    // sum.rs:90-115: SUM adds non-null args; valid-ness latches on.
    // sum validity is implied by cnt>0 (same-column agg set; engine.cpp
    // enforces SUM/COUNT share the argument column)
    sum_accum(&sl->sum, val, t.sum_int);
    atomicAdd(&sl->cnt, 1ull);  // count.rs:90-149: COUNT(arg) non-null
    if (t.mm) {  // maxmin.rs:104-119: MIN/MAX of non-null args
      uint64_t u = val_omap(val, t.sum_int);
      atomicMin(&t.mm[2 * a], u);
      atomicMax(&t.mm[2 * a + 1], u);
    }
  }
  if (t.f_row) {  // pass A: FIRST = any row, FIRST_IGNORES_NULL = valid rows
    atomicMin(&t.f_row[2 * a], (unsigned long long)row);
    if (vvalid) atomicMin(&t.f_row[2 * a + 1], (unsigned long long)row);
  }
  if (t.c_key && vvalid) coll_append(t, key, knull, row, val);
}

// The dominant kernel. PMC evidence (profiles/): one-row-at-a-time leaves
// waves 63% parked on the ~L3-latency probe chain (79% L2 miss on the 128 MB
// slot array). Strip-mined RPT rows per thread: the RPT key/val loads and the
// RPT first-probe loads issue back-to-back (independent), so each lane keeps
// RPT random lines in flight instead of one; the fast path (first probe hits
// or claims) resolves in-line and atomics are fire-and-forget.
static constexpr int RPT = 8;

__global__ void k_agg_update(const AggTable t, const int64_t* __restrict__ keys,
                             const uint8_t* __restrict__ key_valid,
                             const double* __restrict__ vals,
                             const uint8_t* __restrict__ val_valid, int64_t n,
                             uint64_t row_offset) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t mask = t.cap - 1;
  int64_t i0 = tid;
  for (; i0 + (RPT - 1) * stride < n; i0 += RPT * stride) {
    int64_t key[RPT];
    double val[RPT];
    bool knull[RPT], vvalid[RPT];
    int64_t slot[RPT];
#pragma unroll
    for (int k = 0; k < RPT; k++) {
      int64_t i = i0 + k * stride;
      key[k] = keys[i];
      val[k] = vals[i];
      knull[k] = key_valid && !bit_get_dev(key_valid, i);
      vvalid[k] = !val_valid || bit_get_dev(val_valid, i);
    }
#pragma unroll
    for (int k = 0; k < RPT; k++)
      slot[k] = (int64_t)(mix64((uint64_t)key[k]) & (uint64_t)mask);
    long long probe[RPT];
#pragma unroll
    for (int k = 0; k < RPT; k++) probe[k] = t.slots[slot[k]].key;  // in flight
#pragma unroll
    for (int k = 0; k < RPT; k++) {
      int64_t i = i0 + k * stride;
      uint64_t row = row_offset + (uint64_t)i;
      if (!knull[k] && key[k] != KEY_EMPTY && probe[k] == key[k]) {
        // fast path: first probe hit
        AggSlot* sl = &t.slots[slot[k]];
        if (sl->first_row > row) atomicMin(&sl->first_row, row);
        if (vvalid[k]) {
          sum_accum(&sl->sum, val[k], t.sum_int);
          atomicAdd(&sl->cnt, 1ull);
          if (t.mm) {
            uint64_t u = val_omap(val[k], t.sum_int);
            atomicMin(&t.mm[2 * slot[k]], u);
            atomicMax(&t.mm[2 * slot[k] + 1], u);
          }
        }
        if (t.f_row) {
          atomicMin(&t.f_row[2 * slot[k]], (unsigned long long)row);
          if (vvalid[k])
            atomicMin(&t.f_row[2 * slot[k] + 1], (unsigned long long)row);
        }
        if (t.c_key && vvalid[k])
          coll_append(t, key[k], false, row, val[k]);
      } else {
        agg_accum_row(t, knull[k] ? 0 : key[k], knull[k], val[k], vvalid[k],
                      row);
      }
    }
  }
  for (int64_t i = i0; i < n; i += stride) {
    bool kn = key_valid && !bit_get_dev(key_valid, i);
    bool vv = !val_valid || bit_get_dev(val_valid, i);
    agg_accum_row(t, kn ? 0 : keys[i], kn, vals[i], vv,
                  row_offset + (uint64_t)i);
  }
}

__global__ void k_agg_merge_frozen(const AggTable t,
                                   const int64_t* __restrict__ keys,
                                   const uint8_t* __restrict__ key_valid,
                                   const uint8_t* __restrict__ acc_data,
                                   const int32_t* __restrict__ acc_offsets,
                                   int64_t n, uint64_t row_offset,
                                   uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get_dev(key_valid, i);
    int64_t a = agg_upsert_slot(t, knull ? 0 : keys[i], knull);
    if (a < 0) continue;
    AggSlot* sl = &t.slots[a];
    uint64_t row = row_offset + (uint64_t)i;
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
    AccSnap acc;
    agg_parse_frozen(layout, acc_data + acc_offsets[i], &acc, t.sum_int);
    if (acc.valid) sum_accum(&sl->sum, acc.sum, t.sum_int);  // sum.rs:117-145
    if (acc.cnt) atomicAdd(&sl->cnt, acc.cnt);
    if (t.mm) {  // maxmin.rs:196-216 partial_merge; sentinels are no-ops
      atomicMin(&t.mm[2 * a], acc.minu);
      atomicMax(&t.mm[2 * a + 1], acc.maxu);
    }
    if (t.f_row) {  // pass A: earliest TOUCHED record wins (first.rs:198-207)
      if (acc.f_st) atomicMin(&t.f_row[2 * a], (unsigned long long)row);
      if (acc.fn_st) atomicMin(&t.f_row[2 * a + 1], (unsigned long long)row);
    }
    if (t.c_key && acc.c_cnt) {  // collect.rs:139-158: concatenate in merge
      for (uint32_t j = 0; j < acc.c_cnt; j++) {  // order, items in order
        double v;
        memcpy(&v, acc.c_raw + (size_t)j * 8, 8);
        coll_append(t, knull ? 0 : keys[i], knull, (row << 20) | j, v);
      }
    }
  }
}

__global__ void k_agg_merge_spill(const AggTable t,
                                  const int64_t* __restrict__ keys,
                                  const uint8_t* __restrict__ acc_data,
                                  const int32_t* __restrict__ acc_offsets,
                                  const unsigned long long* __restrict__ first_rows,
                                  int64_t n, uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    // special groups (null-key / i64::MIN-key) are never spilled — they stay
    // resident in the table's trailing slots across spill resets — so every
    // spilled key is a regular value here.
    int64_t key = keys[i];
    int32_t off = acc_offsets[i];
    int64_t a = agg_upsert_slot(t, key, false);
    if (a < 0) continue;
    AggSlot* sl = &t.slots[a];
    unsigned long long fr = first_rows[i];
    if (sl->first_row > fr) atomicMin(&sl->first_row, fr);
    AccSnap acc;
    agg_parse_frozen(layout, acc_data + off, &acc, t.sum_int);
    if (acc.valid) sum_accum(&sl->sum, acc.sum, t.sum_int);
    if (acc.cnt) atomicAdd(&sl->cnt, acc.cnt);
    if (t.mm) {
      atomicMin(&t.mm[2 * a], acc.minu);
      atomicMax(&t.mm[2 * a + 1], acc.maxu);
    }
    if (t.f_row) {  // priority = the record's preserved group first_row
      if (acc.f_st) atomicMin(&t.f_row[2 * a], fr);
      if (acc.fn_st) atomicMin(&t.f_row[2 * a + 1], fr);
    }
    if (t.c_key && acc.c_cnt) {  // spilled records carry the item runs;
      for (uint32_t j = 0; j < acc.c_cnt; j++) {  // prio keeps record order
        double v;
        memcpy(&v, acc.c_raw + (size_t)j * 8, 8);
        coll_append(t, key, false, (fr << 20) | j, v);
      }
    }
  }
}

// refill the pool with the resident special groups' gathered items
// (engine.cpp reset_collect_pool): front = i64::MIN-key group, back =
// null-key group, prios re-assigned 0..m (order preserved)
__global__ void k_coll_refill(const AggTable t,
                              const unsigned long long* __restrict__ items,
                              int64_t m0, int64_t m1) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < m0 + m1; i += (int64_t)gridDim.x * blockDim.x) {
    if (i < m0) {
      t.c_key[i] = INT64_MIN;
      t.c_prio[i] = (unsigned long long)i;
      t.c_val[i] = items[i];
    } else {
      int64_t p = i - m0;
      int64_t at = t.c_cap - 1 - p;
      t.c_key[at] = 0;
      t.c_prio[at] = (unsigned long long)p;
      t.c_val[at] = items[i];
    }
  }
}

void launch_coll_refill(const AggTable& t, const unsigned long long* items,
                        int64_t m0, int64_t m1, hipStream_t s) {
  if (m0 + m1 == 0) return;
  hipLaunchKernelGGL(k_coll_refill, dim3(grid_for(m0 + m1)), dim3(BLOCK), 0,
                     s, t, items, m0, m1);
  check_launch("k_coll_refill");
}

void launch_agg_merge_spill(const AggTable& t, const int64_t* keys,
                            const uint8_t* acc_data, const int32_t* acc_offsets,
                            const unsigned long long* first_rows, int64_t n,
                            uint32_t layout, hipStream_t s) {
  hipLaunchKernelGGL(k_agg_merge_spill, dim3(grid_for(n)), dim3(BLOCK), 0, s, t,
                     keys, acc_data, acc_offsets, first_rows, n, layout);
  check_launch("k_agg_merge_spill");
}

__global__ void k_agg_compact(const AggTable t, uint32_t* __restrict__ out_slot,
                              unsigned long long* __restrict__ out_first_row,
                              unsigned long long* __restrict__ num_out) {
  int64_t total = t.cap + 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool used = (i < t.cap) ? (t.slots[i].key != KEY_EMPTY)
                            : (t.special_used[i - t.cap] != 0);
    if (used) {
      unsigned long long idx = atomicAdd(num_out, 1ull);
      out_slot[idx] = (uint32_t)i;
      out_first_row[idx] = t.slots[i].first_row;
    }
  }
}

__global__ void k_agg_gather_out(const AggTable t,
                                 const uint32_t* __restrict__ order_slots,
                                 int64_t num_groups, int64_t* __restrict__ keys,
                                 uint8_t* __restrict__ key_validity,
                                 double* __restrict__ sums,
                                 uint8_t* __restrict__ sum_validity,
                                 long long* __restrict__ counts,
                                 double* __restrict__ mins,
                                 uint8_t* __restrict__ min_validity,
                                 double* __restrict__ maxs,
                                 uint8_t* __restrict__ max_validity) {
  // one thread per output byte-group of 8 rows for validity bitmaps
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < num_groups;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t s = order_slots[i];
    if (keys) keys[i] = (s < t.cap) ? t.slots[s].key
                                    : (s == t.cap ? KEY_EMPTY : 0);
    if (sums) sums[i] = t.slots[s].sum;
    if (counts) counts[i] = (long long)t.slots[s].cnt;
    if (mins) mins[i] = val_omap_inv(t.mm[2 * s], t.sum_int);
    if (maxs) maxs[i] = val_omap_inv(t.mm[2 * s + 1], t.sum_int);
    if ((i & 7) == 0) {
      uint8_t kb = 0, sb = 0, mb = 0, xb = 0;
      for (int j = 0; j < 8 && i + j < num_groups; j++) {
        uint32_t sj = order_slots[i + j];
        bool knull = (sj == t.cap + 1);
        if (!knull) kb |= (uint8_t)(1u << j);
        if (t.slots[sj].cnt != 0) sb |= (uint8_t)(1u << j);
        if (t.mm) {
          if (t.mm[2 * sj] != MM_MIN_INIT) mb |= (uint8_t)(1u << j);
          if (t.mm[2 * sj + 1] != MM_MAX_INIT) xb |= (uint8_t)(1u << j);
        }
      }
      if (key_validity) key_validity[i >> 3] = kb;
      if (sum_validity) sum_validity[i >> 3] = sb;
      if (min_validity) min_validity[i >> 3] = mb;
      if (max_validity) max_validity[i >> 3] = xb;
    }
  }
}

// snapshot one slot's accumulators for freeze. When the collect pool is
// active (and sorted into {c_key ascending, prio ascending within key} by
// the engine before any freeze launch), the group's value run is located by
// binary search; the null-key special group owns the back segment.
__device__ __forceinline__ AccSnap table_snap(const AggTable& t, uint32_t s) {
  AccSnap a;
  a.valid = t.slots[s].cnt != 0;
  a.sum = t.slots[s].sum;
  a.cnt = t.slots[s].cnt;
  if (t.mm) {
    a.minu = t.mm[2 * s];
    a.maxu = t.mm[2 * s + 1];
  }
  if (t.f_row) {
    a.f_st = t.f_st[2 * s];
    a.f_val = t.f_val[2 * s];
    a.fn_st = t.f_st[2 * s + 1];
    a.fn_val = t.f_val[2 * s + 1];
  }
  if (t.c_key) {
    if (s == (uint32_t)t.cap + 1) {  // null-key group = back segment,
      int64_t n1 = (int64_t)t.c_n[1];  // already prio-sorted in place
      a.c_vals = t.c_val + (t.c_cap - n1);
      a.c_cnt = (uint32_t)n1;
    } else {
      long long key = (s == (uint32_t)t.cap) ? (long long)INT64_MIN
                                             : t.slots[s].key;
      int64_t n0 = (int64_t)t.c_n[0];
      int64_t lo = 0, hi = n0;
      while (lo < hi) {  // lower bound
        int64_t mid = (lo + hi) >> 1;
        if (t.c_key[mid] < key) lo = mid + 1; else hi = mid;
      }
      int64_t lo2 = lo, hi2 = n0;
      while (lo2 < hi2) {  // upper bound
        int64_t mid = (lo2 + hi2) >> 1;
        if (t.c_key[mid] <= key) lo2 = mid + 1; else hi2 = mid;
      }
      a.c_vals = t.c_val + lo;
      a.c_cnt = (uint32_t)(lo2 - lo);
    }
  }
  return a;
}

__global__ void k_agg_freeze_len(const AggTable t,
                                 const uint32_t* __restrict__ order_slots,
                                 int64_t num_groups, int32_t* __restrict__ lens,
                                 uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < num_groups;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t s = order_slots[i];
    lens[i] = agg_freeze_len(layout, table_snap(t, s));  // len is type-free
  }
}

__global__ void k_agg_freeze_write(const AggTable t,
                                   const uint32_t* __restrict__ order_slots,
                                   int64_t num_groups,
                                   const int32_t* __restrict__ offsets,
                                   uint8_t* __restrict__ out, uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < num_groups;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t s = order_slots[i];
    agg_freeze_write_rec(layout, table_snap(t, s), out + offsets[i],
                         t.sum_int);
  }
}

void launch_agg_update(const AggTable& t, const int64_t* keys,
                       const uint8_t* key_valid, const double* vals,
                       const uint8_t* val_valid, int64_t n, uint64_t row_offset,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_agg_update, dim3(grid_for(n)), dim3(BLOCK), 0, s, t, keys,
                     key_valid, vals, val_valid, n, row_offset);
  check_launch("k_agg_update");
}
void launch_agg_merge_frozen(const AggTable& t, const int64_t* keys,
                             const uint8_t* key_valid, const uint8_t* acc_data,
                             const int32_t* acc_offsets, int64_t n,
                             uint64_t row_offset, uint32_t layout,
                             hipStream_t s) {
  hipLaunchKernelGGL(k_agg_merge_frozen, dim3(grid_for(n)), dim3(BLOCK), 0, s, t,
                     keys, key_valid, acc_data, acc_offsets, n, row_offset,
                     layout);
  check_launch("k_agg_merge_frozen");
}
void launch_agg_compact(const AggTable& t, uint32_t* out_slot,
                        unsigned long long* out_first_row,
                        unsigned long long* num_out, hipStream_t s) {
  hipLaunchKernelGGL(k_agg_compact, dim3(grid_for(t.cap + 2)), dim3(BLOCK), 0, s,
                     t, out_slot, out_first_row, num_out);
  check_launch("k_agg_compact");
}
void launch_agg_gather_out(const AggTable& t, const uint32_t* order_slots,
                           int64_t num_groups, int64_t* out_keys,
                           uint8_t* out_key_validity, double* out_sums,
                           uint8_t* out_sum_validity, long long* out_counts,
                           double* out_mins, uint8_t* out_min_validity,
                           double* out_maxs, uint8_t* out_max_validity,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_agg_gather_out, dim3(grid_for(num_groups)), dim3(BLOCK),
                     0, s, t, order_slots, num_groups, out_keys,
                     out_key_validity, out_sums, out_sum_validity, out_counts,
                     out_mins, out_min_validity, out_maxs, out_max_validity);
  check_launch("k_agg_gather_out");
}
void launch_agg_freeze_len(const AggTable& t, const uint32_t* order_slots,
                           int64_t num_groups, int32_t* lens, uint32_t layout,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_agg_freeze_len, dim3(grid_for(num_groups)), dim3(BLOCK),
                     0, s, t, order_slots, num_groups, lens, layout);
  check_launch("k_agg_freeze_len");
}
void launch_agg_freeze_write(const AggTable& t, const uint32_t* order_slots,
                             int64_t num_groups, const int32_t* offsets,
                             uint8_t* out, uint32_t layout, hipStream_t s) {
  hipLaunchKernelGGL(k_agg_freeze_write, dim3(grid_for(num_groups)), dim3(BLOCK),
                     0, s, t, order_slots, num_groups, offsets, out, layout);
  check_launch("k_agg_freeze_write");
}

// partial-skipping pass-through freeze (agg_ctx.rs:428-462
// process_partial_skipped): each input row becomes its own single-row group;
// SUM acc = value when valid, COUNT = valid ? 1 : 0, MIN = MAX = value,
// FIRST = value-or-null (touched), FIRST_IGNORES_NULL = value when valid.
__device__ __forceinline__ AccSnap row_snap(bool v, double val,
                                            bool is_int = false) {
  AccSnap a;
  a.valid = v;
  a.sum = val;
  a.cnt = v ? 1 : 0;
  uint64_t u = val_omap(val, is_int);
  a.minu = v ? u : MM_MIN_INIT;
  a.maxu = v ? u : MM_MAX_INIT;
  a.f_st = v ? 2 : 1;  // a row always touches its own group's FIRST
  a.f_val = val;
  a.fn_st = v ? 2 : 0;
  a.fn_val = val;
  a.c_cnt = v ? 1 : 0;  // caller points c_vals at the row's value bits
  return a;
}

__global__ void k_skip_freeze_len(const uint8_t* __restrict__ val_valid,
                                  int64_t n, int32_t* __restrict__ lens,
                                  uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool v = !val_valid || bit_get_dev(val_valid, i);
    lens[i] = agg_freeze_len(layout, row_snap(v, 0.0));
  }
}

__global__ void k_skip_freeze_write(const double* __restrict__ vals,
                                    const uint8_t* __restrict__ val_valid,
                                    int64_t n, const int32_t* __restrict__ offsets,
                                    uint8_t* __restrict__ out, uint32_t layout,
                                    int is_int) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool v = !val_valid || bit_get_dev(val_valid, i);
    AccSnap a = row_snap(v, v ? vals[i] : 0.0, is_int);
    unsigned long long vb;
    memcpy(&vb, &vals[i], 8);
    a.c_vals = &vb;  // single-item list for the row's own group
    agg_freeze_write_rec(layout, a, out + offsets[i], is_int);
  }
}

void launch_skip_freeze_len(const uint8_t* val_valid, int64_t n, int32_t* lens,
                            uint32_t layout, hipStream_t s) {
  hipLaunchKernelGGL(k_skip_freeze_len, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     val_valid, n, lens, layout);
  check_launch("k_skip_freeze_len");
}
void launch_skip_freeze_write(const double* vals, const uint8_t* val_valid,
                              int64_t n, const int32_t* offsets, uint8_t* out,
                              uint32_t layout, int is_int, hipStream_t s) {
  hipLaunchKernelGGL(k_skip_freeze_write, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     vals, val_valid, n, offsets, out, layout, is_int);
  check_launch("k_skip_freeze_write");
}

// ---- shuffle partition -----------------------------------------------------
__global__ void k_histogram(const uint32_t* __restrict__ part_ids, int64_t n,
                            uint32_t P, uint32_t* __restrict__ counts) {
  extern __shared__ uint32_t lds_counts[];
  for (uint32_t p = threadIdx.x; p < P; p += blockDim.x) lds_counts[p] = 0;
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    atomicAdd(&lds_counts[part_ids[i]], 1u);
  __syncthreads();
  for (uint32_t p = threadIdx.x; p < P; p += blockDim.x)
    if (lds_counts[p]) atomicAdd(&counts[p], lds_counts[p]);
}

void launch_histogram(const uint32_t* part_ids, int64_t n, uint32_t P,
                      uint32_t* counts, hipStream_t s) {
  size_t lds = P * sizeof(uint32_t);
  hipLaunchKernelGGL(k_histogram, dim3(grid_for(n)), dim3(BLOCK), lds, s,
                     part_ids, n, P, counts);
  check_launch("k_histogram");
}

__global__ void k_slots_init(AggSlot* __restrict__ slots, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    slots[i].key = KEY_EMPTY;
    slots[i].cnt = 0;
    slots[i].sum = 0.0;
    slots[i].first_row = ~0ull;
  }
}

void launch_slots_init(AggSlot* slots, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_slots_init, dim3(grid_for(n)), dim3(BLOCK), 0, s, slots,
                     n);
  check_launch("k_slots_init");
}

__global__ void k_mm_init(unsigned long long* __restrict__ mm, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    mm[2 * i] = MM_MIN_INIT;
    mm[2 * i + 1] = MM_MAX_INIT;
  }
}

void launch_mm_init(unsigned long long* mm, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_mm_init, dim3(grid_for(n)), dim3(BLOCK), 0, s, mm, n);
  check_launch("k_mm_init");
}

__global__ void k_first_init(unsigned long long* __restrict__ f_row,
                             double* __restrict__ f_val,
                             uint8_t* __restrict__ f_st, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < 2 * n;
       i += (int64_t)gridDim.x * blockDim.x) {
    f_row[i] = ~0ull;
    f_val[i] = 0.0;
    f_st[i] = 0;
  }
}

void launch_first_init(unsigned long long* f_row, double* f_val, uint8_t* f_st,
                       int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_first_init, dim3(grid_for(2 * n)), dim3(BLOCK), 0, s,
                     f_row, f_val, f_st, n);
  check_launch("k_first_init");
}

// FIRST-family pass B (see kernels.h): the unique row whose priority equals
// the slot's captured f_row plain-stores its value and state. Runs after the
// chunk's pass A on the same stream.
__global__ void k_first_capture_update(const AggTable t,
                                       const int64_t* __restrict__ keys,
                                       const uint8_t* __restrict__ key_valid,
                                       const double* __restrict__ vals,
                                       const uint8_t* __restrict__ val_valid,
                                       int64_t n, uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get_dev(key_valid, i);
    int64_t a = agg_upsert_slot(t, knull ? 0 : keys[i], knull);
    if (a < 0) continue;
    unsigned long long row = row_offset + (uint64_t)i;
    bool vv = !val_valid || bit_get_dev(val_valid, i);
    if (t.f_row[2 * a] == row) {  // first.rs:91-148: first row latches,
      t.f_val[2 * a] = vv ? vals[i] : 0.0;  // null first value => touched-null
      t.f_st[2 * a] = vv ? 2 : 1;
    }
    if (vv && t.f_row[2 * a + 1] == row) {  // first_ignores_null.rs:83-117
      t.f_val[2 * a + 1] = vals[i];
      t.f_st[2 * a + 1] = 2;
    }
  }
}

void launch_first_capture_update(const AggTable& t, const int64_t* keys,
                                 const uint8_t* key_valid, const double* vals,
                                 const uint8_t* val_valid, int64_t n,
                                 uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_first_capture_update, dim3(grid_for(n)), dim3(BLOCK), 0,
                     s, t, keys, key_valid, vals, val_valid, n, row_offset);
  check_launch("k_first_capture_update");
}

__global__ void k_first_capture_frozen(const AggTable t,
                                       const int64_t* __restrict__ keys,
                                       const uint8_t* __restrict__ key_valid,
                                       const uint8_t* __restrict__ acc_data,
                                       const int32_t* __restrict__ acc_offsets,
                                       const unsigned long long* __restrict__ prio,
                                       int64_t n, uint64_t row_offset,
                                       uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get_dev(key_valid, i);
    int64_t a = agg_upsert_slot(t, knull ? 0 : keys[i], knull);
    if (a < 0) continue;
    unsigned long long pr = prio ? prio[i] : row_offset + (uint64_t)i;
    AccSnap acc;
    agg_parse_frozen(layout, acc_data + acc_offsets[i], &acc, t.sum_int);
    if (acc.f_st && t.f_row[2 * a] == pr) {
      t.f_val[2 * a] = acc.f_val;
      t.f_st[2 * a] = acc.f_st;
    }
    if (acc.fn_st && t.f_row[2 * a + 1] == pr) {
      t.f_val[2 * a + 1] = acc.fn_val;
      t.f_st[2 * a + 1] = acc.fn_st;
    }
  }
}

void launch_first_capture_frozen(const AggTable& t, const int64_t* keys,
                                 const uint8_t* key_valid,
                                 const uint8_t* acc_data,
                                 const int32_t* acc_offsets,
                                 const unsigned long long* prio, int64_t n,
                                 uint64_t row_offset, uint32_t layout,
                                 hipStream_t s) {
  hipLaunchKernelGGL(k_first_capture_frozen, dim3(grid_for(n)), dim3(BLOCK), 0,
                     s, t, keys, key_valid, acc_data, acc_offsets, prio, n,
                     row_offset, layout);
  check_launch("k_first_capture_frozen");
}

__global__ void k_first_gather(const AggTable t,
                               const uint32_t* __restrict__ order_slots,
                               int64_t num_groups, int which,
                               double* __restrict__ out_vals,
                               uint8_t* __restrict__ out_validity) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < num_groups; i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t s = order_slots[i];
    out_vals[i] = t.f_val[2 * s + which];
    if ((i & 7) == 0) {
      uint8_t b = 0;
      for (int j = 0; j < 8 && i + j < num_groups; j++)
        if (t.f_st[2 * order_slots[i + j] + which] == 2)
          b |= (uint8_t)(1u << j);
      out_validity[i >> 3] = b;
    }
  }
}

void launch_first_gather(const AggTable& t, const uint32_t* order_slots,
                         int64_t num_groups, int which, double* out_vals,
                         uint8_t* out_validity, hipStream_t s) {
  hipLaunchKernelGGL(k_first_gather, dim3(grid_for(num_groups)), dim3(BLOCK),
                     0, s, t, order_slots, num_groups, which, out_vals,
                     out_validity);
  check_launch("k_first_gather");
}

__global__ void k_gather_u64_idx(const unsigned long long* __restrict__ src,
                                 const uint32_t* __restrict__ idx, int64_t n,
                                 unsigned long long* __restrict__ dst) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = src[idx[i]];
}

void launch_gather_u64_idx(const unsigned long long* src, const uint32_t* idx,
                           int64_t n, unsigned long long* dst, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_u64_idx, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     src, idx, n, dst);
  check_launch("k_gather_u64_idx");
}

// sign-bias i64 keys so unsigned radix order == signed order
__global__ void k_bias_i64(const unsigned long long* __restrict__ src,
                           int64_t n, unsigned long long* __restrict__ dst) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = src[i] ^ 0x8000000000000000ull;
}

void launch_bias_i64(const unsigned long long* src, int64_t n,
                     unsigned long long* dst, hipStream_t s) {
  hipLaunchKernelGGL(k_bias_i64, dim3(grid_for(n)), dim3(BLOCK), 0, s, src, n,
                     dst);
  check_launch("k_bias_i64");
}

// per-group collect counts / values (emit side; pool sorted by the engine)
__global__ void k_coll_counts(const AggTable t,
                              const uint32_t* __restrict__ order_slots,
                              int64_t num_groups, int32_t* __restrict__ cnts) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < num_groups; i += (int64_t)gridDim.x * blockDim.x)
    cnts[i] = (int32_t)table_snap(t, order_slots[i]).c_cnt;
}

void launch_coll_counts(const AggTable& t, const uint32_t* order_slots,
                        int64_t num_groups, int32_t* cnts, hipStream_t s) {
  hipLaunchKernelGGL(k_coll_counts, dim3(grid_for(num_groups)), dim3(BLOCK), 0,
                     s, t, order_slots, num_groups, cnts);
  check_launch("k_coll_counts");
}

__global__ void k_coll_gather(const AggTable t,
                              const uint32_t* __restrict__ order_slots,
                              int64_t num_groups,
                              const int32_t* __restrict__ offsets,
                              unsigned long long* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < num_groups; i += (int64_t)gridDim.x * blockDim.x) {
    AccSnap a = table_snap(t, order_slots[i]);
    unsigned long long* dst = out + offsets[i];
    for (uint32_t j = 0; j < a.c_cnt; j++) dst[j] = a.c_vals[j];
  }
}

void launch_coll_gather(const AggTable& t, const uint32_t* order_slots,
                        int64_t num_groups, const int32_t* offsets,
                        unsigned long long* out, hipStream_t s) {
  hipLaunchKernelGGL(k_coll_gather, dim3(grid_for(num_groups)), dim3(BLOCK), 0,
                     s, t, order_slots, num_groups, offsets, out);
  check_launch("k_coll_gather");
}

// COLLECT_SET dedup helpers: entries sorted {key, value, prio} — mark each
// (key,value) run head (it carries the minimum prio = first occurrence,
// AccSet.append semantics: only novel values append), then compact by the
// scanned positions.
__global__ void k_coll_mark_heads(const long long* __restrict__ key,
                                  const unsigned long long* __restrict__ val,
                                  int64_t n, int use_key,
                                  uint8_t* __restrict__ mark) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    mark[i] = (i == 0 || val[i] != val[i - 1] ||
               (use_key && key[i] != key[i - 1]))
                  ? 1
                  : 0;
}

void launch_coll_mark_heads(const long long* key,
                            const unsigned long long* val, int64_t n,
                            int use_key, uint8_t* mark, hipStream_t s) {
  hipLaunchKernelGGL(k_coll_mark_heads, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     key, val, n, use_key, mark);
  check_launch("k_coll_mark_heads");
}

__global__ void k_compact_u64(const unsigned long long* __restrict__ src,
                              const uint8_t* __restrict__ mark,
                              const uint32_t* __restrict__ pos, int64_t n,
                              unsigned long long* __restrict__ dst) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    if (mark[i]) dst[pos[i]] = src[i];
}

void launch_compact_u64(const unsigned long long* src, const uint8_t* mark,
                        const uint32_t* pos, int64_t n,
                        unsigned long long* dst, hipStream_t s) {
  hipLaunchKernelGGL(k_compact_u64, dim3(grid_for(n)), dim3(BLOCK), 0, s, src,
                     mark, pos, n, dst);
  check_launch("k_compact_u64");
}

__global__ void k_iota_u32(uint32_t* __restrict__ dst, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    dst[i] = (uint32_t)i;
}

void launch_iota_u32(uint32_t* dst, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_iota_u32, dim3(grid_for(n)), dim3(BLOCK), 0, s, dst, n);
  check_launch("k_iota_u32");
}

__global__ void k_agg_rebuild(const AggTable dst, const AggTable src) {
  int64_t total = src.cap + 2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool used = (i < src.cap) ? (src.slots[i].key != KEY_EMPTY)
                              : (src.special_used[i - src.cap] != 0);
    if (!used) continue;
    int64_t a;
    if (i >= src.cap) {
      int which = (int)(i - src.cap);
      if (atomicCAS(&dst.special_used[which], 0u, 1u) == 0u)
        atomicAdd(dst.num_groups, 1ull);
      a = dst.cap + which;
    } else {
      a = agg_upsert_slot(dst, src.slots[i].key, false);
      if (a < 0) continue;
    }
    // distinct keys per source slot: single writer, plain stores
    dst.slots[a].sum = src.slots[i].sum;
    dst.slots[a].cnt = src.slots[i].cnt;
    dst.slots[a].first_row = src.slots[i].first_row;
    if (src.mm) {
      dst.mm[2 * a] = src.mm[2 * i];
      dst.mm[2 * a + 1] = src.mm[2 * i + 1];
    }
    if (src.f_row) {
      for (int j = 0; j < 2; j++) {
        dst.f_row[2 * a + j] = src.f_row[2 * i + j];
        dst.f_val[2 * a + j] = src.f_val[2 * i + j];
        dst.f_st[2 * a + j] = src.f_st[2 * i + j];
      }
    }
  }
}

void launch_agg_rebuild(const AggTable& dst, const AggTable& src, hipStream_t s) {
  hipLaunchKernelGGL(k_agg_rebuild, dim3(grid_for(src.cap + 2)), dim3(BLOCK), 0,
                     s, dst, src);
  check_launch("k_agg_rebuild");
}

void sort_pairs_u64_u32(const unsigned long long* keys_in, const uint32_t* vals_in,
                        unsigned long long* keys_out, uint32_t* vals_out,
                        int64_t n, void* temp, size_t* temp_bytes, hipStream_t s) {
  hipError_t e = rocprim::radix_sort_pairs(temp, *temp_bytes, keys_in, keys_out,
                                           vals_in, vals_out, (size_t)n, 0, 64, s);
  if (e != hipSuccess) abort();
}

void sort_pairs_u32_u32(const uint32_t* keys_in, const uint32_t* vals_in,
                        uint32_t* keys_out, uint32_t* vals_out, int64_t n,
                        int end_bit, void* temp, size_t* temp_bytes,
                        hipStream_t s) {
  hipError_t e = rocprim::radix_sort_pairs(temp, *temp_bytes, keys_in, keys_out,
                                           vals_in, vals_out, (size_t)n, 0,
                                           end_bit, s);
  if (e != hipSuccess) abort();
}

__global__ void k_gather_8(const uint8_t* __restrict__ src,
                           const uint32_t* __restrict__ perm, int64_t n,
                           uint8_t* __restrict__ dst) {
  const uint64_t* s64 = (const uint64_t*)src;
  uint64_t* d64 = (uint64_t*)dst;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    d64[i] = s64[perm[i]];
}

__global__ void k_gather_bits(const uint8_t* __restrict__ src_bits,
                              const uint32_t* __restrict__ perm, int64_t n,
                              uint8_t* __restrict__ dst_bits) {
  int64_t nbytes = (n + 7) / 8;
  for (int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; b < nbytes;
       b += (int64_t)gridDim.x * blockDim.x) {
    uint8_t out = 0;
    for (int j = 0; j < 8; j++) {
      int64_t i = b * 8 + j;
      if (i < n && bit_get_dev(src_bits, perm[i])) out |= (uint8_t)(1u << j);
    }
    dst_bits[b] = out;
  }
}

__global__ void k_gather_lens(const int32_t* __restrict__ src_offsets,
                              const uint32_t* __restrict__ perm, int64_t n,
                              int32_t* __restrict__ dst_lens) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t p = perm[i];
    dst_lens[i] = src_offsets[p + 1] - src_offsets[p];
  }
}

__global__ void k_gather_bytes(const uint8_t* __restrict__ src_data,
                               const int32_t* __restrict__ src_offsets,
                               const uint32_t* __restrict__ perm,
                               const int32_t* __restrict__ dst_offsets,
                               int64_t n, uint8_t* __restrict__ dst_data) {
  // one wave per row run; byte-copy with lane stride
  for (int64_t i = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
       i < n; i += (int64_t)gridDim.x * (blockDim.x / 64)) {
    int lane = threadIdx.x & 63;
    uint32_t p = perm[i];
    int32_t len = src_offsets[p + 1] - src_offsets[p];
    const uint8_t* s = src_data + src_offsets[p];
    uint8_t* d = dst_data + dst_offsets[i];
    for (int32_t b = lane; b < len; b += 64) d[b] = s[b];
  }
}

void launch_gather_8(const uint8_t* src, const uint32_t* perm, int64_t n,
                     uint8_t* dst, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_8, dim3(grid_for(n)), dim3(BLOCK), 0, s, src, perm,
                     n, dst);
  check_launch("k_gather_8");
}
void launch_gather_bits(const uint8_t* src_bits, const uint32_t* perm, int64_t n,
                        uint8_t* dst_bits, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_bits, dim3(grid_for((n + 7) / 8)), dim3(BLOCK), 0,
                     s, src_bits, perm, n, dst_bits);
  check_launch("k_gather_bits");
}
void launch_gather_lens(const int32_t* src_offsets, const uint32_t* perm,
                        int64_t n, int32_t* dst_lens, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_lens, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     src_offsets, perm, n, dst_lens);
  check_launch("k_gather_lens");
}
void launch_gather_bytes(const uint8_t* src_data, const int32_t* src_offsets,
                         const uint32_t* perm, const int32_t* dst_offsets,
                         int64_t n, uint8_t* dst_data, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_bytes, dim3(grid_for(n * 64)), dim3(BLOCK), 0, s,
                     src_data, src_offsets, perm, dst_offsets, n, dst_data);
  check_launch("k_gather_bytes");
}

__global__ void k_avg_div(const double* __restrict__ sums,
                          const long long* __restrict__ cnts, int64_t n,
                          double* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = cnts[i] ? sums[i] / (double)cnts[i] : 0.0;
}

void launch_avg_div(const double* sums, const long long* cnts, int64_t n,
                    double* out, hipStream_t s) {
  hipLaunchKernelGGL(k_avg_div, dim3(grid_for(n)), dim3(BLOCK), 0, s, sums,
                     cnts, n, out);
  check_launch("k_avg_div");
}

// ---- filter ----------------------------------------------------------------
template <typename T>
__global__ void k_cmp_lit(const T* __restrict__ vals,
                          const uint8_t* __restrict__ valid, int64_t n,
                          int32_t op, T lit, uint8_t* __restrict__ mask,
                          int first) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool v = !valid || bit_get_dev(valid, i);
    bool r = false;
    if (v) {
      T x = vals[i];
      switch (op) {
        case CMP_EQ: r = x == lit; break;
        case CMP_NE: r = x != lit; break;
        case CMP_LT: r = x < lit; break;
        case CMP_LE: r = x <= lit; break;
        case CMP_GT: r = x > lit; break;
        case CMP_GE: r = x >= lit; break;
      }
    }
    mask[i] = first ? (uint8_t)r : (uint8_t)(mask[i] & (uint8_t)r);
  }
}

__global__ void k_is_not_null(const uint8_t* __restrict__ valid, int64_t n,
                              uint8_t* __restrict__ mask, int first,
                              int negate) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool v = !valid || bit_get_dev(valid, i);
    bool r = negate ? !v : v;
    mask[i] = first ? (uint8_t)r : (uint8_t)(mask[i] & (uint8_t)r);
  }
}

__global__ void k_sel_rows(const uint8_t* __restrict__ mask,
                           const uint32_t* __restrict__ positions, int64_t n,
                           uint32_t* __restrict__ sel) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (mask[i]) sel[positions[i]] = (uint32_t)i;
  }
}

__global__ void k_gather_4(const uint8_t* __restrict__ src,
                           const uint32_t* __restrict__ perm, int64_t n,
                           uint8_t* __restrict__ dst) {
  const uint32_t* s32 = (const uint32_t*)src;
  uint32_t* d32 = (uint32_t*)dst;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    d32[i] = s32[perm[i]];
}

void launch_cmp_lit(DType dt, const void* vals, const uint8_t* valid,
                    int64_t n, CmpOp op, int64_t lit_i, double lit_f,
                    uint8_t* mask, bool first, hipStream_t s) {
  switch (dt) {
    case DType::Int32:
      hipLaunchKernelGGL(k_cmp_lit<int32_t>, dim3(grid_for(n)), dim3(BLOCK), 0,
                         s, (const int32_t*)vals, valid, n, op, (int32_t)lit_i,
                         mask, first);
      break;
    case DType::Int64:
      hipLaunchKernelGGL(k_cmp_lit<int64_t>, dim3(grid_for(n)), dim3(BLOCK), 0,
                         s, (const int64_t*)vals, valid, n, op, lit_i, mask,
                         first);
      break;
    case DType::Float64:
      hipLaunchKernelGGL(k_cmp_lit<double>, dim3(grid_for(n)), dim3(BLOCK), 0,
                         s, (const double*)vals, valid, n, op, lit_f, mask,
                         first);
      break;
    default:
      throw std::runtime_error("cmp_lit: unsupported dtype");
  }
  check_launch("k_cmp_lit");
}

void launch_is_not_null(const uint8_t* valid, int64_t n, uint8_t* mask,
                        bool first, bool negate, hipStream_t s) {
  hipLaunchKernelGGL(k_is_not_null, dim3(grid_for(n)), dim3(BLOCK), 0, s, valid,
                     n, mask, first, negate);
  check_launch("k_is_not_null");
}

void scan_mask_u8(const uint8_t* mask, uint32_t* positions, int64_t n,
                  void* temp, size_t* temp_bytes, hipStream_t s) {
  hipError_t e = rocprim::exclusive_scan(
      temp, *temp_bytes, mask, positions, 0u, (size_t)(n + 1),
      rocprim::plus<uint32_t>(), s);
  if (e != hipSuccess) abort();
}

void launch_sel_rows(const uint8_t* mask, const uint32_t* positions, int64_t n,
                     uint32_t* sel, hipStream_t s) {
  hipLaunchKernelGGL(k_sel_rows, dim3(grid_for(n)), dim3(BLOCK), 0, s, mask,
                     positions, n, sel);
  check_launch("k_sel_rows");
}

void launch_gather_4(const uint8_t* src, const uint32_t* perm, int64_t n,
                     uint8_t* dst, hipStream_t s) {
  hipLaunchKernelGGL(k_gather_4, dim3(grid_for(n)), dim3(BLOCK), 0, s, src,
                     perm, n, dst);
  check_launch("k_gather_4");
}

__global__ void k_bits_to_mask(const uint8_t* __restrict__ bits, int64_t n,
                               uint8_t* __restrict__ mask) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    mask[i] = bit_get_dev(bits, i);
}

template <typename T>
__global__ void k_scatter_packed(const T* __restrict__ packed,
                                 const uint32_t* __restrict__ positions,
                                 const uint8_t* __restrict__ mask, int64_t n,
                                 T* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = mask[i] ? packed[positions[i]] : (T)0;
}

void launch_bits_to_mask(const uint8_t* bits, int64_t n, uint8_t* mask,
                         hipStream_t s) {
  hipLaunchKernelGGL(k_bits_to_mask, dim3(grid_for(n)), dim3(BLOCK), 0, s, bits,
                     n, mask);
  check_launch("k_bits_to_mask");
}

void launch_scatter_packed(int width, const uint8_t* packed,
                           const uint32_t* positions, const uint8_t* mask,
                           int64_t n, uint8_t* out, hipStream_t s) {
  if (width == 8) {
    hipLaunchKernelGGL(k_scatter_packed<uint64_t>, dim3(grid_for(n)),
                       dim3(BLOCK), 0, s, (const uint64_t*)packed, positions,
                       mask, n, (uint64_t*)out);
  } else if (width == 4) {
    hipLaunchKernelGGL(k_scatter_packed<uint32_t>, dim3(grid_for(n)),
                       dim3(BLOCK), 0, s, (const uint32_t*)packed, positions,
                       mask, n, (uint32_t*)out);
  } else {
    throw std::runtime_error("scatter_packed: unsupported width");
  }
  check_launch("k_scatter_packed");
}

__global__ void k_cast_i32_f64(const int32_t* __restrict__ in, int64_t n,
                               double* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (double)in[i];
}

void launch_cast_i32_f64(const int32_t* in, int64_t n, double* out,
                         hipStream_t s) {
  hipLaunchKernelGGL(k_cast_i32_f64, dim3(grid_for(n)), dim3(BLOCK), 0, s, in,
                     n, out);
  check_launch("k_cast_i32_f64");
}

__global__ void k_cast_i64_f64(const int64_t* __restrict__ in, int64_t n,
                               double* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (double)in[i];
}

void launch_cast_i64_f64(const int64_t* in, int64_t n, double* out,
                         hipStream_t s) {
  hipLaunchKernelGGL(k_cast_i64_f64, dim3(grid_for(n)), dim3(BLOCK), 0, s, in,
                     n, out);
  check_launch("k_cast_i64_f64");
}

// ---- device repartition helpers (in-memory exchange prep) ------------------
// ord = (pid % world) * P + pid: sorting by ord groups rows by destination
// rank, partition-ordered within each rank (buffered_data.rs:284-351 analog
// with the rank grouping the RCCL all-to-all needs).
__global__ void k_exchange_ord(const uint32_t* __restrict__ pids, int64_t n,
                               uint32_t P, uint32_t world,
                               uint32_t* __restrict__ ord) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t p = pids[i];
    ord[i] = (p % world) * P + p;
  }
}

void launch_exchange_ord(const uint32_t* pids, int64_t n, uint32_t P,
                         uint32_t world, uint32_t* ord, hipStream_t s) {
  hipLaunchKernelGGL(k_exchange_ord, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     pids, n, P, world, ord);
  check_launch("k_exchange_ord");
}

// per-destination-rank row and byte counts (the all-to-all splits); LDS
// staged, world <= 8 on one node
__global__ void k_dest_counts(const uint32_t* __restrict__ pids,
                              const int32_t* __restrict__ offsets, int64_t n,
                              uint32_t world,
                              unsigned long long* __restrict__ rows,
                              unsigned long long* __restrict__ bytes) {
  __shared__ unsigned long long l_rows[64], l_bytes[64];
  for (uint32_t w = threadIdx.x; w < world; w += blockDim.x) {
    l_rows[w] = 0;
    l_bytes[w] = 0;
  }
  __syncthreads();
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t d = pids[i] % world;
    atomicAdd(&l_rows[d], 1ull);
    if (offsets)
      atomicAdd(&l_bytes[d],
                (unsigned long long)(offsets[i + 1] - offsets[i]));
  }
  __syncthreads();
  for (uint32_t w = threadIdx.x; w < world; w += blockDim.x) {
    if (l_rows[w]) atomicAdd(&rows[w], l_rows[w]);
    if (l_bytes[w]) atomicAdd(&bytes[w], l_bytes[w]);
  }
}

void launch_dest_counts(const uint32_t* pids, const int32_t* offsets,
                        int64_t n, uint32_t world, unsigned long long* rows,
                        unsigned long long* bytes, hipStream_t s) {
  if (world > 64) throw std::runtime_error("dest_counts: world > 64");
  hipLaunchKernelGGL(k_dest_counts, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     pids, offsets, n, world, rows, bytes);
  check_launch("k_dest_counts");
}

// ---- parquet RLE/bit-packed run expansion ----------------------------------
// The host walks run HEADERS only (parquet.cpp rle_bp_runs); these kernels do
// the wide work: each output element binary-searches the run table by output
// position and extracts its value — coalesced writes, ~log2(#runs) cheap
// reads. Run layout mirrors struct PqRun (parquet.h, 16 B).
struct RunView {
  uint32_t out_pos, count, src_off;
  uint8_t kind, bw;
  uint16_t _pad;
};

__device__ __forceinline__ uint32_t run_value(const RunView& r,
                                              const uint8_t* __restrict__ bytes,
                                              uint32_t j) {
  if (r.kind == 0) return r.src_off;  // RLE: the value itself
  // bit-packed, LSB-first: extract bw bits at bit position j*bw
  uint64_t bitpos = (uint64_t)j * r.bw;
  uint64_t word;
  memcpy(&word, bytes + r.src_off + (bitpos >> 3), 8);  // 8B tail pad (host)
  uint32_t mask = r.bw >= 32 ? 0xFFFFFFFFu : ((1u << r.bw) - 1u);
  return (uint32_t)(word >> (bitpos & 7)) & mask;
}

__device__ __forceinline__ int run_find(const RunView* __restrict__ runs,
                                        int nruns, uint32_t i) {
  int lo = 0, hi = nruns - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (runs[mid].out_pos <= i) lo = mid; else hi = mid - 1;
  }
  return lo;
}

__global__ void k_runs_expand_u32(const RunView* __restrict__ runs, int nruns,
                                  const uint8_t* __restrict__ bytes, int64_t n,
                                  uint32_t* __restrict__ out) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const RunView& r = runs[run_find(runs, nruns, (uint32_t)i)];
    out[i] = run_value(r, bytes, (uint32_t)i - r.out_pos);
  }
}

void launch_runs_expand_u32(const void* runs, int nruns, const uint8_t* bytes,
                            int64_t n, uint32_t* out, hipStream_t s) {
  hipLaunchKernelGGL(k_runs_expand_u32, dim3(grid_for(n)), dim3(BLOCK), 0, s,
                     (const RunView*)runs, nruns, bytes, n, out);
  check_launch("k_runs_expand_u32");
}

// def levels (bw=1) -> LSB validity bitmap; one thread per output BYTE
__global__ void k_def_expand_validity(const RunView* __restrict__ runs,
                                      int nruns,
                                      const uint8_t* __restrict__ bytes,
                                      int64_t n, uint8_t* __restrict__ bitmap) {
  int64_t nbytes = (n + 7) / 8;
  for (int64_t byte = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       byte < nbytes; byte += (int64_t)gridDim.x * blockDim.x) {
    uint8_t out = 0;
    int64_t base = byte * 8;
    int ri = run_find(runs, nruns, (uint32_t)base);
    for (int b = 0; b < 8 && base + b < n; b++) {
      uint32_t i = (uint32_t)(base + b);
      while (ri + 1 < nruns && runs[ri + 1].out_pos <= i) ri++;
      const RunView& r = runs[ri];
      if (run_value(r, bytes, i - r.out_pos)) out |= (uint8_t)(1u << b);
    }
    bitmap[byte] = out;
  }
}

void launch_def_expand_validity(const void* runs, int nruns,
                                const uint8_t* bytes, int64_t n,
                                uint8_t* bitmap, hipStream_t s) {
  hipLaunchKernelGGL(k_def_expand_validity, dim3(grid_for((n + 7) / 8)),
                     dim3(BLOCK), 0, s, (const RunView*)runs, nruns, bytes, n,
                     bitmap);
  check_launch("k_def_expand_validity");
}

}  // namespace auron
