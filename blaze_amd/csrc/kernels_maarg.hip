// kernels_maarg.hip — multi-argument aggregation: per-agg argument columns
// and per-agg accumulator types, the reference's actual aggregate evaluation
// model (agg.rs:73-169 evaluates independent partial args per agg;
// sum.rs:78-88 / maxmin.rs:81-83 give each agg an accumulator column of ITS
// declared type). The shared-argument fast path (kernels.hip/agg2/agg3)
// remains the north-star pipeline; this mode handles arbitrary agg lists —
// e.g. the reference's own end-to-end golden (agg_exec.rs:493-681): ten
// aggregates over seven distinct argument columns with Int64/Float64/Int32
// accumulators, NULL-literal collect args, and FIRST_IGNORES_NULL.
//
// Slot resolution happens once per row (launch_slots_upsert); every kernel
// here is slot-indexed. Accumulator bank per agg j: acc[j*stride + slot]
// (sum bits / order-mapped min-max / first value bits), meta[.] (count /
// first-row priority), st[.] (FIRST state). i32 accumulators are stored
// widened to i64; freeze/emit narrow to 4-byte LE (acc.rs:335-347 prim
// format at the declared width).
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "dev_agg.h"
#include "kernels.h"

namespace auron {

namespace {
inline void check_launchm(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("kernel launch failed: ") + name +
                             ": " + hipGetErrorString(e));
}
constexpr int MBLOCK = 256;
constexpr int64_t MMAX_BLOCKS = 256 * 8;
inline int mgrid(int64_t n) {
  int64_t b = (n + MBLOCK - 1) / MBLOCK;
  if (b > MMAX_BLOCKS) b = MMAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}
}  // namespace

// ---- slot resolution -------------------------------------------------------
__global__ void k_slots_upsert(const AggTable t,
                               const int64_t* __restrict__ keys,
                               const uint8_t* __restrict__ key_valid,
                               int64_t n, uint64_t row_offset,
                               uint32_t* __restrict__ slots) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    bool knull = key_valid && !bit_get_dev(key_valid, i);
    int64_t a = agg_upsert_slot(t, knull ? 0 : keys[i], knull);
    if (a < 0) {
      slots[i] = 0xFFFFFFFFu;
      continue;
    }
    slots[i] = (uint32_t)a;
    AggSlot* sl = &t.slots[a];
    uint64_t row = row_offset + (uint64_t)i;
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
  }
}

void launch_slots_upsert(const AggTable& t, const int64_t* keys,
                         const uint8_t* key_valid, int64_t n,
                         uint64_t row_offset, uint32_t* slots, hipStream_t s) {
  hipLaunchKernelGGL(k_slots_upsert, dim3(mgrid(n)), dim3(MBLOCK), 0, s, t,
                     keys, key_valid, n, row_offset, slots);
  check_launchm("k_slots_upsert");
}

// ---- bank init -------------------------------------------------------------
__global__ void k_ma_init(const MaDesc d, const MaAcc m, int64_t cap2) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < (int64_t)d.n * cap2; i += (int64_t)gridDim.x * blockDim.x) {
    int j = (int)(i / cap2);
    uint32_t k = d.a[j].kind;
    unsigned long long a0 = 0;
    if (k == AGGL_MIN) a0 = MM_MIN_INIT;
    if (k == AGGL_MAX) a0 = MM_MAX_INIT;
    m.acc[i] = a0;
    m.meta[i] = (k == AGGL_FIRST || k == AGGL_FIRSTIN) ? ~0ull : 0ull;
    m.st[i] = 0;
  }
}

void launch_ma_init(const MaDesc& d, const MaAcc& m, int64_t cap2,
                    hipStream_t s) {
  hipLaunchKernelGGL(k_ma_init, dim3(mgrid(d.n * cap2)), dim3(MBLOCK), 0, s,
                     d, m, cap2);
  check_launchm("k_ma_init");
}

// ---- per-row argument read: value in BOTH domains --------------------------
// Loads agg j's argument for row i: returns validity; fv = value as f64
// (acc_t 0), bits = value widened to i64 (acc_t 1/2). NULL literal => false.
__device__ __forceinline__ bool ma_read_arg(const MaAgg& a, const void* vals,
                                            const uint8_t* valid, int64_t i,
                                            double* fv, long long* bits) {
  if (a.arg_dt == 3 || !vals) return false;
  if (valid && !bit_get_dev(valid, i)) return false;
  if (a.arg_dt == 0) {
    double v = ((const double*)vals)[i];
    *fv = v;
    memcpy(bits, &v, 8);  // raw bits ride through for int accs (unused)
  } else if (a.arg_dt == 1) {
    long long v = ((const long long*)vals)[i];
    *bits = v;
    *fv = (double)v;
  } else {
    int v = ((const int*)vals)[i];
    *bits = (long long)v;
    *fv = (double)v;
  }
  return true;
}

// the 8 accumulator bytes for value-kind aggs, in the ACC domain
__device__ __forceinline__ unsigned long long ma_val_bits(const MaAgg& a,
                                                          double fv,
                                                          long long bits) {
  if (a.acc_t == 0) {
    unsigned long long b;
    memcpy(&b, &fv, 8);
    return b;
  }
  return (unsigned long long)bits;
}

// ---- update ----------------------------------------------------------------
__global__ void k_ma_update(const AggTable t, const MaDesc d, const MaAcc m,
                            const MaPools p, const MaArgs args,
                            const int64_t* __restrict__ keys,
                            const uint8_t* __restrict__ key_valid,
                            const uint32_t* __restrict__ slots, int64_t n,
                            uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == 0xFFFFFFFFu) continue;
    uint64_t row = row_offset + (uint64_t)i;
    for (int j = 0; j < d.n; j++) {
      const MaAgg& ag = d.a[j];
      double fv = 0;
      long long bits = 0;
      bool valid = ma_read_arg(ag, args.vals[j], args.valid[j], i, &fv,
                               &bits);
      unsigned long long* acc = &m.acc[(int64_t)j * m.stride + a];
      unsigned long long* meta = &m.meta[(int64_t)j * m.stride + a];
      switch (ag.kind) {
        case AGGL_SUM:
        case AGGL_AVG:
          if (valid) {
            if (ag.acc_t == 0)
              unsafeAtomicAdd((double*)acc, fv);
            else
              atomicAdd(acc, (unsigned long long)bits);  // wrapping int
            atomicAdd(meta, 1ull);  // non-null count = sum validity / avg cnt
          }
          break;
        case AGGL_CNT:
          if (valid) atomicAdd(meta, 1ull);
          break;
        case AGGL_MIN:
          if (valid)
            atomicMin(acc, val_omap_bits(ma_val_bits(ag, fv, bits),
                                         ag.acc_t != 0));
          break;
        case AGGL_MAX:
          if (valid)
            atomicMax(acc, val_omap_bits(ma_val_bits(ag, fv, bits),
                                         ag.acc_t != 0));
          break;
        case AGGL_FIRST:  // first.rs: latch the FIRST ROW, null included
          atomicMin(meta, (unsigned long long)row);
          break;
        case AGGL_FIRSTIN:  // first_ignores_null.rs: first NON-NULL row
          if (valid) atomicMin(meta, (unsigned long long)row);
          break;
        case AGGL_CLIST:
        case AGGL_CSET:
          if (valid && ag.pool < MA_MAX_POOLS) {
            bool knull = key_valid && !bit_get_dev(key_valid, i);
            int pi = ag.pool;
            unsigned long long vb = ma_val_bits(ag, fv, bits);
            if (knull) {
              unsigned long long q = atomicAdd(&p.n[2 * pi + 1], 1ull);
              if ((int64_t)(q + p.n[2 * pi]) >= p.cap) {
                atomicOr(t.error_flag, 4u);
              } else {
                int64_t at = p.cap - 1 - (int64_t)q;
                p.key[pi][at] = 0;
                p.prio[pi][at] = row;
                p.val[pi][at] = vb;
              }
            } else {
              unsigned long long q = atomicAdd(&p.n[2 * pi], 1ull);
              if ((int64_t)(q + p.n[2 * pi + 1]) >= p.cap) {
                atomicOr(t.error_flag, 4u);
              } else {
                p.key[pi][q] = keys[i];
                p.prio[pi][q] = row;
                p.val[pi][q] = vb;
              }
            }
          }
          break;
      }
    }
  }
}

void launch_ma_update(const AggTable& t, const MaDesc& d, const MaAcc& m,
                      const MaPools& p, const MaArgs& args,
                      const int64_t* keys, const uint8_t* key_valid,
                      const uint32_t* slots, int64_t n, uint64_t row_offset,
                      hipStream_t s) {
  hipLaunchKernelGGL(k_ma_update, dim3(mgrid(n)), dim3(MBLOCK), 0, s, t, d, m,
                     p, args, keys, key_valid, slots, n, row_offset);
  check_launchm("k_ma_update");
}

// ---- FIRST pass B: the winning row stores its value ------------------------
__global__ void k_ma_first_capture(const AggTable t, const MaDesc d,
                                   const MaAcc m, const MaArgs args,
                                   const uint32_t* __restrict__ slots,
                                   int64_t n, uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == 0xFFFFFFFFu) continue;
    uint64_t row = row_offset + (uint64_t)i;
    for (int j = 0; j < d.n; j++) {
      const MaAgg& ag = d.a[j];
      if (ag.kind != AGGL_FIRST && ag.kind != AGGL_FIRSTIN) continue;
      int64_t off = (int64_t)j * m.stride + a;
      if (m.meta[off] != (unsigned long long)row) continue;
      double fv = 0;
      long long bits = 0;
      bool valid = ma_read_arg(ag, args.vals[j], args.valid[j], i, &fv,
                               &bits);
      if (ag.kind == AGGL_FIRSTIN && !valid) continue;  // cannot happen
      if (valid) {
        m.acc[off] = ma_val_bits(ag, fv, bits);
        m.st[off] = 2;
      } else {
        m.st[off] = 1;  // FIRST touched, first value null
      }
    }
  }
}

void launch_ma_first_capture(const AggTable& t, const MaDesc& d,
                             const MaAcc& m, const MaArgs& args,
                             const uint32_t* slots, int64_t n,
                             uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_ma_first_capture, dim3(mgrid(n)), dim3(MBLOCK), 0, s,
                     t, d, m, args, slots, n, row_offset);
  check_launchm("k_ma_first_capture");
}

// ---- a8 wire: freeze / parse per agg at its declared width -----------------
__device__ __forceinline__ int ma_prim_w(const MaAgg& a) {
  return a.acc_t == 2 ? 4 : 8;
}

// group's item run in a SORTED pool view (installed by the engine before
// freezing: front = {key asc, prio asc} over [0,n0), null-key group at
// [n0, n0+n1) — the prepare_collect convention)
__device__ __forceinline__ uint32_t ma_pool_run(
    const long long* __restrict__ keys, unsigned long long n0,
    unsigned long long n1, long long key, bool knull, int64_t* beg) {
  if (knull) {
    *beg = (int64_t)n0;
    return (uint32_t)n1;
  }
  int64_t lo = 0, hi = (int64_t)n0;
  while (lo < hi) {
    int64_t mid = (lo + hi) >> 1;
    if (keys[mid] < key) lo = mid + 1; else hi = mid;
  }
  int64_t lo2 = lo, hi2 = (int64_t)n0;
  while (lo2 < hi2) {
    int64_t mid = (lo2 + hi2) >> 1;
    if (keys[mid] <= key) lo2 = mid + 1; else hi2 = mid;
  }
  *beg = lo;
  return (uint32_t)(lo2 - lo);
}

// per-agg frozen-part length for group slot `a`
__device__ __forceinline__ int ma_part_len(const MaDesc& d, const MaAcc& m,
                                           const MaPools& p, int j,
                                           int64_t a, int64_t key_for_pool,
                                           bool knull) {
  const MaAgg& ag = d.a[j];
  int64_t off = (int64_t)j * m.stride + a;
  int w = ma_prim_w(ag);
  switch (ag.kind) {
    case AGGL_SUM:
      return 1 + (m.meta[off] ? w : 0);
    case AGGL_AVG:
      return 1 + (m.meta[off] ? 8 : 0) + varint_len_dev(m.meta[off]);
    case AGGL_CNT:
      return varint_len_dev(m.meta[off]);
    case AGGL_MIN:
      return 1 + (m.acc[off] != MM_MIN_INIT ? w : 0);
    case AGGL_MAX:
      return 1 + (m.acc[off] != MM_MAX_INIT ? w : 0);
    case AGGL_FIRST:
      return 2 + (m.st[off] == 2 ? w : 0);
    case AGGL_FIRSTIN:
      return 1 + (m.st[off] == 2 ? w : 0);
    case AGGL_CLIST:
    case AGGL_CSET: {
      if (ag.pool >= MA_MAX_POOLS) return varint_len_dev(0);
      int64_t beg;
      uint32_t cnt = ma_pool_run(p.key[ag.pool], p.n[2 * ag.pool],
                                 p.n[2 * ag.pool + 1], key_for_pool, knull,
                                 &beg);
      uint64_t raw = (uint64_t)cnt * ma_prim_w(ag);
      return varint_len_dev(raw) + (int)raw;
    }
  }
  return 0;
}

__global__ void k_ma_freeze_len(const AggTable t, const MaDesc d,
                                const MaAcc m, const MaPools p,
                                const uint32_t* __restrict__ order_slots,
                                int64_t n, int32_t* __restrict__ lens) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = order_slots[i];
    bool knull = a == t.cap + 1;
    int64_t key = (a == t.cap) ? INT64_MIN : t.slots[a].key;
    int len = 0;
    for (int j = 0; j < d.n; j++)
      len += ma_part_len(d, m, p, j, a, key, knull);
    lens[i] = len;
  }
}

void launch_ma_freeze_len(const AggTable& t, const MaDesc& d, const MaAcc& m,
                          const MaPools& p, const uint32_t* order_slots,
                          int64_t n, int32_t* lens, hipStream_t s) {
  hipLaunchKernelGGL(k_ma_freeze_len, dim3(mgrid(n)), dim3(MBLOCK), 0, s, t,
                     d, m, p, order_slots, n, lens);
  check_launchm("k_ma_freeze_len");
}

__device__ __forceinline__ uint8_t* ma_write_prim(uint8_t* q, bool valid,
                                                  unsigned long long bits,
                                                  int w) {
  if (!valid) {
    *q++ = 0;
    return q;
  }
  *q++ = 1;
  for (int b = 0; b < w; b++) *q++ = (uint8_t)(bits >> (8 * b));
  return q;
}

__device__ __forceinline__ uint8_t* ma_write_varint(uint8_t* q, uint64_t v) {
  while (v >= 128) {
    *q++ = (uint8_t)(128 + v % 128);
    v /= 128;
  }
  *q++ = (uint8_t)v;
  return q;
}

__global__ void k_ma_freeze_write(const AggTable t, const MaDesc d,
                                  const MaAcc m, const MaPools p,
                                  const uint32_t* __restrict__ order_slots,
                                  int64_t n,
                                  const int32_t* __restrict__ offsets,
                                  uint8_t* __restrict__ data) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = order_slots[i];
    bool knull = a == t.cap + 1;
    int64_t key = (a == t.cap) ? INT64_MIN : t.slots[a].key;
    uint8_t* q = data + offsets[i];
    for (int j = 0; j < d.n; j++) {
      const MaAgg& ag = d.a[j];
      int64_t off = (int64_t)j * m.stride + a;
      int w = ma_prim_w(ag);
      switch (ag.kind) {
        case AGGL_SUM:
          q = ma_write_prim(q, m.meta[off] != 0, m.acc[off], w);
          break;
        case AGGL_AVG:
          q = ma_write_prim(q, m.meta[off] != 0, m.acc[off], 8);
          q = ma_write_varint(q, m.meta[off]);
          break;
        case AGGL_CNT:
          q = ma_write_varint(q, m.meta[off]);
          break;
        case AGGL_MIN:
          q = ma_write_prim(q, m.acc[off] != MM_MIN_INIT,
                            val_omap_inv_bits(m.acc[off], ag.acc_t != 0), w);
          break;
        case AGGL_MAX:
          q = ma_write_prim(q, m.acc[off] != MM_MAX_INIT,
                            val_omap_inv_bits(m.acc[off], ag.acc_t != 0), w);
          break;
        case AGGL_FIRST:
          q = ma_write_prim(q, m.st[off] == 2, m.acc[off], w);
          *q++ = m.st[off] ? 2 : 0;
          break;
        case AGGL_FIRSTIN:
          q = ma_write_prim(q, m.st[off] == 2, m.acc[off], w);
          break;
        case AGGL_CLIST:
        case AGGL_CSET: {
          uint32_t cnt = 0;
          int64_t beg = 0;
          if (ag.pool < MA_MAX_POOLS)
            cnt = ma_pool_run(p.key[ag.pool], p.n[2 * ag.pool],
                              p.n[2 * ag.pool + 1], key, knull, &beg);
          q = ma_write_varint(q, (uint64_t)cnt * w);
          for (uint32_t c = 0; c < cnt; c++) {
            unsigned long long vb = p.val[ag.pool][beg + c];
            for (int b = 0; b < w; b++) *q++ = (uint8_t)(vb >> (8 * b));
          }
          break;
        }
      }
    }
  }
}

void launch_ma_freeze_write(const AggTable& t, const MaDesc& d,
                            const MaAcc& m, const MaPools& p,
                            const uint32_t* order_slots, int64_t n,
                            const int32_t* offsets, uint8_t* data,
                            hipStream_t s) {
  hipLaunchKernelGGL(k_ma_freeze_write, dim3(mgrid(n)), dim3(MBLOCK), 0, s, t,
                     d, m, p, order_slots, n, offsets, data);
  check_launchm("k_ma_freeze_write");
}

// ---- merge frozen records (Final / PartialMerge input) ---------------------
__global__ void k_ma_merge_frozen(const AggTable t, const MaDesc d,
                                  const MaAcc m, const MaPools p,
                                  const int64_t* __restrict__ keys,
                                  const uint8_t* __restrict__ key_valid,
                                  const uint32_t* __restrict__ slots,
                                  const uint8_t* __restrict__ acc_data,
                                  const int32_t* __restrict__ acc_offsets,
                                  int64_t n, uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == 0xFFFFFFFFu) continue;
    uint64_t row = row_offset + (uint64_t)i;
    const uint8_t* q = acc_data + acc_offsets[i];
    for (int j = 0; j < d.n; j++) {
      const MaAgg& ag = d.a[j];
      int64_t off = (int64_t)j * m.stride + a;
      int w = ma_prim_w(ag);
      switch (ag.kind) {
        case AGGL_SUM: {
          uint8_t v = *q++;
          if (v) {
            unsigned long long bits = 0;
            for (int b = 0; b < w; b++)
              bits |= (unsigned long long)(*q++) << (8 * b);
            if (ag.acc_t == 2) bits = (unsigned long long)(long long)(int)bits;
            if (ag.acc_t == 0) {
              double x;
              memcpy(&x, &bits, 8);
              unsafeAtomicAdd((double*)&m.acc[off], x);
            } else {
              atomicAdd(&m.acc[off], bits);
            }
            atomicAdd(&m.meta[off], 1ull);  // validity latch (count of parts)
          }
          break;
        }
        case AGGL_AVG: {
          uint8_t v = *q++;
          if (v) {
            unsigned long long bits = 0;
            for (int b = 0; b < 8; b++)
              bits |= (unsigned long long)(*q++) << (8 * b);
            double x;
            memcpy(&x, &bits, 8);
            unsafeAtomicAdd((double*)&m.acc[off], x);
          }
          int used;
          uint64_t c = read_varint_dev(q, &used);
          q += used;
          if (c) atomicAdd(&m.meta[off], c);
          break;
        }
        case AGGL_CNT: {
          int used;
          uint64_t c = read_varint_dev(q, &used);
          q += used;
          if (c) atomicAdd(&m.meta[off], c);
          break;
        }
        case AGGL_MIN:
        case AGGL_MAX: {
          uint8_t v = *q++;
          if (v) {
            unsigned long long bits = 0;
            for (int b = 0; b < w; b++)
              bits |= (unsigned long long)(*q++) << (8 * b);
            if (ag.acc_t == 2) bits = (unsigned long long)(long long)(int)bits;
            unsigned long long u = val_omap_bits(bits, ag.acc_t != 0);
            if (ag.kind == AGGL_MIN)
              atomicMin(&m.acc[off], u);
            else
              atomicMax(&m.acc[off], u);
          }
          break;
        }
        case AGGL_FIRST: {
          uint8_t v = *q++;
          if (v) q += w;
          uint8_t fl = *q++;
          if (fl) atomicMin(&m.meta[off], (unsigned long long)row);
          break;
        }
        case AGGL_FIRSTIN: {
          uint8_t v = *q++;
          if (v) {
            q += w;
            atomicMin(&m.meta[off], (unsigned long long)row);
          }
          break;
        }
        case AGGL_CLIST:
        case AGGL_CSET: {
          int used;
          uint64_t raw = read_varint_dev(q, &used);
          q += used;
          uint32_t cnt = (uint32_t)(raw / w);
          if (ag.pool < MA_MAX_POOLS) {
            bool knull = key_valid && !bit_get_dev(key_valid, i);
            int pi = ag.pool;
            for (uint32_t c = 0; c < cnt; c++) {
              unsigned long long bits = 0;
              for (int b = 0; b < w; b++)
                bits |= (unsigned long long)(q[(size_t)c * w + b]) << (8 * b);
              if (ag.acc_t == 2)
                bits = (unsigned long long)(long long)(int)bits;
              unsigned long long prio = (row << 20) | c;
              if (knull) {
                unsigned long long qq = atomicAdd(&p.n[2 * pi + 1], 1ull);
                if ((int64_t)(qq + p.n[2 * pi]) >= p.cap) {
                  atomicOr(t.error_flag, 4u);
                } else {
                  int64_t at = p.cap - 1 - (int64_t)qq;
                  p.key[pi][at] = 0;
                  p.prio[pi][at] = prio;
                  p.val[pi][at] = bits;
                }
              } else {
                unsigned long long qq = atomicAdd(&p.n[2 * pi], 1ull);
                if ((int64_t)(qq + p.n[2 * pi + 1]) >= p.cap) {
                  atomicOr(t.error_flag, 4u);
                } else {
                  p.key[pi][qq] = keys[i];
                  p.prio[pi][qq] = prio;
                  p.val[pi][qq] = bits;
                }
              }
            }
          }
          q += raw;
          break;
        }
      }
    }
  }
}

void launch_ma_merge_frozen(const AggTable& t, const MaDesc& d,
                            const MaAcc& m, const MaPools& p,
                            const int64_t* keys, const uint8_t* key_valid,
                            const uint32_t* slots, const uint8_t* acc_data,
                            const int32_t* acc_offsets, int64_t n,
                            uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_ma_merge_frozen, dim3(mgrid(n)), dim3(MBLOCK), 0, s, t,
                     d, m, p, keys, key_valid, slots, acc_data, acc_offsets,
                     n, row_offset);
  check_launchm("k_ma_merge_frozen");
}

// pass B over frozen records: the winning record stores its FIRST value
__global__ void k_ma_first_capture_frozen(
    const AggTable t, const MaDesc d, const MaAcc m,
    const uint32_t* __restrict__ slots, const uint8_t* __restrict__ acc_data,
    const int32_t* __restrict__ acc_offsets, int64_t n, uint64_t row_offset) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == 0xFFFFFFFFu) continue;
    uint64_t row = row_offset + (uint64_t)i;
    const uint8_t* q = acc_data + acc_offsets[i];
    for (int j = 0; j < d.n; j++) {
      const MaAgg& ag = d.a[j];
      int64_t off = (int64_t)j * m.stride + a;
      int w = ma_prim_w(ag);
      switch (ag.kind) {
        case AGGL_SUM: {
          uint8_t v = *q++;
          if (v) q += w;
          break;
        }
        case AGGL_AVG: {
          uint8_t v = *q++;
          if (v) q += 8;
          int used;
          read_varint_dev(q, &used);
          q += used;
          break;
        }
        case AGGL_CNT: {
          int used;
          read_varint_dev(q, &used);
          q += used;
          break;
        }
        case AGGL_MIN:
        case AGGL_MAX: {
          uint8_t v = *q++;
          if (v) q += w;
          break;
        }
        case AGGL_FIRST:
        case AGGL_FIRSTIN: {
          uint8_t v = *q++;
          unsigned long long bits = 0;
          if (v) {
            for (int b = 0; b < w; b++)
              bits |= (unsigned long long)(*q++) << (8 * b);
            if (ag.acc_t == 2) bits = (unsigned long long)(long long)(int)bits;
          }
          uint8_t fl = v;
          if (ag.kind == AGGL_FIRST) fl = *q++;
          if (m.meta[off] == (unsigned long long)row) {
            if (v) {
              m.acc[off] = bits;
              m.st[off] = 2;
            } else if (fl) {
              m.st[off] = 1;
            }
          }
          break;
        }
        case AGGL_CLIST:
        case AGGL_CSET: {
          int used;
          uint64_t raw = read_varint_dev(q, &used);
          q += used + raw;
          break;
        }
      }
    }
  }
}

void launch_ma_first_capture_frozen(const AggTable& t, const MaDesc& d,
                                    const MaAcc& m, const uint32_t* slots,
                                    const uint8_t* acc_data,
                                    const int32_t* acc_offsets, int64_t n,
                                    uint64_t row_offset, hipStream_t s) {
  hipLaunchKernelGGL(k_ma_first_capture_frozen, dim3(mgrid(n)), dim3(MBLOCK),
                     0, s, t, d, m, slots, acc_data, acc_offsets, n,
                     row_offset);
  check_launchm("k_ma_first_capture_frozen");
}

// ---- final-output gather ---------------------------------------------------
__global__ void k_ma_gather_out(const MaDesc d, const MaAcc m, int agg,
                                const uint32_t* __restrict__ order_slots,
                                int64_t n, uint8_t* __restrict__ values,
                                uint8_t* __restrict__ valid_bitmap) {
  const MaAgg& ag = d.a[agg];
  int w = ma_prim_w(ag);
  int64_t nbytes = (n + 7) / 8;
  for (int64_t byte = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       byte < nbytes; byte += (int64_t)gridDim.x * blockDim.x) {
    uint8_t bm = 0;
    for (int b = 0; b < 8 && byte * 8 + b < n; b++) {
      int64_t i = byte * 8 + b;
      int64_t off = (int64_t)agg * m.stride + order_slots[i];
      bool valid = false;
      unsigned long long bits = 0;
      switch (ag.kind) {
        case AGGL_SUM:
          valid = m.meta[off] != 0;
          bits = m.acc[off];
          break;
        case AGGL_AVG: {
          valid = m.meta[off] != 0;
          if (valid) {
            double sum;
            memcpy(&sum, &m.acc[off], 8);
            double avg = sum / (double)m.meta[off];
            memcpy(&bits, &avg, 8);
          }
          break;
        }
        case AGGL_CNT:
          valid = true;
          bits = m.meta[off];
          break;
        case AGGL_MIN:
          valid = m.acc[off] != MM_MIN_INIT;
          if (valid) bits = val_omap_inv_bits(m.acc[off], ag.acc_t != 0);
          break;
        case AGGL_MAX:
          valid = m.acc[off] != MM_MAX_INIT;
          if (valid) bits = val_omap_inv_bits(m.acc[off], ag.acc_t != 0);
          break;
        case AGGL_FIRST:
        case AGGL_FIRSTIN:
          valid = m.st[off] == 2;
          if (valid) bits = m.acc[off];
          break;
      }
      if (valid) bm |= (uint8_t)(1u << b);
      int ow = (ag.kind == AGGL_CNT) ? 8 : w;
      for (int x = 0; x < ow; x++)
        values[i * ow + x] = valid ? (uint8_t)(bits >> (8 * x)) : 0;
    }
    valid_bitmap[byte] = bm;
  }
}

void launch_ma_gather_out(const MaDesc& d, const MaAcc& m, int agg,
                          const uint32_t* order_slots, int64_t n,
                          uint8_t* values, uint8_t* valid_bitmap,
                          hipStream_t s) {
  hipLaunchKernelGGL(k_ma_gather_out, dim3(mgrid((n + 7) / 8)), dim3(MBLOCK),
                     0, s, d, m, agg, order_slots, n, values, valid_bitmap);
  check_launchm("k_ma_gather_out");
}

// ---- growth ----------------------------------------------------------------
__global__ void k_ma_rebuild(const AggTable dst, const MaAcc dm,
                             const AggTable src, const MaAcc sm, int naggs) {
  const int64_t mask = dst.cap - 1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < src.cap + 2; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t a = -1;
    if (i >= src.cap) {  // specials keep their trailing slots
      if (!src.special_used[i - src.cap]) continue;
      a = dst.cap + (i - src.cap);
      if (atomicCAS(&dst.special_used[i - src.cap], 0u, 1u) == 0u)
        atomicAdd(dst.num_groups, 1ull);
      dst.slots[a] = src.slots[i];
    } else {
      long long key = src.slots[i].key;
      if (key == KEY_EMPTY) continue;
      uint64_t h = mix64((uint64_t)key);
      int64_t s = (int64_t)(h & (uint64_t)mask);
      for (int64_t probe = 0; probe <= mask; probe++) {
        long long prev =
            (long long)atomicCAS((unsigned long long*)&dst.slots[s].key,
                                 (unsigned long long)KEY_EMPTY,
                                 (unsigned long long)key);
        if (prev == KEY_EMPTY) {
          atomicAdd(dst.num_groups, 1ull);
          a = s;
          break;
        }
        s = (s + 1) & mask;
      }
      if (a < 0) {
        atomicOr(dst.error_flag, 1u);
        continue;
      }
      dst.slots[a].cnt = src.slots[i].cnt;
      dst.slots[a].sum = src.slots[i].sum;
      dst.slots[a].first_row = src.slots[i].first_row;
    }
    for (int j = 0; j < naggs; j++) {
      dm.acc[(int64_t)j * dm.stride + a] = sm.acc[(int64_t)j * sm.stride + i];
      dm.meta[(int64_t)j * dm.stride + a] =
          sm.meta[(int64_t)j * sm.stride + i];
      dm.st[(int64_t)j * dm.stride + a] = sm.st[(int64_t)j * sm.stride + i];
    }
  }
}

void launch_ma_rebuild(const AggTable& dst, const MaAcc& dm,
                       const AggTable& src, const MaAcc& sm, int naggs,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_ma_rebuild, dim3(mgrid(src.cap + 2)), dim3(MBLOCK), 0,
                     s, dst, dm, src, sm, naggs);
  check_launchm("k_ma_rebuild");
}

}  // namespace auron
