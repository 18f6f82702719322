// kernels_gkey.hip — generalized grouping keys (Utf8 / multi-column tuples),
// closing SURVEY.md §8 a4's "substituted i64 keys" caveat. The reference
// row-encodes key tuples through the arrow Row format (agg_ctx.rs:219-231);
// per SURVEY §8c(i) the encoding itself is substitutable — only the grouping
// column VALUES must round-trip — so this engine uses its own per-column
// [valid u8][value bytes] encoding (utf8: u32 len + bytes), the same 32-byte
// slot table (slot.key = 64-bit hash of the encoded bytes), and a device
// byte pool holding each group's encoded key for the byte-compare probe.
//
// Claim/publish protocol: the CAS winner on slot.key appends the encoded key
// to the pool and RELEASE-publishes (offset+1) into GKeyTable::off; probers
// that match the hash ACQUIRE-load the offset (bounded spin; the claimer
// makes progress independently, so no deadlock) and byte-compare. Distinct
// keys with equal hashes byte-compare unequal and continue probing.
#include <hip/hip_runtime.h>

#include <stdexcept>
#include <string>

#include "dev_agg.h"
#include "kernels.h"

namespace auron {

namespace {
inline void check_launchg(const char* name) {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess)
    throw std::runtime_error(std::string("kernel launch failed: ") + name +
                             ": " + hipGetErrorString(e));
}
constexpr int GBLOCK = 256;
constexpr int64_t GMAX_BLOCKS = 256 * 8;
inline int ggrid(int64_t n) {
  int64_t b = (n + GBLOCK - 1) / GBLOCK;
  if (b > GMAX_BLOCKS) b = GMAX_BLOCKS;
  if (b < 1) b = 1;
  return (int)b;
}
}  // namespace

// ---- murmur3 byte fold (mur.rs:19-30: 4-byte words, then signed trailing
// bytes, fmix by total length); null rows leave the hash unchanged
// (spark_hash.rs hash_array semantics, matching the i64/i32 folds)
__global__ void k_hash_fold_bytes(const int32_t* __restrict__ offsets,
                                  const uint8_t* __restrict__ data,
                                  const uint8_t* __restrict__ valid, int64_t n,
                                  int32_t* __restrict__ hashes) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (valid && !bit_get_dev(valid, i)) continue;
    const uint8_t* p = data + offsets[i];
    int32_t len = offsets[i + 1] - offsets[i];
    int32_t h1 = hashes[i];
    int32_t aligned = len - (len & 3);
    for (int32_t j = 0; j < aligned; j += 4) {
      int32_t w;
      memcpy(&w, p + j, 4);
      h1 = mur_mix_h1(h1, mur_mix_k1(w));
    }
    for (int32_t j = aligned; j < len; j++)
      h1 = mur_mix_h1(h1, mur_mix_k1((int32_t)(int8_t)p[j]));
    hashes[i] = mur_fmix(h1, len);
  }
}

void launch_hash_fold_bytes(const int32_t* offsets, const uint8_t* data,
                            const uint8_t* valid, int64_t n, int32_t* hashes,
                            hipStream_t s) {
  hipLaunchKernelGGL(k_hash_fold_bytes, dim3(ggrid(n)), dim3(GBLOCK), 0, s,
                     offsets, data, valid, n, hashes);
  check_launchg("k_hash_fold_bytes");
}

// ---- key tuple encoding ----------------------------------------------------
__device__ __forceinline__ int gk_col_width(uint8_t dt) {
  return dt == GK_I32 ? 4 : 8;  // i64/f64 = 8
}

__global__ void k_gkey_enc_lens(const GKeyCols c, int64_t n,
                                uint32_t* __restrict__ lens) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t len = 0;
    for (int k = 0; k < c.ncols; k++) {
      bool v = !c.validity[k] || bit_get_dev(c.validity[k], i);
      len += 1;
      if (!v) continue;
      if (c.dt[k] == GK_UTF8)
        len += 4 + (uint32_t)(c.offsets[k][i + 1] - c.offsets[k][i]);
      else
        len += gk_col_width(c.dt[k]);
    }
    lens[i] = len;
  }
}

void launch_gkey_enc_lens(const GKeyCols& c, int64_t n, uint32_t* lens,
                          hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_enc_lens, dim3(ggrid(n)), dim3(GBLOCK), 0, s, c, n,
                     lens);
  check_launchg("k_gkey_enc_lens");
}

__global__ void k_gkey_enc_write(const GKeyCols c,
                                 const uint32_t* __restrict__ enc_offsets,
                                 uint8_t* __restrict__ enc_bytes, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint8_t* p = enc_bytes + enc_offsets[i];
    for (int k = 0; k < c.ncols; k++) {
      bool v = !c.validity[k] || bit_get_dev(c.validity[k], i);
      *p++ = v ? 1 : 0;
      if (!v) continue;
      if (c.dt[k] == GK_UTF8) {
        int32_t beg = c.offsets[k][i], end = c.offsets[k][i + 1];
        uint32_t sl = (uint32_t)(end - beg);
        memcpy(p, &sl, 4);
        p += 4;
        const uint8_t* sp = (const uint8_t*)c.values[k] + beg;
        for (uint32_t j = 0; j < sl; j++) p[j] = sp[j];
        p += sl;
      } else if (c.dt[k] == GK_I32) {
        memcpy(p, (const int32_t*)c.values[k] + i, 4);
        p += 4;
      } else {
        memcpy(p, (const uint8_t*)c.values[k] + (size_t)i * 8, 8);
        p += 8;
      }
    }
  }
}

void launch_gkey_enc_write(const GKeyCols& c, const uint32_t* enc_offsets,
                           uint8_t* enc_bytes, int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_enc_write, dim3(ggrid(n)), dim3(GBLOCK), 0, s, c,
                     enc_offsets, enc_bytes, n);
  check_launchg("k_gkey_enc_write");
}

// ---- probe-or-insert -------------------------------------------------------
__device__ __forceinline__ uint64_t gk_hash(const uint8_t* p, uint32_t len) {
  uint64_t h = 0x9E3779B97F4A7C15ull ^ (uint64_t)len;
  uint32_t i = 0;
  for (; i + 8 <= len; i += 8) {
    uint64_t w;
    memcpy(&w, p + i, 8);
    h = mix64(h ^ w);
  }
  if (i < len) {
    uint64_t w = 0;
    for (uint32_t j = 0; i + j < len; j++) w |= (uint64_t)p[i + j] << (8 * j);
    h = mix64(h ^ w);
  }
  // never 0 / KEY_EMPTY (low bit set); used as slot.key
  return h | 1ull;
}

__device__ __forceinline__ bool gk_bytes_eq(const uint8_t* a, const uint8_t* b,
                                            uint32_t len) {
  for (uint32_t i = 0; i < len; i++)
    if (a[i] != b[i]) return false;
  return true;
}

static constexpr uint32_t GK_NO_SLOT = 0xFFFFFFFFu;
static constexpr int GK_SPIN = 1 << 22;

__global__ void k_gkey_upsert(const AggTable t, const GKeyTable g,
                              const uint32_t* __restrict__ enc_offsets,
                              const uint8_t* __restrict__ enc_bytes, int64_t n,
                              uint64_t row_offset,
                              uint32_t* __restrict__ slots) {
  const int64_t mask = t.cap - 1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* kb = enc_bytes + enc_offsets[i];
    uint32_t klen = enc_offsets[i + 1] - enc_offsets[i];
    uint64_t h = gk_hash(kb, klen);
    int64_t s = (int64_t)(h & (uint64_t)mask);
    int64_t a = -1;
    for (int64_t probe = 0; probe <= mask; probe++) {
      long long cur = t.slots[s].key;
      if (cur == KEY_EMPTY) {
        long long prev =
            (long long)atomicCAS((unsigned long long*)&t.slots[s].key,
                                 (unsigned long long)KEY_EMPTY,
                                 (unsigned long long)h);
        if (prev == KEY_EMPTY) {
          // claimed: append [u32 len][bytes] to the pool and publish
          unsigned long long off =
              atomicAdd(g.pool_n, (unsigned long long)(klen + 4));
          if ((int64_t)(off + klen + 4) > g.pool_cap) {
            // host ensures capacity; this is the loud-failure backstop
            atomicOr(t.error_flag, 8u);
          } else {
            memcpy(g.pool + off, &klen, 4);
            for (uint32_t j = 0; j < klen; j++) g.pool[off + 4 + j] = kb[j];
          }
          atomicAdd(t.num_groups, 1ull);
          __hip_atomic_store(&g.off[s], off + 1, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          a = s;
          break;
        }
        cur = prev;
      }
      if (cur == (long long)h) {
        unsigned long long off = 0;
        for (int spin = 0; spin < GK_SPIN; spin++) {
          off = __hip_atomic_load(&g.off[s], __ATOMIC_ACQUIRE,
                                  __HIP_MEMORY_SCOPE_AGENT);
          if (off) break;
        }
        if (!off) {
          atomicOr(t.error_flag, 16u);  // publish starvation (never expected)
          break;
        }
        const uint8_t* pk = g.pool + (off - 1);
        uint32_t plen;
        memcpy(&plen, pk, 4);
        if (plen == klen && gk_bytes_eq(pk + 4, kb, klen)) {
          a = s;
          break;
        }
        // same hash, different key: keep probing
      }
      s = (s + 1) & mask;
    }
    if (a < 0) {
      atomicOr(t.error_flag, 1u);
      slots[i] = GK_NO_SLOT;
      continue;
    }
    slots[i] = (uint32_t)a;
    AggSlot* sl = &t.slots[a];
    uint64_t row = row_offset + (uint64_t)i;
    if (sl->first_row > row) atomicMin(&sl->first_row, row);
  }
}

void launch_gkey_upsert(const AggTable& t, const GKeyTable& g,
                        const uint32_t* enc_offsets, const uint8_t* enc_bytes,
                        int64_t n, uint64_t row_offset, uint32_t* slots,
                        hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_upsert, dim3(ggrid(n)), dim3(GBLOCK), 0, s, t, g,
                     enc_offsets, enc_bytes, n, row_offset, slots);
  check_launchg("k_gkey_upsert");
}

// ---- slot-indexed accumulate / merge --------------------------------------
__global__ void k_gkey_update_idx(const AggTable t,
                                  const uint32_t* __restrict__ slots,
                                  const double* __restrict__ vals,
                                  const uint8_t* __restrict__ val_valid,
                                  int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == GK_NO_SLOT) continue;
    bool vvalid = !val_valid || bit_get_dev(val_valid, i);
    if (!vvalid) continue;
    AggSlot* sl = &t.slots[a];
    sum_accum(&sl->sum, vals[i], t.sum_int);
    atomicAdd(&sl->cnt, 1ull);
    if (t.mm) {
      uint64_t u = val_omap(vals[i], t.sum_int);
      atomicMin(&t.mm[2 * (int64_t)a], u);
      atomicMax(&t.mm[2 * (int64_t)a + 1], u);
    }
  }
}

void launch_gkey_update_idx(const AggTable& t, const uint32_t* slots,
                            const double* vals, const uint8_t* val_valid,
                            int64_t n, hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_update_idx, dim3(ggrid(n)), dim3(GBLOCK), 0, s, t,
                     slots, vals, val_valid, n);
  check_launchg("k_gkey_update_idx");
}

__global__ void k_gkey_merge_frozen_idx(const AggTable t,
                                        const uint32_t* __restrict__ slots,
                                        const uint8_t* __restrict__ acc_data,
                                        const int32_t* __restrict__ acc_offsets,
                                        int64_t n, uint32_t layout) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint32_t a = slots[i];
    if (a == GK_NO_SLOT) continue;
    AggSlot* sl = &t.slots[a];
    AccSnap acc;
    agg_parse_frozen(layout, acc_data + acc_offsets[i], &acc, t.sum_int);
    if (acc.valid) sum_accum(&sl->sum, acc.sum, t.sum_int);
    if (acc.cnt) atomicAdd(&sl->cnt, acc.cnt);
    if (t.mm) {
      atomicMin(&t.mm[2 * (int64_t)a], acc.minu);
      atomicMax(&t.mm[2 * (int64_t)a + 1], acc.maxu);
    }
  }
}

void launch_gkey_merge_frozen_idx(const AggTable& t, const uint32_t* slots,
                                  const uint8_t* acc_data,
                                  const int32_t* acc_offsets, int64_t n,
                                  uint32_t layout, hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_merge_frozen_idx, dim3(ggrid(n)), dim3(GBLOCK), 0,
                     s, t, slots, acc_data, acc_offsets, n, layout);
  check_launchg("k_gkey_merge_frozen_idx");
}

// ---- growth: re-probe every occupied src slot into dst by stored hash ------
__global__ void k_gkey_rebuild(const AggTable dst, const GKeyTable dg,
                               const AggTable src, const GKeyTable sg) {
  const int64_t mask = dst.cap - 1;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < src.cap; i += (int64_t)gridDim.x * blockDim.x) {
    long long h = src.slots[i].key;
    if (h == KEY_EMPTY) continue;
    unsigned long long my_off = sg.off[i];  // published before growth
    int64_t s = (int64_t)((uint64_t)h & (uint64_t)mask);
    for (int64_t probe = 0; probe <= mask; probe++) {
      long long cur = dst.slots[s].key;
      if (cur == KEY_EMPTY) {
        long long prev =
            (long long)atomicCAS((unsigned long long*)&dst.slots[s].key,
                                 (unsigned long long)KEY_EMPTY,
                                 (unsigned long long)h);
        if (prev == KEY_EMPTY) {
          dst.slots[s].cnt = src.slots[i].cnt;
          dst.slots[s].sum = src.slots[i].sum;
          dst.slots[s].first_row = src.slots[i].first_row;
          if (dst.mm && src.mm) {
            dst.mm[2 * s] = src.mm[2 * i];
            dst.mm[2 * s + 1] = src.mm[2 * i + 1];
          }
          atomicAdd(dst.num_groups, 1ull);
          __hip_atomic_store(&dg.off[s], my_off, __ATOMIC_RELEASE,
                             __HIP_MEMORY_SCOPE_AGENT);
          break;
        }
        cur = prev;
      }
      if (cur == h) {
        // same hash: distinct groups have distinct pool offsets — wait for
        // the other rebuilder's publish and compare identity
        unsigned long long off = 0;
        for (int spin = 0; spin < GK_SPIN; spin++) {
          off = __hip_atomic_load(&dg.off[s], __ATOMIC_ACQUIRE,
                                  __HIP_MEMORY_SCOPE_AGENT);
          if (off) break;
        }
        if (off == my_off) break;  // cannot happen (each group rebuilt once)
        if (!off) {
          atomicOr(dst.error_flag, 16u);
          break;
        }
      }
      s = (s + 1) & mask;
      if (probe == mask) atomicOr(dst.error_flag, 1u);
    }
  }
}

void launch_gkey_rebuild(const AggTable& dst, const GKeyTable& dg,
                         const AggTable& src, const GKeyTable& sg,
                         hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_rebuild, dim3(ggrid(src.cap)), dim3(GBLOCK), 0, s,
                     dst, dg, src, sg);
  check_launchg("k_gkey_rebuild");
}

// ---- emit: decode grouping columns from the pool ---------------------------
// walk the encoded record to column `col`; returns pointer to its valid byte
__device__ __forceinline__ const uint8_t* gk_seek_col(const uint8_t* rec,
                                                      const uint8_t* dts,
                                                      int col) {
  const uint8_t* p = rec;
  for (int k = 0; k < col; k++) {
    uint8_t v = *p++;
    if (!v) continue;
    if (dts[k] == GK_UTF8) {
      uint32_t sl;
      memcpy(&sl, p, 4);
      p += 4 + sl;
    } else {
      p += gk_col_width(dts[k]);
    }
  }
  return p;
}

__global__ void k_gkey_out_fixed(const GKeyTable g,
                                 const uint32_t* __restrict__ order_slots,
                                 int64_t n, const uint8_t* __restrict__ dts,
                                 int ncols, int col,
                                 uint8_t* __restrict__ values,
                                 uint8_t* __restrict__ valid_bitmap) {
  const int w = gk_col_width(dts[col]);
  int64_t nbytes = (n + 7) / 8;
  for (int64_t byte = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       byte < nbytes; byte += (int64_t)gridDim.x * blockDim.x) {
    uint8_t bm = 0;
    for (int b = 0; b < 8 && byte * 8 + b < n; b++) {
      int64_t i = byte * 8 + b;
      const uint8_t* rec = g.pool + (g.off[order_slots[i]] - 1) + 4;
      const uint8_t* p = gk_seek_col(rec, dts, col);
      if (*p) {
        bm |= (uint8_t)(1u << b);
        for (int j = 0; j < w; j++) values[i * w + j] = p[1 + j];
      } else {
        for (int j = 0; j < w; j++) values[i * w + j] = 0;
      }
    }
    valid_bitmap[byte] = bm;
  }
}

void launch_gkey_out_fixed(const GKeyTable& g, const uint32_t* order_slots,
                           int64_t n, const uint8_t* dts, int ncols, int col,
                           uint8_t* values, uint8_t* valid_bitmap,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_out_fixed, dim3(ggrid((n + 7) / 8)), dim3(GBLOCK),
                     0, s, g, order_slots, n, dts, ncols, col, values,
                     valid_bitmap);
  check_launchg("k_gkey_out_fixed");
}

__global__ void k_gkey_out_lens(const GKeyTable g,
                                const uint32_t* __restrict__ order_slots,
                                int64_t n, const uint8_t* __restrict__ dts,
                                int ncols, int col, uint32_t* __restrict__ lens,
                                uint8_t* __restrict__ valid_bitmap) {
  int64_t nbytes = (n + 7) / 8;
  for (int64_t byte = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       byte < nbytes; byte += (int64_t)gridDim.x * blockDim.x) {
    uint8_t bm = 0;
    for (int b = 0; b < 8 && byte * 8 + b < n; b++) {
      int64_t i = byte * 8 + b;
      const uint8_t* rec = g.pool + (g.off[order_slots[i]] - 1) + 4;
      const uint8_t* p = gk_seek_col(rec, dts, col);
      uint32_t sl = 0;
      if (*p) {
        bm |= (uint8_t)(1u << b);
        memcpy(&sl, p + 1, 4);
      }
      lens[i] = sl;
    }
    valid_bitmap[byte] = bm;
  }
}

void launch_gkey_out_lens(const GKeyTable& g, const uint32_t* order_slots,
                          int64_t n, const uint8_t* dts, int ncols, int col,
                          uint32_t* lens, uint8_t* valid_bitmap,
                          hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_out_lens, dim3(ggrid((n + 7) / 8)), dim3(GBLOCK),
                     0, s, g, order_slots, n, dts, ncols, col, lens,
                     valid_bitmap);
  check_launchg("k_gkey_out_lens");
}

__global__ void k_gkey_out_bytes(const GKeyTable g,
                                 const uint32_t* __restrict__ order_slots,
                                 int64_t n, const uint8_t* __restrict__ dts,
                                 int ncols, int col,
                                 const int32_t* __restrict__ out_offsets,
                                 uint8_t* __restrict__ out_data) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint8_t* rec = g.pool + (g.off[order_slots[i]] - 1) + 4;
    const uint8_t* p = gk_seek_col(rec, dts, col);
    if (!*p) continue;
    uint32_t sl;
    memcpy(&sl, p + 1, 4);
    uint8_t* dst = out_data + out_offsets[i];
    for (uint32_t j = 0; j < sl; j++) dst[j] = p[5 + j];
  }
}

void launch_gkey_out_bytes(const GKeyTable& g, const uint32_t* order_slots,
                           int64_t n, const uint8_t* dts, int ncols, int col,
                           const int32_t* out_offsets, uint8_t* out_data,
                           hipStream_t s) {
  hipLaunchKernelGGL(k_gkey_out_bytes, dim3(ggrid(n)), dim3(GBLOCK), 0, s, g,
                     order_slots, n, dts, ncols, col, out_offsets, out_data);
  check_launchg("k_gkey_out_bytes");
}

}  // namespace auron
