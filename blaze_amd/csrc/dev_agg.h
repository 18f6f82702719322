// dev_agg.h — shared device-side helpers for the aggregation kernels:
// murmur3 restatement (mur.rs:19-87), slot-table probe, typed accumulate,
// the a8 Binary agg-buf freeze/parse wire logic (acc.rs / count.rs /
// first.rs / collect.rs formats). Inline-only; included by kernels.hip and
// kernels_gkey.hip so the WIRE logic has exactly one definition.
#pragma once

#include <hip/hip_runtime.h>

#include "kernels.h"

namespace auron {

// ---- murmur3 (bit-exact restatement of mur.rs:19-87) ----------------------
__device__ __forceinline__ uint32_t rotl32(uint32_t x, int r) {
  return (x << r) | (x >> (32 - r));
}

__device__ __forceinline__ int32_t mur_mix_k1(int32_t k1) {
  uint32_t k = (uint32_t)k1;
  k *= 0xcc9e2d51u;
  k = rotl32(k, 15);
  k *= 0x1b873593u;
  return (int32_t)k;
}

__device__ __forceinline__ int32_t mur_mix_h1(int32_t h1, int32_t k1) {
  uint32_t h = (uint32_t)h1 ^ (uint32_t)k1;
  h = rotl32(h, 13);
  h = h * 5u + 0xe6546b64u;
  return (int32_t)h;
}

__device__ __forceinline__ int32_t mur_fmix(int32_t h1, int32_t len) {
  uint32_t h = (uint32_t)h1 ^ (uint32_t)len;
  h ^= h >> 16;
  h *= 0x85ebca6bu;
  h ^= h >> 13;
  h *= 0xc2b2ae35u;
  h ^= h >> 16;
  return (int32_t)h;
}

__device__ __forceinline__ int32_t murmur3_long(int64_t value, int32_t seed) {
  int32_t low = (int32_t)value;
  int32_t high = (int32_t)((uint64_t)value >> 32);
  int32_t h1 = mur_mix_h1(seed, mur_mix_k1(low));
  h1 = mur_mix_h1(h1, mur_mix_k1(high));
  return mur_fmix(h1, 8);
}

__device__ __forceinline__ bool bit_get_dev(const uint8_t* bm, int64_t i) {
  return (bm[i >> 3] >> (i & 7)) & 1;
}


static constexpr int64_t KEY_EMPTY = INT64_MIN;

// internal slot hash — NOT part of the parity contract (agg_hash_map.rs's
// foldhash only shapes its private layout); a strong 64-bit mix keeps probes
// short at any key distribution.
__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

// probe-or-insert; returns acc index (slot, or cap/cap+1 for the special
// groups), or -1 when the probe exhausts the table (table full / corrupt) —
// the caller skips the row and the host fails the task loudly via the error
// flag. The probe is a BOUNDED loop: an unbounded no-side-effect loop
// iteration is UB under C++ forward-progress rules and may be miscompiled.
__device__ __forceinline__ int64_t agg_upsert_slot(const AggTable t, int64_t key,
                                                   bool key_null) {
  if (key_null || key == KEY_EMPTY) {
    int which = key_null ? 1 : 0;
    if (atomicCAS(&t.special_used[which], 0u, 1u) == 0u)
      atomicAdd(t.num_groups, 1ull);
    return t.cap + which;
  }
  uint64_t h = mix64((uint64_t)key);
  int64_t mask = t.cap - 1;
  int64_t i = (int64_t)(h & (uint64_t)mask);
  for (int64_t probe = 0; probe <= mask; probe++) {
    long long cur = t.slots[i].key;
    if (cur == key) return i;
    if (cur == KEY_EMPTY) {
      long long prev = (long long)atomicCAS((unsigned long long*)&t.slots[i].key,
                                            (unsigned long long)KEY_EMPTY,
                                            (unsigned long long)key);
      if (prev == KEY_EMPTY) {
        atomicAdd(t.num_groups, 1ull);
        return i;
      }
      if (prev == key) return i;
    }
    i = (i + 1) & mask;
  }
  atomicOr(t.error_flag, 1u);
  return -1;
}

// order-preserving f64 <-> u64 map: monotone, so u64 atomicMin/atomicMax
// implement f64 min/max (maxmin.rs:104-119 compare-and-keep semantics for
// all comparable values; see kernels.h on the NaN-sentinel edge)
// typed order map: i64 mode flips the sign bit (monotone over int64).
// Sentinel collisions in i64 mode: min acc == i64::MAX maps to ~0 and
// max acc == i64::MIN maps to 0 (the init sentinels) — a group whose every
// value is that extreme reads as empty; documented edge like the f64 NaN
// payloads. `v` carries the 8 raw value bytes through a double register
// (loads/stores are bit-preserving; no FP arithmetic touches it).
__device__ __forceinline__ uint64_t val_omap(double v, bool is_int) {
  uint64_t b;
  memcpy(&b, &v, 8);
  if (is_int) return b ^ 0x8000000000000000ull;
  return (b >> 63) ? ~b : (b | 0x8000000000000000ull);
}
// bit-domain forms (value already as raw 8-byte bits)
__device__ __forceinline__ uint64_t val_omap_bits(uint64_t b, bool is_int) {
  if (is_int) return b ^ 0x8000000000000000ull;
  return (b >> 63) ? ~b : (b | 0x8000000000000000ull);
}
__device__ __forceinline__ uint64_t val_omap_inv_bits(uint64_t u,
                                                      bool is_int) {
  return is_int ? (u ^ 0x8000000000000000ull)
                : ((u >> 63) ? (u & 0x7FFFFFFFFFFFFFFFull) : ~u);
}

__device__ __forceinline__ double val_omap_inv(uint64_t u, bool is_int) {
  uint64_t b = is_int ? (u ^ 0x8000000000000000ull)
                      : ((u >> 63) ? (u & 0x7FFFFFFFFFFFFFFFull) : ~u);
  double x;
  memcpy(&x, &b, 8);
  return x;
}

// typed sum accumulate: i64 mode adds the raw bits with wrapping integer
// arithmetic (sum.rs release-mode `v + x` wraps), f64 mode uses the native
// global f64 atomic add
__device__ __forceinline__ void sum_accum(double* acc, double v, bool is_int) {
  if (is_int) {
    uint64_t b;
    memcpy(&b, &v, 8);
    atomicAdd(reinterpret_cast<unsigned long long*>(acc),
              (unsigned long long)b);
  } else {
    unsafeAtomicAdd(acc, v);
  }
}

// COLLECT_LIST pool append (collect.rs:119-138: every non-null arg).
// Regular keys from the front, null-key rows from the back; overflow raises
// error_flag bit 4 and drops the item (host fails the task loudly).
__device__ __forceinline__ void coll_append(const AggTable& t, int64_t key,
                                            bool knull, uint64_t prio,
                                            double val) {
  uint64_t vb;
  memcpy(&vb, &val, 8);
  if (knull) {
    unsigned long long p = atomicAdd(&t.c_n[1], 1ull);
    if ((int64_t)(p + t.c_n[0]) >= t.c_cap) {  // approximate guard; exact
      atomicOr(t.error_flag, 4u);              // check on host readback
      return;
    }
    int64_t at = t.c_cap - 1 - (int64_t)p;
    t.c_key[at] = 0;
    t.c_prio[at] = prio;
    t.c_val[at] = vb;
  } else {
    unsigned long long p = atomicAdd(&t.c_n[0], 1ull);
    if ((int64_t)(p + t.c_n[1]) >= t.c_cap) {
      atomicOr(t.error_flag, 4u);
      return;
    }
    t.c_key[p] = key;
    t.c_prio[p] = prio;
    t.c_val[p] = vb;
  }
}


// varint read (io/mod.rs:69-79)
__device__ __forceinline__ uint64_t read_varint_dev(const uint8_t* p, int* used) {
  uint64_t len = 0, factor = 1;
  int k = 0;
  while (true) {
    uint8_t v = p[k++];
    if (v < 128) {
      len += (uint64_t)v * factor;
      break;
    }
    len += (uint64_t)(v - 128) * factor;
    factor *= 128;
  }
  *used = k;
  return len;
}

__device__ __forceinline__ int varint_len_dev(uint64_t v) {
  int k = 1;
  while (v >= 128) {
    v /= 128;
    k++;
  }
  return k;
}

// agg layout: 3 bits per agg, LSB-first, 0-terminated. All aggs share one
// argument column, so SUM/AVG parts duplicate the same sum and COUNT/AVG
// parts the same count (avg.rs:208-217: AVG freeze = sum ++ count).
// AGGL_SUM=1, AGGL_CNT=2, AGGL_AVG=3, AGGL_MIN=4, AGGL_MAX=5; MIN/MAX parts
// are prim freezes [u8 valid][8B LE f64]? like SUM's (acc.rs:335-347 — the
// maxmin accumulator is the same generic prim column, maxmin.rs:91-93).
// minu/maxu are order-mapped u64 accumulators (sentinel = invalid).
// per-record accumulator snapshot for freeze/parse. FIRST freeze = prim
// value part ++ flag byte 0|2 (first.rs:315-319 freeze_to_rows = values ++
// flags; acc.rs:156-166 AccBooleanColumn byte 0 = None / 1+v = Some(v), and
// the FIRST flag is only ever Some(true) = 2). FIRST_IGNORES_NULL freeze =
// prim value part only (first_ignores_null.rs:79-81 generic prim column).
struct AccSnap {
  bool valid = false;    // sum part valid
  double sum = 0.0;
  uint64_t cnt = 0;
  uint64_t minu = MM_MIN_INIT, maxu = MM_MAX_INIT;
  uint8_t f_st = 0;      // FIRST state: 0 untouched / 1 first-null / 2 valid
  double f_val = 0.0;
  uint8_t fn_st = 0;     // FIRST_IGNORES_NULL state: 0 untouched / 2 valid
  double fn_val = 0.0;
  // COLLECT_LIST part: run of 8-byte values (freeze reads them, parse
  // exposes the frozen bytes; collect.rs:237-241 save_raw)
  const unsigned long long* c_vals = nullptr;  // 8-aligned? frozen bytes are
  const uint8_t* c_raw = nullptr;              // not aligned — use c_raw
  uint32_t c_cnt = 0;
};

__device__ __forceinline__ int agg_freeze_len(uint32_t layout,
                                              const AccSnap& a) {
  int len = 0;
  for (uint32_t l = layout; l & 15u; l >>= 4) {
    uint32_t k = l & 15u;
    if (k == 1 || k == 3) len += 1 + (a.valid ? 8 : 0);
    if (k == 2 || k == 3) len += varint_len_dev(a.cnt);
    if (k == 4) len += 1 + (a.minu != MM_MIN_INIT ? 8 : 0);
    if (k == 5) len += 1 + (a.maxu != MM_MAX_INIT ? 8 : 0);
    if (k == 6) len += 2 + (a.f_st == 2 ? 8 : 0);  // prim part + flag byte
    if (k == 7) len += 1 + (a.fn_st == 2 ? 8 : 0);
    if (k == 8 || k == 9)  // collect.rs:237-241: varint(raw_len) ++ values
      len += varint_len_dev((uint64_t)a.c_cnt * 8) + (int)a.c_cnt * 8;
  }
  return len;
}

__device__ __forceinline__ uint8_t* agg_prim_freeze_part(bool valid, double v,
                                                         uint8_t* p) {
  if (valid) {
    *p++ = 1;
    memcpy(p, &v, 8);
    p += 8;
  } else {
    *p++ = 0;
  }
  return p;
}

__device__ __forceinline__ uint8_t* agg_freeze_write_rec(uint32_t layout,
                                                         const AccSnap& a,
                                                         uint8_t* p,
                                                         bool is_int = false) {
  const bool valid = a.valid;
  const double sum = a.sum;
  const uint64_t cnt = a.cnt;
  for (uint32_t l = layout; l & 15u; l >>= 4) {
    uint32_t k = l & 15u;
    if (k == 1 || k == 3)  // acc.rs:335-347 prim freeze
      p = agg_prim_freeze_part(valid, sum, p);
    if (k == 4)
      p = agg_prim_freeze_part(a.minu != MM_MIN_INIT,
                               val_omap_inv(a.minu, is_int), p);
    if (k == 5)
      p = agg_prim_freeze_part(a.maxu != MM_MAX_INIT,
                               val_omap_inv(a.maxu, is_int), p);
    if (k == 6) {  // FIRST: prim value ++ flag byte (first.rs:315-319)
      p = agg_prim_freeze_part(a.f_st == 2, a.f_val, p);
      *p++ = a.f_st ? 2 : 0;
    }
    if (k == 7)  // FIRST_IGNORES_NULL: prim value only
      p = agg_prim_freeze_part(a.fn_st == 2, a.fn_val, p);
    if (k == 8 || k == 9) {  // COLLECT raw list (collect.rs:237-241)
      uint64_t raw = (uint64_t)a.c_cnt * 8;
      int used = varint_len_dev(raw);
      for (uint64_t v = raw; ; v /= 128) {
        *p++ = (uint8_t)(v >= 128 ? (v % 128) + 128 : v);
        if (v < 128) break;
      }
      (void)used;
      for (uint32_t i = 0; i < a.c_cnt; i++) {
        if (a.c_vals) {
          memcpy(p, &a.c_vals[i], 8);
        } else if (a.c_raw) {
          memcpy(p, a.c_raw + (size_t)i * 8, 8);
        }
        p += 8;
      }
    }
    if (k == 2 || k == 3) {  // count.rs:193-203 varint
      uint64_t c = cnt;
      while (c >= 128) {
        *p++ = (uint8_t)(128 + c % 128);
        c /= 128;
      }
      *p++ = (uint8_t)c;
    }
  }
  return p;
}

// parse one frozen record; accumulates only the FIRST part of each family
// (the rest are duplicates of the same shared-column accs). minu/maxu come
// back order-mapped, at their sentinels when the part is absent or invalid —
// so merging them with atomicMin/atomicMax needs no guard.
__device__ __forceinline__ void agg_parse_frozen(uint32_t layout,
                                                 const uint8_t* p, AccSnap* a,
                                                 bool is_int = false) {
  *a = AccSnap{};
  uint32_t got = 0;  // bit per family
  for (uint32_t l = layout; l & 15u; l >>= 4) {
    uint32_t k = l & 15u;
    if (k != 2 && k != 8 && k != 9) {  // prim-headed parts
      uint8_t v = *p++;
      double x = 0;
      if (v) {
        memcpy(&x, p, 8);
        p += 8;
      }
      if ((k == 1 || k == 3) && !(got & 1u)) {
        a->valid = v != 0;
        a->sum = x;
        got |= 1u;
      } else if (k == 4 && !(got & 2u)) {
        if (v) a->minu = val_omap(x, is_int);
        got |= 2u;
      } else if (k == 5 && !(got & 4u)) {
        if (v) a->maxu = val_omap(x, is_int);
        got |= 4u;
      } else if (k == 6 && !(got & 8u)) {
        a->f_val = x;
        a->f_st = v ? 2 : 0;  // refined by the flag byte below
        got |= 8u;
      } else if (k == 7 && !(got & 16u)) {
        a->fn_val = x;
        a->fn_st = v ? 2 : 0;
        got |= 16u;
      }
    }
    if (k == 6) {  // FIRST flag byte: 0 = untouched, 2 = touched
      uint8_t fl = *p++;
      if (a->f_st == 0 && fl) a->f_st = 1;  // touched but first value null
    }
    if (k == 8 || k == 9) {  // COLLECT raw list: expose, then skip
      int used;
      uint64_t raw = read_varint_dev(p, &used);
      p += used;
      if (!a->c_raw) {
        a->c_raw = p;
        a->c_cnt = (uint32_t)(raw / 8);
      }
      p += raw;
    }
    if (k == 2 || k == 3) {
      int used;
      uint64_t c = read_varint_dev(p, &used);
      p += used;
      if (!(got & 32u)) {
        a->cnt = c;
        got |= 32u;
      }
    }
  }
}


}  // namespace auron
