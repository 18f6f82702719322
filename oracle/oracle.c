/* oracle.c — CPU restatement of the Auron hash-agg + shuffle hot path.
 * TEST INFRASTRUCTURE ONLY (see oracle.h header comment).
 * Built with: gcc -O2 -shared -fPIC oracle.c -o liboracle.so -ldl
 */
#include "oracle.h"

#include <dlfcn.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

/* ======================= hashes ======================= */
/* mur.rs:38-63 */
static inline uint32_t rotl32(uint32_t x, int r) { return (x << r) | (x >> (32 - r)); }

static inline int32_t mix_k1(int32_t k1) {
    uint32_t k = (uint32_t)k1;
    k *= 0xcc9e2d51u;
    k = rotl32(k, 15);
    k *= 0x1b873593u;
    return (int32_t)k;
}

static inline int32_t mix_h1(int32_t h1, int32_t k1) {
    uint32_t h = (uint32_t)h1 ^ (uint32_t)k1;
    h = rotl32(h, 13);
    h = h * 5u + 0xe6546b64u;
    return (int32_t)h;
}

static inline int32_t fmix(int32_t h1, int32_t len) {
    uint32_t h = (uint32_t)h1 ^ (uint32_t)len;
    h ^= h >> 16;
    h *= 0x85ebca6bu;
    h ^= h >> 13;
    h *= 0xc2b2ae35u;
    h ^= h >> 16;
    return (int32_t)h;
}

static inline uint32_t read32le(const uint8_t* p) {
    uint32_t v;
    memcpy(&v, p, 4);
    return v; /* x86-64 is little-endian; read32 in hash/mod.rs is LE */
}

static inline uint64_t read64le(const uint8_t* p) {
    uint64_t v;
    memcpy(&v, p, 8);
    return v;
}

/* mur.rs:19-30 spark_compatible_murmur3_hash */
int32_t oracle_murmur3(const uint8_t* data, size_t len, int32_t seed) {
    size_t aligned = len - len % 4;
    int32_t h1 = seed;
    for (size_t i = 0; i < aligned; i += 4) { /* mur.rs:65-74 hash_bytes_by_int */
        h1 = mix_h1(h1, mix_k1((int32_t)read32le(data + i)));
    }
    for (size_t i = aligned; i < len; i++) { /* mur.rs:25-28: sign-extended byte */
        int32_t half_word = (int32_t)(int8_t)data[i];
        h1 = mix_h1(h1, mix_k1(half_word));
    }
    return fmix(h1, (int32_t)len);
}

/* mur.rs:76-87 hash_long */
int32_t oracle_murmur3_long(int64_t value, int32_t seed) {
    int32_t low = (int32_t)value;
    int32_t high = (int32_t)((uint64_t)value >> 32);
    int32_t h1 = mix_h1(seed, mix_k1(low));
    h1 = mix_h1(h1, mix_k1(high));
    return fmix(h1, 8);
}

/* xxhash.rs:17-96 */
#define PRIME64_1 0x9E3779B185EBCA87ull
#define PRIME64_2 0xC2B2AE3D27D4EB4Full
#define PRIME64_3 0x165667B19E3779F9ull
#define PRIME64_4 0x85EBCA77C2B2AE63ull
#define PRIME64_5 0x27D4EB2F165667C5ull

static inline uint64_t rotl64(uint64_t x, int r) { return (x << r) | (x >> (64 - r)); }

static inline uint64_t xxh64_round(uint64_t acc, uint64_t input) {
    acc += input * PRIME64_2;
    acc = rotl64(acc, 31);
    acc *= PRIME64_1;
    return acc;
}

static inline uint64_t xxh64_merge_round(uint64_t hash, uint64_t acc) {
    hash ^= xxh64_round(0, acc);
    hash = hash * PRIME64_1 + PRIME64_4;
    return hash;
}

int64_t oracle_xxhash64(const uint8_t* input, size_t len, int64_t seed_) {
    uint64_t seed = (uint64_t)seed_;
    uint64_t hash;
    size_t remaining = len, offset = 0;
    if (remaining >= 32) {
        uint64_t acc1 = seed + PRIME64_1 + PRIME64_2;
        uint64_t acc2 = seed + PRIME64_2;
        uint64_t acc3 = seed;
        uint64_t acc4 = seed - PRIME64_1;
        while (remaining >= 32) {
            acc1 = xxh64_round(acc1, read64le(input + offset)); offset += 8;
            acc2 = xxh64_round(acc2, read64le(input + offset)); offset += 8;
            acc3 = xxh64_round(acc3, read64le(input + offset)); offset += 8;
            acc4 = xxh64_round(acc4, read64le(input + offset)); offset += 8;
            remaining -= 32;
        }
        hash = rotl64(acc1, 1) + rotl64(acc2, 7) + rotl64(acc3, 12) + rotl64(acc4, 18);
        hash = xxh64_merge_round(hash, acc1);
        hash = xxh64_merge_round(hash, acc2);
        hash = xxh64_merge_round(hash, acc3);
        hash = xxh64_merge_round(hash, acc4);
    } else {
        hash = seed + PRIME64_5;
    }
    hash += (uint64_t)len;
    while (remaining >= 8) {
        hash ^= xxh64_round(0, read64le(input + offset));
        hash = rotl64(hash, 27) * PRIME64_1 + PRIME64_4;
        offset += 8; remaining -= 8;
    }
    if (remaining >= 4) {
        hash ^= (uint64_t)read32le(input + offset) * PRIME64_1;
        hash = rotl64(hash, 23) * PRIME64_2 + PRIME64_3;
        offset += 4; remaining -= 4;
    }
    while (remaining != 0) {
        hash ^= (uint64_t)input[offset] * PRIME64_5;
        hash = rotl64(hash, 11) * PRIME64_1;
        offset += 1; remaining -= 1;
    }
    /* xxh64_avalanche */
    hash ^= hash >> 33;
    hash *= PRIME64_2;
    hash ^= hash >> 29;
    hash *= PRIME64_3;
    hash ^= hash >> 32;
    return (int64_t)hash;
}

/* ---- column folds: spark_hash.rs:28-57 + hash_array_primitive ---- */
static inline int bit_get(const uint8_t* bm, size_t i) {
    return (bm[i >> 3] >> (i & 7)) & 1;
}

void oracle_hash_col_i64(const int64_t* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes) {
    for (size_t i = 0; i < n; i++) {
        if (!valid || bit_get(valid, i)) {
            hashes[i] = oracle_murmur3_long(vals[i], hashes[i]);
        }
    }
}

void oracle_hash_col_i32(const int32_t* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes) {
    for (size_t i = 0; i < n; i++) {
        if (!valid || bit_get(valid, i)) {
            hashes[i] = oracle_murmur3((const uint8_t*)&vals[i], 4, hashes[i]);
        }
    }
}

void oracle_hash_col_f64(const double* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes) {
    for (size_t i = 0; i < n; i++) {
        if (!valid || bit_get(valid, i)) {
            hashes[i] = oracle_murmur3((const uint8_t*)&vals[i], 8, hashes[i]);
        }
    }
}

void oracle_xxhash_col_i64(const int64_t* vals, const uint8_t* valid, size_t n,
                           int64_t* hashes) {
    for (size_t i = 0; i < n; i++) {
        if (!valid || bit_get(valid, i)) {
            hashes[i] = oracle_xxhash64((const uint8_t*)&vals[i], 8, hashes[i]);
        }
    }
}

/* shuffle/mod.rs:178-188: rem_euclid */
void oracle_partition_ids(const int32_t* hashes, size_t n, uint32_t num_partitions,
                          uint32_t* out) {
    int32_t m = (int32_t)num_partitions;
    for (size_t i = 0; i < n; i++) {
        int32_t r = hashes[i] % m;
        if (r < 0) r += m;
        out[i] = (uint32_t)r;
    }
}

/* ======================= radix_sort_by_key ======================= */
/* rdx_sort.rs:24-74: exact restatement (American-flag, unstable) over
 * (part_id, batch_idx, row_idx) triples with key = item[0]. */
typedef struct { size_t cur, end; } RdxPart;

void oracle_radix_sort_triples(uint32_t* items, size_t n, size_t num_keys,
                               size_t* counts) {
    RdxPart* parts = calloc(num_keys, sizeof(RdxPart));
    size_t* inexhausted = malloc(num_keys * sizeof(size_t));
    for (size_t i = 0; i < n; i++) counts[items[i * 3]]++;
    size_t beg = 0;
    for (size_t idx = 0; idx < num_keys; idx++) {
        if (counts[idx] > 0) {
            parts[idx].cur = beg;
            parts[idx].end = beg + counts[idx];
            beg += counts[idx];
        }
    }
    size_t num_inex = num_keys;
    for (size_t i = 0; i < num_keys; i++) inexhausted[i] = i;
    for (;;) {
        /* retain parts with cur < end */
        size_t m = 0;
        for (size_t i = 0; i < num_inex; i++) {
            size_t p = inexhausted[i];
            if (parts[p].cur < parts[p].end) inexhausted[m++] = p;
        }
        num_inex = m;
        if (num_inex <= 1) break;
        for (size_t i = 0; i < num_inex; i++) {
            size_t part_idx = inexhausted[i];
            size_t cur = parts[part_idx].cur;
            size_t end = parts[part_idx].end;
            for (size_t item_idx = cur; item_idx < end; item_idx++) {
                uint32_t target = items[item_idx * 3];
                RdxPart* tp = &parts[target];
                /* swap items[item_idx] <-> items[tp->cur] */
                size_t j = tp->cur;
                uint32_t t0 = items[item_idx * 3], t1 = items[item_idx * 3 + 1],
                         t2 = items[item_idx * 3 + 2];
                items[item_idx * 3] = items[j * 3];
                items[item_idx * 3 + 1] = items[j * 3 + 1];
                items[item_idx * 3 + 2] = items[j * 3 + 2];
                items[j * 3] = t0; items[j * 3 + 1] = t1; items[j * 3 + 2] = t2;
                tp->cur++;
            }
        }
    }
    free(parts);
    free(inexhausted);
}

/* ======================= varint ======================= */
/* io/mod.rs:60-79 */
size_t oracle_write_len(uint64_t len, uint8_t* out) {
    size_t k = 0;
    while (len >= 128) {
        out[k++] = (uint8_t)(128 + len % 128);
        len /= 128;
    }
    out[k++] = (uint8_t)len;
    return k;
}

size_t oracle_read_len(const uint8_t* in, size_t avail, uint64_t* out) {
    uint64_t len = 0, factor = 1;
    size_t k = 0;
    for (;;) {
        if (k >= avail) return 0;
        uint8_t v = in[k++];
        if (v < 128) { len += (uint64_t)v * factor; break; }
        len += (uint64_t)(v - 128) * factor;
        factor *= 128;
    }
    *out = len;
    return k;
}

/* ======================= hash aggregation ======================= */
/* Insertion-ordered open-addressing map over (key_is_null, key). The internal
 * hash function is NOT part of the parity contract (the reference's foldhash
 * seed 0x3F6F1B93, agg_hash_map.rs:228-234, only shapes its table layout);
 * record order = first-occurrence order is the observable the reference
 * exposes, and that is preserved here. */
typedef struct {
    int64_t key;
    uint8_t key_null;
    uint8_t sum_valid;
    double sum;
    int64_t count;
} AggRecord;

struct OracleAgg {
    AggRecord* records;
    size_t num_records, cap_records;
    /* open addressing: slot -> record idx+1, 0 = empty */
    uint32_t* slots;
    size_t cap_slots; /* power of 2 */
    long long null_record; /* record idx of null-key group, -1 if none */
};

static inline uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

OracleAgg* oracle_agg_new(void) {
    OracleAgg* a = calloc(1, sizeof(OracleAgg));
    a->cap_slots = 1 << 16;
    a->slots = calloc(a->cap_slots, sizeof(uint32_t));
    a->cap_records = 1 << 14;
    a->records = malloc(a->cap_records * sizeof(AggRecord));
    a->null_record = -1;
    return a;
}

void oracle_agg_free(OracleAgg* a) {
    if (!a) return;
    free(a->records);
    free(a->slots);
    free(a);
}

static void agg_rehash(OracleAgg* a) {
    size_t ncap = a->cap_slots * 2;
    uint32_t* nslots = calloc(ncap, sizeof(uint32_t));
    for (size_t r = 0; r < a->num_records; r++) {
        if (a->records[r].key_null) continue;
        uint64_t h = splitmix64((uint64_t)a->records[r].key);
        size_t i = h & (ncap - 1);
        while (nslots[i]) i = (i + 1) & (ncap - 1);
        nslots[i] = (uint32_t)(r + 1);
    }
    free(a->slots);
    a->slots = nslots;
    a->cap_slots = ncap;
}

static size_t agg_upsert(OracleAgg* a, int64_t key, int key_null) {
    if (key_null) {
        if (a->null_record < 0) {
            a->null_record = (long long)a->num_records;
            goto new_record;
        }
        return (size_t)a->null_record;
    }
    {
        uint64_t h = splitmix64((uint64_t)key);
        size_t i = h & (a->cap_slots - 1);
        while (a->slots[i]) {
            AggRecord* r = &a->records[a->slots[i] - 1];
            if (!r->key_null && r->key == key) return (size_t)(a->slots[i] - 1);
            i = (i + 1) & (a->cap_slots - 1);
        }
        if ((a->num_records + 1) * 2 > a->cap_slots) {
            agg_rehash(a);
            i = h & (a->cap_slots - 1);
            while (a->slots[i]) i = (i + 1) & (a->cap_slots - 1);
        }
        a->slots[i] = (uint32_t)(a->num_records + 1);
    }
new_record:
    if (a->num_records == a->cap_records) {
        a->cap_records *= 2;
        a->records = realloc(a->records, a->cap_records * sizeof(AggRecord));
    }
    {
        AggRecord* r = &a->records[a->num_records];
        r->key = key;
        r->key_null = (uint8_t)key_null;
        r->sum_valid = 0;
        r->sum = 0.0;
        r->count = 0;
        return a->num_records++;
    }
}

/* sum.rs:90-115 partial_update + count.rs:90-149 partial_update, row order */
void oracle_agg_update(OracleAgg* a, const int64_t* keys, const uint8_t* key_valid,
                       const double* vals, const uint8_t* val_valid, size_t n) {
    for (size_t i = 0; i < n; i++) {
        int knull = key_valid && !bit_get(key_valid, i);
        size_t r = agg_upsert(a, knull ? 0 : keys[i], knull);
        int vvalid = !val_valid || bit_get(val_valid, i);
        if (vvalid) {
            AggRecord* rec = &a->records[r];
            if (rec->sum_valid) {
                rec->sum += vals[i];
            } else { /* acc.rs:272-280 update_value: first value replaces */
                rec->sum = vals[i];
                rec->sum_valid = 1;
            }
            rec->count += 1; /* COUNT(val): all args non-null */
        }
    }
}

/* a8 wire format: per row, concat of per-agg freeze bytes:
 * SUM(f64): acc.rs:335-347 — u8 valid, then 8B LE value iff valid
 * COUNT:    count.rs:193-203 — write_len(count) */
void oracle_agg_merge_frozen(OracleAgg* a, const int64_t* keys,
                             const uint8_t* key_valid, const uint8_t* acc_data,
                             const int64_t* acc_offsets, size_t n) {
    for (size_t i = 0; i < n; i++) {
        int knull = key_valid && !bit_get(key_valid, i);
        size_t r = agg_upsert(a, knull ? 0 : keys[i], knull);
        const uint8_t* p = acc_data + acc_offsets[i];
        size_t avail = (size_t)(acc_offsets[i + 1] - acc_offsets[i]);
        uint8_t valid = p[0];
        size_t off = 1;
        double mval = 0.0;
        if (valid) {
            memcpy(&mval, p + 1, 8);
            off += 8;
        }
        uint64_t mcount = 0;
        off += oracle_read_len(p + off, avail - off, &mcount);
        AggRecord* rec = &a->records[r];
        if (valid) { /* sum.rs:117-145 partial_merge */
            if (rec->sum_valid) rec->sum += mval;
            else { rec->sum = mval; rec->sum_valid = 1; }
        }
        rec->count += (int64_t)mcount; /* count.rs:152-175 partial_merge */
    }
}

size_t oracle_agg_num_groups(const OracleAgg* a) { return a->num_records; }

void oracle_agg_output(const OracleAgg* a, int64_t* keys, uint8_t* key_valid,
                       double* sums, uint8_t* sum_valid, int64_t* counts) {
    for (size_t r = 0; r < a->num_records; r++) {
        const AggRecord* rec = &a->records[r];
        if (keys) keys[r] = rec->key;
        if (key_valid) key_valid[r] = !rec->key_null;
        if (sums) sums[r] = rec->sum;
        if (sum_valid) sum_valid[r] = rec->sum_valid;
        if (counts) counts[r] = rec->count;
    }
}

size_t oracle_agg_freeze(const OracleAgg* a, uint8_t* data, int64_t* offsets) {
    size_t total = 0;
    uint8_t tmp[20];
    for (size_t r = 0; r < a->num_records; r++) {
        const AggRecord* rec = &a->records[r];
        if (offsets) offsets[r] = (int64_t)total;
        size_t k = 0;
        tmp[k++] = rec->sum_valid ? 1 : 0;
        if (rec->sum_valid) {
            memcpy(tmp + k, &rec->sum, 8);
            k += 8;
        }
        k += oracle_write_len((uint64_t)rec->count, tmp + k);
        if (data) memcpy(data + total, tmp, k);
        total += k;
    }
    if (offsets) offsets[a->num_records] = (int64_t)total;
    return total;
}

/* ======================= batch_serde ======================= */
/* byte-plane transpose: batch_serde.rs:271-306 uses transpose::transpose with
 * (byte_width, len) -> output plane-major: out[b*n + i] = in[i*w + b] */
static void transpose_bytes(const uint8_t* in, uint8_t* out, size_t w, size_t n) {
    for (size_t i = 0; i < n; i++)
        for (size_t b = 0; b < w; b++)
            out[b * n + i] = in[i * w + b];
}

size_t oracle_serde_col_prim(const uint8_t* values, size_t byte_width, size_t n,
                             const uint8_t* valid, uint8_t* out) {
    size_t total = 0;
    uint8_t hdr[10];
    size_t bm_len = (n + 7) / 8;
    if (valid) {
        size_t k = oracle_write_len(1, hdr);
        if (out) memcpy(out + total, hdr, k);
        total += k;
        if (out) memcpy(out + total, valid, bm_len);
        total += bm_len;
    } else {
        size_t k = oracle_write_len(0, hdr);
        if (out) memcpy(out + total, hdr, k);
        total += k;
    }
    if (byte_width > 1) {
        if (out) transpose_bytes(values, out + total, byte_width, n);
        total += byte_width * n;
    } else {
        if (out) memcpy(out + total, values, n);
        total += n;
    }
    return total;
}

size_t oracle_serde_col_bytes(const uint8_t* data, const int64_t* offsets, size_t n,
                              const uint8_t* valid, uint8_t* out) {
    size_t total = 0;
    uint8_t hdr[10];
    size_t bm_len = (n + 7) / 8;
    if (valid) {
        size_t k = oracle_write_len(1, hdr);
        if (out) memcpy(out + total, hdr, k);
        total += k;
        if (out) memcpy(out + total, valid, bm_len);
        total += bm_len;
    } else {
        size_t k = oracle_write_len(0, hdr);
        if (out) memcpy(out + total, hdr, k);
        total += k;
    }
    /* lens as i32, byte-transposed (batch_serde.rs:217-240 write_offsets) */
    if (out) {
        int32_t* lens = malloc(n * sizeof(int32_t));
        for (size_t i = 0; i < n; i++) lens[i] = (int32_t)(offsets[i + 1] - offsets[i]);
        transpose_bytes((const uint8_t*)lens, out + total, 4, n);
        free(lens);
    }
    total += 4 * n;
    size_t data_len = (size_t)(offsets[n] - offsets[0]);
    if (out) memcpy(out + total, data + offsets[0], data_len);
    total += data_len;
    return total;
}

/* ======================= IPC compression ======================= */
/* LZ4F frame API bound from liblz4.so.1 at run time (no lz4 headers in this
 * image). Prototypes restated from the public lz4frame.h API. */
typedef size_t (*LZ4F_compressBound_t)(size_t srcSize, const void* prefsPtr);
typedef size_t (*LZ4F_compressFrame_t)(void* dst, size_t dstCap, const void* src,
                                       size_t srcSize, const void* prefsPtr);
typedef unsigned (*LZ4F_isError_t)(size_t code);
typedef size_t (*LZ4F_createDecompressionContext_t)(void** ctx, unsigned version);
typedef size_t (*LZ4F_freeDecompressionContext_t)(void* ctx);
typedef size_t (*LZ4F_decompress_t)(void* ctx, void* dst, size_t* dstSize,
                                    const void* src, size_t* srcSize, const void* opt);

static struct {
    void* handle;
    LZ4F_compressBound_t compressBound;
    LZ4F_compressFrame_t compressFrame;
    LZ4F_isError_t isError;
    LZ4F_createDecompressionContext_t createDCtx;
    LZ4F_freeDecompressionContext_t freeDCtx;
    LZ4F_decompress_t decompress;
} g_lz4;

static int lz4_init(void) {
    if (g_lz4.handle) return 1;
    void* h = dlopen("liblz4.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("liblz4.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) return 0;
    g_lz4.compressBound = (LZ4F_compressBound_t)dlsym(h, "LZ4F_compressBound");
    g_lz4.compressFrame = (LZ4F_compressFrame_t)dlsym(h, "LZ4F_compressFrame");
    g_lz4.isError = (LZ4F_isError_t)dlsym(h, "LZ4F_isError");
    g_lz4.createDCtx = (LZ4F_createDecompressionContext_t)dlsym(
        h, "LZ4F_createDecompressionContext");
    g_lz4.freeDCtx = (LZ4F_freeDecompressionContext_t)dlsym(
        h, "LZ4F_freeDecompressionContext");
    g_lz4.decompress = (LZ4F_decompress_t)dlsym(h, "LZ4F_decompress");
    if (!g_lz4.compressBound || !g_lz4.compressFrame || !g_lz4.isError ||
        !g_lz4.createDCtx || !g_lz4.freeDCtx || !g_lz4.decompress)
        return 0;
    g_lz4.handle = h;
    return 1;
}

struct OracleIpcWriter {
    uint8_t* out;
    size_t out_len, out_cap;
    uint8_t* staged; /* uncompressed payload of the current block */
    size_t staged_len, staged_cap;
    size_t target;
    int codec;      /* 0 = lz4 frames, 1 = zstd (ipc_compression.rs:189-196) */
    int zstd_level; /* SPARK_IO_COMPRESSION_ZSTD_LEVEL, reference default 1 */
};

/* zstd simple API via dlopen (no headers in image) */
typedef size_t (*ZSTD_compressBound_t)(size_t);
typedef size_t (*ZSTD_compress_t)(void*, size_t, const void*, size_t, int);
typedef unsigned long long (*ZSTD_getFrameContentLength_t)(const void*, size_t);
typedef size_t (*ZSTD_decompress_t)(void*, size_t, const void*, size_t);
typedef unsigned (*ZSTD_isError_t)(size_t);
static struct {
    void* handle;
    ZSTD_compressBound_t bound;
    ZSTD_compress_t compress;
    ZSTD_getFrameContentLength_t clen;
    ZSTD_decompress_t decompress;
    ZSTD_isError_t iserr;
} g_zstd;

static int zstd_init(void) {
    if (g_zstd.handle) return 1;
    void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) h = dlopen("libzstd.so", RTLD_NOW | RTLD_GLOBAL);
    if (!h) return 0;
    g_zstd.handle = h;
    g_zstd.bound = (ZSTD_compressBound_t)dlsym(h, "ZSTD_compressBound");
    g_zstd.compress = (ZSTD_compress_t)dlsym(h, "ZSTD_compress");
    g_zstd.clen = (ZSTD_getFrameContentLength_t)dlsym(h,
                                                 "ZSTD_getDecompressedSize");
    g_zstd.decompress = (ZSTD_decompress_t)dlsym(h, "ZSTD_decompress");
    g_zstd.iserr = (ZSTD_isError_t)dlsym(h, "ZSTD_isError");
    return g_zstd.bound && g_zstd.compress && g_zstd.clen &&
           g_zstd.decompress && g_zstd.iserr;
}

static void buf_reserve(uint8_t** buf, size_t* cap, size_t need) {
    if (need <= *cap) return;
    size_t ncap = *cap ? *cap : 4096;
    while (ncap < need) ncap *= 2;
    *buf = realloc(*buf, ncap);
    *cap = ncap;
}

OracleIpcWriter* oracle_ipc_writer_new2(size_t target_block_size, int codec,
                                        int zstd_level);

OracleIpcWriter* oracle_ipc_writer_new(size_t target_block_size) {
    return oracle_ipc_writer_new2(target_block_size, 0, 1);
}

OracleIpcWriter* oracle_ipc_writer_new2(size_t target_block_size, int codec,
                                        int zstd_level) {
    OracleIpcWriter* w = calloc(1, sizeof(*w));
    w->target = target_block_size ? target_block_size : 4194304; /* conf default */
    w->codec = codec;
    w->zstd_level = zstd_level ? zstd_level : 1;
    return w;
}

void oracle_ipc_writer_free(OracleIpcWriter* w) {
    if (!w) return;
    free(w->out);
    free(w->staged);
    free(w);
}

int oracle_ipc_finish_block(OracleIpcWriter* w) {
    size_t clen, bound;
    if (w->staged_len == 0) return 0;
    if (w->codec == 1) {
        if (!zstd_init()) return -1;
        bound = g_zstd.bound(w->staged_len) + 64;
        buf_reserve(&w->out, &w->out_cap, w->out_len + 4 + bound);
        clen = g_zstd.compress(w->out + w->out_len + 4, bound, w->staged,
                               w->staged_len, w->zstd_level);
        if (g_zstd.iserr(clen)) return -1;
        goto framed;
    }
    if (!lz4_init()) return -1;
    bound = g_lz4.compressBound(w->staged_len, NULL) + 64;
    buf_reserve(&w->out, &w->out_cap, w->out_len + 4 + bound);
    clen = g_lz4.compressFrame(w->out + w->out_len + 4, bound, w->staged,
                               w->staged_len, NULL);
    if (g_lz4.isError(clen)) return -1;
framed:;
    uint32_t len32 = (uint32_t)clen;
    memcpy(w->out + w->out_len, &len32, 4); /* u32-LE, ipc_compression.rs:87-92 */
    w->out_len += 4 + clen;
    w->staged_len = 0;
    return 0;
}

int oracle_ipc_write_payload(OracleIpcWriter* w, const uint8_t* payload, size_t len) {
    buf_reserve(&w->staged, &w->staged_cap, w->staged_len + len);
    memcpy(w->staged + w->staged_len, payload, len);
    w->staged_len += len;
    /* ipc_compression.rs:72-79: flush at 0.9 * target (approximated on the
     * uncompressed side; block split points are not part of the parity
     * contract, only the framing is) */
    if ((double)w->staged_len >= (double)w->target * 0.9) {
        return oracle_ipc_finish_block(w);
    }
    return 0;
}

size_t oracle_ipc_bytes(OracleIpcWriter* w, const uint8_t** data) {
    *data = w->out;
    return w->out_len;
}

/* zstd variant of the block-stream decoder */
size_t oracle_ipc_decode_zstd(const uint8_t* in, size_t in_len, uint8_t* out,
                              size_t out_cap) {
    size_t pos = 0, out_len = 0;
    if (!zstd_init()) return (size_t)-1;
    while (pos + 4 <= in_len) {
        uint32_t block_len;
        unsigned long long need;
        memcpy(&block_len, in + pos, 4);
        pos += 4;
        if (pos + block_len > in_len) return (size_t)-1;
        need = g_zstd.clen(in + pos, block_len);
        if (need == 0 && block_len > 0) return (size_t)-1; /* unknown size */
        if (out && out_len + need <= out_cap) {
            size_t rc = g_zstd.decompress(out + out_len, (size_t)need,
                                          in + pos, block_len);
            if (g_zstd.iserr(rc) || rc != (size_t)need) return (size_t)-1;
        }
        out_len += (size_t)need;
        pos += block_len;
    }
    return out_len;
}

size_t oracle_ipc_decode(const uint8_t* in, size_t in_len, uint8_t* out,
                         size_t out_cap) {
    if (!lz4_init()) return (size_t)-1;
    size_t pos = 0, out_len = 0;
    while (pos + 4 <= in_len) {
        uint32_t block_len;
        memcpy(&block_len, in + pos, 4);
        pos += 4;
        if (pos + block_len > in_len) return (size_t)-1;
        void* ctx = NULL;
        if (g_lz4.isError(g_lz4.createDCtx(&ctx, 100))) return (size_t)-1;
        size_t src_pos = 0;
        while (src_pos < block_len) {
            uint8_t sink[1 << 16];
            size_t dst_size = out ? (out_cap - out_len) : sizeof(sink);
            uint8_t* dst = out ? out + out_len : sink;
            if (out && dst_size == 0) { g_lz4.freeDCtx(ctx); return (size_t)-1; }
            if (!out && dst_size > sizeof(sink)) dst_size = sizeof(sink);
            size_t src_size = block_len - src_pos;
            size_t rc = g_lz4.decompress(ctx, dst, &dst_size, in + pos + src_pos,
                                         &src_size, NULL);
            if (g_lz4.isError(rc)) { g_lz4.freeDCtx(ctx); return (size_t)-1; }
            src_pos += src_size;
            out_len += dst_size;
            if (src_size == 0 && dst_size == 0) break;
        }
        g_lz4.freeDCtx(ctx);
        pos += block_len;
    }
    return out_len;
}
