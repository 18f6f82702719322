/* oracle.h — CPU restatement of the kwai/blaze (Apache Auron) hash-agg + shuffle
 * hot path, used ONLY as the parity oracle and CPU baseline.
 *
 * TEST INFRASTRUCTURE ONLY: this library may be imported/linked/executed only by
 * tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg. The product
 * path (blaze_amd + libauron_hip.so) must never route through it.
 *
 * Every function cites the reference file:line (under /root/reference) whose
 * behavior it restates. The reference is Rust (not buildable in this container:
 * no cargo/rustc); parity of this restatement is pinned by the reference's own
 * golden vectors transcribed into tests/golden/ (see each citation).
 */
#ifndef AURON_ORACLE_H
#define AURON_ORACLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- Spark-compatible hashes ----------------------------------------- */
/* native-engine/datafusion-ext-commons/src/hash/mur.rs:19-87
 * (golden vectors: spark_hash.rs:377-482, mur.rs:94-103) */
int32_t oracle_murmur3(const uint8_t* data, size_t len, int32_t seed);
int32_t oracle_murmur3_long(int64_t value, int32_t seed);
/* native-engine/datafusion-ext-commons/src/hash/xxhash.rs:17-96 */
int64_t oracle_xxhash64(const uint8_t* data, size_t len, int64_t seed);

/* Column-fold loops: spark_hash.rs:28-57 create_murmur3_hashes /
 * create_xxhash64_hashes + hash_array_primitive (null rows keep previous
 * hash). `hashes` must be pre-filled with the seed by the caller for the
 * first column. `valid` is an Arrow validity bitmap (LSB-first) or NULL. */
void oracle_hash_col_i64(const int64_t* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes);
void oracle_hash_col_i32(const int32_t* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes);
void oracle_hash_col_f64(const double* vals, const uint8_t* valid, size_t n,
                         int32_t* hashes);
void oracle_xxhash_col_i64(const int64_t* vals, const uint8_t* valid, size_t n,
                           int64_t* hashes);

/* ---- Shuffle partition ids -------------------------------------------- */
/* shuffle/mod.rs:163-188: murmur3 seed 42 over partition exprs, then
 * part_id = hash.rem_euclid(num_partitions). This computes only the pmod
 * step from precomputed hashes. */
void oracle_partition_ids(const int32_t* hashes, size_t n, uint32_t num_partitions,
                          uint32_t* out);

/* ---- radix_sort_by_key ------------------------------------------------- */
/* datafusion-ext-commons/src/algorithm/rdx_sort.rs:24-74 — in-place
 * American-flag counting sort; UNSTABLE, restated swap-for-swap so the
 * resulting order is bit-identical to the reference's.
 * items are (part_id, batch_idx, row_idx) u32 triples as in
 * buffered_data.rs:298-331; counts has num_keys entries, zeroed by caller. */
void oracle_radix_sort_triples(uint32_t* items /* n*3 */, size_t n,
                               size_t num_keys, size_t* counts);

/* ---- Hash aggregation (north-star shape) ------------------------------ */
/* GROUP BY int64 key -> SUM(float64), COUNT(val).
 * Semantics restated from:
 *  - record order: first-occurrence insertion order
 *    (agg_hash_map.rs:77-168 upsert_many assigns dense record indices in
 *     arrival order; agg_table.rs output iterates records in index order)
 *  - SUM: sum.rs:90-115 partial_update — f64 added in row-arrival order,
 *    null args skipped, acc starts invalid, becomes value on first add
 *  - COUNT: count.rs:90-149 — +1 when all args non-null, never null
 *  - merge: sum.rs:117-145 partial_merge / count.rs partial_merge
 *  - null grouping key is a normal group (arrow-row encodes null distinctly;
 *    agg_ctx.rs:219-231) */
typedef struct OracleAgg OracleAgg;
OracleAgg* oracle_agg_new(void);
void oracle_agg_free(OracleAgg* a);
void oracle_agg_update(OracleAgg* a, const int64_t* keys, const uint8_t* key_valid,
                       const double* vals, const uint8_t* val_valid, size_t n);
/* merge rows of frozen partial state (the Binary agg-buf column, a8):
 * per row: [u8 valid][8B LE f64 sum]? ++ [varint count]
 * acc.rs:335-347 freeze_to_rows + count.rs:193-211 + io/mod.rs:60-79 */
void oracle_agg_merge_frozen(OracleAgg* a, const int64_t* keys,
                             const uint8_t* key_valid, const uint8_t* acc_data,
                             const int64_t* acc_offsets, size_t n);
size_t oracle_agg_num_groups(const OracleAgg* a);
/* outputs in record (insertion) order; any pointer may be NULL to skip */
void oracle_agg_output(const OracleAgg* a, int64_t* keys, uint8_t* key_valid,
                       double* sums, uint8_t* sum_valid, int64_t* counts);
/* freeze all records to the Binary agg-buf wire format; returns total bytes.
 * offsets must hold num_groups+1 entries; data may be NULL for sizing pass. */
size_t oracle_agg_freeze(const OracleAgg* a, uint8_t* data, int64_t* offsets);

/* ---- varint (write_len/read_len) -------------------------------------- */
/* io/mod.rs:60-79: little-endian base-128, continuation bit on high bit */
size_t oracle_write_len(uint64_t len, uint8_t* out); /* returns bytes written */
size_t oracle_read_len(const uint8_t* in, size_t avail, uint64_t* out);

/* ---- batch_serde (columnar wire format) -------------------------------- */
/* io/batch_serde.rs:66-99 write_batch: varint(num_rows) then per column:
 * varint(has_nulls) [null bitmap ceil(n/8)] then byte-TRANSPOSED value
 * bytes for primitives wider than 1 byte (batch_serde.rs:271-306).
 * These helpers serialize single columns; the caller concatenates. */
size_t oracle_serde_col_prim(const uint8_t* values, size_t byte_width, size_t n,
                             const uint8_t* valid, uint8_t* out /* NULL=size */);
/* Binary/Utf8 column: batch_serde.rs:595-660 write_bytes_array
 * (varint has_nulls, bitmap, transposed i32 lens, then raw data bytes) */
size_t oracle_serde_col_bytes(const uint8_t* data, const int64_t* offsets, size_t n,
                              const uint8_t* valid, uint8_t* out);

/* ---- IPC compression block stream -------------------------------------- */
/* common/ipc_compression.rs:64-112: stream of [u32-LE frame_len][lz4 frame],
 * each frame holding >=1 batch_serde batches, target 4 MB uncompressed.
 * Uses liblz4.so.1 (LZ4F frame API) via dlopen — spec-conformant lz4 frames;
 * byte-parity with the reference's lz4_flex encoder is NOT claimed (different
 * compressor), round-trip + frame-spec conformance is (SURVEY.md §8c iii). */
typedef struct OracleIpcWriter OracleIpcWriter;
OracleIpcWriter* oracle_ipc_writer_new(size_t target_block_size /* 0=4MB */);
void oracle_ipc_writer_free(OracleIpcWriter* w);
/* append one serialized batch_serde payload (already encoded) */
int oracle_ipc_write_payload(OracleIpcWriter* w, const uint8_t* payload, size_t len);
int oracle_ipc_finish_block(OracleIpcWriter* w);
/* take ownership of the output bytes written so far */
size_t oracle_ipc_bytes(OracleIpcWriter* w, const uint8_t** data);
/* decompress a whole block stream back to concatenated payload bytes;
 * returns bytes written (or needed if out==NULL), (size_t)-1 on error */
size_t oracle_ipc_decode(const uint8_t* in, size_t in_len, uint8_t* out,
                         size_t out_cap);

#ifdef __cplusplus
}
#endif
#endif /* AURON_ORACLE_H */
