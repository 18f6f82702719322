// kernels.h — host-side launch API for the CDNA4 hot-path kernels.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

#include "plan.h"

namespace auron {

// ---- Spark murmur3 (mur.rs:19-87) on device -------------------------------
// hashes[i] must be pre-initialized with the seed (k_hash_init); each fold
// call applies one column (spark_hash.rs:28-57 create_murmur3_hashes).
void launch_hash_init(int32_t* hashes, int32_t seed, int64_t n, hipStream_t s);
void launch_hash_fold_i64(const int64_t* vals, const uint8_t* valid, int64_t n,
                          int32_t* hashes, hipStream_t s);
// 4-byte int column fold (hash_array_primitive Int32, spark_hash.rs)
void launch_hash_fold_i32(const int32_t* vals, const uint8_t* valid, int64_t n,
                          int32_t* hashes, hipStream_t s);
// dtype bridges for Int32 grouping keys (internally widened to the i64 table)
void launch_widen_i32_i64(const int32_t* in, int64_t n, int64_t* out,
                          hipStream_t s);
void launch_narrow_i64_i32(const int64_t* in, int64_t n, int32_t* out,
                           hipStream_t s);
// numeric agg-argument casts (sum.rs:78-88 prepare_partial_args analog)
void launch_cast_i32_f64(const int32_t* in, int64_t n, double* out,
                         hipStream_t s);
void launch_cast_i64_f64(const int64_t* in, int64_t n, double* out,
                         hipStream_t s);
// round-robin partition ids (evaluate_robin_partition_ids,
// shuffle/mod.rs:190-202): part[i] = (start + i) % P
void launch_robin_ids(int64_t n, uint32_t start, uint32_t P, uint32_t* out,
                      hipStream_t s);
// part_id = pmod(hash, P) (shuffle/mod.rs:178-188)
void launch_pmod(const int32_t* hashes, int64_t n, int32_t num_partitions,
                 uint32_t* part_ids, hipStream_t s);

// ---- hash aggregation ------------------------------------------------------
// GPU replacement for AggHashMap::upsert_records + per-agg partial_update
// (agg_hash_map.rs:77-168, sum.rs:90-115, count.rs:90-149).
// Open-addressing table of 32-byte slots over i64 keys (key i64::MIN = empty
// sentinel): key probe and the row's atomics all land in ONE 64-byte cache
// line (2 slots/line), keeping the random traffic to one line per row. The
// two trailing slots (index cap, cap+1) hold the i64::MIN-key group and the
// null-key group.
struct AggSlot {
  long long key;                 // INT64_MIN = empty
  unsigned long long cnt;        // COUNT accumulator (also: sum valid <=> >0)
  double sum;                    // SUM(f64) accumulator
  unsigned long long first_row;  // global arrival index, for record order
};
static_assert(sizeof(AggSlot) == 32, "AggSlot must be 32 bytes");

// MIN/MAX accumulators (maxmin.rs:104-119 partial_update compare-and-keep)
// live in an OPTIONAL side array `mm` so the 32-byte hot-path slot is
// untouched when no MIN/MAX agg is present. Values are stored under the
// order-preserving f64<->u64 map (sign ? ~bits : bits|MSB), so device
// atomicMin/atomicMax on u64 implement f64 min/max directly and the init
// sentinels (min=~0ull, max=0ull) make merging guard-free: merging an
// absent/invalid part is a no-op atomic. Validity = "accumulator left its
// sentinel"; the only values colliding with a sentinel are the two extreme
// NaN payloads (0x7FFFFFFFFFFFFFFF / 0xFFFFFFFFFFFFFFFF), documented edge
// (the reference's own NaN ordering is arrival-order-dependent:
// maxmin.rs:110-116 partial_cmp==None takes the new value).
constexpr unsigned long long MM_MIN_INIT = ~0ull;
constexpr unsigned long long MM_MAX_INIT = 0ull;

struct AggTable {
  int64_t cap = 0;                    // power of two
  AggSlot* slots = nullptr;           // [cap + 2]
  uint32_t* special_used = nullptr;   // [2]: {min-key group, null-key group}
  unsigned long long* num_groups = nullptr; // [1] device counter
  uint32_t* error_flag = nullptr;           // [1] raised on probe exhaustion
  // interleaved {min_u64, max_u64} per slot, [2*(cap+2)]; null when the agg
  // list has no MIN/MAX (the north-star path) — kernels skip it then
  unsigned long long* mm = nullptr;
  // FIRST / FIRST_IGNORES_NULL accumulators (first.rs / first_ignores_null.rs)
  // — optional side arrays, [2*(cap+2)] each, even index = FIRST, odd =
  // FIRST_IGNORES_NULL. A parallel reduction can't latch "first arrival"
  // directly, so it runs in TWO passes per chunk on the same stream:
  // pass A (inside update/merge kernels) takes atomicMin of each candidate
  // row's priority into f_row; pass B (k_first_capture_*) re-probes each row
  // and the unique row whose priority equals f_row[..] plain-stores the
  // value and state. Priorities are global arrival rows (update/merge) or
  // the record's preserved first_row (spill merge), matching the reference's
  // sequential update/merge order.
  unsigned long long* f_row = nullptr;  // candidate priority, init ~0ull
  double* f_val = nullptr;              // captured value
  uint8_t* f_st = nullptr;              // 0 = untouched, 1 = first-was-null
                                        // (FIRST only), 2 = valid
  // COLLECT_LIST pool (collect.rs:119-138 restated; DESIGN §8 round-2 item
  // 2, landed early): every non-null arg appends a (key, prio, value-bits)
  // triple; prio = global arrival row (update) or record_row<<20|item
  // (frozen-list merge), so a sort by prio then by key reproduces the
  // reference's per-group arrival order. Regular keys fill from the front
  // (counter n[0]); null-key rows fill from the BACK (counter n[1], slot
  // cap-1-i) — two segments, one allocation, no partition pass. Keys (not
  // slot indices) make table grows safe. error_flag bit 4 = pool overflow.
  long long* c_key = nullptr;            // [c_cap]
  unsigned long long* c_prio = nullptr;  // [c_cap]
  unsigned long long* c_val = nullptr;   // [c_cap] raw value bits
  unsigned long long* c_n = nullptr;     // [2] device counters
  int64_t c_cap = 0;
  // accumulator arithmetic type (sum.rs:78-88: the acc column is the agg's
  // declared data type and inputs are cast to it). 0 = f64 (north star);
  // 1 = i64 — sums use wrapping integer atomicAdd on the same 8-byte slot
  // field, MIN/MAX use the i64 order map (sign-bit flip), and all
  // freeze/parse paths move the raw 8 bytes either way, so the wire format
  // is the typed prim save in both modes.
  uint32_t sum_int = 0;
};

// grid-stride row update: keys/vals length n, rows globally numbered starting
// at row_offset (for cross-batch first-occurrence order).
void launch_agg_update(const AggTable& t, const int64_t* keys,
                       const uint8_t* key_valid, const double* vals,
                       const uint8_t* val_valid, int64_t n, uint64_t row_offset,
                       hipStream_t s);

// agg layout descriptor: 4 bits per agg LSB-first, 0-terminated below 8
// aggs (8 aggs fill the u32 exactly; the walker self-terminates); all aggs
// share one argument column. AVG freeze = sum ++
// count (avg.rs:208-217); MIN/MAX freeze = the same prim [u8 valid][8B LE]
// format as SUM (both are AccPrimColumn saves, acc.rs:335-347 — maxmin.rs
// create_acc_column:91-93 uses the same generic prim column as sum.rs), so
// duplicated parts repeat the shared accumulators.
enum AggLayoutKind : uint32_t {
  AGGL_SUM = 1, AGGL_CNT = 2, AGGL_AVG = 3, AGGL_MIN = 4, AGGL_MAX = 5,
  AGGL_FIRST = 6, AGGL_FIRSTIN = 7, AGGL_CLIST = 8, AGGL_CSET = 9
};

// merge rows of frozen partial state per the layout
// (acc.rs:335-347 + count.rs:193-211 + io/mod.rs:60-79)
void launch_agg_merge_frozen(const AggTable& t, const int64_t* keys,
                             const uint8_t* key_valid, const uint8_t* acc_data,
                             const int32_t* acc_offsets, int64_t n,
                             uint64_t row_offset, uint32_t layout,
                             hipStream_t s);

// merge spilled records back (a9 analog): like merge_frozen but each record
// carries its preserved global first_row explicitly
void launch_coll_refill(const AggTable& t, const unsigned long long* items,
                        int64_t m0, int64_t m1, hipStream_t s);
void launch_agg_merge_spill(const AggTable& t, const int64_t* keys,
                            const uint8_t* acc_data, const int32_t* acc_offsets,
                            const unsigned long long* first_rows, int64_t n,
                            uint32_t layout, hipStream_t s);

// compact occupied slots to dense arrays (unordered); returns count via
// num_out (device). out_slot holds the source slot index per group.
void launch_agg_compact(const AggTable& t, uint32_t* out_slot,
                        unsigned long long* out_first_row,
                        unsigned long long* num_out, hipStream_t s);

// divide sums by counts for AVG output columns (avg final = sum/count)
void launch_avg_div(const double* sums, const long long* cnts, int64_t n,
                    double* out, hipStream_t s);

// gather ordered group outputs given order[] (group -> slot):
// keys + key validity bits, sums + validity bits, counts, and (when t.mm is
// set and the out pointers are non-null) min/max values + validity bits
void launch_agg_gather_out(const AggTable& t, const uint32_t* order_slots,
                           int64_t num_groups, int64_t* out_keys,
                           uint8_t* out_key_validity, double* out_sums,
                           uint8_t* out_sum_validity, long long* out_counts,
                           double* out_mins, uint8_t* out_min_validity,
                           double* out_maxs, uint8_t* out_max_validity,
                           hipStream_t s);

// freeze ordered groups into the Binary agg-buf wire format (a8):
// pass 1 computes per-group byte length, host scans, pass 2 writes bytes.
void launch_agg_freeze_len(const AggTable& t, const uint32_t* order_slots,
                           int64_t num_groups, int32_t* lens, uint32_t layout,
                           hipStream_t s);
void launch_agg_freeze_write(const AggTable& t, const uint32_t* order_slots,
                             int64_t num_groups, const int32_t* offsets,
                             uint8_t* out, uint32_t layout, hipStream_t s);

// partial-skipping pass-through freeze (agg_ctx.rs:428-462); is_int selects
// the i64 order map for MIN/MAX parts (values themselves move as raw bytes)
void launch_skip_freeze_len(const uint8_t* val_valid, int64_t n, int32_t* lens,
                            uint32_t layout, hipStream_t s);
void launch_skip_freeze_write(const double* vals, const uint8_t* val_valid,
                              int64_t n, const int32_t* offsets, uint8_t* out,
                              uint32_t layout, int is_int, hipStream_t s);

// initialize table slots (key = empty sentinel, accs zero, first_row = ~0)
void launch_slots_init(AggSlot* slots, int64_t n, hipStream_t s);

// initialize n interleaved {min,max} accumulator pairs to their sentinels
void launch_mm_init(unsigned long long* mm, int64_t n, hipStream_t s);

// initialize n {f_row, f_val, f_st} pair-slots (2 entries each)
void launch_first_init(unsigned long long* f_row, double* f_val, uint8_t* f_st,
                       int64_t n, hipStream_t s);

// gather u64 elements by u32 index (pool reorder after a sort pass)
void launch_gather_u64_idx(const unsigned long long* src, const uint32_t* idx,
                           int64_t n, unsigned long long* dst, hipStream_t s);
void launch_bias_i64(const unsigned long long* src, int64_t n,
                     unsigned long long* dst, hipStream_t s);
// emit side: per-group collect counts and value runs (pool sorted first)
void launch_coll_counts(const AggTable& t, const uint32_t* order_slots,
                        int64_t num_groups, int32_t* cnts, hipStream_t s);
void launch_coll_gather(const AggTable& t, const uint32_t* order_slots,
                        int64_t num_groups, const int32_t* offsets,
                        unsigned long long* out, hipStream_t s);
// COLLECT_SET dedup: mark (key,value)-run heads; compact by scanned marks
void launch_coll_mark_heads(const long long* key,
                            const unsigned long long* val, int64_t n,
                            int use_key, uint8_t* mark, hipStream_t s);
void launch_compact_u64(const unsigned long long* src, const uint8_t* mark,
                        const uint32_t* pos, int64_t n,
                        unsigned long long* dst, hipStream_t s);


// FIRST/FIRST_IGNORES_NULL pass B over update-mode rows: the unique row
// whose global row index equals the slot's captured f_row priority stores
// its value/state (first.rs:91-148 partial_update latch semantics)
void launch_first_capture_update(const AggTable& t, const int64_t* keys,
                                 const uint8_t* key_valid, const double* vals,
                                 const uint8_t* val_valid, int64_t n,
                                 uint64_t row_offset, hipStream_t s);

// pass B over frozen records (merge mode / spill merge): prio = explicit
// per-record priorities (spill), or null for row_offset + i (merge)
void launch_first_capture_frozen(const AggTable& t, const int64_t* keys,
                                 const uint8_t* key_valid,
                                 const uint8_t* acc_data,
                                 const int32_t* acc_offsets,
                                 const unsigned long long* prio, int64_t n,
                                 uint64_t row_offset, uint32_t layout,
                                 hipStream_t s);

// gather one FIRST-family output column (which: 0 = FIRST, 1 = IGNORES_NULL)
void launch_first_gather(const AggTable& t, const uint32_t* order_slots,
                         int64_t num_groups, int which, double* out_vals,
                         uint8_t* out_validity, hipStream_t s);

// rebuild `src` table into the (larger, initialized) `dst` table; each source
// slot holds a distinct key so plain stores after the CAS claim are race-free.
void launch_agg_rebuild(const AggTable& dst, const AggTable& src, hipStream_t s);

// ---- filter (filter_exec.rs:174-198, cached_exprs_evaluator.rs:82-95) ------
// comparison ops (auron-serde/src/lib.rs:70-96 names)
enum CmpOp : int32_t { CMP_EQ, CMP_NE, CMP_LT, CMP_LE, CMP_GT, CMP_GE };
// mask[i] &= (col[i] op literal), null compares false; `first` initializes
// the mask instead of ANDing
void launch_cmp_lit(DType dt, const void* vals, const uint8_t* valid,
                    int64_t n, CmpOp op, int64_t lit_i, double lit_f,
                    uint8_t* mask, bool first, hipStream_t s);
void launch_is_not_null(const uint8_t* valid, int64_t n, uint8_t* mask,
                        bool first, bool negate, hipStream_t s);
// positions of selected rows, stable: needs exclusive scan of mask
void scan_mask_u8(const uint8_t* mask, uint32_t* positions /* n+1 */, int64_t n,
                  void* temp, size_t* temp_bytes, hipStream_t s);
void launch_sel_rows(const uint8_t* mask, const uint32_t* positions, int64_t n,
                     uint32_t* sel, hipStream_t s);
void launch_gather_4(const uint8_t* src, const uint32_t* perm, int64_t n,
                     uint8_t* dst, hipStream_t s);  // 4-byte elements

// ---- parquet decode helpers (parquet.cpp host side feeds these) ------------
// expand an LSB validity bitmap into a per-row u8 mask
void launch_bits_to_mask(const uint8_t* bits, int64_t n, uint8_t* mask,
                         hipStream_t s);
// scatter densely-packed non-null values into row slots:
// out[i] = mask[i] ? packed[positions[i]] : 0  (positions = excl. scan of mask)
void launch_scatter_packed(int width, const uint8_t* packed,
                           const uint32_t* positions, const uint8_t* mask,
                           int64_t n, uint8_t* out, hipStream_t s);

// ---- two-phase aggregation (kernels_agg2.hip) ------------------------------
// radix-partition rows into buckets whose groups fit in LDS, aggregate each
// bucket in LDS, merge the counted per-bucket group lists into the table.
struct StagedGroup {
  int64_t key;
  double sum;
  unsigned long long cnt_first;  // cnt<<32 | chunk-local first_row
};
// partitioned row records. The hot streams are split 16B + 4B: the {key,val}
// pair is one aligned 16-byte store (4 per cache line, never straddling),
// the rowv stream coalesces as plain dwords — 20 B/row vs the 24 B padded
// AoS record (measured faster), while still touching at most two line runs
// per row (the original 3×8B SoA split made scatter 64% of GPU time).
struct PartKV {
  int64_t key;
  double val;
};
static_assert(sizeof(PartKV) == 16, "PartKV must be 16 bytes");
// leftover rows (LDS-window overflow) keep the self-contained AoS record
struct PartRow {
  int64_t key;
  double val;
  uint32_t rowv;  // chunk-local row | valid<<31
  uint32_t _pad;
};
static_assert(sizeof(PartRow) == 24, "PartRow must be 24 bytes");
constexpr int AGG2_LSLOTS = 2048;  // LDS table entries per bucket

constexpr int AGG2_GRID_LOG2_MAX = 10;  // matrix sizing bound for the
                                        // runtime-tunable hist/scatter grid
// per-BLOCK bucket histogram into the bucket-major counts matrix
// [nbuck << AGG2_GRID_LOG2]
void launch_agg2_hist(const int64_t* keys, const uint8_t* key_valid, int64_t n,
                      int nbuck_log2, int grid_log2, uint32_t* counts_matrix,
                      uint32_t* special_rows, int block, hipStream_t s,
                      unsigned long long* kminmax = nullptr);
// exclusive scan over the flat counts matrix -> per-(block,bucket) bases
void scan_counts_matrix(const uint32_t* counts, uint32_t* scanned, int64_t n,
                        void* temp, size_t* temp_bytes, hipStream_t s);
void launch_agg2_offsets(const uint32_t* scanned, int nbuck_log2,
                         int grid_log2, uint32_t* offsets, hipStream_t s);
void launch_agg2_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* scanned,
                         PartKV* out_kv, uint32_t* out_rowv, int block,
                         hipStream_t s);
void launch_agg2_scatter24(const int64_t* keys, const uint8_t* key_valid,
                           const double* vals, const uint8_t* val_valid,
                           int64_t n, int nbuck_log2, int grid_log2,
                           const uint32_t* scanned,
                           PartRow* out, int block, hipStream_t s);
void launch_agg2_bucket24(const PartRow* part, const uint32_t* offsets,
                          int is_int, int nbuckets, StagedGroup* staged,
                          unsigned long long* staged_n, int64_t staged_cap,
                          PartRow* leftover, unsigned long long* lo_n,
                          uint32_t* error_flag, int block, hipStream_t s);
void launch_agg2_specials(const AggTable& t, const int64_t* keys,
                          const uint8_t* key_valid, const double* vals,
                          const uint8_t* val_valid, int64_t n,
                          uint64_t row_offset, hipStream_t s);
void launch_agg2_bucket(const PartKV* part_kv, const uint32_t* part_rowv,
                        const uint32_t* offsets, int is_int,
                        int nbuckets, StagedGroup* staged,
                        unsigned long long* staged_n, int64_t staged_cap,
                        PartRow* leftover, unsigned long long* lo_n,
                        uint32_t* error_flag, int block, hipStream_t s);
void launch_agg2_merge_groups(const AggTable& t, const StagedGroup* staged,
                              int64_t n, uint64_t row_offset, hipStream_t s);
void launch_agg2_leftovers(const AggTable& t, const PartRow* rows, int64_t n,
                           uint64_t row_offset, hipStream_t s);

// ---- shuffle partition + gather -------------------------------------------
// per-partition histogram of part_ids (LDS-staged, one global atomic per
// (block, partition))
void launch_histogram(const uint32_t* part_ids, int64_t n, uint32_t P,
                      uint32_t* counts /* zeroed [P] */, hipStream_t s);
// stable partition permutation = rocprim stable radix sort of
// (part_id, row_idx) pairs over ceil(log2 P) bits
void sort_pairs_u32_u32(const uint32_t* keys_in, const uint32_t* vals_in,
                        uint32_t* keys_out, uint32_t* vals_out, int64_t n,
                        int end_bit, void* temp, size_t* temp_bytes,
                        hipStream_t s);
void launch_iota_u32(uint32_t* dst, int64_t n, hipStream_t s);

void launch_gather_8(const uint8_t* src, const uint32_t* perm, int64_t n,
                     uint8_t* dst, hipStream_t s);  // 8-byte elements
void launch_gather_bits(const uint8_t* src_bits, const uint32_t* perm, int64_t n,
                        uint8_t* dst_bits, hipStream_t s);
// binary column gather: lens then bytes (offsets host-scanned)
void launch_gather_lens(const int32_t* src_offsets, const uint32_t* perm,
                        int64_t n, int32_t* dst_lens, hipStream_t s);
void launch_gather_bytes(const uint8_t* src_data, const int32_t* src_offsets,
                         const uint32_t* perm, const int32_t* dst_offsets,
                         int64_t n, uint8_t* dst_data, hipStream_t s);

// device repartition (exchange prep): dest-rank-major partition order key
// and per-rank all-to-all split counts
void launch_exchange_ord(const uint32_t* pids, int64_t n, uint32_t P,
                         uint32_t world, uint32_t* ord, hipStream_t s);
void launch_dest_counts(const uint32_t* pids, const int32_t* offsets,
                        int64_t n, uint32_t world, unsigned long long* rows,
                        unsigned long long* bytes, hipStream_t s);

// ---- generalized grouping keys (a4 closure: Utf8 / multi-column) ---------
// The reference row-encodes key tuples (agg_ctx.rs:219-231 arrow-row); the
// key ENCODING is substitutable as long as the grouping
// column VALUES round-trip — this engine uses its own per-column
// [valid u8][value bytes] encoding (utf8: u32 len + bytes), hashed into the
// same 32-byte slot table (slot.key holds the 64-bit key-bytes hash; a side
// `off` array points into a device byte pool holding each group's encoded
// key for the byte-compare probe).
constexpr int GKEY_MAX_COLS = 4;
enum GKeyDt : uint8_t { GK_I64 = 0, GK_I32 = 1, GK_F64 = 2, GK_UTF8 = 3 };
struct GKeyCols {
  int ncols = 0;
  const void* values[GKEY_MAX_COLS] = {};
  const int32_t* offsets[GKEY_MAX_COLS] = {};   // utf8 only
  const uint8_t* validity[GKEY_MAX_COLS] = {};
  uint8_t dt[GKEY_MAX_COLS] = {};
};
struct GKeyTable {
  unsigned long long* off = nullptr;  // [cap]: 0 = pending/absent, else
                                      // pool offset + 1 (release-published)
  uint8_t* pool = nullptr;            // [u32 len][encoded bytes] records
  unsigned long long* pool_n = nullptr;  // [1] byte cursor
  int64_t pool_cap = 0;
};
// row murmur3 fold over a Utf8/Binary column (mur.rs:19-30 byte path)
void launch_hash_fold_bytes(const int32_t* offsets, const uint8_t* data,
                            const uint8_t* valid, int64_t n, int32_t* hashes,
                            hipStream_t s);
// encoded key length per row (u32), then the byte write after a scan
void launch_gkey_enc_lens(const GKeyCols& c, int64_t n, uint32_t* lens,
                          hipStream_t s);
void launch_gkey_enc_write(const GKeyCols& c, const uint32_t* enc_offsets,
                           uint8_t* enc_bytes, int64_t n, hipStream_t s);
// probe-or-insert each row's encoded key; writes the slot index per row and
// takes the first_row atomicMin. Pool capacity must be host-ensured.
void launch_gkey_upsert(const AggTable& t, const GKeyTable& g,
                        const uint32_t* enc_offsets, const uint8_t* enc_bytes,
                        int64_t n, uint64_t row_offset, uint32_t* slots,
                        hipStream_t s);
// slot-indexed accumulate / frozen-record merge (no per-row probing)
void launch_gkey_update_idx(const AggTable& t, const uint32_t* slots,
                            const double* vals, const uint8_t* val_valid,
                            int64_t n, hipStream_t s);
void launch_gkey_merge_frozen_idx(const AggTable& t, const uint32_t* slots,
                                  const uint8_t* acc_data,
                                  const int32_t* acc_offsets, int64_t n,
                                  uint32_t layout, hipStream_t s);
// table growth: re-probe every occupied slot of src into dst by stored hash
void launch_gkey_rebuild(const AggTable& dst, const GKeyTable& dg,
                         const AggTable& src, const GKeyTable& sg,
                         hipStream_t s);
// emit: decode grouping column `col` for the ordered groups
void launch_gkey_out_fixed(const GKeyTable& g, const uint32_t* order_slots,
                           int64_t n, const uint8_t* dts, int ncols, int col,
                           uint8_t* values, uint8_t* valid_bitmap,
                           hipStream_t s);
void launch_gkey_out_lens(const GKeyTable& g, const uint32_t* order_slots,
                          int64_t n, const uint8_t* dts, int ncols, int col,
                          uint32_t* lens, uint8_t* valid_bitmap, hipStream_t s);
void launch_gkey_out_bytes(const GKeyTable& g, const uint32_t* order_slots,
                           int64_t n, const uint8_t* dts, int ncols, int col,
                           const int32_t* out_offsets, uint8_t* out_data,
                           hipStream_t s);

// ---- multi-argument aggregation (per-agg arg columns / acc types) --------
// The reference evaluates INDEPENDENT argument expressions per aggregate
// (agg.rs:73-169 prepare_partial_args) with per-agg accumulator columns of
// the declared type (sum.rs:78-88, maxmin.rs:81-83). This mode drops the
// shared-argument restriction: each agg owns an accumulator bank column
// (acc/meta/st arrays indexed by slot) and reads its own argument column
// (or a NULL literal, collect.rs-style empty behavior). Single-phase only.
constexpr int MA_MAX_AGGS = 12;
constexpr int MA_MAX_POOLS = 4;
struct MaAgg {
  uint8_t kind = 0;    // AGGL_* code
  uint8_t acc_t = 0;   // 0 = f64, 1 = i64, 2 = i32 (stored widened to i64;
                       //     freeze/emit narrow to 4 bytes)
  uint8_t arg_dt = 0;  // 0 = f64, 1 = i64, 2 = i32, 3 = NULL literal
  uint8_t pool = 0xFF; // collect pool index (0..MA_MAX_POOLS) or 0xFF
};
struct MaDesc {
  int n = 0;
  MaAgg a[MA_MAX_AGGS];
};
struct MaArgs {  // per-chunk argument column pointers (null = NULL literal)
  const void* vals[MA_MAX_AGGS] = {};
  const uint8_t* valid[MA_MAX_AGGS] = {};
};
struct MaAcc {   // accumulator bank: [agg][cap + 2] each
  unsigned long long* acc = nullptr;   // sum bits / omap value / first value
  unsigned long long* meta = nullptr;  // count / first-priority
  uint8_t* st = nullptr;               // FIRST state 0/1/2
  int64_t stride = 0;                  // cap + 2
};
struct MaPools {  // per-collect-agg pools (single-key i64 mode only)
  long long* key[MA_MAX_POOLS] = {};
  unsigned long long* prio[MA_MAX_POOLS] = {};
  unsigned long long* val[MA_MAX_POOLS] = {};
  unsigned long long* n = nullptr;  // [MA_MAX_POOLS][2] counters
  int64_t cap = 0;                  // per pool
};
// probe-or-insert only: slot index per row + first_row min (the accumulate
// runs slot-indexed afterwards)
void launch_slots_upsert(const AggTable& t, const int64_t* keys,
                         const uint8_t* key_valid, int64_t n,
                         uint64_t row_offset, uint32_t* slots, hipStream_t s);
void launch_ma_init(const MaDesc& d, const MaAcc& m, int64_t cap2,
                    hipStream_t s);
void launch_ma_update(const AggTable& t, const MaDesc& d, const MaAcc& m,
                      const MaPools& p, const MaArgs& args,
                      const int64_t* keys, const uint8_t* key_valid,
                      const uint32_t* slots, int64_t n, uint64_t row_offset,
                      hipStream_t s);
void launch_ma_first_capture(const AggTable& t, const MaDesc& d,
                             const MaAcc& m, const MaArgs& args,
                             const uint32_t* slots, int64_t n,
                             uint64_t row_offset, hipStream_t s);
void launch_ma_merge_frozen(const AggTable& t, const MaDesc& d,
                            const MaAcc& m, const MaPools& p,
                            const int64_t* keys, const uint8_t* key_valid,
                            const uint32_t* slots,
                            const uint8_t* acc_data,
                            const int32_t* acc_offsets, int64_t n,
                            uint64_t row_offset, hipStream_t s);
void launch_ma_first_capture_frozen(const AggTable& t, const MaDesc& d,
                                    const MaAcc& m,
                                    const uint32_t* slots,
                                    const uint8_t* acc_data,
                                    const int32_t* acc_offsets, int64_t n,
                                    uint64_t row_offset, hipStream_t s);
// freeze: per-record lens then bytes (sorted pool views per collect agg are
// installed into `p` by the engine before freezing)
void launch_ma_freeze_len(const AggTable& t, const MaDesc& d, const MaAcc& m,
                          const MaPools& p, const uint32_t* order_slots,
                          int64_t n, int32_t* lens, hipStream_t s);
void launch_ma_freeze_write(const AggTable& t, const MaDesc& d,
                            const MaAcc& m, const MaPools& p,
                            const uint32_t* order_slots, int64_t n,
                            const int32_t* offsets, uint8_t* data,
                            hipStream_t s);
// final-output gather for agg j: values (width 8, or 4 for acc_t i32) +
// validity bitmap; collect aggs use launch_coll_counts/gather with the pool
void launch_ma_gather_out(const MaDesc& d, const MaAcc& m, int agg,
                          const uint32_t* order_slots, int64_t n,
                          uint8_t* values, uint8_t* valid_bitmap,
                          hipStream_t s);
// growth: re-probe occupied src slots into dst, carrying the bank columns
void launch_ma_rebuild(const AggTable& dst, const MaAcc& dm,
                       const AggTable& src, const MaAcc& sm, int naggs,
                       hipStream_t s);

// parquet RLE/bit-packed run expansion (runs = PqRun[] from parquet.h)
void launch_runs_expand_u32(const void* runs, int nruns, const uint8_t* bytes,
                            int64_t n, uint32_t* out, hipStream_t s);
void launch_def_expand_validity(const void* runs, int nruns,
                                const uint8_t* bytes, int64_t n,
                                uint8_t* bitmap, hipStream_t s);

// v3 two-phase partition pipeline (kernels_agg3.hip): LDS-staged packet
// scatter into 64B-aligned per-(block,bucket) ranges + 4096-slot bucket agg
void launch_agg3_line_sizes(const uint32_t* counts, int64_t n,
                            uint32_t* sizes, int rec, hipStream_t s);
void launch_keys_minmax(const int64_t* keys, const uint8_t* key_valid,
                        int64_t n, unsigned long long* kminmax,
                        hipStream_t s);
// rec = 24 (i64 key) or 16 (u32 key offset from key_base — rows whose key
// falls outside [key_base, key_base+2^32) bypass to the leftover list)
void launch_agg3_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* line_scan, uint8_t* out,
                         PartRow* leftover, unsigned long long* lo_n,
                         uint32_t* bypass_matrix, uint32_t* err_flag,
                         int rec, int64_t key_base, hipStream_t s);
// v4 scatter: barrier-free per-bucket LDS rings drained by dedicated
// flusher waves (same in/out contract as launch_agg3_scatter). The hist
// feeding it must run with agg4_worker_waves()*64 threads per block so the
// row->block mapping matches the v4 worker traversal.
int agg4_worker_waves();
void launch_agg4_scatter(const int64_t* keys, const uint8_t* key_valid,
                         const double* vals, const uint8_t* val_valid,
                         int64_t n, int nbuck_log2, int grid_log2,
                         const uint32_t* line_scan, uint8_t* out,
                         PartRow* leftover, unsigned long long* lo_n,
                         uint32_t* bypass_matrix, uint32_t* err_flag,
                         hipStream_t s);
void launch_agg3_bucket(const uint8_t* part, const uint32_t* counts,
                        const uint32_t* bypass,
                        const uint32_t* line_scan, int grid_log2, int is_int,
                        int nbuckets, StagedGroup* staged,
                        unsigned long long* staged_n, int64_t staged_cap,
                        PartRow* leftover, unsigned long long* lo_n,
                        uint32_t* error_flag, int rec, int64_t key_base,
                        hipStream_t s);

// GPU Parquet page staging (kernels_pq.hip): wave-per-page snappy decompress
// + on-device def-level parse + dense compaction. Built by the host pre-scan
// in parquet.cpp; consumed by engine pump_parquet.
struct PqGpuPage {
  uint64_t comp_off;    // into the uploaded compressed blob
  uint64_t uncomp_off;  // into the scratch blob (8-aligned)
  uint32_t comp_len;
  uint32_t uncomp_len;
  uint32_t num_values;  // value slots this page covers (incl nulls)
  uint32_t value_base;  // first value slot within the chunk (incl prefix)
  uint32_t has_def;     // 1 = page begins with [u32 ll][RLE def levels]
  uint32_t chunk_id;
};
struct PqGpuChunk {
  uint64_t valid_base;  // byte offset of this chunk's bitmap (4-aligned)
  uint64_t dense_base;  // byte offset of this chunk's dense values blob
  uint32_t page0;       // first page index
  uint32_t npages;
  uint32_t prefix_nn;   // host-decoded dense prefix length (values)
  uint32_t vw;          // fixed value width (4 or 8)
};
void launch_pq_pages_decode(const uint8_t* comp, const PqGpuPage* pages,
                            int npages, uint8_t* scratch, uint8_t* valid_blob,
                            const PqGpuChunk* chunks, uint32_t* nn_counts,
                            uint32_t* val_offs, uint32_t* err, hipStream_t s);
void launch_pq_page_offsets(const PqGpuChunk* chunks, int nchunks,
                            const uint32_t* nn_counts, uint32_t* page_dense,
                            uint32_t* chunk_nn, hipStream_t s);
void launch_pq_pages_compact(const PqGpuPage* pages, int npages,
                             const uint8_t* scratch, const PqGpuChunk* chunks,
                             const uint32_t* nn_counts,
                             const uint32_t* val_offs,
                             const uint32_t* page_dense,
                             uint8_t* dense_blob, hipStream_t s);

// sort groups by first_row: rocprim radix sort pairs wrapper
void sort_pairs_u64_u32(const unsigned long long* keys_in, const uint32_t* vals_in,
                        unsigned long long* keys_out, uint32_t* vals_out,
                        int64_t n, void* temp, size_t* temp_bytes, hipStream_t s);

}  // namespace auron
