// parquet.h — restricted Parquet reader for the config-3 scan path
// (SURVEY.md §8 a15 / f.1): flat non-repeated columns of INT32/INT64/FLOAT/
// DOUBLE, PLAIN + dictionary encodings, RLE/bit-packed definition levels,
// UNCOMPRESSED/SNAPPY/ZSTD/LZ4_RAW page codecs. The reference delegates
// decode to the third-party parquet crate (parquet_exec.rs:174-196, SURVEY.md
// §8c ii — parity pinned against files written and read back by pyarrow).
//
// Host side: footer/page-header parsing (Thrift compact protocol, hand-rolled
// — no thrift in this image), page decompression, RLE/bit-packed run
// extraction. GPU side (engine): run expansion to validity bitmaps and
// dictionary indices, dictionary gather, packed-value scatter.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include "plan.h"

namespace auron {

// decoded RLE/bit-packed hybrid stream as run list (host): either a repeated
// run (count, value) or a bit-packed literal span (values materialized)
struct RunList {
  // flattened: literal values for every position, materialized on host.
  // Levels and dict indices are small (u32); full materialization keeps the
  // host side simple, the GPU does the wide work (gather/scatter).
  std::vector<uint32_t> values;
};

// one decoded RLE/bit-packed hybrid run HEADER: the host walks headers only
// (sequential varints, ~#runs work) and the GPU expands values — the wide
// work of the decode (binary-search-over-output-position kernels)
struct PqRun {
  uint32_t out_pos;  // first output slot this run fills
  uint32_t count;    // values in this run (clipped to the page's remainder)
  uint32_t src_off;  // bit-packed: byte offset into the span bytes buffer;
                     // RLE: the repeated VALUE itself (<= 32 bits)
  uint8_t kind;      // 0 = RLE, 1 = bit-packed
  uint8_t bw;        // bit width
  uint16_t _pad = 0;
};
static_assert(sizeof(PqRun) == 16, "PqRun must be 16 bytes");

struct PqColumnChunkData {
  // per row-group column results, host-resident, ready for GPU upload
  int64_t num_values = 0;              // value slots incl. nulls
  std::vector<uint8_t> plain;          // PLAIN-encoded non-null values (LE)
  std::vector<uint32_t> dict_indices;  // non-null dictionary indices
  std::vector<uint8_t> dict_values;    // dictionary page PLAIN values
  int64_t dict_count = 0;
  std::vector<uint8_t> validity;       // LSB bitmap, empty = all valid
  int64_t null_count = 0;
  bool uses_dict = false;
  // BYTE_ARRAY columns: row-aligned (null rows zero-length), host-assembled
  std::vector<int32_t> bin_offsets;    // num_values + 1
  std::vector<uint8_t> bin_data;
  // GPU run-expansion staging (fixed-width dict chunks; parquet.cpp sets
  // gpu_dict/gpu_def when every page qualified). Span byte buffers carry 8
  // bytes of tail padding for unaligned u64 bit extraction on device.
  bool gpu_dict = false;   // dict_indices replaced by idx_runs/idx_bytes
  std::vector<PqRun> idx_runs;
  std::vector<uint8_t> idx_bytes;
  int64_t nn_count = 0;    // non-null value count (== dense index count)
  bool gpu_def = false;    // validity replaced by def_runs/def_bytes
  std::vector<PqRun> def_runs;
  std::vector<uint8_t> def_bytes;
  // GPU page decompression (kernels_pq.hip): SNAPPY fixed-width chunks
  // whose data pages after a dictionary-encoded prefix are all v1 PLAIN.
  // The prefix is host-decoded into plain/validity as usual; the suffix
  // pages are recorded RAW (pointers into the file mapping — valid only
  // while the ParquetFile is alive) and decompressed+decoded on device.
  bool gpu_comp = false;
  struct GpuPageRef {
    const uint8_t* src;      // compressed page bytes (mmap)
    uint32_t comp_len;
    uint32_t uncomp_len;
    uint32_t num_values;     // value slots (incl nulls)
  };
  std::vector<GpuPageRef> comp_pages;
  int64_t prefix_values = 0;  // host-decoded value slots (incl nulls)
  int64_t suffix_values = 0;  // sum over comp_pages
  bool nullable = false;      // pages carry a def-level section
  int value_width = 0;        // fixed-width byte size (0 for BYTE_ARRAY)
};

// host materialization of GPU-staged runs (debug / CPU-test path): expands
// idx_runs -> dict_indices and def_runs -> validity bitmap exactly as the
// device kernels (k_runs_expand_u32 / k_def_expand_validity) would, so CPU
// tests pin the header walk without a GPU.
void pq_materialize_gpu_staging(PqColumnChunkData* cd);

struct PqColumnInfo {
  std::string name;
  int physical_type = -1;  // 1=INT32, 2=INT64, 4=FLOAT, 5=DOUBLE, 6=BYTE_ARRAY
  bool utf8 = false;       // converted_type UTF8
  bool nullable = false;   // max_def == 1
  DType dtype() const {
    switch (physical_type) {
      case 1: return DType::Int32;
      case 2: return DType::Int64;
      case 4: return DType::Float32;
      case 5: return DType::Float64;
      case 6: return utf8 ? DType::Utf8 : DType::Binary;
      default: return DType::Unsupported;
    }
  }
};

// row-group column statistics for pruning (parquet Statistics min_value=6 /
// max_value=5, PLAIN-encoded; parquet_exec.rs row-group stats pruning analog)
struct PqColStats {
  bool has_minmax = false;
  int64_t min_i = 0, max_i = 0;  // INT32/INT64 widened
  double min_f = 0, max_f = 0;   // FLOAT/DOUBLE
};

class ParquetFile {
 public:
  // reads and parses the footer; throws std::runtime_error on malformed or
  // out-of-scope features
  explicit ParquetFile(const std::string& path);
  ~ParquetFile();
  ParquetFile(const ParquetFile&) = delete;
  ParquetFile& operator=(const ParquetFile&) = delete;

  int num_row_groups() const { return (int)row_groups_.size(); }
  int64_t row_group_rows(int rg) const { return row_groups_[rg].num_rows; }
  const std::vector<PqColumnInfo>& columns() const { return columns_; }
  PqColStats column_stats(int rg, int col) const;

  // decode one column chunk of one row group
  PqColumnChunkData read_chunk(int rg, int col) const;

 private:
  struct ChunkMeta {
    int64_t data_page_offset = -1;
    int64_t dict_page_offset = -1;
    int64_t total_compressed_size = 0;
    int64_t num_values = 0;
    int codec = 0;
    std::vector<uint8_t> stat_min, stat_max;  // PLAIN-encoded min/max_value
  };
  struct RowGroupMeta {
    int64_t num_rows = 0;
    std::vector<ChunkMeta> chunks;
  };

  std::string path_;
  // mmap'd whole file: page-cache-backed, demand-paged IN PARALLEL by the
  // per-chunk decode threads (a 10 GB config-3 file slurped into a vector
  // cost ~3-5 s of single-thread read per task)
  struct MappedFile {
    const uint8_t* data = nullptr;
    size_t size = 0;
    const uint8_t* data_ptr() const { return data; }
  } map_;
  struct FileView {
    const MappedFile* m;
    const uint8_t* data() const { return m->data; }
    size_t size() const { return m->size; }
  } file_{&map_};
  std::vector<PqColumnInfo> columns_;
  std::vector<RowGroupMeta> row_groups_;
};

// snappy block decompression (format: varint uncompressed len, then
// literal/copy tags) — no libsnappy in this image
bool snappy_uncompress(const uint8_t* src, size_t n, std::vector<uint8_t>* out,
                       std::string* err);
// core over caller storage: dst needs out_len + 32 writable bytes (the hot
// paths overshoot with fixed-size copies); lets the page loop reuse one
// grow-only buffer instead of an alloc + zero-fill per page
bool snappy_uncompress_raw(const uint8_t* src, size_t n, uint8_t* dst,
                           size_t out_len, std::string* err);

}  // namespace auron
