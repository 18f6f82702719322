// serde_host.h — host-side batch_serde writer + IPC block framing + shuffle
// data/index file emit. Wire formats restated from:
//   batch_serde.rs:66-99 (write_batch), :271-306 (primitive byte-transpose),
//   :595-660 (bytes array), io/mod.rs:60-79 (varint),
//   ipc_compression.rs:64-112 (4 MB lz4 blocks, u32-LE length framing),
//   buffered_data.rs:123-158 + sort_repartitioner.rs:166-253 (data + index).
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace auron {

struct HostCol {
  int byte_width = 0;                  // >0: primitive, 0: binary
  const uint8_t* values = nullptr;     // primitives: n*w; binary: data bytes
  const uint8_t* validity = nullptr;   // LSB bitmap or null
  const int32_t* offsets = nullptr;    // binary: n+1
};

void serde_write_len(uint64_t v, std::vector<uint8_t>* out);
// append write_batch(rows [row_beg, row_end) of cols) to out
void serde_write_batch(const std::vector<HostCol>& cols, int64_t row_beg,
                       int64_t row_end, std::vector<uint8_t>* out);

class IpcBlockWriter {
 public:
  // codec 0 = lz4 frames, 1 = zstd frames (level per
  // SPARK_IO_COMPRESSION_ZSTD_LEVEL; ipc_compression.rs:189-196)
  explicit IpcBlockWriter(size_t target = 4194304, int codec = 0,
                          int zstd_level = 1)
      : target_(target), codec_(codec), zstd_level_(zstd_level) {}
  bool write_payload(const uint8_t* p, size_t len, std::string* err);
  bool finish_block(std::string* err);
  const std::vector<uint8_t>& bytes() const { return out_; }
  std::vector<uint8_t> take() { return std::move(out_); }

 private:
  size_t target_;
  int codec_ = 0;
  int zstd_level_ = 1;
  std::vector<uint8_t> staged_;
  std::vector<uint8_t> out_;
};

// ---- read side (IpcReaderExec, ipc_reader_exec.rs:62-120) ------------------
struct OwnedCol {
  int byte_width = 0;              // >0 primitive, 0 binary
  std::vector<uint8_t> values;     // prim values / binary data
  std::vector<uint8_t> validity;   // empty = no nulls
  std::vector<int32_t> offsets;    // binary: rows+1
};

// parse ONE batch_serde batch from p; *used advances past it. dtype_widths:
// >0 for primitives, 0 for binary/utf8.
bool serde_read_batch(const uint8_t* p, size_t len, size_t* used,
                      const std::vector<int>& dtype_widths, int64_t* rows,
                      std::vector<OwnedCol>* cols, std::string* err);

// decode a [u32-LE len][lz4 frame] block stream into the concatenated
// uncompressed payload (ipc_compression.rs:64-112 read side)
bool ipc_decode_blocks(const uint8_t* p, size_t len,
                       std::vector<uint8_t>* payload, std::string* err,
                       int codec = 0);

// write the shuffle data + index files: per partition, rows
// [part_offsets[p], part_offsets[p+1]) of the partition-sorted cols, framed
// as IPC blocks finished at each partition boundary; index = (P+1) u64-LE.
bool write_shuffle_files(const std::vector<HostCol>& sorted_cols,
                         const std::vector<int64_t>& part_offsets,
                         int64_t batch_size, const std::string& data_file,
                         const std::string& index_file, std::string* err,
                         int codec = 0, int zstd_level = 1);

}  // namespace auron
