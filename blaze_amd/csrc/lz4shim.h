// lz4shim.h — LZ4F frame API bound from liblz4.so.1 at run time (no lz4
// headers ship in this image). Produces/consumes spec-conformant lz4 frames —
// the block payload format of ipc_compression.rs:64-112 (the reference's
// lz4_flex frames are also spec frames; byte-level compressor parity is not
// claimed, round-trip + framing parity is — SURVEY.md §8c iii).
#pragma once

#include <cstddef>
#include <cstdint>
#include <string>
#include <vector>

namespace auron {

// compress whole payload into one lz4 frame appended to out; returns false on
// failure (error text in err)
bool lz4_compress_frame(const uint8_t* src, size_t len, std::vector<uint8_t>* out,
                        std::string* err);
// decompress a whole frame blob (single frame) appending to out
bool lz4_decompress_frame(const uint8_t* src, size_t len,
                          std::vector<uint8_t>* out, std::string* err);

// zstd frames (libzstd.so.1 via dlopen): the shuffle codec alternative the
// reference selects through spark.io.compression.codec
// (ipc_compression.rs:189-196; level conf SPARK_IO_COMPRESSION_ZSTD_LEVEL,
// default 1)
bool zstd_compress_frame(const uint8_t* src, size_t len, int level,
                         std::vector<uint8_t>* out, std::string* err);
bool zstd_decompress_frame(const uint8_t* src, size_t len,
                           std::vector<uint8_t>* out, std::string* err);

}  // namespace auron
