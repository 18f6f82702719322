#include "parquet.h"

#include <dlfcn.h>
#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstdio>
#include <cstring>
#include <stdexcept>

#include "lz4shim.h"

namespace auron {

constexpr size_t kSnappyPad = 64;

// AURON_PARQUET_PROF=1: per-thread phase timing for the host decode path,
// dumped by pq_prof_dump() (tools/pq_prof harness; nanoseconds, aggregated)
std::atomic<long long> g_pq_ns_decomp{0}, g_pq_ns_levels{0},
    g_pq_ns_values{0}, g_pq_ns_assemble{0};
void pq_prof_dump() {
  fprintf(stderr,
          "[pq prof] decompress %.2fs  levels %.2fs  values %.2fs  "
          "assemble %.2fs\n",
          g_pq_ns_decomp.load() / 1e9, g_pq_ns_levels.load() / 1e9,
          g_pq_ns_values.load() / 1e9, g_pq_ns_assemble.load() / 1e9);
}
namespace {
struct PqTimer {
  std::atomic<long long>* acc;
  std::chrono::steady_clock::time_point t0;
  explicit PqTimer(std::atomic<long long>* a)
      : acc(a), t0(std::chrono::steady_clock::now()) {}
  ~PqTimer() {
    *acc += std::chrono::duration_cast<std::chrono::nanoseconds>(
                std::chrono::steady_clock::now() - t0)
                .count();
  }
};
}  // namespace
namespace {

[[noreturn]] void fail(const std::string& m) { throw std::runtime_error(m); }

// ---- Thrift compact protocol (only what parquet.thrift needs) --------------
struct TReader {
  const uint8_t* p;
  const uint8_t* end;

  uint64_t uvarint() {
    uint64_t v = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
    }
    fail("thrift: truncated varint");
  }
  int64_t zigzag() {
    uint64_t v = uvarint();
    return (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
  }

  // returns (field_id, type); type 0 = STOP
  std::pair<int, int> field_header(int* last_id) {
    if (p >= end) fail("thrift: truncated struct");
    uint8_t b = *p++;
    int type = b & 0x0f;
    if (type == 0) return {0, 0};
    int delta = (b >> 4) & 0x0f;
    int id = delta ? *last_id + delta : (int)zigzag();
    *last_id = id;
    return {id, type};
  }

  void skip(int type) {
    switch (type) {
      case 1: case 2: break;                      // bool encoded in type
      case 3: p++; break;                         // byte
      case 4: case 5: case 6: zigzag(); break;    // i16/i32/i64
      case 7: p += 8; break;                      // double
      case 8: {                                   // binary
        uint64_t n = uvarint();
        p += n;
        break;
      }
      case 9: case 10: {                          // list/set
        uint8_t h = *p++;
        uint64_t n = (h >> 4) == 15 ? uvarint() : (h >> 4);
        int et = h & 0x0f;
        for (uint64_t i = 0; i < n; i++) skip(et);
        break;
      }
      case 11: {                                  // map
        uint64_t n = uvarint();
        if (n > 0) {
          uint8_t kv = *p++;
          for (uint64_t i = 0; i < n; i++) {
            skip((kv >> 4) & 0x0f);
            skip(kv & 0x0f);
          }
        }
        break;
      }
      case 12: {                                  // struct
        int last = 0;
        while (true) {
          auto [id, t] = field_header(&last);
          if (t == 0) break;
          skip(t);
        }
        break;
      }
      default: fail("thrift: unknown type " + std::to_string(type));
    }
    if (p > end) fail("thrift: overrun");
  }

  std::pair<uint64_t, int> list_header() {
    uint8_t h = *p++;
    uint64_t n = (h >> 4) == 15 ? uvarint() : (h >> 4);
    return {n, h & 0x0f};
  }

  std::string binary() {
    uint64_t n = uvarint();
    if ((uint64_t)(end - p) < n) fail("thrift: truncated binary");
    std::string s((const char*)p, n);
    p += n;
    return s;
  }
};

struct SchemaElem {
  int type = -1;
  int repetition = -1;  // 0=REQUIRED 1=OPTIONAL 2=REPEATED
  std::string name;
  int num_children = 0;
  int converted_type = -1;  // 0 = UTF8
};

SchemaElem read_schema_element(TReader& r) {
  SchemaElem e;
  int last = 0;
  while (true) {
    auto [id, t] = r.field_header(&last);
    if (t == 0) break;
    switch (id) {
      case 1: e.type = (int)r.zigzag(); break;
      case 3: e.repetition = (int)r.zigzag(); break;
      case 4: e.name = r.binary(); break;
      case 5: e.num_children = (int)r.zigzag(); break;
      case 6: e.converted_type = (int)r.zigzag(); break;
      default: r.skip(t);
    }
  }
  return e;
}

struct PageHeader {
  int type = -1;  // 0 data, 2 dict, 3 data v2
  int64_t uncompressed_size = 0, compressed_size = 0;
  // data page v1
  int num_values = 0, encoding = -1, def_encoding = -1;
  // v2
  int num_nulls = 0, num_rows = 0, def_len = 0, rep_len = 0;
  bool v2_compressed = true;
  // dict
  int dict_num_values = 0;
};

PageHeader read_page_header(TReader& r) {
  PageHeader h;
  int last = 0;
  while (true) {
    auto [id, t] = r.field_header(&last);
    if (t == 0) break;
    switch (id) {
      case 1: h.type = (int)r.zigzag(); break;
      case 2: h.uncompressed_size = r.zigzag(); break;
      case 3: h.compressed_size = r.zigzag(); break;
      case 5: {  // DataPageHeader
        int l2 = 0;
        while (true) {
          auto [id2, t2] = r.field_header(&l2);
          if (t2 == 0) break;
          switch (id2) {
            case 1: h.num_values = (int)r.zigzag(); break;
            case 2: h.encoding = (int)r.zigzag(); break;
            case 3: h.def_encoding = (int)r.zigzag(); break;
            default: r.skip(t2);
          }
        }
        break;
      }
      case 7: {  // DictionaryPageHeader
        int l2 = 0;
        while (true) {
          auto [id2, t2] = r.field_header(&l2);
          if (t2 == 0) break;
          if (id2 == 1) h.dict_num_values = (int)r.zigzag();
          else r.skip(t2);
        }
        break;
      }
      case 8: {  // DataPageHeaderV2
        int l2 = 0;
        h.v2_compressed = true;
        while (true) {
          auto [id2, t2] = r.field_header(&l2);
          if (t2 == 0) break;
          switch (id2) {
            case 1: h.num_values = (int)r.zigzag(); break;
            case 2: h.num_nulls = (int)r.zigzag(); break;
            case 3: h.num_rows = (int)r.zigzag(); break;
            case 4: h.encoding = (int)r.zigzag(); break;
            case 5: h.def_len = (int)r.zigzag(); break;
            case 6: h.rep_len = (int)r.zigzag(); break;
            case 7: h.v2_compressed = (t2 == 1); break;  // BOOL_TRUE
            default: r.skip(t2);
          }
        }
        h.type = 3;
        break;
      }
      default: r.skip(t);
    }
  }
  return h;
}

// ---- zstd (runtime binding; no headers in image) ---------------------------
typedef size_t (*ZSTD_decompress_t)(void*, size_t, const void*, size_t);
typedef unsigned (*ZSTD_isError_t)(size_t);

bool zstd_uncompress(const uint8_t* src, size_t n, size_t out_size,
                     std::vector<uint8_t>* out, std::string* err) {
  static ZSTD_decompress_t dec = nullptr;
  static ZSTD_isError_t iserr = nullptr;
  if (!dec) {
    void* h = dlopen("libzstd.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) {
      *err = "libzstd.so.1 not available";
      return false;
    }
    dec = (ZSTD_decompress_t)dlsym(h, "ZSTD_decompress");
    iserr = (ZSTD_isError_t)dlsym(h, "ZSTD_isError");
  }
  out->resize(out_size);
  size_t rc = dec(out->data(), out_size, src, n);
  if (iserr(rc) || rc != out_size) {
    *err = "zstd decompress failed";
    return false;
  }
  return true;
}

typedef int (*LZ4_decompress_safe_t)(const char*, char*, int, int);

bool lz4raw_uncompress(const uint8_t* src, size_t n, size_t out_size,
                       std::vector<uint8_t>* out, std::string* err) {
  static LZ4_decompress_safe_t dec = nullptr;
  if (!dec) {
    void* h = dlopen("liblz4.so.1", RTLD_NOW | RTLD_GLOBAL);
    if (!h) {
      *err = "liblz4.so.1 not available";
      return false;
    }
    dec = (LZ4_decompress_safe_t)dlsym(h, "LZ4_decompress_safe");
  }
  out->resize(out_size);
  int rc = dec((const char*)src, (char*)out->data(), (int)n, (int)out_size);
  if (rc < 0 || (size_t)rc != out_size) {
    *err = "lz4 raw decompress failed";
    return false;
  }
  return true;
}

// ---- RLE/bit-packed hybrid (parquet encoding spec) -------------------------
void rle_bp_decode(const uint8_t* p, size_t len, int bit_width, int64_t count,
                   std::vector<uint32_t>* out) {
  out->resize(count);
  uint32_t* dst = out->data();
  int64_t filled = 0;
  size_t pos = 0;
  const int byte_w = (bit_width + 7) / 8;
  while (filled < count) {
    if (pos >= len) fail("rle: truncated stream");
    uint64_t header = 0;
    int shift = 0;
    while (true) {
      if (pos >= len) fail("rle: truncated varint");
      uint8_t b = p[pos++];
      header |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    if (header & 1) {
      // bit-packed group: (header>>1) groups of 8 values
      int64_t n = (int64_t)(header >> 1) * 8;
      size_t bytes = (size_t)n * bit_width / 8;
      if (pos + bytes > len) fail("rle: truncated bit-packed run");
      int64_t take = n < count - filled ? n : count - filled;
      if (bit_width == 1) {  // def-level fast path: 1 byte -> 8 values
        int64_t i = 0;
        for (; i + 8 <= take; i += 8) {
          uint8_t b = p[pos + (size_t)(i >> 3)];
          for (int j = 0; j < 8; j++) dst[filled + i + j] = (b >> j) & 1;
        }
        for (; i < take; i++)
          dst[filled + i] = (p[pos + (size_t)(i >> 3)] >> (i & 7)) & 1;
      } else {
        uint64_t acc = 0;
        int bits = 0;
        size_t bp = pos;
        const uint32_t mask = (uint32_t)((1ull << bit_width) - 1);
        for (int64_t i = 0; i < take; i++) {
          while (bits < bit_width) {
            acc |= (uint64_t)p[bp++] << bits;
            bits += 8;
          }
          dst[filled + i] = (uint32_t)acc & mask;
          acc >>= bit_width;
          bits -= bit_width;
        }
      }
      filled += take;
      pos += bytes;
    } else {
      int64_t n = (int64_t)(header >> 1);
      uint32_t v = 0;
      if (pos + byte_w > len) fail("rle: truncated repeated run");
      for (int b = 0; b < byte_w; b++) v |= (uint32_t)p[pos + b] << (8 * b);
      pos += byte_w;
      int64_t take = n < count - filled ? n : count - filled;
      for (int64_t i = 0; i < take; i++) dst[filled + i] = v;
      filled += take;
    }
  }
}

// walk an RLE/bit-packed hybrid stream's run HEADERS only (sequential
// varints — ~#runs of host work instead of ~#values): append PqRun records
// plus the raw bit-packed span bytes; the GPU expands values
// (binary-search-over-output-position kernels in kernels.hip). When `ones`
// is non-null (def levels, bit_width 1) it accumulates the 1-count via RLE
// headers + span popcounts. Returns the number of values consumed.
int64_t rle_bp_runs(const uint8_t* p, size_t len, int bit_width,
                    int64_t count, uint32_t out_base, std::vector<PqRun>* runs,
                    std::vector<uint8_t>* span_bytes, int64_t* ones) {
  int64_t filled = 0;
  size_t pos = 0;
  const int byte_w = (bit_width + 7) / 8;
  while (filled < count) {
    if (pos >= len) fail("rle: truncated stream");
    uint64_t header = 0;
    int shift = 0;
    while (true) {
      if (pos >= len) fail("rle: truncated varint");
      uint8_t b = p[pos++];
      header |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    PqRun r;
    r.out_pos = out_base + (uint32_t)filled;
    r.bw = (uint8_t)bit_width;
    if (header & 1) {
      int64_t n = (int64_t)(header >> 1) * 8;
      size_t bytes = (size_t)n * bit_width / 8;
      if (pos + bytes > len) fail("rle: truncated bit-packed run");
      int64_t take = n < count - filled ? n : count - filled;
      r.kind = 1;
      r.count = (uint32_t)take;
      r.src_off = (uint32_t)span_bytes->size();
      span_bytes->insert(span_bytes->end(), p + pos, p + pos + bytes);
      if (ones) {  // def levels: popcount the span, bounded by `take` bits
        int64_t full = take >> 3, rem = take & 7;
        for (int64_t i = 0; i < full; i++)
          *ones += __builtin_popcount(p[pos + (size_t)i]);
        if (rem)
          *ones += __builtin_popcount(p[pos + (size_t)full] &
                                      ((1u << rem) - 1));
      }
      filled += take;
      pos += bytes;
    } else {
      int64_t n = (int64_t)(header >> 1);
      uint32_t v = 0;
      if (pos + byte_w > len) fail("rle: truncated repeated run");
      for (int b = 0; b < byte_w; b++) v |= (uint32_t)p[pos + b] << (8 * b);
      pos += byte_w;
      int64_t take = n < count - filled ? n : count - filled;
      r.kind = 0;
      r.count = (uint32_t)take;
      r.src_off = v;
      if (ones && v) *ones += take;
      filled += take;
    }
    if (r.count) runs->push_back(r);
  }
  return filled;
}

// host expansion of staged runs (the fallback when a chunk turns out to mix
// dict and PLAIN pages and must be materialized after all)
void expand_runs_host(const std::vector<PqRun>& runs,
                      const std::vector<uint8_t>& span_bytes,
                      std::vector<uint32_t>* out) {
  for (const PqRun& r : runs) {
    size_t base = out->size();
    out->resize(base + r.count);
    uint32_t* dst = out->data() + base;
    if (r.kind == 0) {
      for (uint32_t i = 0; i < r.count; i++) dst[i] = r.src_off;
    } else {
      uint64_t acc = 0;
      int bits = 0;
      size_t bp = r.src_off;
      const uint32_t mask =
          r.bw >= 32 ? 0xFFFFFFFFu : (uint32_t)((1ull << r.bw) - 1);
      for (uint32_t i = 0; i < r.count; i++) {
        while (bits < r.bw) {
          acc |= (uint64_t)span_bytes[bp++] << bits;
          bits += 8;
        }
        dst[i] = (uint32_t)acc & mask;
        acc >>= r.bw;
        bits -= r.bw;
      }
    }
  }
}

}  // namespace

// ---- DELTA_BINARY_PACKED (encoding 5; parquet spec delta encoding) ---------
// <block_size><miniblocks_per_block><total_count><first: zigzag> then per
// block: <min_delta: zigzag><miniblock bitwidth bytes><bit-packed deltas>.
// v[i+1] = v[i] + min_delta + packed[i]; arithmetic wraps in i64 (arrow-cpp
// semantics; INT32 columns truncate per value). Data bytes exist only for
// miniblocks that contain remaining values (arrow-cpp writer behavior).
// core reader: ONE delta stream starting at *pos; advances *pos exactly past
// the stream (arrow-cpp writes full padded miniblocks only while values
// remain, so the cursor lands on the next stream for the composed byte-array
// encodings below)
void delta_bp_stream(const uint8_t* p, size_t len, size_t* posp, int64_t count,
                     std::vector<int64_t>* vals) {
  size_t pos = *posp;
  auto uleb = [&]() -> uint64_t {
    uint64_t v = 0;
    int shift = 0;
    while (true) {
      if (pos >= len) fail("delta: truncated varint");
      uint8_t b = p[pos++];
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
    }
  };
  auto zigzag = [&]() -> int64_t {
    uint64_t u = uleb();
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
  };
  uint64_t block_size = uleb();
  uint64_t mini_per_block = uleb();
  uint64_t total = uleb();
  int64_t value = zigzag();
  if (mini_per_block == 0 || block_size % mini_per_block != 0)
    fail("delta: bad miniblock config");
  uint64_t mini_size = block_size / mini_per_block;
  if (mini_size % 8 != 0) fail("delta: miniblock size not multiple of 8");
  if ((int64_t)total < count) fail("delta: fewer values than expected");
  vals->reserve(vals->size() + (size_t)count);
  int64_t remaining = count;
  if (remaining > 0) {
    vals->push_back(value);
    remaining--;
  }
  std::vector<uint8_t> widths(mini_per_block);
  while (remaining > 0) {
    int64_t min_delta = zigzag();
    if (pos + mini_per_block > len) fail("delta: truncated widths");
    memcpy(widths.data(), p + pos, mini_per_block);
    pos += mini_per_block;
    for (uint64_t m = 0; m < mini_per_block && remaining > 0; m++) {
      int bw = widths[m];
      if (bw > 64) fail("delta: bad bit width");
      size_t bytes = mini_size * (size_t)bw / 8;
      if (pos + bytes > len) fail("delta: truncated miniblock");
      uint64_t acc = 0;
      int bits = 0;
      size_t bp = pos;
      const uint64_t mask = bw == 64 ? ~0ull : ((1ull << bw) - 1);
      for (uint64_t i = 0; i < mini_size && remaining > 0; i++) {
        uint64_t d = 0;
        if (bw > 56) {  // bit-addressed path: the 64-bit acc can't stage it
          uint64_t bit = (uint64_t)i * bw;
          for (int bbit = 0; bbit < bw; bbit++) {
            uint64_t idx = bit + bbit;
            d |= (uint64_t)((p[pos + idx / 8] >> (idx % 8)) & 1) << bbit;
          }
        } else if (bw > 0) {
          while (bits < bw) {
            acc |= (uint64_t)p[bp++] << bits;
            bits += 8;
          }
          d = acc & mask;
          acc >>= bw;
          bits -= bw;
        }
        value += min_delta + (int64_t)d;
        vals->push_back(value);
        remaining--;
      }
      pos += bytes;
      acc = 0;
      bits = 0;
    }
  }
  *posp = pos;
}

void delta_bp_decode(const uint8_t* p, size_t len, int64_t count, int vw,
                     std::vector<uint8_t>* out) {
  std::vector<int64_t> vals;
  size_t pos = 0;
  delta_bp_stream(p, len, &pos, count, &vals);
  out->reserve(out->size() + vals.size() * (size_t)vw);
  for (int64_t v : vals) {
    uint8_t b[8];
    memcpy(b, &v, 8);
    out->insert(out->end(), b, b + vw);
  }
}

// ---- snappy ----------------------------------------------------------------
// core decoder over caller-provided storage: dst must have out_len +
// kSnappyPad writable bytes (the hot paths overshoot with fixed 16/32-byte
// copies). Separated from the vector API so the parquet page loop can REUSE
// one grow-only buffer: a fresh vector per page cost an alloc + zero-fill +
// page-fault storm per page and was ~90% of the measured chunk-decode time
// (tools/pq_prof, 100M-row snappy file).
// 256-entry tag table: bits 0-7 = token length, bits 8-10 = copy-offset
// high bits (<<8), bits 11-13 = trailer byte count. Lets every token decode
// with one table load + one unconditional 4-byte trailer load and a SINGLE
// literal-vs-copy branch — the naive nested-branch decoder measured ~30
// cycles/token on short-match-heavy int columns (~2 mispredicts/token:
// tools/pq_prof + token census).
static const uint16_t* snappy_tag_table() {
  static uint16_t t[256];
  static bool init = [] {
    for (int c = 0; c < 256; c++) {
      int type = c & 3;
      uint16_t e = 0;
      if (type == 0) {
        int lm1 = c >> 2;
        if (lm1 < 60) {
          e = (uint16_t)(lm1 + 1);            // literal, length inline
        } else {
          e = (uint16_t)((lm1 - 59) << 11);   // long literal: 1-4 len bytes
        }
      } else if (type == 1) {
        e = (uint16_t)((1 << 11) | (((c >> 5) & 7) << 8) |
                       (((c >> 2) & 7) + 4));
      } else if (type == 2) {
        e = (uint16_t)((2 << 11) | ((c >> 2) + 1));
      } else {
        e = (uint16_t)((4 << 11) | ((c >> 2) + 1));
      }
      t[c] = e;
    }
    return true;
  }();
  (void)init;
  return t;
}

bool snappy_uncompress_raw(const uint8_t* src, size_t n, uint8_t* dst,
                           size_t out_len, std::string* err) {
  size_t pos = 0;
  uint64_t hdr_len = 0;
  int shift = 0;
  bool got = false;
  while (pos < n) {
    uint8_t b = src[pos++];
    hdr_len |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) {
      got = true;
      break;
    }
    shift += 7;
  }
  if (!got || hdr_len != out_len) {
    *err = "snappy: bad header length";
    return false;
  }
  static const uint32_t kWordMask[5] = {0, 0xffu, 0xffffu, 0xffffffu,
                                        0xffffffffu};
  const uint16_t* table = snappy_tag_table();
  size_t op = 0;

  auto do_copy = [&](size_t off, size_t len) __attribute__((always_inline)) -> bool {
    // every path copies in FIXED-size chunks that overshoot into the slack
    // (all writes < op+len+16 <= out_len+kSnappyPad; overshot bytes are
    // rewritten by later tokens) — a variable-length memcpy is a library
    // call per token
    if (off == 0 || off > op || op + len > out_len) return false;
    size_t start = op - off;
    if (off >= 16 && len <= 32) {
      memcpy(dst + op, dst + start, 16);
      memcpy(dst + op + 16, dst + start + 16, 16);
    } else if (off >= 8) {
      for (size_t d = 0; d < len; d += 8)
        memcpy(dst + op + d, dst + start + d, 8);
    } else if ((8 % off) == 0) {  // off 1/2/4: period divides 8
      for (size_t j = 0, s = 0; j < 8; j++) {
        dst[op + j] = dst[start + s];
        if (++s == off) s = 0;
      }
      for (size_t d = 8; d < len; d += 8)
        memcpy(dst + op + d, dst + op + d - 8, 8);
    } else {  // off 3/5/6/7: byte replication (rare)
      for (size_t j = 0, s = 0; j < len; j++) {
        dst[op + j] = dst[start + s];
        if (++s == off) s = 0;
      }
    }
    op += len;
    return true;
  };

  // fast loop: enough input margin for the unconditional 4-byte trailer
  // load and the 64-byte literal overshoot copy
  while (pos + 68 <= n) {
    uint8_t c = src[pos++];
    uint32_t entry = table[c];
    uint32_t tb = entry >> 11;
    uint32_t trailer;
    memcpy(&trailer, src + pos, 4);
    trailer &= kWordMask[tb];
    uint32_t len = entry & 0xff;
    if ((c & 3) == 0) {  // literal
      if (__builtin_expect(len == 0, 0)) {  // long literal: tb length bytes
        pos += tb;
        uint64_t ll = (uint64_t)trailer + 1;
        if (pos + ll > n || op + ll > out_len) {
          *err = "snappy: truncated literal";
          return false;
        }
        memcpy(dst + op, src + pos, ll);
        pos += ll;
        op += ll;
        continue;
      }
      if (op + len > out_len) {
        *err = "snappy: literal overruns output";
        return false;
      }
      memcpy(dst + op, src + pos, 16);  // len <= 60: fixed chunks overshoot
      if (__builtin_expect(len > 16, 0)) {
        memcpy(dst + op + 16, src + pos + 16, 16);
        memcpy(dst + op + 32, src + pos + 32, 16);
        memcpy(dst + op + 48, src + pos + 48, 16);
      }
      pos += len;
      op += len;
    } else {
      pos += tb;
      size_t off = (entry & 0x700) + trailer;
      if (!do_copy(off, len)) {
        *err = "snappy: bad copy";
        return false;
      }
    }
  }

  // tail loop: exact bounds checks near the end of the input
  while (pos < n) {
    uint8_t tag = src[pos++];
    int type = tag & 3;
    if (type == 0) {
      uint64_t len = (tag >> 2) + 1;
      if (len > 60) {
        int nb = (int)len - 60;
        if (pos + (size_t)nb > n) {
          *err = "snappy: truncated literal length";
          return false;
        }
        len = 0;
        for (int i = 0; i < nb; i++) len |= (uint64_t)src[pos++] << (8 * i);
        len += 1;
      }
      if (pos + len > n || op + len > out_len) {
        *err = "snappy: truncated literal";
        return false;
      }
      memcpy(dst + op, src + pos, len);
      op += len;
      pos += len;
    } else {
      uint64_t len, off;
      const size_t offw = (type == 1) ? 1 : (type == 2) ? 2 : 4;
      if (pos + offw > n) {
        *err = "snappy: truncated copy offset";
        return false;
      }
      if (type == 1) {
        len = ((tag >> 2) & 7) + 4;
        off = ((uint64_t)(tag >> 5) << 8) | src[pos++];
      } else if (type == 2) {
        len = (tag >> 2) + 1;
        off = (uint64_t)src[pos] | ((uint64_t)src[pos + 1] << 8);
        pos += 2;
      } else {
        len = (tag >> 2) + 1;
        off = 0;
        for (int i = 0; i < 4; i++) off |= (uint64_t)src[pos++] << (8 * i);
      }
      if (!do_copy(off, len)) {
        *err = "snappy: bad copy offset";
        return false;
      }
    }
  }
  if (op != out_len) {
    *err = "snappy: length mismatch";
    return false;
  }
  return true;
}

bool snappy_uncompress(const uint8_t* src, size_t n, std::vector<uint8_t>* out,
                       std::string* err) {
  size_t pos = 0;
  uint64_t out_len = 0;
  int shift = 0;
  while (pos < n) {
    uint8_t b = src[pos++];
    out_len |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  if (out->size() < out_len + kSnappyPad) out->resize(out_len + kSnappyPad);
  if (!snappy_uncompress_raw(src, n, out->data(), out_len, err)) return false;
  out->resize(out_len);
  return true;
}

// ---- ParquetFile ------------------------------------------------------------
ParquetFile::~ParquetFile() {
  if (map_.data) munmap((void*)map_.data, map_.size);
}

ParquetFile::ParquetFile(const std::string& path) : path_(path) {
  int fd = open(path.c_str(), O_RDONLY);
  if (fd < 0) fail("parquet: cannot open " + path);
  struct stat st;
  if (fstat(fd, &st) != 0) {
    close(fd);
    fail("parquet: cannot stat " + path);
  }
  long sz = (long)st.st_size;
  void* p = mmap(nullptr, sz, PROT_READ, MAP_PRIVATE, fd, 0);
  close(fd);
  if (p == MAP_FAILED) fail("parquet: mmap failed for " + path);
  (void)madvise(p, sz, MADV_WILLNEED);
  map_.data = (const uint8_t*)p;
  map_.size = (size_t)sz;
  if (sz < 12 || memcmp(file_.data() + sz - 4, "PAR1", 4) != 0)
    fail("parquet: bad magic in " + path);
  uint32_t meta_len;
  memcpy(&meta_len, file_.data() + sz - 8, 4);
  if ((long)meta_len + 12 > sz) fail("parquet: bad footer length");
  const uint8_t* meta = file_.data() + sz - 8 - meta_len;

  TReader r{meta, meta + meta_len};
  int last = 0;
  std::vector<SchemaElem> schema;
  while (true) {
    auto [id, t] = r.field_header(&last);
    if (t == 0) break;
    switch (id) {
      case 2: {  // schema: list<SchemaElement>
        auto [n, et] = r.list_header();
        (void)et;
        for (uint64_t i = 0; i < n; i++) schema.push_back(read_schema_element(r));
        break;
      }
      case 4: {  // row_groups
        auto [n, et] = r.list_header();
        (void)et;
        for (uint64_t i = 0; i < n; i++) {
          RowGroupMeta rg;
          int lrg = 0;
          while (true) {
            auto [idr, tr] = r.field_header(&lrg);
            if (tr == 0) break;
            if (idr == 1) {  // columns: list<ColumnChunk>
              auto [nc, cet] = r.list_header();
              (void)cet;
              for (uint64_t c = 0; c < nc; c++) {
                ChunkMeta cm;
                int lcc = 0;
                while (true) {
                  auto [idc, tc] = r.field_header(&lcc);
                  if (tc == 0) break;
                  if (idc == 3) {  // ColumnMetaData
                    int lmd = 0;
                    while (true) {
                      auto [idm, tm] = r.field_header(&lmd);
                      if (tm == 0) break;
                      switch (idm) {
                        case 4: cm.codec = (int)r.zigzag(); break;
                        case 5: cm.num_values = r.zigzag(); break;
                        case 7: cm.total_compressed_size = r.zigzag(); break;
                        case 9: cm.data_page_offset = r.zigzag(); break;
                        case 11: cm.dict_page_offset = r.zigzag(); break;
                        case 12: {  // Statistics{5:max_value, 6:min_value}
                          int lst = 0;
                          while (true) {
                            auto [ids, ts] = r.field_header(&lst);
                            if (ts == 0) break;
                            if (ids == 5) {
                              std::string v = r.binary();
                              cm.stat_max.assign(v.begin(), v.end());
                            } else if (ids == 6) {
                              std::string v = r.binary();
                              cm.stat_min.assign(v.begin(), v.end());
                            } else {
                              r.skip(ts);
                            }
                          }
                          break;
                        }
                        default: r.skip(tm);
                      }
                    }
                  } else {
                    r.skip(tc);
                  }
                }
                rg.chunks.push_back(cm);
              }
            } else if (idr == 3) {
              rg.num_rows = r.zigzag();
            } else {
              r.skip(tr);
            }
          }
          row_groups_.push_back(std::move(rg));
        }
        break;
      }
      default: r.skip(t);
    }
  }

  if (schema.empty()) fail("parquet: empty schema");
  // flat schema: root + leaves only
  for (size_t i = 1; i < schema.size(); i++) {
    const SchemaElem& e = schema[i];
    if (e.num_children > 0) fail("parquet: nested schema unsupported");
    if (e.repetition == 2) fail("parquet: repeated fields unsupported");
    PqColumnInfo ci;
    ci.name = e.name;
    ci.physical_type = e.type;
    ci.utf8 = e.converted_type == 0;
    ci.nullable = e.repetition == 1;
    if (ci.dtype() == DType::Unsupported)
      fail("parquet: unsupported physical type for column " + e.name);
    columns_.push_back(std::move(ci));
  }
  for (const auto& rg : row_groups_)
    if (rg.chunks.size() != columns_.size())
      fail("parquet: row group column count mismatch");
}

void pq_materialize_gpu_staging(PqColumnChunkData* cd) {
  if (cd->gpu_dict) {
    expand_runs_host(cd->idx_runs, cd->idx_bytes, &cd->dict_indices);
    cd->gpu_dict = false;
    cd->uses_dict = true;
    cd->idx_runs.clear();
    cd->idx_bytes.clear();
  }
  if (cd->gpu_def) {
    std::vector<uint32_t> bits;
    expand_runs_host(cd->def_runs, cd->def_bytes, &bits);
    cd->validity.assign((bits.size() + 7) / 8, 0);
    for (size_t i = 0; i < bits.size(); i++)
      if (bits[i]) cd->validity[i >> 3] |= (uint8_t)(1u << (i & 7));
    cd->gpu_def = false;
    cd->def_runs.clear();
    cd->def_bytes.clear();
  }
  if (cd->gpu_comp) {
    // host re-statement of kernels_pq.hip: decompress each recorded page,
    // parse its def levels, append dense values — CPU tests pin the
    // pre-scan/split logic without a GPU
    const int w = cd->value_width;
    if (w <= 0) fail("parquet: gpu_comp on non-fixed-width column");
    std::vector<uint8_t> vb;  // byte per value
    if (cd->nullable) {
      vb.assign((size_t)cd->prefix_values, 1);
      if (!cd->validity.empty())
        for (int64_t i = 0; i < cd->prefix_values; i++)
          vb[(size_t)i] = (cd->validity[i >> 3] >> (i & 7)) & 1;
    }
    std::vector<uint8_t> buf;
    std::string err;
    for (const auto& pg : cd->comp_pages) {
      if (!snappy_uncompress(pg.src, pg.comp_len, &buf, &err))
        fail("parquet: " + err);
      if (buf.size() != pg.uncomp_len) fail("parquet: page size mismatch");
      const uint8_t* data = buf.data();
      int64_t dlen = (int64_t)pg.uncomp_len;
      int64_t nn = pg.num_values;
      if (cd->nullable) {
        uint32_t ll;
        if (dlen < 4) fail("parquet: truncated def-level length");
        memcpy(&ll, data, 4);
        if ((int64_t)ll > dlen - 4) fail("parquet: def levels overrun page");
        std::vector<uint32_t> def;
        rle_bp_decode(data + 4, ll, 1, pg.num_values, &def);
        nn = 0;
        for (uint32_t d : def) {
          vb.push_back((uint8_t)(d != 0));
          nn += d != 0;
        }
        cd->null_count += pg.num_values - nn;
        data += 4 + ll;
        dlen -= 4 + ll;
      }
      if (dlen < nn * w) fail("parquet: short PLAIN data in gpu page");
      cd->plain.insert(cd->plain.end(), data, data + nn * w);
      cd->num_values += pg.num_values;
    }
    if (cd->nullable && cd->null_count > 0) {
      cd->validity.assign((vb.size() + 7) / 8, 0);
      for (size_t i = 0; i < vb.size(); i++)
        if (vb[i]) cd->validity[i >> 3] |= (uint8_t)(1u << (i & 7));
    } else {
      cd->validity.clear();
    }
    cd->gpu_comp = false;
    cd->comp_pages.clear();
    cd->suffix_values = 0;
  }
}

PqColStats ParquetFile::column_stats(int rg, int col) const {
  const ChunkMeta& cm = row_groups_.at(rg).chunks.at(col);
  const PqColumnInfo& ci = columns_.at(col);
  PqColStats s;
  if (cm.stat_min.empty() || cm.stat_max.empty()) return s;
  auto rd = [&](const std::vector<uint8_t>& v, int64_t* i, double* f) -> bool {
    switch (ci.physical_type) {
      case 1: { int32_t x; if (v.size() < 4) return false; memcpy(&x, v.data(), 4); *i = x; return true; }
      case 2: { int64_t x; if (v.size() < 8) return false; memcpy(&x, v.data(), 8); *i = x; return true; }
      case 4: { float x; if (v.size() < 4) return false; memcpy(&x, v.data(), 4); *f = x; return true; }
      case 5: { double x; if (v.size() < 8) return false; memcpy(&x, v.data(), 8); *f = x; return true; }
      default: return false;
    }
  };
  s.has_minmax = rd(cm.stat_min, &s.min_i, &s.min_f) &&
                 rd(cm.stat_max, &s.max_i, &s.max_f);
  return s;
}

PqColumnChunkData ParquetFile::read_chunk(int rg, int col) const {
  const RowGroupMeta& g = row_groups_.at(rg);
  const ChunkMeta& cm = g.chunks.at(col);
  const PqColumnInfo& ci = columns_.at(col);
  const int vw = (int)dtype_width(ci.dtype());

  PqColumnChunkData out;
  out.num_values = 0;
  out.value_width = ci.physical_type == 6 ? 0 : (int)dtype_width(ci.dtype());
  // GPU run-expansion mode (fixed-width columns): the host walks RLE/
  // bit-packed run headers only and the GPU expands def levels and dict
  // indices (kernels.hip k_runs_expand_u32 / k_def_expand_validity).
  // AURON_PARQUET_GPU=0 forces the all-host decode for A/B comparison.
  const char* pg = getenv("AURON_PARQUET_GPU");
  const bool gpu_ok = ci.physical_type != 6 && !(pg && pg[0] == '0');
  // GPU page-decompression mode: pre-scan the page headers (cheap — no
  // decompression); if every data page from the first v1 PLAIN page onward
  // qualifies, the host decodes only the dictionary-encoded prefix and the
  // device decompresses + decodes the PLAIN suffix (kernels_pq.hip)
  const char* pc = getenv("AURON_PARQUET_GPUCOMP");
  bool gpu_comp_mode = gpu_ok && cm.codec == 1 && !(pc && pc[0] == '0');
  int64_t stop_pos = -1;  // file offset of the first GPU-handled page
  if (gpu_comp_mode) {
    gpu_comp_mode = false;
    int64_t p0 = cm.dict_page_offset >= 0 ? std::min(cm.dict_page_offset,
                                                     cm.data_page_offset)
                                          : cm.data_page_offset;
    int64_t pend = p0 + cm.total_compressed_size;
    int64_t split = -1;
    int64_t sfx_values = 0;
    std::vector<PqColumnChunkData::GpuPageRef> refs;
    bool ok = pend <= (int64_t)file_.size();
    int64_t sp = p0;
    while (ok && sp < pend) {
      TReader hr{file_.data() + sp, file_.data() + pend};
      PageHeader ph;
      try {
        ph = read_page_header(hr);
      } catch (...) {
        ok = false;
        break;
      }
      const uint8_t* body = hr.p;
      if (ph.compressed_size < 0 || ph.uncompressed_size < 0 ||
          body - file_.data() + ph.compressed_size > pend) {
        ok = false;
        break;
      }
      bool suffix_ok = ph.type == 0 && ph.encoding == 0 &&
                       (!ci.nullable || ph.def_encoding == 3) &&
                       ph.num_values > 0 && ph.num_values <= (1 << 18) &&
                       ph.uncompressed_size <= (1 << 20) + 65536;
      if (split < 0) {
        if (suffix_ok) split = sp;  // first qualifying PLAIN page
      } else if (!suffix_ok) {
        ok = false;  // non-qualifying page AFTER the split: give up
        break;
      }
      if (split >= 0 && suffix_ok) {
        refs.push_back({body, (uint32_t)ph.compressed_size,
                        (uint32_t)ph.uncompressed_size,
                        (uint32_t)ph.num_values});
        sfx_values += ph.num_values;
      }
      sp = (body - file_.data()) + ph.compressed_size;
    }
    if (ok && split >= 0 && sfx_values > 0) {
      gpu_comp_mode = true;
      stop_pos = split;
      out.gpu_comp = true;
      out.comp_pages = std::move(refs);
      out.suffix_values = sfx_values;
      out.nullable = ci.nullable;
    }
  }
  const bool gpu_def_mode = gpu_ok && ci.nullable && !gpu_comp_mode;
  if (ci.physical_type != 6)
    out.plain.reserve((size_t)cm.num_values * vw);
  std::vector<uint8_t> valid_bits;  // byte per value (bit-packed at the end)
  valid_bits.reserve((size_t)cm.num_values);
  const bool is_bytes = ci.physical_type == 6;
  // BYTE_ARRAY values accumulate as non-null (len, bytes) pairs; row-aligned
  // offsets are assembled at the end against the validity bits
  std::vector<int32_t> nn_lens;
  std::vector<uint8_t> nn_data;
  std::vector<int32_t> dict_lens;  // dict entry lens for byte-array dicts

  int64_t pos = cm.dict_page_offset >= 0 ? cm.dict_page_offset
                                         : cm.data_page_offset;
  if (cm.dict_page_offset >= 0 && cm.data_page_offset < cm.dict_page_offset)
    pos = cm.data_page_offset;  // defensive: some writers order differently
  int64_t chunk_end = pos + cm.total_compressed_size;
  if (chunk_end > (int64_t)file_.size()) fail("parquet: chunk overruns file");

  auto decompress = [&](const uint8_t* src, size_t n, size_t out_size,
                        std::vector<uint8_t>* buf) -> const uint8_t* {
    PqTimer _t(&g_pq_ns_decomp);
    std::string err;
    switch (cm.codec) {
      case 0: return src;  // UNCOMPRESSED
      case 1:
        // grow-only reuse: the caller passes the same buffer for every page
        if (buf->size() < out_size + kSnappyPad)
          buf->resize(out_size + kSnappyPad);
        if (!snappy_uncompress_raw(src, n, buf->data(), out_size, &err))
          fail("parquet: " + err);
        return buf->data();
      case 6:
        if (!zstd_uncompress(src, n, out_size, buf, &err))
          fail("parquet: " + err);
        return buf->data();
      case 7:
        if (!lz4raw_uncompress(src, n, out_size, buf, &err))
          fail("parquet: " + err);
        return buf->data();
      default:
        fail("parquet: unsupported codec " + std::to_string(cm.codec));
    }
  };

    // writers may FALL BACK from dictionary to PLAIN mid-chunk (e.g. pyarrow
  // once the dictionary page hits its size limit); flatten the accumulated
  // dict-encoded prefix to PLAIN on host when a mix appears (the dict
  // portion is small by construction in that case)
  // expand dict indices to PLAIN values appended to out.plain (typed
  // copies: the per-value vector::insert this replaces dominated the
  // whole chunk decode)
  auto expand_dict = [&](const uint32_t* idx, size_t cnt) {
    size_t base = out.plain.size();
    out.plain.resize(base + cnt * (size_t)vw);
    if (vw == 8) {
      uint64_t* d = reinterpret_cast<uint64_t*>(out.plain.data() + base);
      const uint64_t* dict =
          reinterpret_cast<const uint64_t*>(out.dict_values.data());
      for (size_t i = 0; i < cnt; i++) {
        if ((int64_t)idx[i] >= out.dict_count)
          fail("parquet: dict index range");
        d[i] = dict[idx[i]];
      }
    } else if (vw == 4) {
      uint32_t* d = reinterpret_cast<uint32_t*>(out.plain.data() + base);
      const uint32_t* dict =
          reinterpret_cast<const uint32_t*>(out.dict_values.data());
      for (size_t i = 0; i < cnt; i++) {
        if ((int64_t)idx[i] >= out.dict_count)
          fail("parquet: dict index range");
        d[i] = dict[idx[i]];
      }
    } else {
      for (size_t i = 0; i < cnt; i++) {
        if ((int64_t)idx[i] >= out.dict_count)
          fail("parquet: dict index range");
        memcpy(out.plain.data() + base + i * (size_t)vw,
               out.dict_values.data() + (size_t)idx[i] * vw, (size_t)vw);
      }
    }
  };

  std::vector<uint8_t> page_buf;  // reused across pages (grow-only)
  const int64_t loop_end = gpu_comp_mode ? stop_pos : chunk_end;
  while (pos < loop_end && out.num_values < cm.num_values) {
    TReader hr{file_.data() + pos, file_.data() + chunk_end};
    PageHeader ph = read_page_header(hr);
    const uint8_t* page = hr.p;
    // every size below comes from the (untrusted) page header: validate
    // against the chunk bounds before any read
    if (ph.compressed_size < 0 || ph.uncompressed_size < 0 ||
        page - file_.data() + ph.compressed_size > chunk_end)
      fail("parquet: page overruns chunk");
    pos = (page - file_.data()) + ph.compressed_size;

    if (ph.type == 2) {  // dictionary page
      std::vector<uint8_t>& buf = page_buf;
      const uint8_t* data =
          decompress(page, ph.compressed_size, ph.uncompressed_size, &buf);
      out.dict_values.assign(data, data + ph.uncompressed_size);
      out.dict_count = ph.dict_num_values;
      if (is_bytes) {
        // index the dictionary's (u32 len, bytes) entries
        size_t pos2 = 0;
        dict_lens.clear();
        std::vector<int32_t> offs;
        offs.push_back(0);
        for (int64_t i = 0; i < out.dict_count; i++) {
          if (pos2 + 4 > out.dict_values.size())
            fail("parquet: truncated byte-array dict");
          uint32_t l;
          memcpy(&l, out.dict_values.data() + pos2, 4);
          pos2 += 4;
          if (l > out.dict_values.size() - pos2)
            fail("parquet: byte-array dict entry overruns dict");
          dict_lens.push_back((int32_t)pos2);  // start of bytes
          dict_lens.push_back((int32_t)l);     // length
          pos2 += l;
        }
      } else if (out.dict_count > 0 &&
                 out.dict_values.size() < (size_t)out.dict_count * vw) {
        fail("parquet: dict page shorter than dict_count * width");
      }
      continue;
    }
    if (ph.type != 0 && ph.type != 3)
      fail("parquet: unsupported page type " + std::to_string(ph.type));

    std::vector<uint8_t>& buf = page_buf;
    const uint8_t* data;
    int64_t dlen;
    std::vector<uint32_t> def_levels;
    int64_t page_ones = 0;
    bool page_def_staged = false;
    if (ph.type == 0) {  // V1: whole page compressed, def levels inside
      data = decompress(page, ph.compressed_size, ph.uncompressed_size, &buf);
      dlen = ph.uncompressed_size;
      if (ci.nullable) {
        PqTimer _t(&g_pq_ns_levels);
        if (ph.def_encoding != 3) fail("parquet: def levels must be RLE");
        uint32_t ll;
        if (dlen < 4) fail("parquet: truncated def-level length");
        memcpy(&ll, data, 4);
        if ((int64_t)ll > dlen - 4) fail("parquet: def levels overrun page");
        if (gpu_def_mode) {
          page_ones = 0;
          rle_bp_runs(data + 4, ll, 1, ph.num_values,
                      (uint32_t)out.num_values, &out.def_runs, &out.def_bytes,
                      &page_ones);
          page_def_staged = true;
        } else {
          rle_bp_decode(data + 4, ll, 1, ph.num_values, &def_levels);
        }
        data += 4 + ll;
        dlen -= 4 + ll;
      }
    } else {  // V2: def levels uncompressed before the (compressed) values
      const uint8_t* dp = page;
      if (ph.def_len < 0 || ph.rep_len < 0 ||
          ph.def_len + ph.rep_len > ph.compressed_size)
        fail("parquet: V2 level lengths overrun page");
      if (ci.nullable) {
        if (gpu_def_mode) {
          page_ones = 0;
          if (ph.def_len > 0) {
            rle_bp_runs(dp, ph.def_len, 1, ph.num_values,
                        (uint32_t)out.num_values, &out.def_runs,
                        &out.def_bytes, &page_ones);
          } else {  // whole page valid: synthetic all-ones run
            out.def_runs.push_back(PqRun{(uint32_t)out.num_values,
                                         (uint32_t)ph.num_values, 1u, 0, 1,
                                         0});
            page_ones = ph.num_values;
          }
          page_def_staged = true;
        } else if (ph.def_len > 0) {
          rle_bp_decode(dp, ph.def_len, 1, ph.num_values, &def_levels);
        }
      }
      dp += ph.def_len + ph.rep_len;
      int64_t comp = ph.compressed_size - ph.def_len - ph.rep_len;
      int64_t uncomp = ph.uncompressed_size - ph.def_len - ph.rep_len;
      if (uncomp < 0) fail("parquet: V2 level lengths exceed uncompressed size");
      data = ph.v2_compressed ? decompress(dp, comp, uncomp, &buf) : dp;
      dlen = uncomp;
    }

    int64_t nvals = ph.num_values;
    int64_t non_null = nvals;
    if (page_def_staged) {
      non_null = page_ones;
      out.null_count += nvals - page_ones;
    } else if (ci.nullable && !def_levels.empty()) {
      non_null = 0;
      size_t base = valid_bits.size();
      valid_bits.resize(base + (size_t)nvals);
      for (int64_t i = 0; i < nvals; i++) {
        uint8_t v = def_levels[i] != 0;
        valid_bits[base + (size_t)i] = v;
        non_null += v;
      }
      out.null_count += nvals - non_null;
    } else if (!gpu_def_mode) {
      valid_bits.resize(valid_bits.size() + (size_t)nvals, 1);
    }

        auto flatten_dict = [&]() {
      if (out.gpu_dict) {
        // staged GPU runs must materialize after all (dict/PLAIN mix)
        std::vector<uint32_t> idx;
        expand_runs_host(out.idx_runs, out.idx_bytes, &idx);
        out.gpu_dict = false;
        out.idx_runs.clear();
        out.idx_bytes.clear();
        out.nn_count = 0;
        expand_dict(idx.data(), idx.size());
      }
      if (!out.uses_dict) return;
      std::vector<uint32_t> idx = std::move(out.dict_indices);
      out.dict_indices.clear();
      out.uses_dict = false;
      expand_dict(idx.data(), idx.size());
    };

    PqTimer _tv(&g_pq_ns_values);
    switch (ph.encoding) {
      case 0: {  // PLAIN
        if (is_bytes) {
          int64_t pos2 = 0;
          for (int64_t i = 0; i < non_null; i++) {
            if (pos2 + 4 > dlen) fail("parquet: short BYTE_ARRAY data");
            uint32_t l;
            memcpy(&l, data + pos2, 4);
            pos2 += 4;
            if (pos2 + l > dlen) fail("parquet: short BYTE_ARRAY value");
            nn_lens.push_back((int32_t)l);
            nn_data.insert(nn_data.end(), data + pos2, data + pos2 + l);
            pos2 += l;
          }
          break;
        }
        if (dlen < non_null * vw) fail("parquet: short PLAIN data");
        flatten_dict();
        out.plain.insert(out.plain.end(), data, data + non_null * vw);
        break;
      }
      case 5: {  // DELTA_BINARY_PACKED (v2 writers; INT32/INT64 only)
        if (is_bytes) fail("parquet: delta encoding on byte arrays");
        flatten_dict();
        delta_bp_decode(data, (size_t)dlen, non_null, vw, &out.plain);
        break;
      }
      case 9: {  // BYTE_STREAM_SPLIT: w byte planes, de-interleave
        if (is_bytes) fail("parquet: BYTE_STREAM_SPLIT on byte arrays");
        if (dlen < non_null * vw) fail("parquet: short BYTE_STREAM_SPLIT");
        flatten_dict();
        size_t base = out.plain.size();
        out.plain.resize(base + (size_t)non_null * vw);
        for (int b = 0; b < vw; b++) {
          const uint8_t* plane = data + (size_t)b * non_null;
          uint8_t* dst = out.plain.data() + base + b;
          for (int64_t i = 0; i < non_null; i++) dst[(size_t)i * vw] = plane[i];
        }
        break;
      }
      case 6: {  // DELTA_LENGTH_BYTE_ARRAY: delta lens ++ concatenated bytes
        if (!is_bytes) fail("parquet: DELTA_LENGTH on non-byte column");
        std::vector<int64_t> lens;
        size_t pos2 = 0;
        delta_bp_stream(data, (size_t)dlen, &pos2, non_null, &lens);
        for (int64_t l : lens) {
          if (l < 0 || pos2 + (size_t)l > (size_t)dlen)
            fail("parquet: DELTA_LENGTH overrun");
          nn_lens.push_back((int32_t)l);
          nn_data.insert(nn_data.end(), data + pos2, data + pos2 + l);
          pos2 += (size_t)l;
        }
        break;
      }
      case 7: {  // DELTA_BYTE_ARRAY: prefix-len stream ++ suffix-len stream
                 // ++ suffix bytes; value[i] = value[i-1][:prefix] + suffix
        if (!is_bytes) fail("parquet: DELTA_BYTE_ARRAY on non-byte column");
        std::vector<int64_t> plens, slens;
        size_t pos2 = 0;
        delta_bp_stream(data, (size_t)dlen, &pos2, non_null, &plens);
        delta_bp_stream(data, (size_t)dlen, &pos2, non_null, &slens);
        std::vector<uint8_t> prev;
        for (int64_t i = 0; i < non_null; i++) {
          int64_t pl = plens[(size_t)i], sl = slens[(size_t)i];
          if (pl < 0 || sl < 0 || (size_t)pl > prev.size() ||
              pos2 + (size_t)sl > (size_t)dlen)
            fail("parquet: DELTA_BYTE_ARRAY overrun");
          std::vector<uint8_t> cur(prev.begin(), prev.begin() + pl);
          cur.insert(cur.end(), data + pos2, data + pos2 + sl);
          pos2 += (size_t)sl;
          nn_lens.push_back((int32_t)cur.size());
          nn_data.insert(nn_data.end(), cur.begin(), cur.end());
          prev = std::move(cur);
        }
        break;
      }
      case 2:    // PLAIN_DICTIONARY
      case 8: {  // RLE_DICTIONARY
        if (out.dict_values.empty()) fail("parquet: dict page missing");
        if (dlen < 1) fail("parquet: empty dict-index page");
        int bw = data[0];
        if (bw > 32) fail("parquet: dict index bit width > 32");
        if (gpu_ok && out.plain.empty() && !out.uses_dict) {
          // stage run headers for GPU expansion (all pages so far dict)
          out.gpu_dict = true;
          rle_bp_runs(data + 1, dlen - 1, bw, non_null,
                      (uint32_t)out.nn_count, &out.idx_runs, &out.idx_bytes,
                      nullptr);
          out.nn_count += non_null;
          break;
        }
        std::vector<uint32_t> idx;
        rle_bp_decode(data + 1, dlen - 1, bw, non_null, &idx);
        if (is_bytes) {
          for (uint32_t ix : idx) {
            if ((int64_t)ix >= out.dict_count)
              fail("parquet: dict index range");
            int32_t start = dict_lens[(size_t)ix * 2];
            int32_t l = dict_lens[(size_t)ix * 2 + 1];
            nn_lens.push_back(l);
            nn_data.insert(nn_data.end(), out.dict_values.begin() + start,
                           out.dict_values.begin() + start + l);
          }
          break;
        }
        if (!out.plain.empty()) {
          // already flattened: expand this page directly
          expand_dict(idx.data(), idx.size());
        } else {
          out.dict_indices.insert(out.dict_indices.end(), idx.begin(),
                                  idx.end());
          out.uses_dict = true;
        }
        break;
      }
      default:
        fail("parquet: unsupported value encoding " +
             std::to_string(ph.encoding));
    }
    out.num_values += nvals;
  }

  if (out.gpu_comp) {
    out.prefix_values = out.num_values;
    // the engine needs the prefix as DENSE PLAIN values + packed validity;
    // flatten any dict-encoded prefix on host (it is small by construction)
    if (out.gpu_dict) {
      std::vector<uint32_t> idx;
      expand_runs_host(out.idx_runs, out.idx_bytes, &idx);
      out.gpu_dict = false;
      out.idx_runs.clear();
      out.idx_bytes.clear();
      out.nn_count = 0;
      expand_dict(idx.data(), idx.size());
    }
    if (out.uses_dict) {
      std::vector<uint32_t> idx = std::move(out.dict_indices);
      out.dict_indices.clear();
      out.uses_dict = false;
      expand_dict(idx.data(), idx.size());
    }
    if (!out.def_runs.empty())
      fail("parquet: gpu_comp prefix must not stage def runs");
  }
  if (out.num_values + out.suffix_values != cm.num_values)
    fail("parquet: value count mismatch in chunk");
  if (is_bytes) {
    // assemble row-aligned offsets (null rows zero-length)
    out.bin_offsets.assign(out.num_values + 1, 0);
    out.bin_data = std::move(nn_data);
    size_t nn = 0;
    for (int64_t i = 0; i < out.num_values; i++) {
      int32_t l = valid_bits[i] ? nn_lens[nn++] : 0;
      out.bin_offsets[i + 1] = out.bin_offsets[i] + l;
    }
    if (nn != nn_lens.size()) fail("parquet: byte-array count mismatch");
  }
  if (!out.def_runs.empty()) {
    if (out.null_count > 0) {
      out.gpu_def = true;
      out.def_bytes.resize(out.def_bytes.size() + 8, 0);  // u64 tail pad
    } else {  // no nulls after all: no validity needed
      out.def_runs.clear();
      out.def_bytes.clear();
    }
  } else if (out.null_count > 0) {
    out.validity.assign((valid_bits.size() + 7) / 8, 0);
    for (size_t i = 0; i < valid_bits.size(); i++)
      if (valid_bits[i]) out.validity[i >> 3] |= (uint8_t)(1u << (i & 7));
  }
  if (out.gpu_dict) out.idx_bytes.resize(out.idx_bytes.size() + 8, 0);
  return out;
}

}  // namespace auron
