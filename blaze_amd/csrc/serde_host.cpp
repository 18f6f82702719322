#include "serde_host.h"

#include <cstdio>
#include <cstring>

#include "lz4shim.h"

namespace auron {

void serde_write_len(uint64_t v, std::vector<uint8_t>* out) {
  // io/mod.rs:60-67
  while (v >= 128) {
    out->push_back((uint8_t)(128 + v % 128));
    v /= 128;
  }
  out->push_back((uint8_t)v);
}

namespace {

// batch_serde.rs:271-306: byte-plane transpose out[b*n+i] = in[i*w+b]
void append_transposed(const uint8_t* in, int64_t n, int w,
                       std::vector<uint8_t>* out) {
  size_t base = out->size();
  out->resize(base + (size_t)n * w);
  uint8_t* o = out->data() + base;
  for (int64_t i = 0; i < n; i++)
    for (int b = 0; b < w; b++) o[(int64_t)b * n + i] = in[i * w + b];
}

// slice a validity bitmap [row_beg, row_end) into a fresh LSB bitmap
void append_bitmap_slice(const uint8_t* bm, int64_t row_beg, int64_t row_end,
                         std::vector<uint8_t>* out) {
  int64_t n = row_end - row_beg;
  size_t base = out->size();
  out->resize(base + (size_t)((n + 7) / 8), 0);
  uint8_t* o = out->data() + base;
  for (int64_t i = 0; i < n; i++) {
    int64_t src = row_beg + i;
    if ((bm[src >> 3] >> (src & 7)) & 1) o[i >> 3] |= (uint8_t)(1u << (i & 7));
  }
}

}  // namespace

void serde_write_batch(const std::vector<HostCol>& cols, int64_t row_beg,
                       int64_t row_end, std::vector<uint8_t>* out) {
  int64_t n = row_end - row_beg;
  serde_write_len((uint64_t)n, out);  // batch_serde.rs:66-68
  for (const HostCol& c : cols) {
    if (c.validity) {  // batch_serde.rs:276-289 null header
      serde_write_len(1, out);
      append_bitmap_slice(c.validity, row_beg, row_end, out);
    } else {
      serde_write_len(0, out);
    }
    if (c.byte_width > 0) {
      append_transposed(c.values + row_beg * c.byte_width, n, c.byte_width, out);
    } else {
      // bytes array (batch_serde.rs:595-660): transposed i32 lens, then data
      std::vector<int32_t> lens((size_t)n);
      for (int64_t i = 0; i < n; i++)
        lens[i] = c.offsets[row_beg + i + 1] - c.offsets[row_beg + i];
      append_transposed((const uint8_t*)lens.data(), n, 4, out);
      const uint8_t* d = c.values + c.offsets[row_beg];
      out->insert(out->end(), d,
                  d + (c.offsets[row_end] - c.offsets[row_beg]));
    }
  }
}

namespace {

bool read_len_at(const uint8_t* p, size_t len, size_t* pos, uint64_t* out) {
  // io/mod.rs:69-79
  uint64_t v = 0, factor = 1;
  while (*pos < len) {
    uint8_t b = p[(*pos)++];
    if (b < 128) {
      v += (uint64_t)b * factor;
      *out = v;
      return true;
    }
    v += (uint64_t)(b - 128) * factor;
    factor *= 128;
  }
  return false;
}

void untranspose(const uint8_t* in, uint8_t* out, int64_t n, int w) {
  // inverse of batch_serde.rs:271-306 byte-plane transpose
  for (int64_t i = 0; i < n; i++)
    for (int b = 0; b < w; b++) out[i * w + b] = in[(int64_t)b * n + i];
}

}  // namespace

bool serde_read_batch(const uint8_t* p, size_t len, size_t* used,
                      const std::vector<int>& dtype_widths, int64_t* rows,
                      std::vector<OwnedCol>* cols, std::string* err) {
  size_t pos = *used;
  uint64_t n = 0;
  if (!read_len_at(p, len, &pos, &n)) {
    *err = "serde: truncated row count";
    return false;
  }
  // adversarial-input guards: a batch cannot carry more rows than input
  // bytes (every row costs >=1 byte in any column), and n*width arithmetic
  // below must not overflow size_t
  if (n > len) {
    *err = "serde: row count exceeds input size";
    return false;
  }
  *rows = (int64_t)n;
  cols->clear();
  for (int w : dtype_widths) {
    OwnedCol c;
    c.byte_width = w;
    uint64_t has_null = 0;
    if (!read_len_at(p, len, &pos, &has_null)) {
      *err = "serde: truncated null header";
      return false;
    }
    size_t bm = (n + 7) / 8;
    if (has_null) {
      if (pos + bm > len) {
        *err = "serde: truncated bitmap";
        return false;
      }
      c.validity.assign(p + pos, p + pos + bm);
      pos += bm;
    }
    if (w > 0) {
      size_t nb = (size_t)n * w;
      if (pos + nb > len) {
        *err = "serde: truncated values";
        return false;
      }
      c.values.resize(nb);
      if (w > 1)
        untranspose(p + pos, c.values.data(), (int64_t)n, w);
      else
        memcpy(c.values.data(), p + pos, nb);
      pos += nb;
    } else {
      // binary: transposed i32 lens then raw data (batch_serde.rs:595-660)
      if (pos + 4 * n > len) {
        *err = "serde: truncated lens";
        return false;
      }
      std::vector<int32_t> lens(n);
      untranspose(p + pos, (uint8_t*)lens.data(), (int64_t)n, 4);
      pos += 4 * n;
      c.offsets.resize(n + 1);
      c.offsets[0] = 0;
      for (uint64_t i = 0; i < n; i++) {
        if (lens[i] < 0 || (size_t)lens[i] > len ||
            (int64_t)c.offsets[i] + lens[i] > (int64_t)INT32_MAX) {
          *err = "serde: bad binary length";  // negative/overflowing offsets
          return false;                       // would walk out of bounds
        }
        c.offsets[i + 1] = c.offsets[i] + lens[i];
      }
      size_t db = (size_t)c.offsets[n];
      if (pos + db > len) {
        *err = "serde: truncated binary data";
        return false;
      }
      c.values.assign(p + pos, p + pos + db);
      pos += db;
    }
    cols->push_back(std::move(c));
  }
  *used = pos;
  return true;
}

bool ipc_decode_blocks(const uint8_t* p, size_t len,
                       std::vector<uint8_t>* payload, std::string* err,
                       int codec) {
  size_t pos = 0;
  while (pos + 4 <= len) {
    uint32_t block_len;
    memcpy(&block_len, p + pos, 4);
    pos += 4;
    if (pos + block_len > len) {
      *err = "ipc: truncated block";
      return false;
    }
    bool ok = codec == 1
                  ? zstd_decompress_frame(p + pos, block_len, payload, err)
                  : lz4_decompress_frame(p + pos, block_len, payload, err);
    if (!ok) return false;
    pos += block_len;
  }
  if (pos != len) {  // 1-3 trailing bytes = a truncated length prefix
    *err = "ipc: truncated block length prefix";
    return false;
  }
  return true;
}

bool IpcBlockWriter::write_payload(const uint8_t* p, size_t len,
                                   std::string* err) {
  staged_.insert(staged_.end(), p, p + len);
  // ipc_compression.rs:72-79: flush at 0.9 × target
  if ((double)staged_.size() >= (double)target_ * 0.9) return finish_block(err);
  return true;
}

bool IpcBlockWriter::finish_block(std::string* err) {
  if (staged_.empty()) return true;
  size_t base = out_.size();
  out_.resize(base + 4);
  bool ok = codec_ == 1
                ? zstd_compress_frame(staged_.data(), staged_.size(),
                                      zstd_level_, &out_, err)
                : lz4_compress_frame(staged_.data(), staged_.size(), &out_,
                                     err);
  if (!ok) return false;
  uint32_t block_len = (uint32_t)(out_.size() - base - 4);
  memcpy(out_.data() + base, &block_len, 4);  // u32-LE, ipc_compression.rs:87-92
  staged_.clear();
  return true;
}

bool write_shuffle_files(const std::vector<HostCol>& sorted_cols,
                         const std::vector<int64_t>& part_offsets,
                         int64_t batch_size, const std::string& data_file,
                         const std::string& index_file, std::string* err,
                         int codec, int zstd_level) {
  size_t P = part_offsets.size() - 1;
  FILE* df = fopen(data_file.c_str(), "wb");
  if (!df) {
    *err = "cannot open " + data_file;
    return false;
  }
  std::vector<uint64_t> index(P + 1, 0);
  uint64_t pos = 0;
  std::vector<uint8_t> payload;
  for (size_t p = 0; p < P; p++) {
    index[p] = pos;
    int64_t beg = part_offsets[p], end = part_offsets[p + 1];
    if (beg == end) continue;
    IpcBlockWriter w(4194304, codec, zstd_level);
    for (int64_t b = beg; b < end; b += batch_size) {
      int64_t e = b + batch_size < end ? b + batch_size : end;
      payload.clear();
      serde_write_batch(sorted_cols, b, e, &payload);
      if (!w.write_payload(payload.data(), payload.size(), err)) {
        fclose(df);
        return false;
      }
    }
    if (!w.finish_block(err)) {  // partition boundary: whole blocks
      fclose(df);
      return false;
    }
    const std::vector<uint8_t>& bytes = w.bytes();
    if (!bytes.empty() && fwrite(bytes.data(), 1, bytes.size(), df) != bytes.size()) {
      *err = "short write to " + data_file;
      fclose(df);
      return false;
    }
    pos += bytes.size();
  }
  index[P] = pos;
  fclose(df);

  FILE* xf = fopen(index_file.c_str(), "wb");
  if (!xf) {
    *err = "cannot open " + index_file;
    return false;
  }
  // (P+1) × u64-LE absolute offsets (AuronShuffleWriterBase.scala:48-99)
  if (fwrite(index.data(), 8, P + 1, xf) != P + 1) {
    *err = "short write to " + index_file;
    fclose(xf);
    return false;
  }
  fclose(xf);
  return true;
}

}  // namespace auron
