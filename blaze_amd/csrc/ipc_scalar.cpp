// Minimal flatbuffers walker for Arrow IPC metadata (format/Message.fbs,
// format/Schema.fbs) — just enough to read the reference's ScalarValue
// literals: a stream of [Schema message][RecordBatch message] with one row
// and one primitive/utf8 column.
#include "ipc_scalar.h"

#include <cstring>

namespace auron {
namespace {

struct FB {
  const uint8_t* buf;
  size_t len;

  bool ok(size_t pos, size_t n) const { return pos + n <= len; }
  uint8_t u8(size_t p) const { return buf[p]; }
  uint16_t u16(size_t p) const {
    uint16_t v;
    memcpy(&v, buf + p, 2);
    return v;
  }
  uint32_t u32(size_t p) const {
    uint32_t v;
    memcpy(&v, buf + p, 4);
    return v;
  }
  int32_t i32(size_t p) const {
    int32_t v;
    memcpy(&v, buf + p, 4);
    return v;
  }
  int64_t i64(size_t p) const {
    int64_t v;
    memcpy(&v, buf + p, 8);
    return v;
  }

  // field position in table at tpos for flatbuffer field id; 0 = missing
  size_t field(size_t tpos, int id) const {
    if (!ok(tpos, 4)) return 0;
    int32_t soff = i32(tpos);
    size_t vt = tpos - (size_t)soff;  // soffset points back to vtable
    if (vt >= len) return 0;
    uint16_t vt_len = u16(vt);
    size_t slot = 4 + 2 * (size_t)id;
    if (slot + 2 > vt_len) return 0;
    uint16_t off = u16(vt + slot);
    if (off == 0) return 0;
    return tpos + off;
  }

  // dereference an offset field (table/string/vector)
  size_t indirect(size_t pos) const {
    if (!ok(pos, 4)) return 0;
    return pos + u32(pos);
  }
};

constexpr uint32_t CONTINUATION = 0xFFFFFFFFu;

struct Msg {
  int header_type = 0;   // MessageHeader union: Schema=1, RecordBatch=3
  size_t header_pos = 0; // table position in metadata flatbuffer
  FB meta{nullptr, 0};
  const uint8_t* body = nullptr;
  size_t body_len = 0;
};

// read one encapsulated message starting at *pos; returns false at end
bool next_message(const uint8_t* data, size_t len, size_t* pos, Msg* m,
                  std::string* err) {
  size_t p = *pos;
  if (p + 4 > len) return false;
  uint32_t first;
  memcpy(&first, data + p, 4);
  uint32_t meta_len;
  if (first == CONTINUATION) {
    if (p + 8 > len) return false;
    memcpy(&meta_len, data + p + 4, 4);
    p += 8;
  } else {
    meta_len = first;
    p += 4;
  }
  if (meta_len == 0) return false;  // end-of-stream marker
  if (p + meta_len > len) {
    *err = "truncated IPC metadata";
    return false;
  }
  FB fb{data + p, meta_len};
  size_t root = fb.indirect(0);
  // Message fields: version=0, header_type=1, header=2, bodyLength=3
  size_t ht = fb.field(root, 1);
  m->header_type = ht ? fb.u8(ht) : 0;
  size_t hdr = fb.field(root, 2);
  m->header_pos = hdr ? fb.indirect(hdr) : 0;
  size_t bl = fb.field(root, 3);
  int64_t body_len = bl ? fb.i64(bl) : 0;
  m->meta = fb;
  p += meta_len;
  // body is 8-byte aligned already in the stream (metadata padded)
  if ((size_t)body_len > 0) {
    if (p + (size_t)body_len > len) {
      *err = "truncated IPC body";
      return false;
    }
    m->body = data + p;
    m->body_len = (size_t)body_len;
    p += (size_t)body_len;
  } else {
    m->body = nullptr;
    m->body_len = 0;
  }
  *pos = p;
  return true;
}

// Schema.fbs Type union members we support
enum ArrowFbType {
  FB_Null = 1,
  FB_Int = 2,
  FB_Float = 3,
  FB_Binary = 4,
  FB_Utf8 = 5,
  FB_Bool = 6,
  FB_Date = 8,
};

}  // namespace

bool decode_ipc_scalar(const uint8_t* data, size_t len, ScalarLit* out,
                       std::string* err) {
  size_t pos = 0;
  Msg schema_msg, batch_msg;
  bool have_schema = false, have_batch = false;
  Msg m;
  while (next_message(data, len, &pos, &m, err)) {
    if (m.header_type == 1) {
      schema_msg = m;
      have_schema = true;
    } else if (m.header_type == 3) {
      batch_msg = m;
      have_batch = true;
      break;
    }
  }
  if (!err->empty()) return false;
  if (!have_schema || !have_batch) {
    *err = "IPC scalar: missing schema or record batch";
    return false;
  }

  // ---- schema: first field's type ----
  FB& sf = schema_msg.meta;
  // Schema fields: endianness=0, fields=1
  size_t fields_off = sf.field(schema_msg.header_pos, 1);
  if (!fields_off) {
    *err = "IPC scalar: schema without fields";
    return false;
  }
  size_t fields_vec = sf.indirect(fields_off);
  uint32_t nfields = sf.u32(fields_vec);
  if (nfields != 1) {
    *err = "IPC scalar: expected single-column batch";
    return false;
  }
  size_t field_tab = sf.indirect(fields_vec + 4);
  // Field: name=0, nullable=1, type_type=2, type=3
  size_t tt = sf.field(field_tab, 2);
  int type_type = tt ? sf.u8(tt) : 0;
  size_t type_off = sf.field(field_tab, 3);
  size_t type_tab = type_off ? sf.indirect(type_off) : 0;

  int bit_width = 0, is_signed = 1, fp_precision = 2;
  switch (type_type) {
    case FB_Int: {
      size_t bw = sf.field(type_tab, 0);
      bit_width = bw ? sf.i32(bw) : 32;
      size_t sg = sf.field(type_tab, 1);
      is_signed = sg ? sf.u8(sg) : 0;
      out->dtype = bit_width == 64 ? (is_signed ? DType::Int64 : DType::UInt64)
                   : bit_width == 32 ? (is_signed ? DType::Int32 : DType::UInt32)
                   : bit_width == 16 ? (is_signed ? DType::Int16 : DType::UInt16)
                                     : (is_signed ? DType::Int8 : DType::UInt8);
      break;
    }
    case FB_Float: {
      size_t pr = sf.field(type_tab, 0);
      fp_precision = pr ? sf.u16(pr) : 2;
      out->dtype = fp_precision == 2 ? DType::Float64
                   : fp_precision == 1 ? DType::Float32
                                       : DType::Float16;
      break;
    }
    case FB_Utf8: out->dtype = DType::Utf8; break;
    case FB_Binary: out->dtype = DType::Binary; break;
    case FB_Bool: out->dtype = DType::Bool; break;
    case FB_Null: out->dtype = DType::Null; break;
    default:
      *err = "IPC scalar: unsupported arrow type " + std::to_string(type_type);
      return false;
  }

  // ---- record batch: nodes + buffers ----
  FB& bf = batch_msg.meta;
  // RecordBatch: length=0, nodes=1 (structs 16B), buffers=2 (structs 16B)
  size_t len_off = bf.field(batch_msg.header_pos, 0);
  int64_t nrows = len_off ? bf.i64(len_off) : 0;
  if (nrows != 1) {
    *err = "IPC scalar: expected 1-row batch";
    return false;
  }
  size_t nodes_off = bf.field(batch_msg.header_pos, 1);
  size_t nodes_vec = nodes_off ? bf.indirect(nodes_off) : 0;
  int64_t null_count = 0;
  if (nodes_vec && bf.u32(nodes_vec) >= 1)
    null_count = bf.i64(nodes_vec + 4 + 8);  // FieldNode{length, null_count}
  size_t bufs_off = bf.field(batch_msg.header_pos, 2);
  if (!bufs_off) {
    *err = "IPC scalar: batch without buffers";
    return false;
  }
  size_t bufs_vec = bf.indirect(bufs_off);
  uint32_t nbufs = bf.u32(bufs_vec);
  auto buffer_at = [&](uint32_t i, int64_t* off, int64_t* blen) {
    size_t p = bufs_vec + 4 + 16 * (size_t)i;  // Buffer{offset, length}
    *off = bf.i64(p);
    *blen = bf.i64(p + 8);
  };

  out->is_null = null_count > 0 || out->dtype == DType::Null;
  if (out->is_null) return true;

  const uint8_t* body = batch_msg.body;
  if (out->dtype == DType::Utf8 || out->dtype == DType::Binary) {
    if (nbufs < 3) {
      *err = "IPC scalar: utf8 needs 3 buffers";
      return false;
    }
    int64_t ooff, olen, doff, dlen;
    buffer_at(1, &ooff, &olen);
    buffer_at(2, &doff, &dlen);
    int32_t beg, end;
    memcpy(&beg, body + ooff, 4);
    memcpy(&end, body + ooff + 4, 4);
    out->utf8.assign((const char*)body + doff + beg, (size_t)(end - beg));
  } else {
    if (nbufs < 2) {
      *err = "IPC scalar: primitive needs 2 buffers";
      return false;
    }
    int64_t doff, dlen;
    buffer_at(1, &doff, &dlen);
    switch (out->dtype) {
      case DType::Int8: out->i64 = (int8_t)body[doff]; break;
      case DType::UInt8: out->i64 = body[doff]; break;
      case DType::Int16: { int16_t v; memcpy(&v, body + doff, 2); out->i64 = v; break; }
      case DType::UInt16: { uint16_t v; memcpy(&v, body + doff, 2); out->i64 = v; break; }
      case DType::Int32: { int32_t v; memcpy(&v, body + doff, 4); out->i64 = v; break; }
      case DType::UInt32: { uint32_t v; memcpy(&v, body + doff, 4); out->i64 = v; break; }
      case DType::Int64: case DType::UInt64: {
        int64_t v; memcpy(&v, body + doff, 8); out->i64 = v; break; }
      case DType::Float32: { float v; memcpy(&v, body + doff, 4); out->f64 = v; break; }
      case DType::Float64: { double v; memcpy(&v, body + doff, 8); out->f64 = v; break; }
      case DType::Bool: out->i64 = (body[doff] & 1); break;
      default:
        *err = "IPC scalar: unsupported primitive";
        return false;
    }
  }
  return true;
}

}  // namespace auron
