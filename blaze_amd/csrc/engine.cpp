// engine.cpp — the MI355X-native Auron hot-path runtime behind the C ABI of
// include/auron_hip.h. Mirrors the reference runtime semantics:
//   NativeExecutionRuntime (auron/src/rt.rs:71-259): decode TaskDefinition →
//   operator tree → pull-model batch pump → Arrow FFI export per nextBatch.
// Operators implemented (the SURVEY.md §8 hot-path subset):
//   FFIReaderExec   (ffi_reader_exec.rs)        — input via callback
//   AggExec         (agg_exec.rs:141-278, agg_table.rs, agg_ctx.rs) — GPU
//   ShuffleWriterExec (shuffle_writer_exec.rs, sort_repartitioner.rs,
//                      buffered_data.rs) — GPU partition + host file emit
// The GPU path is mandatory: any HIP failure aborts the task loudly; there is
// no CPU fallback anywhere in this library.
#include <chrono>
#include <future>
#include <thread>
#include <cstring>
#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "../../include/auron_hip.h"
#include "dev.h"
#include "kernels.h"
#include "plan.h"
#include "ipc_scalar.h"
#include "parquet.h"
#include "serde_host.h"

namespace auron {
namespace {

struct EngineError : std::runtime_error {
  using std::runtime_error::runtime_error;
};

// AURON_DEBUG=1 traces engine stages (with ms timestamps) to stderr
static bool debug_on() {
  static bool v = getenv("AURON_DEBUG") != nullptr;
  return v;
}
static double dbg_ms() {
  static auto t0 = std::chrono::steady_clock::now();
  return std::chrono::duration<double, std::milli>(
             std::chrono::steady_clock::now() - t0)
      .count();
}
#define DBG(...)                                          \
  do {                                                    \
    if (debug_on()) {                                     \
      fprintf(stderr, "[auron %9.2f] ", dbg_ms());        \
      fprintf(stderr, __VA_ARGS__);                       \
      fprintf(stderr, "\n");                              \
      fflush(stderr);                                     \
    }                                                     \
  } while (0)

#define FAIL(msg) throw EngineError(msg)

// ---------------------------------------------------------------- columns --
struct DevColumn {
  DType dt = DType::Unsupported;
  int64_t len = 0;
  // non-owning views (point into own_* when owned)
  const void* values = nullptr;       // prim: len*w; binary: data bytes
  const uint8_t* validity = nullptr;  // LSB bitmap or null
  const int32_t* offsets = nullptr;   // binary: len+1
  int64_t data_len = 0;               // binary data bytes
  DevBuf own_values, own_validity, own_offsets;
};

struct DevBatch {
  int64_t num_rows = 0;
  std::vector<DevColumn> cols;
};

const char* dtype_format(DType t) {
  switch (t) {
    case DType::Int8: return "c";
    case DType::Int16: return "s";
    case DType::Int32: return "i";
    case DType::Int64: return "l";
    case DType::UInt8: return "C";
    case DType::UInt16: return "S";
    case DType::UInt32: return "I";
    case DType::UInt64: return "L";
    case DType::Float32: return "f";
    case DType::Float64: return "g";
    case DType::Utf8: return "u";
    case DType::Binary: return "z";
    default: return nullptr;
  }
}

DType format_dtype(const char* f) {
  if (!f || !f[0] || f[1]) return DType::Unsupported;
  switch (f[0]) {
    case 'c': return DType::Int8;
    case 's': return DType::Int16;
    case 'i': return DType::Int32;
    case 'l': return DType::Int64;
    case 'C': return DType::UInt8;
    case 'S': return DType::UInt16;
    case 'I': return DType::UInt32;
    case 'L': return DType::UInt64;
    case 'f': return DType::Float32;
    case 'g': return DType::Float64;
    case 'u': return DType::Utf8;
    case 'z': return DType::Binary;
    default: return DType::Unsupported;
  }
}

// ------------------------------------------------------------ arrow import --
DevColumn import_child(const ArrowArray* a, DType dt, bool device,
                       hipStream_t stream) {
  DevColumn c;
  c.dt = dt;
  c.len = a->length;
  if (a->offset != 0) FAIL("ArrowArray with nonzero offset unsupported");
  size_t w = dtype_width(dt);
  const void* validity_src = a->n_buffers > 0 ? a->buffers[0] : nullptr;
  if (dt == DType::Binary || dt == DType::Utf8) {
    if (a->n_buffers < 3) FAIL("binary array needs 3 buffers");
    const int32_t* off_src = (const int32_t*)a->buffers[1];
    const void* data_src = a->buffers[2];
    if (device) {
      c.offsets = off_src;
      int32_t dl = 0;
      AURON_HIP(hipMemcpyAsync(&dl, off_src + a->length, 4, hipMemcpyDeviceToHost,
                               stream));
      AURON_HIP(hipStreamSynchronize(stream));
      c.data_len = dl;
      c.values = data_src;
    } else {
      int32_t dl = off_src[a->length];
      c.data_len = dl;
      c.own_offsets.alloc((a->length + 1) * 4);
      PinnedUploader::inst().copy(c.own_offsets.get(), off_src,
                                  (a->length + 1) * 4, stream);
      c.offsets = c.own_offsets.get<int32_t>();
      if (dl > 0) {
        c.own_values.alloc(dl);
        PinnedUploader::inst().copy(c.own_values.get(), data_src, dl, stream);
      }
      c.values = c.own_values.get();
    }
  } else {
    if (w == 0) FAIL("unsupported input dtype");
    const void* val_src = a->n_buffers > 1 ? a->buffers[1] : nullptr;
    if (device) {
      c.values = val_src;
    } else {
      c.own_values.alloc(a->length * w);
      PinnedUploader::inst().copy(c.own_values.get(), val_src,
                                  (size_t)a->length * w, stream);
      c.values = c.own_values.get();
    }
  }
  if (validity_src && a->null_count != 0) {
    size_t bl = (a->length + 7) / 8;
    if (device) {
      c.validity = (const uint8_t*)validity_src;
    } else {
      c.own_validity.alloc(bl);
      AURON_HIP(hipMemcpyAsync(c.own_validity.get(), validity_src, bl,
                               hipMemcpyHostToDevice, stream));
      c.validity = c.own_validity.get<uint8_t>();
    }
  }
  return c;
}

// Import a top-level struct batch (rt.rs:232-241 exports batches as
// StructArray). `schema_hint` gives dtypes when the caller passed no schema.
DevBatch import_batch(const ArrowArray* a, const ArrowSchema* schema,
                      const Schema& schema_hint, bool device,
                      hipStream_t stream) {
  DevBatch b;
  b.num_rows = a->length;
  for (int64_t i = 0; i < a->n_children; i++) {
    DType dt;
    if (schema && schema->n_children == a->n_children) {
      dt = format_dtype(schema->children[i]->format);
    } else if ((size_t)i < schema_hint.fields.size()) {
      dt = schema_hint.fields[i].dtype;
    } else {
      FAIL("cannot determine input column dtype");
    }
    b.cols.push_back(import_child(a->children[i], dt, device, stream));
  }
  if (!device) {
    // host buffers may be freed by the caller after return — drain the H2D
    // copies before handing the borrowed pointers back
    AURON_HIP(hipStreamSynchronize(stream));
  }
  return b;
}

// ------------------------------------------------------------ arrow export --
struct ExportPriv {
  std::vector<void*> host_bufs;
  std::vector<const void*> buffer_ptrs;
  std::vector<ArrowArray> children_store;
  std::vector<ArrowArray*> children_ptrs;
  std::vector<std::vector<const void*>> child_buffer_ptrs;
  // list columns: one item array per column slot
  std::vector<ArrowArray> item_store;
  std::vector<ArrowArray*> item_ptrs;
  std::vector<std::vector<const void*>> item_buffer_ptrs;
};

void release_exported_array(ArrowArray* a) {
  if (!a || !a->release) return;
  ExportPriv* p = (ExportPriv*)a->private_data;
  for (void* b : p->host_bufs) free(b);
  delete p;
  a->release = nullptr;
}

struct SchemaPriv {
  std::vector<ArrowSchema> children_store;
  std::vector<ArrowSchema*> children_ptrs;
  std::vector<std::string> names;
  std::vector<const char*> formats;
  // list fields: one item-schema per field slot (null when not a list)
  std::vector<ArrowSchema> item_store;
  std::vector<ArrowSchema*> item_ptrs;
};

void release_exported_schema(ArrowSchema* s) {
  if (!s || !s->release) return;
  delete (SchemaPriv*)s->private_data;
  s->release = nullptr;
}

struct OutField {
  std::string name;
  DType dt;        // list fields: the ITEM type
  bool nullable;
  bool is_list = false;  // Arrow list<dt> (COLLECT output)
};

void export_schema(const std::vector<OutField>& fields, ArrowSchema* out) {
  auto* priv = new SchemaPriv;
  priv->children_store.resize(fields.size());
  priv->item_store.resize(fields.size());
  priv->item_ptrs.resize(fields.size());
  for (size_t i = 0; i < fields.size(); i++) {
    priv->names.push_back(fields[i].name);
  }
  for (size_t i = 0; i < fields.size(); i++) {
    ArrowSchema& c = priv->children_store[i];
    memset(&c, 0, sizeof(c));
    c.name = priv->names[i].c_str();
    c.flags = fields[i].nullable ? ARROW_FLAG_NULLABLE : 0;
    c.release = [](ArrowSchema* s) { s->release = nullptr; };
    if (fields[i].is_list) {
      c.format = "+l";
      ArrowSchema& it = priv->item_store[i];
      memset(&it, 0, sizeof(it));
      it.format = dtype_format(fields[i].dt);
      it.name = "item";
      it.release = [](ArrowSchema* s) { s->release = nullptr; };
      priv->item_ptrs[i] = &it;
      c.n_children = 1;
      c.children = &priv->item_ptrs[i];
    } else {
      c.format = dtype_format(fields[i].dt);
    }
    priv->children_ptrs.push_back(&c);
  }
  memset(out, 0, sizeof(*out));
  out->format = "+s";
  out->name = "";
  out->n_children = (int64_t)fields.size();
  out->children = priv->children_ptrs.data();
  out->release = release_exported_schema;
  out->private_data = priv;
}

// host-side column staging for export
struct HostOutCol {
  DType dt;                       // list cols: the ITEM type
  std::vector<uint8_t> values;    // list cols: child (item) values
  std::vector<uint8_t> validity;  // empty = no nulls
  std::vector<int32_t> offsets;   // binary / list: n+1
  bool is_list = false;
};

// device-resident column staging: partial-agg output that stays in HBM and
// is exported through import_device_batch (buffers owned by the runtime,
// valid until auron_finalize — the documented contract for this harness).
struct DevOutCol {
  DType dt;
  DevBuf values;    // prim values / binary data bytes
  DevBuf offsets;   // binary only: (n+1) int32
  DevBuf validity;  // LSB bitmap or empty
  int64_t data_len = 0;
};


void export_batch(int64_t num_rows, std::vector<HostOutCol>&& cols,
                  ArrowArray* out) {
  auto* priv = new ExportPriv;
  priv->children_store.resize(cols.size());
  priv->child_buffer_ptrs.resize(cols.size());
  priv->item_store.resize(cols.size());
  priv->item_ptrs.resize(cols.size());
  priv->item_buffer_ptrs.resize(cols.size());
  for (size_t i = 0; i < cols.size(); i++) {
    HostOutCol& c = cols[i];
    ArrowArray& ch = priv->children_store[i];
    memset(&ch, 0, sizeof(ch));
    ch.length = num_rows;
    ch.offset = 0;
    auto copy_out = [&](const void* src, size_t len) -> void* {
      void* b = malloc(len ? len : 1);
      memcpy(b, src, len);
      priv->host_bufs.push_back(b);
      return b;
    };
    std::vector<const void*>& bufs = priv->child_buffer_ptrs[i];
    const void* validity = nullptr;
    if (!c.validity.empty()) {
      validity = copy_out(c.validity.data(), c.validity.size());
      int64_t nulls = 0;
      for (int64_t r = 0; r < num_rows; r++)
        if (!((c.validity[r >> 3] >> (r & 7)) & 1)) nulls++;
      ch.null_count = nulls;
    } else {
      ch.null_count = 0;
    }
    if (c.is_list) {
      // list<prim>: parent {validity, offsets}; child = item values
      bufs = {validity, copy_out(c.offsets.data(), c.offsets.size() * 4)};
      ArrowArray& it = priv->item_store[i];
      memset(&it, 0, sizeof(it));
      it.length = c.offsets.empty() ? 0 : c.offsets.back();
      it.null_count = 0;
      std::vector<const void*>& ibufs = priv->item_buffer_ptrs[i];
      ibufs = {nullptr, copy_out(c.values.data(), c.values.size())};
      it.n_buffers = 2;
      it.buffers = ibufs.data();
      it.release = [](ArrowArray* a) { a->release = nullptr; };
      priv->item_ptrs[i] = &it;
      ch.n_children = 1;
      ch.children = &priv->item_ptrs[i];
    } else if (c.dt == DType::Binary || c.dt == DType::Utf8) {
      bufs = {validity, copy_out(c.offsets.data(), c.offsets.size() * 4),
              copy_out(c.values.data(), c.values.size())};
    } else {
      bufs = {validity, copy_out(c.values.data(), c.values.size())};
    }
    ch.n_buffers = (int64_t)bufs.size();
    ch.buffers = bufs.data();
    ch.release = [](ArrowArray* a) { a->release = nullptr; };
    priv->children_ptrs.push_back(&ch);
  }
  memset(out, 0, sizeof(*out));
  out->length = num_rows;
  out->null_count = 0;
  out->n_buffers = 1;
  priv->buffer_ptrs = {nullptr};
  out->buffers = priv->buffer_ptrs.data();
  out->n_children = (int64_t)cols.size();
  out->children = priv->children_ptrs.data();
  out->release = release_exported_array;
  out->private_data = priv;
}

// ------------------------------------------------------------------- conf --
struct Conf {
  AuronCallbacks* cb;
  std::string get(const char* key, const std::string& dflt) const {
    char buf[256];
    buf[0] = '\0';
    if (cb && cb->get_conf &&
        cb->get_conf(cb->user, key, buf, sizeof(buf)) == 0 && buf[0] != '\0')
      return std::string(buf);
    return dflt;
  }
  int64_t get_i(const char* key, int64_t dflt) const {
    std::string v = get(key, "");
    return v.empty() ? dflt : (int64_t)strtoll(v.c_str(), nullptr, 10);
  }
  double get_d(const char* key, double dflt) const {
    std::string v = get(key, "");
    return v.empty() ? dflt : strtod(v.c_str(), nullptr);
  }
};

// ------------------------------------------------------------------ AggOp --
// GPU AggExec: HashAgg over one Int64 grouping column with the north-star agg
// set [SUM(Float64), COUNT] (agg_exec.rs:141-278 semantics; wider agg/type
// coverage is tracked in DESIGN.md §scope).
class AggOp {
 public:
  AggOp(const AggNode& node, const Conf& conf, hipStream_t stream)
      : stream_(stream) {
    if (node.exec_mode != 0) FAIL("SORT_AGG unsupported (hot path is HASH_AGG)");
    if (node.grouping_exprs.empty() ||
        node.grouping_exprs.size() > (size_t)GKEY_MAX_COLS)
      FAIL("AggExec: 1..4 Column grouping exprs supported");
    for (size_t gi = 0; gi < node.grouping_exprs.size(); gi++) {
      const Expr& ge = node.grouping_exprs[gi];
      if (ge.kind != Expr::Column)
        FAIL("AggExec: grouping exprs must be Columns");
      key_cols_.push_back(ge.col_index);
      key_names_.push_back(gi < node.grouping_names.size()
                               ? node.grouping_names[gi]
                               : "key" + std::to_string(gi));
    }
    key_col_ = key_cols_[0];
    key_name_ = key_names_[0];
    // agg set: any list of SUM/COUNT/AVG/MIN/MAX over ONE shared argument
    // column — {sum, cnt} accumulators (sum.rs, count.rs, avg.rs; AVG freeze
    // = sum ++ count, avg.rs:208-217) plus the optional side {min, max} pair
    // (maxmin.rs:104-119; single-phase path only). Anything else fails
    // loudly at plan build.
    if (node.agg_exprs.empty() ||
        node.agg_exprs.size() > (size_t)MA_MAX_AGGS)
      FAIL("AggExec: 1..12 aggregates supported");
    if (node.modes.size() != node.agg_exprs.size())
      FAIL("AggExec: modes/aggs length mismatch");
    for (const AggMode m : node.modes)
      if (m != node.modes[0]) FAIL("AggExec: mixed agg modes unsupported");
    mode_ = node.modes[0];
    merge_mode_ = (mode_ != AggMode::Partial);
    final_output_ = (mode_ == AggMode::Final);
    layout_ = 0;
    bool have_col = false, have_distinct_cols = false, have_null_lit = false;
    int fam_sum = 0, fam_cnt = 0, fam_min = 0, fam_max = 0, fam_first = 0,
        fam_firstin = 0, npools = 0;
    bool any_i32 = false, mixed_acc = false;
    for (size_t i = 0; i < node.agg_exprs.size(); i++) {
      const Expr& a = node.agg_exprs[i];
      uint32_t k;
      switch (a.agg_function) {
        case AGG_SUM: k = AGGL_SUM; fam_sum++; break;
        case AGG_COUNT: k = AGGL_CNT; fam_cnt++; break;
        case AGG_AVG: k = AGGL_AVG; fam_sum++; fam_cnt++; break;
        case AGG_MIN: k = AGGL_MIN; fam_min++; has_mm_ = true; break;
        case AGG_MAX: k = AGGL_MAX; fam_max++; has_mm_ = true; break;
        case AGG_FIRST: k = AGGL_FIRST; fam_first++; has_first_ = true; break;
        case AGG_FIRST_IGNORES_NULL:
          k = AGGL_FIRSTIN;
          fam_firstin++;
          has_first_ = true;
          break;
        case AGG_COLLECT_LIST:
          k = AGGL_CLIST;
          npools++;
          has_coll_ = true;
          break;
        case AGG_COLLECT_SET:
          k = AGGL_CSET;
          npools++;
          has_coll_ = true;
          coll_set_ = true;
          break;
        default:
          FAIL("AggExec: only SUM/COUNT/AVG/MIN/MAX/FIRST[_IGNORES_NULL]/"
               "COLLECT_LIST aggregates on this path");
      }
      if (i < 8) layout_ |= k << (4 * i);
      // accumulator arithmetic type follows the agg's declared data type
      // (sum.rs:78-88 casts inputs to it; maxmin.rs:81-83 preserves it)
      MaAgg ma;
      ma.kind = (uint8_t)k;
      if (k != AGGL_CNT) {
        bool is_int = (a.return_type == DType::Int64);
        bool is_i32 = (a.return_type == DType::Int32);
        ma.acc_t = is_i32 ? 2 : (is_int ? 1 : 0);
        any_i32 |= is_i32;
        if (!val_typed_seen_) {
          val_is_int_ = is_int;
          val_typed_seen_ = true;
        } else if (val_is_int_ != (is_int || is_i32)) {
          mixed_acc = true;
        }
        if (k == AGGL_AVG && (is_int || is_i32))
          FAIL("AggExec: AVG with integer output unsupported (avg.rs "
               "declares a floating result)");
      }
      ma_desc_.a[ma_desc_.n++] = ma;
      agg_kinds_.push_back(k);
      agg_names_.push_back(i < node.agg_names.size() ? node.agg_names[i]
                                                     : "agg" + std::to_string(i));
      ma_arg_cols_.push_back(-1);
      if (!merge_mode_) {
        const Expr& child = a.children.at(0);
        if (child.kind == Expr::Literal) {
          ScalarLit sl;
          std::string e2;
          if (!decode_ipc_scalar(child.literal_ipc.data(),
                                 child.literal_ipc.size(), &sl, &e2) ||
              !sl.is_null)
            FAIL("aggregate Literal arg must be NULL on this path");
          have_null_lit = true;  // collect.rs-style empty behavior
        } else if (child.kind == Expr::Column) {
          ma_arg_cols_.back() = (int)child.col_index;
          if (!have_col) {
            val_col_ = child.col_index;
            have_col = true;
          } else if (child.col_index != val_col_) {
            have_distinct_cols = true;
          }
        } else {
          FAIL("aggregate arg must be a Column or NULL literal");
        }
      }
    }
    // Multi-argument mode (agg.rs:73-169: independent arg expressions per
    // agg; per-agg accumulator columns at the declared types). The trigger
    // rule is computable from the PLAN alone, so a Partial task and its
    // Final consumer always agree on the a8 wire layout: family duplicates
    // make the legacy shared-accumulator wire ambiguous; Int32/mixed
    // accumulator types and multiple collect pools are not representable in
    // it at all. Distinct arg columns / NULL-literal args additionally
    // trigger ma on the PARTIAL side (the wire stays legacy-parseable when
    // families do not repeat, so the Final side may run either mode).
    ma_mode_ = fam_sum > 1 || fam_cnt > 1 || fam_min > 1 || fam_max > 1 ||
               fam_first > 1 || fam_firstin > 1 || npools > 1 || any_i32 ||
               mixed_acc ||
               (!merge_mode_ && (have_distinct_cols || have_null_lit));
    if (!ma_mode_ && node.agg_exprs.size() > 8)
      FAIL("AggExec: >8 aggregates require the multi-arg layout");
    if (ma_mode_) {
      if (npools > MA_MAX_POOLS)
        FAIL("AggExec: at most 4 COLLECT aggregates");
      int pool = 0;
      for (int j = 0; j < ma_desc_.n; j++)
        if (ma_desc_.a[j].kind == AGGL_CLIST ||
            ma_desc_.a[j].kind == AGGL_CSET)
          ma_desc_.a[j].pool = (uint8_t)pool++;
      // the legacy side-array machinery is replaced by the bank
      has_mm_ = has_first_ = has_coll_ = false;
      skip_enabled_ = false;
    } else {
      bool has_l = false, has_s = false;
      for (uint32_t kk : agg_kinds_) {
        has_l |= (kk == AGGL_CLIST);
        has_s |= (kk == AGGL_CSET);
      }
      if (has_l && has_s)
        FAIL("COLLECT_LIST and COLLECT_SET together share one argument pool "
             "— unsupported on this path (single-pool legacy layout)");
      if (!merge_mode_ && !have_col && !node.agg_exprs.empty()) {
        bool only_counts = true;
        for (uint32_t kk : agg_kinds_) only_counts &= (kk == AGGL_CNT);
        if (!only_counts) FAIL("aggregate arg must be a Column");
      }
    }
    batch_size_ = conf.get_i("BATCH_SIZE", 10000);
    if (batch_size_ <= 0) FAIL("invalid BATCH_SIZE conf");
    skip_enabled_ =
        node.supports_partial_skipping && !merge_mode_ && !ma_mode_;
    skip_ratio_ = conf.get_d("PARTIAL_AGG_SKIPPING_RATIO", 0.999);
    skip_min_rows_ = conf.get_i("PARTIAL_AGG_SKIPPING_MIN_ROWS", 20000);
    // device output: partial freeze output stays in HBM (ArrowDeviceArray
    // export) so the bench/exchange chain never round-trips through host
    device_out_ = conf.get_i("AURON_HIP_DEVICE_OUTPUT", 0) != 0 &&
                  !final_output_;
    int64_t slots = conf.get_i("AURON_HIP_AGG_TABLE_SLOTS", 1 << 23);
    conf_coll_cap_ =
        std::max<int64_t>(1024, conf.get_i("AURON_HIP_COLLECT_POOL", 16 << 20));
    // VRAM budget for the slot table (a9 analog of MemManager's budget,
    // memmgr/mod.rs:36-105): exceeding it spills frozen records to host
    // buckets instead of growing (agg_table.rs:540-588 semantics).
    mem_budget_ = conf.get_i("AURON_HIP_MEM_BUDGET", 8LL << 30);
    num_spill_buckets_ =
        (int)std::max<int64_t>(16, conf.get_i("AURON_HIP_SPILL_BUCKETS", 16));
    max_cap_ = 1;
    while (max_cap_ * 2 * (int64_t)sizeof(AggSlot) <= mem_budget_)
      max_cap_ *= 2;
    if (slots > max_cap_) slots = max_cap_;
    init_table(slots);
    t_.sum_int = val_is_int_ ? 1u : 0u;
    AURON_HIP(hipEventCreate(&ev_start_));
    AURON_HIP(hipEventCreate(&ev_stop_));
  }

  ~AggOp() {
    if (s_aux_) {
      (void)hipStreamSynchronize(s_aux_);
      (void)hipStreamDestroy(s_aux_);
      for (int i = 0; i < 2; i++) {
        (void)hipEventDestroy(ev_scatter_[i]);
        (void)hipEventDestroy(ev_merge_[i]);
        (void)hipEventDestroy(ev_read_[i]);
      }
      (void)hipEventDestroy(ev_pipe_[0]);
    }
    for (auto& [e0, e1] : ev_pairs_) {
      (void)hipEventDestroy(e0);
      (void)hipEventDestroy(e1);
    }
    (void)hipEventDestroy(ev_start_);
    (void)hipEventDestroy(ev_stop_);
  }

  void consume(DevBatch&& b) {
    if (b.num_rows == 0) return;
    DevColumn& key = b.cols.at(key_col_);
    if (!key_mode_set_) {
      // key mode: single numeric column keeps the i64 slot fast path
      // (incl. the two-phase pipeline); anything else — Utf8, Float64, or
      // 2..4-column tuples — runs the generalized-key path (a4 closure,
      // agg_ctx.rs:219-231 substitution per SURVEY.md 8c(i))
      key_mode_set_ = true;
      gkey_ = !(key_cols_.size() == 1 &&
                (key.dt == DType::Int64 || key.dt == DType::Int32));
      if (gkey_) {
        if (has_first_ || has_coll_)
          FAIL("FIRST/COLLECT with Utf8/multi-column grouping keys "
               "unsupported on this path");
        skip_enabled_ = false;  // emit_skipped is single-key-column shaped
        for (uint32_t ci : key_cols_) {
          DType dt = b.cols.at(ci).dt;
          if (dt != DType::Int64 && dt != DType::Int32 &&
              dt != DType::Float64 && dt != DType::Utf8 &&
              dt != DType::Binary)
            FAIL("unsupported grouping key dtype");
          key_dts_.push_back(dt);
        }
        init_gkey();
      }
    }
    if (gkey_) {
      for (size_t k = 0; k < key_cols_.size(); k++)
        if (b.cols.at(key_cols_[k]).dt != key_dts_[k])
          FAIL("grouping key dtype changed mid-stream");
    } else {
    if (key.dt != DType::Int64 && key.dt != DType::Int32)
      FAIL("grouping key must be Int64/Int32");
    if (key_dt_ == DType::Unsupported) key_dt_ = key.dt;
    if (key.dt != key_dt_) FAIL("grouping key dtype changed mid-stream");
    if (key.dt == DType::Int32) {
      // widen to the i64 slot table; the narrow Int32 view is restored at
      // output (values round-trip exactly)
      DevBuf wide(b.num_rows * 8);
      launch_widen_i32_i64((const int32_t*)key.values, b.num_rows,
                           wide.get<int64_t>(), stream_);
      key.own_values = std::move(wide);
      key.values = key.own_values.get();
      key.dt = DType::Int64;
    }
    }
    if (ma_mode_ && gkey_)
      FAIL("multi-argument aggregates with Utf8/multi-column grouping keys "
           "unsupported on this path");
    if (ma_mode_ && !merge_mode_ && !ma_args_checked_) {
      ma_args_checked_ = true;
      for (int j = 0; j < ma_desc_.n; j++) {
        if (ma_arg_cols_[j] < 0) {
          ma_desc_.a[j].arg_dt = 3;  // NULL literal
          continue;
        }
        DType dt = b.cols.at(ma_arg_cols_[j]).dt;
        uint8_t code;
        if (dt == DType::Float64) code = 0;
        else if (dt == DType::Int64) code = 1;
        else if (dt == DType::Int32) code = 2;
        else FAIL("unsupported aggregate argument dtype");
        if (ma_desc_.a[j].acc_t != 0 && code == 0)
          FAIL("float argument with integer accumulator unsupported");
        ma_desc_.a[j].arg_dt = code;
      }
    }
    if (!merge_mode_ && !ma_mode_) {
      // sum.rs:78-88 prepare_partial_args: the argument is CAST to the
      // accumulator type before update. Narrower numeric args widen on
      // device; anything else fails loudly.
      DevColumn& val = b.cols.at(val_col_);
      const DType want = val_is_int_ ? DType::Int64 : DType::Float64;
      if (val.dt != want) {
        DevBuf wide(b.num_rows * 8);
        if (val.dt == DType::Int32 && want == DType::Int64)
          launch_widen_i32_i64((const int32_t*)val.values, b.num_rows,
                               wide.get<int64_t>(), stream_);
        else if (val.dt == DType::Int32 && want == DType::Float64)
          launch_cast_i32_f64((const int32_t*)val.values, b.num_rows,
                              wide.get<double>(), stream_);
        else if (val.dt == DType::Int64 && want == DType::Float64)
          launch_cast_i64_f64((const int64_t*)val.values, b.num_rows,
                              wide.get<double>(), stream_);
        else
          FAIL("unsupported agg argument cast");
        val.own_values = std::move(wide);
        val.values = val.own_values.get();
        val.dt = want;
      }
    }
    if (skipping_) {
      skipped_.push_back(std::move(b));
      return;
    }
    // Process in chunks with a HARD no-overflow guarantee: before each chunk,
    // capacity is ensured for the conservative bound ng_bound_ (last readback
    // + every row since being a new key), so a probe can never exhaust the
    // table mid-launch. Readbacks (stream syncs) happen only when the bound
    // nears the 3/4 load threshold — rare once cardinality saturates.
    auto slice_valid = [](const uint8_t* v, int64_t done) {
      return v ? v + done / 8 : nullptr;  // done is a multiple of 8
    };
    int64_t done = 0;
    DBG("agg.consume n=%lld merge=%d", (long long)b.num_rows, (int)merge_mode_);

    while (done < b.num_rows) {
      // two-phase path (update mode, large chunks): its table inserts are
      // bounded by counted staged/leftover lists, not by chunk rows, so it
      // chunks on partition-buffer size instead of table free slots.
      init_agg2_conf();  // chunk bound must be read BEFORE sizing the chunk
      // MIN/MAX agg sets stay single-phase: the LDS bucket kernel's slot
      // holds {key,cnt,sum,first} only (perf note in DESIGN.md)
      if (!gkey_ && !ma_mode_ && !merge_mode_ && !has_mm_ && !has_first_ &&
          !has_coll_ && b.num_rows - done >= AGG2_MIN_CHUNK) {
        init_agg2_conf();
        if (agg2_v3_ && agg2_pipe_ && !skip_enabled_ &&
            b.num_rows - done >= 2 * agg2_chunk_max_) {
          // multi-chunk batch: cross-chunk pipeline — scatter(k+1) on the
          // aux stream overlaps bucket/merge(k) on the engine stream
          done = two_phase_pipelined(b, done);
          continue;
        }
        int64_t chunk2 = std::min(b.num_rows - done, agg2_chunk_max_);
        if (done + chunk2 < b.num_rows) chunk2 &= ~(int64_t)7;
        two_phase_chunk(b, done, chunk2);
        done += chunk2;
        row_cursor_ += (uint64_t)chunk2;
        if (maybe_enter_skipping(b, done)) break;
        continue;
      }
      int64_t free_slots = t_.cap * 3 / 4 - (int64_t)ng_bound_;
      if (free_slots < (1 << 16)) {
        // the conservative bound is exhausted: read the true cardinality
        // first; grow only if the table is genuinely near 3/4 load
        refresh_ng();
        free_slots = t_.cap * 3 / 4 - (int64_t)ng_true_;
        if (free_slots < (1 << 16)) {
          ensure_capacity((int64_t)ng_true_ + (1 << 20));
          free_slots = t_.cap * 3 / 4 - (int64_t)ng_bound_;
        }
      }
      int64_t chunk = std::min(b.num_rows - done, free_slots);
      if (done + chunk < b.num_rows) chunk &= ~(int64_t)7;  // bitmap-sliceable
      if (ma_mode_) {
        ma_chunk(b, done, chunk);
      } else if (gkey_) {
        gkey_chunk(b, done, chunk);
      } else if (merge_mode_) {
        const DevColumn& buf = b.cols.at(1);
        if (buf.dt != DType::Binary) FAIL("agg-buf column must be Binary");
        // acc_offsets are absolute into buf.values, so only the index shifts
        launch_agg_merge_frozen(t_, (const int64_t*)key.values + done,
                                slice_valid(key.validity, done),
                                (const uint8_t*)buf.values, buf.offsets + done,
                                chunk, row_cursor_, layout_, stream_);
        if (has_first_)  // pass B: capture the winning records' values
          launch_first_capture_frozen(t_, (const int64_t*)key.values + done,
                                      slice_valid(key.validity, done),
                                      (const uint8_t*)buf.values,
                                      buf.offsets + done, nullptr, chunk,
                                      row_cursor_, layout_, stream_);
      } else {
        const DevColumn& val = b.cols.at(val_col_);
        const DType want = val_is_int_ ? DType::Int64 : DType::Float64;
        if (val.dt != want)
          FAIL("agg arg dtype must match the declared accumulator type "
               "(narrower args are widened at batch import)");
        // HIP-event timing on the launch stream (roofline evidence for the
        // dominant kernel; pairs are drained once at finish() so the hot
        // loop never synchronizes for timing)
        hipEvent_t e0, e1;
        AURON_HIP(hipEventCreate(&e0));
        AURON_HIP(hipEventCreate(&e1));
        AURON_HIP(hipEventRecord(e0, stream_));
        launch_agg_update(t_, (const int64_t*)key.values + done,
                          slice_valid(key.validity, done),
                          (const double*)val.values + done,
                          slice_valid(val.validity, done), chunk, row_cursor_,
                          stream_);
        if (has_first_)  // pass B: same stream, so it runs after pass A
          launch_first_capture_update(t_, (const int64_t*)key.values + done,
                                      slice_valid(key.validity, done),
                                      (const double*)val.values + done,
                                      slice_valid(val.validity, done), chunk,
                                      row_cursor_, stream_);
        AURON_HIP(hipEventRecord(e1, stream_));
        ev_pairs_.push_back({e0, e1});
        update_rows_ += chunk;
      }
      done += chunk;
      row_cursor_ += (uint64_t)chunk;
      ng_bound_ += (uint64_t)chunk;
      DBG("agg.chunk done=%lld/%lld cap=%lld ng_bound=%llu", (long long)done,
          (long long)b.num_rows, (long long)t_.cap,
          (unsigned long long)ng_bound_);
      if (maybe_enter_skipping(b, done)) break;
    }
    held_.push_back(std::move(b));  // keep borrowed buffers alive
  }

  // partial skipping (agg_table.rs:109-120): needs the true cardinality.
  // Returns true (and slices this batch's unprocessed tail into skipped_)
  // when the policy flips to pass-through.
  bool maybe_enter_skipping(const DevBatch& b, int64_t done) {
    if (!skip_enabled_ || skipping_ || row_cursor_ < (uint64_t)skip_min_rows_)
      return false;
    refresh_ng();
    if ((double)ng_true_ / (double)row_cursor_ < skip_ratio_) return false;
    skipping_ = true;
    if (done < b.num_rows) {
      // the un-aggregated tail of this batch passes through, like the
      // reference flipping between batches
      DevBatch tail;
      tail.num_rows = b.num_rows - done;
      for (const DevColumn& src : b.cols) {
        DevColumn sc;
        sc.dt = src.dt;
        sc.len = tail.num_rows;
        if (src.dt == DType::Binary || src.dt == DType::Utf8)
          FAIL("binary columns unsupported in skip-tail slice");
        sc.values = (const uint8_t*)src.values + done * dtype_width(src.dt);
        sc.validity = src.validity ? src.validity + done / 8 : nullptr;
        tail.cols.push_back(std::move(sc));
      }
      skipped_.push_back(std::move(tail));
    }
    return true;
  }

  std::vector<OutField> key_fields() const {
    std::vector<OutField> f;
    for (size_t k = 0; k < key_cols_.size(); k++) {
      DType kd = k < key_dts_.size() ? key_dts_[k]
                 : (key_dt_ == DType::Unsupported ? DType::Int64 : key_dt_);
      f.push_back({key_names_[k], kd, true});
    }
    return f;
  }

  DType ma_out_dt(size_t i) const {  // per-agg declared type (ma mode)
    switch (ma_desc_.a[i].acc_t) {
      case 1: return DType::Int64;
      case 2: return DType::Int32;
      default: return DType::Float64;
    }
  }

  std::vector<OutField> output_fields() const {
    if (final_output_) {
      std::vector<OutField> f = key_fields();
      const DType svdt = val_is_int_ ? DType::Int64 : DType::Float64;
      for (size_t i = 0; i < agg_kinds_.size(); i++) {
        DType vdt = ma_mode_ ? ma_out_dt(i) : svdt;
        if (agg_kinds_[i] == AGGL_CNT)
          f.push_back({agg_names_[i], DType::Int64, false});
        else if (agg_kinds_[i] == AGGL_AVG)
          f.push_back({agg_names_[i], DType::Float64, true});
        else if (agg_kinds_[i] == AGGL_CLIST || agg_kinds_[i] == AGGL_CSET)
          // collect.rs:110-112: nullable() = false (empty lists, not nulls)
          f.push_back({agg_names_[i], vdt, false, /*is_list=*/true});
        else
          f.push_back({agg_names_[i], vdt, true});
      }
      return f;
    }
    // partial/partial-merge: grouping + AGG_BUF (agg/mod.rs:37)
    std::vector<OutField> f = key_fields();
    f.push_back({"#9223372036854775807", DType::Binary, false});
    return f;
  }

  // drain: produce all output batches (host-staged)
  // Sort the COLLECT pool into {key ascending (signed), prio ascending
  // within key} + the null-key back segment into prio order, then point the
  // table's c_key/c_val views at the sorted arrays (binary-searched by the
  // freeze/emit kernels). One-shot before any emit.
  void prepare_collect() {
    if (!has_coll_ || coll_sorted_) return;
    coll_sorted_ = true;
    if (pinned_meta_.size() < 32) pinned_meta_.alloc(32);
    AURON_HIP(hipMemcpyAsync(pinned_meta_.get(), d_cn_.get(), 16,
                             hipMemcpyDeviceToHost, stream_));
    uint32_t* errp = pinned_meta_.get<uint32_t>() + 4;
    AURON_HIP(hipMemcpyAsync(errp, t_.error_flag, 4, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    int64_t n0 = (int64_t)pinned_meta_.get<unsigned long long>()[0];
    int64_t n1 = (int64_t)pinned_meta_.get<unsigned long long>()[1];
    // the per-append guard reads the opposing counter non-atomically (an
    // approximate early trip); this is the exact post-hoc check
    if ((*errp & 4u) || n0 + n1 > coll_cap_)
      FAIL("collect pool overflow: raise AURON_HIP_COLLECT_POOL");
    coll_n0_ = n0;
    coll_n1_ = n1;
    int64_t nmax = std::max<int64_t>(std::max(n0, n1), 1);
    size_t tb0 = 0;
    sort_pairs_u64_u32(nullptr, nullptr, nullptr, nullptr, nmax, nullptr,
                       &tb0, stream_);
    DevBuf tmp(tb0), idx(nmax * 4), idxo(nmax * 4), scr(nmax * 8);

    // one stable radix pass over `by`, permuting key/prio/val in place
    // (ping-pong through scratch buffers)
    auto sort_pass = [&](unsigned long long* by, int64_t n,
                         unsigned long long* key, unsigned long long* prio,
                         unsigned long long* val, DevBuf& alt_key,
                         DevBuf& alt_prio, DevBuf& alt_val) {
      launch_iota_u32(idx.get<uint32_t>(), n, stream_);
      size_t tb = tmp.size();
      sort_pairs_u64_u32(by, idx.get<uint32_t>(),
                         scr.get<unsigned long long>(), idxo.get<uint32_t>(),
                         n, tmp.get(), &tb, stream_);
      if (key)
        launch_gather_u64_idx(key, idxo.get<uint32_t>(), n,
                              alt_key.get<unsigned long long>(), stream_);
      launch_gather_u64_idx(prio, idxo.get<uint32_t>(), n,
                            alt_prio.get<unsigned long long>(), stream_);
      launch_gather_u64_idx(val, idxo.get<uint32_t>(), n,
                            alt_val.get<unsigned long long>(), stream_);
    };

    d_cskey_.alloc(std::max<int64_t>(n0, 1) * 8);
    d_csval_.alloc(std::max<int64_t>(n0 + n1, 1) * 8);
    if (n0 > 0) {
      // working copies (ping-pong pairs)
      DevBuf ka(n0 * 8), kb(n0 * 8), pa(n0 * 8), pb(n0 * 8), va(n0 * 8),
          vb(n0 * 8), biased(n0 * 8);
      AURON_HIP(hipMemcpyAsync(ka.get(), t_.c_key, n0 * 8,
                               hipMemcpyDeviceToDevice, stream_));
      AURON_HIP(hipMemcpyAsync(pa.get(), t_.c_prio, n0 * 8,
                               hipMemcpyDeviceToDevice, stream_));
      AURON_HIP(hipMemcpyAsync(va.get(), t_.c_val, n0 * 8,
                               hipMemcpyDeviceToDevice, stream_));
      auto K = [&](DevBuf& b) { return b.get<unsigned long long>(); };
      // pass 1: prio order
      sort_pass(K(pa), n0, K(ka), K(pa), K(va), kb, pb, vb);
      std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
      if (coll_set_) {
        // AccSet dedup (collect.rs AccSet.append: only novel values):
        // sort by value then key (stable keeps prio asc inside runs),
        // keep each (key,value) run head = first occurrence
        sort_pass(K(va), n0, K(ka), K(pa), K(va), kb, pb, vb);
        std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
        launch_bias_i64(K(ka), n0, biased.get<unsigned long long>(), stream_);
        sort_pass(biased.get<unsigned long long>(), n0, K(ka), K(pa), K(va),
                  kb, pb, vb);
        std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
        DevBuf mark(n0), pos((n0 + 1) * 4);
        launch_coll_mark_heads((const long long*)K(ka), K(va), n0, 1,
                               mark.get<uint8_t>(), stream_);
        size_t stb = 0;
        scan_mask_u8(mark.get<uint8_t>(), pos.get<uint32_t>(), n0, nullptr,
                     &stb, stream_);
        DevBuf stmp(stb);
        scan_mask_u8(mark.get<uint8_t>(), pos.get<uint32_t>(), n0, stmp.get(),
                     &stb, stream_);
        for (auto arr : {&ka, &pa, &va}) {
          launch_compact_u64(K(*arr), mark.get<uint8_t>(),
                             pos.get<uint32_t>(), n0,
                             scr.get<unsigned long long>(), stream_);
          AURON_HIP(hipMemcpyAsync(arr->get(), scr.get(), n0 * 8,
                                   hipMemcpyDeviceToDevice, stream_));
        }
        // new count = scan total (slot n0 of pos)
        AURON_HIP(hipMemcpyAsync(pinned_meta_.get(),
                                 pos.get<uint32_t>() + n0, 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        n0 = (int64_t)*pinned_meta_.get<uint32_t>();
        // device counter must match the binary-search bound
        unsigned long long nn = (unsigned long long)n0;
        AURON_HIP(hipMemcpyAsync(d_cn_.get(), &nn, 8, hipMemcpyHostToDevice,
                                 stream_));
        if (n0 > 0) {
          // restore prio order for the final key pass
          sort_pass(K(pa), n0, K(ka), K(pa), K(va), kb, pb, vb);
          std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
        }
      }
      if (n0 > 0) {
        // final pass: sign-biased key order, stable (prio asc within key)
        launch_bias_i64(K(ka), n0, biased.get<unsigned long long>(), stream_);
        launch_iota_u32(idx.get<uint32_t>(), n0, stream_);
        size_t tb = tmp.size();
        sort_pairs_u64_u32(biased.get<unsigned long long>(),
                           idx.get<uint32_t>(),
                           scr.get<unsigned long long>(),
                           idxo.get<uint32_t>(), n0, tmp.get(), &tb, stream_);
        launch_gather_u64_idx(K(ka), idxo.get<uint32_t>(), n0,
                              d_cskey_.get<unsigned long long>(), stream_);
        launch_gather_u64_idx(K(va), idxo.get<uint32_t>(), n0,
                              d_csval_.get<unsigned long long>(), stream_);
      }
      coll_n0_ = n0;
    }
    if (n1 > 0) {
      // null-key back segment: prio order (+ value dedup for SET)
      DevBuf pa(n1 * 8), pb(n1 * 8), va(n1 * 8), vb(n1 * 8);
      AURON_HIP(hipMemcpyAsync(pa.get(), t_.c_prio + (coll_cap_ - n1), n1 * 8,
                               hipMemcpyDeviceToDevice, stream_));
      AURON_HIP(hipMemcpyAsync(va.get(), t_.c_val + (coll_cap_ - n1), n1 * 8,
                               hipMemcpyDeviceToDevice, stream_));
      auto K = [&](DevBuf& b) { return b.get<unsigned long long>(); };
      sort_pass(K(pa), n1, nullptr, K(pa), K(va), pb, pb, vb);
      std::swap(pa, pb); std::swap(va, vb);
      if (coll_set_) {
        sort_pass(K(va), n1, nullptr, K(pa), K(va), pb, pb, vb);
        std::swap(pa, pb); std::swap(va, vb);
        DevBuf mark(n1), pos((n1 + 1) * 4);
        launch_coll_mark_heads(nullptr, K(va), n1, 0, mark.get<uint8_t>(),
                               stream_);
        size_t stb = 0;
        scan_mask_u8(mark.get<uint8_t>(), pos.get<uint32_t>(), n1, nullptr,
                     &stb, stream_);
        DevBuf stmp(stb);
        scan_mask_u8(mark.get<uint8_t>(), pos.get<uint32_t>(), n1, stmp.get(),
                     &stb, stream_);
        for (auto arr : {&pa, &va}) {
          launch_compact_u64(K(*arr), mark.get<uint8_t>(),
                             pos.get<uint32_t>(), n1,
                             scr.get<unsigned long long>(), stream_);
          AURON_HIP(hipMemcpyAsync(arr->get(), scr.get(), n1 * 8,
                                   hipMemcpyDeviceToDevice, stream_));
        }
        AURON_HIP(hipMemcpyAsync(pinned_meta_.get(),
                                 pos.get<uint32_t>() + n1, 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        n1 = (int64_t)*pinned_meta_.get<uint32_t>();
        unsigned long long nn = (unsigned long long)n1;
        AURON_HIP(hipMemcpyAsync(d_cn_.get<uint8_t>() + 8, &nn, 8,
                                 hipMemcpyHostToDevice, stream_));
        if (n1 > 0) {
          sort_pass(K(pa), n1, nullptr, K(pa), K(va), pb, pb, vb);
          std::swap(pa, pb); std::swap(va, vb);
        }
      }
      if (n1 > 0)
        AURON_HIP(hipMemcpyAsync(
            d_csval_.get<unsigned long long>() + coll_n0_, va.get(), n1 * 8,
            hipMemcpyDeviceToDevice, stream_));
      coll_n1_ = n1;
    }
    AURON_HIP(hipStreamSynchronize(stream_));  // temps return to the pool
    t_.c_key = d_cskey_.get<long long>();
    t_.c_val = d_csval_.get<unsigned long long>();
    t_.c_cap = coll_n0_ + coll_n1_;  // null segment ends the sorted view
  }

  // Sort each multi-arg collect pool exactly like prepare_collect sorts the
  // legacy pool: {key asc (signed), prio asc within key} + null-key back
  // segment in prio order appended at [n0, n0+n1); COLLECT_SET pools dedup
  // by (key, value) keeping first occurrence. The sorted views replace the
  // raw pool pointers in mp_ (no further appends happen after finish) and
  // the device counters are updated to the post-dedup counts.
  void prepare_ma_pools() {
    if (!ma_mode_ || ma_pools_sorted_) return;
    ma_pools_sorted_ = true;
    if (pinned_meta_.size() < MA_MAX_POOLS * 16 + 16)
      pinned_meta_.alloc(MA_MAX_POOLS * 16 + 16);
    unsigned long long* h_n = pinned_meta_.get<unsigned long long>();
    AURON_HIP(hipMemcpyAsync(h_n, d_mp_n_.get(), MA_MAX_POOLS * 16,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    for (int j = 0; j < ma_desc_.n; j++) {
      uint8_t p = ma_desc_.a[j].pool;
      if (p >= MA_MAX_POOLS) continue;
      bool dedup = ma_desc_.a[j].kind == AGGL_CSET;
      int64_t n0 = (int64_t)h_n[2 * p], n1 = (int64_t)h_n[2 * p + 1];
      if (n0 + n1 > mp_.cap) FAIL("collect pool overflow (ma mode)");
      int64_t nmax = std::max<int64_t>(std::max(n0, n1), 1);
      size_t tb0 = 0;
      sort_pairs_u64_u32(nullptr, nullptr, nullptr, nullptr, nmax, nullptr,
                         &tb0, stream_);
      DevBuf tmp(tb0), idx(nmax * 4), idxo(nmax * 4), scr(nmax * 8);
      auto sort_pass = [&](unsigned long long* by, int64_t n,
                           unsigned long long* key, unsigned long long* prio,
                           unsigned long long* val, DevBuf& ak, DevBuf& ap,
                           DevBuf& av) {
        launch_iota_u32(idx.get<uint32_t>(), n, stream_);
        size_t tb = tmp.size();
        sort_pairs_u64_u32(by, idx.get<uint32_t>(),
                           scr.get<unsigned long long>(),
                           idxo.get<uint32_t>(), n, tmp.get(), &tb, stream_);
        if (key)
          launch_gather_u64_idx(key, idxo.get<uint32_t>(), n,
                                ak.get<unsigned long long>(), stream_);
        launch_gather_u64_idx(prio, idxo.get<uint32_t>(), n,
                              ap.get<unsigned long long>(), stream_);
        launch_gather_u64_idx(val, idxo.get<uint32_t>(), n,
                              av.get<unsigned long long>(), stream_);
      };
      d_mp_skey_[p].alloc(std::max<int64_t>(n0 + n1, 1) * 8);
      d_mp_sval_[p].alloc(std::max<int64_t>(n0 + n1, 1) * 8);
      auto K = [&](DevBuf& bb) { return bb.get<unsigned long long>(); };
      if (n0 > 0) {
        DevBuf ka(n0 * 8), kb(n0 * 8), pa(n0 * 8), pb(n0 * 8), va(n0 * 8),
            vb(n0 * 8), biased(n0 * 8);
        AURON_HIP(hipMemcpyAsync(ka.get(), mp_.key[p], n0 * 8,
                                 hipMemcpyDeviceToDevice, stream_));
        AURON_HIP(hipMemcpyAsync(pa.get(), mp_.prio[p], n0 * 8,
                                 hipMemcpyDeviceToDevice, stream_));
        AURON_HIP(hipMemcpyAsync(va.get(), mp_.val[p], n0 * 8,
                                 hipMemcpyDeviceToDevice, stream_));
        sort_pass(K(pa), n0, K(ka), K(pa), K(va), kb, pb, vb);
        std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
        if (dedup) {
          sort_pass(K(va), n0, K(ka), K(pa), K(va), kb, pb, vb);
          std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
          launch_bias_i64(K(ka), n0, biased.get<unsigned long long>(),
                          stream_);
          sort_pass(biased.get<unsigned long long>(), n0, K(ka), K(pa),
                    K(va), kb, pb, vb);
          std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
          DevBuf mark(n0), pos2((n0 + 1) * 4);
          launch_coll_mark_heads((const long long*)K(ka), K(va), n0, 1,
                                 mark.get<uint8_t>(), stream_);
          size_t stb = 0;
          scan_mask_u8(mark.get<uint8_t>(), pos2.get<uint32_t>(), n0,
                       nullptr, &stb, stream_);
          DevBuf stmp(stb);
          scan_mask_u8(mark.get<uint8_t>(), pos2.get<uint32_t>(), n0,
                       stmp.get(), &stb, stream_);
          for (auto arr : {&ka, &pa, &va}) {
            launch_compact_u64(K(*arr), mark.get<uint8_t>(),
                               pos2.get<uint32_t>(), n0,
                               scr.get<unsigned long long>(), stream_);
            AURON_HIP(hipMemcpyAsync(arr->get(), scr.get(), n0 * 8,
                                     hipMemcpyDeviceToDevice, stream_));
          }
          uint32_t* hc = (uint32_t*)(h_n + MA_MAX_POOLS * 2);
          AURON_HIP(hipMemcpyAsync(hc, pos2.get<uint32_t>() + n0, 4,
                                   hipMemcpyDeviceToHost, stream_));
          AURON_HIP(hipStreamSynchronize(stream_));
          n0 = (int64_t)*hc;
          if (n0 > 0) {  // restore prio order for the final key pass
            sort_pass(K(pa), n0, K(ka), K(pa), K(va), kb, pb, vb);
            std::swap(ka, kb); std::swap(pa, pb); std::swap(va, vb);
          }
        }
        if (n0 > 0) {
          launch_bias_i64(K(ka), n0, biased.get<unsigned long long>(),
                          stream_);
          launch_iota_u32(idx.get<uint32_t>(), n0, stream_);
          size_t tb = tmp.size();
          sort_pairs_u64_u32(biased.get<unsigned long long>(),
                             idx.get<uint32_t>(),
                             scr.get<unsigned long long>(),
                             idxo.get<uint32_t>(), n0, tmp.get(), &tb,
                             stream_);
          launch_gather_u64_idx(K(ka), idxo.get<uint32_t>(), n0,
                                d_mp_skey_[p].get<unsigned long long>(),
                                stream_);
          launch_gather_u64_idx(K(va), idxo.get<uint32_t>(), n0,
                                d_mp_sval_[p].get<unsigned long long>(),
                                stream_);
        }
      }
      if (n1 > 0) {  // null-key back segment -> [n0, n0+n1) of the view
        DevBuf pa(n1 * 8), pb(n1 * 8), va(n1 * 8), vb(n1 * 8);
        // back segment occupies [cap-n1, cap); arrival order is reversed in
        // memory but the prio sort below restores it regardless
        AURON_HIP(hipMemcpyAsync(pa.get(), mp_.prio[p] + (mp_.cap - n1),
                                 n1 * 8, hipMemcpyDeviceToDevice, stream_));
        AURON_HIP(hipMemcpyAsync(va.get(), mp_.val[p] + (mp_.cap - n1),
                                 n1 * 8, hipMemcpyDeviceToDevice, stream_));
        sort_pass(K(pa), n1, nullptr, K(pa), K(va), pb, pb, vb);
        std::swap(pa, pb); std::swap(va, vb);
        if (dedup) {
          sort_pass(K(va), n1, nullptr, K(pa), K(va), pb, pb, vb);
          std::swap(pa, pb); std::swap(va, vb);
          DevBuf mark(n1), pos2((n1 + 1) * 4);
          launch_coll_mark_heads(nullptr, K(va), n1, 0, mark.get<uint8_t>(),
                                 stream_);
          size_t stb = 0;
          scan_mask_u8(mark.get<uint8_t>(), pos2.get<uint32_t>(), n1,
                       nullptr, &stb, stream_);
          DevBuf stmp(stb);
          scan_mask_u8(mark.get<uint8_t>(), pos2.get<uint32_t>(), n1,
                       stmp.get(), &stb, stream_);
          for (auto arr : {&pa, &va}) {
            launch_compact_u64(K(*arr), mark.get<uint8_t>(),
                               pos2.get<uint32_t>(), n1,
                               scr.get<unsigned long long>(), stream_);
            AURON_HIP(hipMemcpyAsync(arr->get(), scr.get(), n1 * 8,
                                     hipMemcpyDeviceToDevice, stream_));
          }
          uint32_t* hc = (uint32_t*)(h_n + MA_MAX_POOLS * 2);
          AURON_HIP(hipMemcpyAsync(hc, pos2.get<uint32_t>() + n1, 4,
                                   hipMemcpyDeviceToHost, stream_));
          AURON_HIP(hipStreamSynchronize(stream_));
          n1 = (int64_t)*hc;
          if (n1 > 0) {
            sort_pass(K(pa), n1, nullptr, K(pa), K(va), pb, pb, vb);
            std::swap(pa, pb); std::swap(va, vb);
          }
        }
        if (n1 > 0)
          AURON_HIP(hipMemcpyAsync(
              d_mp_sval_[p].get<unsigned long long>() + n0, va.get(), n1 * 8,
              hipMemcpyDeviceToDevice, stream_));
      }
      AURON_HIP(hipStreamSynchronize(stream_));  // temps return to the pool
      mp_.key[p] = d_mp_skey_[p].get<long long>();
      mp_.val[p] = d_mp_sval_[p].get<unsigned long long>();
      mp_counts_[p][0] = n0;
      mp_counts_[p][1] = n1;
      unsigned long long nn[2] = {(unsigned long long)n0,
                                  (unsigned long long)n1};
      AURON_HIP(hipMemcpyAsync(d_mp_n_.get<unsigned long long>() + 2 * p, nn,
                               16, hipMemcpyHostToDevice, stream_));
    }
    AURON_HIP(hipStreamSynchronize(stream_));
  }

  std::vector<std::pair<int64_t, std::vector<HostOutCol>>> finish() {
    std::vector<std::pair<int64_t, std::vector<HostOutCol>>> out;
    drain_timing();
    if (spill_.empty()) {
      emit_table(&out, false);
    } else {
      // a9 analog (agg_table.rs:540-588 spill + RadixQueue merge): drain the
      // live table into the buckets, then merge+emit bucket by bucket so
      // peak table size is one bucket's groups. Output order is per-bucket
      // (the reference's own spill path also gives up global insertion
      // order), first-occurrence-ordered within each bucket.
      spill_table();
      for (size_t b = 0; b < spill_.size(); b++) {
        SpillBucket& sb = spill_[b];
        int64_t n = (int64_t)sb.keys.size();
        if (n > 0) {
          std::vector<int32_t> offs(n + 1, 0);
          for (int64_t i = 0; i < n; i++) offs[i + 1] = offs[i] + sb.lens[i];
          DevBuf d_keys(n * 8), d_first(n * 8), d_offs((n + 1) * 4),
              d_data(offs[n] ? offs[n] : 1);
          AURON_HIP(hipMemcpyAsync(d_keys.get(), sb.keys.data(), n * 8,
                                   hipMemcpyHostToDevice, stream_));
          AURON_HIP(hipMemcpyAsync(d_first.get(), sb.first_rows.data(), n * 8,
                                   hipMemcpyHostToDevice, stream_));
          AURON_HIP(hipMemcpyAsync(d_offs.get(), offs.data(), (n + 1) * 4,
                                   hipMemcpyHostToDevice, stream_));
          if (offs[n])
            AURON_HIP(hipMemcpyAsync(d_data.get(), sb.data.data(), offs[n],
                                     hipMemcpyHostToDevice, stream_));
          for (int64_t done2 = 0; done2 < n;) {
            ensure_capacity((int64_t)ng_bound_ +
                            std::min<int64_t>(n - done2, 1 << 20));
            int64_t free2 = t_.cap * 3 / 4 - (int64_t)ng_bound_;
            if (free2 <= 0)
              FAIL("spill bucket exceeds VRAM budget: raise "
                   "AURON_HIP_SPILL_BUCKETS");
            int64_t piece = std::min(n - done2, free2);
            launch_agg_merge_spill(t_, d_keys.get<int64_t>() + done2,
                                   d_data.get<uint8_t>(),
                                   d_offs.get<int32_t>() + done2,
                                   d_first.get<unsigned long long>() + done2,
                                   piece, layout_, stream_);
            if (has_first_)  // pass B with the records' preserved priorities
              launch_first_capture_frozen(
                  t_, d_keys.get<int64_t>() + done2, nullptr,
                  d_data.get<uint8_t>(), d_offs.get<int32_t>() + done2,
                  d_first.get<unsigned long long>() + done2, piece, 0,
                  layout_, stream_);
            done2 += piece;
            ng_bound_ += (uint64_t)piece;
          }
          AURON_HIP(hipStreamSynchronize(stream_));
        }
        bool last = (b + 1 == spill_.size());
        emit_table(&out, /*exclude_specials=*/!last);
        if (!last) {
          reset_main();
          // the pool holds this bucket's merged items + the resident
          // specials; keep only the specials for the next bucket
          if (has_coll_) reset_collect_pool();
        }
      }
      spill_.clear();
    }
    // 2) partial-skipped pass-through batches (agg_ctx.rs:428-462)
    for (DevBatch& b : skipped_) {
      out.emplace_back(emit_skipped(b));
    }
    skipped_.clear();
    held_.clear();
    return out;
  }

  // device-resident variant of finish(): partial (key + a8 Binary) batches
  // stay in HBM. Spill / partial-skipping fall outside this mode (the bench
  // and exchange paths that use it never enter them) and FAIL loudly.
  std::vector<std::pair<int64_t, std::vector<DevOutCol>>> finish_dev() {
    std::vector<std::pair<int64_t, std::vector<DevOutCol>>> out;
    drain_timing();
    prepare_collect();
    if (gkey_ || ma_mode_)
      FAIL("AURON_HIP_DEVICE_OUTPUT with generalized keys / multi-argument "
           "aggregates unsupported (unset the conf)");
    if (!spill_.empty())
      FAIL("AURON_HIP_DEVICE_OUTPUT with spill unsupported (unset the conf)");
    if (!skipped_.empty())
      FAIL("AURON_HIP_DEVICE_OUTPUT with partial skipping unsupported");
    DevBuf order, first;
    int64_t ng = table_order(&order, &first);
    for (int64_t beg = 0; beg < ng; beg += batch_size_) {
      int64_t len = std::min(batch_size_, ng - beg);
      out.emplace_back(emit_groups_dev(order.get<uint32_t>() + beg, len));
    }
    AURON_HIP(hipStreamSynchronize(stream_));  // order/first return to pool
    held_.clear();
    return out;
  }

  std::pair<int64_t, std::vector<DevOutCol>> emit_groups_dev(
      const uint32_t* order_slots, int64_t n) {
    size_t bm = (n + 7) / 8;
    std::vector<DevOutCol> cols(2);
    DevOutCol& kc = cols[0];
    DevOutCol& bc = cols[1];
    kc.dt = key_dt_ == DType::Unsupported ? DType::Int64 : key_dt_;
    DevBuf keys(n * 8), sums(n * 8), svalid(bm), cnts(n * 8);
    kc.validity.alloc(bm);
    launch_agg_gather_out(t_, order_slots, n, keys.get<int64_t>(),
                          kc.validity.get<uint8_t>(), sums.get<double>(),
                          svalid.get<uint8_t>(), cnts.get<long long>(),
                          nullptr, nullptr, nullptr, nullptr, stream_);
    if (kc.dt == DType::Int32) {
      kc.values.alloc(n * 4);
      launch_narrow_i64_i32(keys.get<int64_t>(), n, kc.values.get<int32_t>(),
                            stream_);
    } else {
      kc.values = std::move(keys);
    }
    // freeze lens -> device exclusive scan -> offsets (+ total in slot n)
    bc.dt = DType::Binary;
    DevBuf lens((n + 1) * 4);
    bc.offsets.alloc((n + 1) * 4);
    launch_agg_freeze_len(t_, order_slots, n, lens.get<int32_t>(), layout_,
                          stream_);
    size_t tb = 0;
    scan_counts_matrix(lens.get<uint32_t>(), bc.offsets.get<uint32_t>(),
                       n + 1, nullptr, &tb, stream_);
    DevBuf scan_tmp(tb);
    scan_counts_matrix(lens.get<uint32_t>(), bc.offsets.get<uint32_t>(),
                       n + 1, scan_tmp.get(), &tb, stream_);
    if (pinned_meta_.size() < 16) pinned_meta_.alloc(16);
    AURON_HIP(hipMemcpyAsync(pinned_meta_.get(),
                             bc.offsets.get<uint32_t>() + n, 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    bc.data_len = (int64_t)*pinned_meta_.get<uint32_t>();
    bc.values.alloc(bc.data_len ? bc.data_len : 1);
    launch_agg_freeze_write(t_, order_slots, n, bc.offsets.get<int32_t>(),
                            bc.values.get<uint8_t>(), layout_, stream_);
    return {n, std::move(cols)};
  }

  uint64_t num_groups_host() {
    if (!pinned_meta_.get()) pinned_meta_.alloc(16);
    uint64_t* ng = pinned_meta_.get<uint64_t>();
    uint32_t* err = (uint32_t*)(pinned_meta_.get<uint8_t>() + 8);
    AURON_HIP(hipMemcpyAsync(ng, t_.num_groups, 8, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipMemcpyAsync(err, t_.error_flag, 4, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    if (*err) {
      // bit 1: table probe exhausted; 16: gkey pool publish spin; 32: v3
      // scatter tile bound; 64: v4 scatter ring-space retry bound
      if (*err & ~1u)
        FAIL("agg device error flags 0x" +
             ([](uint32_t v) {
               char b[16];
               snprintf(b, sizeof b, "%x", v);
               return std::string(b);
             })(*err));
      FAIL("agg hash table probe exhausted (table full/corrupt)");
    }
    return *ng;
  }

  int64_t batch_size() const { return batch_size_; }

 private:
  void init_table(int64_t slots) {
    int64_t cap = 1;
    while (cap < slots) cap <<= 1;
    t_.cap = cap;
    d_slots_.alloc((cap + 2) * sizeof(AggSlot));
    d_special_.alloc(2 * 4);
    d_ng_.alloc(8);
    d_err_.alloc(4);
    t_.slots = d_slots_.get<AggSlot>();
    t_.special_used = d_special_.get<uint32_t>();
    t_.num_groups = d_ng_.get<unsigned long long>();
    t_.error_flag = d_err_.get<uint32_t>();
    AURON_HIP(hipMemsetAsync(d_err_.get(), 0, 4, stream_));
    AURON_HIP(hipMemsetAsync(d_special_.get(), 0, 2 * 4, stream_));
    AURON_HIP(hipMemsetAsync(d_ng_.get(), 0, 8, stream_));
    launch_slots_init(t_.slots, cap + 2, stream_);
    if (has_mm_) {
      d_mm_.alloc((cap + 2) * 16);
      t_.mm = d_mm_.get<unsigned long long>();
      launch_mm_init(t_.mm, cap + 2, stream_);
    }
    if (has_first_) {
      d_frow_.alloc((cap + 2) * 16);
      d_fval_.alloc((cap + 2) * 16);
      d_fst_.alloc((cap + 2) * 2);
      t_.f_row = d_frow_.get<unsigned long long>();
      t_.f_val = d_fval_.get<double>();
      t_.f_st = d_fst_.get<uint8_t>();
      launch_first_init(t_.f_row, t_.f_val, t_.f_st, cap + 2, stream_);
    }
    if (ma_mode_) {  // per-agg accumulator bank (growth re-allocs)
      int64_t cap2 = cap + 2;
      d_ma_acc_.alloc((int64_t)ma_desc_.n * cap2 * 8);
      d_ma_meta_.alloc((int64_t)ma_desc_.n * cap2 * 8);
      d_ma_st_.alloc((int64_t)ma_desc_.n * cap2);
      ma_.acc = d_ma_acc_.get<unsigned long long>();
      ma_.meta = d_ma_meta_.get<unsigned long long>();
      ma_.st = d_ma_st_.get<uint8_t>();
      ma_.stride = cap2;
      launch_ma_init(ma_desc_, ma_, cap2, stream_);
      if (!d_mp_n_) {  // pools survive growth (keyed by VALUE not slot)
        int64_t pc = std::max<int64_t>(1024, conf_coll_cap_ / 4);
        for (int j = 0; j < ma_desc_.n; j++) {
          uint8_t p = ma_desc_.a[j].pool;
          if (p >= MA_MAX_POOLS) continue;
          d_mp_key_[p].alloc(pc * 8);
          d_mp_prio_[p].alloc(pc * 8);
          d_mp_val_[p].alloc(pc * 8);
          mp_.cap = pc;
        }
        d_mp_n_.alloc(MA_MAX_POOLS * 16);
        AURON_HIP(hipMemsetAsync(d_mp_n_.get(), 0, MA_MAX_POOLS * 16,
                                 stream_));
      }
      for (int p = 0; p < MA_MAX_POOLS; p++) {
        if (!d_mp_key_[p]) continue;
        mp_.key[p] = d_mp_key_[p].get<long long>();
        mp_.prio[p] = d_mp_prio_[p].get<unsigned long long>();
        mp_.val[p] = d_mp_val_[p].get<unsigned long long>();
      }
      mp_.n = d_mp_n_.get<unsigned long long>();
    }
    if (gkey_) {  // growth re-allocs the off array; pool survives
      d_gkoff_.alloc(cap * 8);
      AURON_HIP(hipMemsetAsync(d_gkoff_.get(), 0, cap * 8, stream_));
      g_.off = d_gkoff_.get<unsigned long long>();
    }
    if (has_coll_ && !d_ckey_) {  // pool survives grows (keys, not slots)
      coll_cap_ = conf_coll_cap_;
      d_ckey_.alloc(coll_cap_ * 8);
      d_cprio_.alloc(coll_cap_ * 8);
      d_cval_.alloc(coll_cap_ * 8);
      d_cn_.alloc(16);
      AURON_HIP(hipMemsetAsync(d_cn_.get(), 0, 16, stream_));
    }
    if (has_coll_) {
      t_.c_key = d_ckey_.get<long long>();
      t_.c_prio = d_cprio_.get<unsigned long long>();
      t_.c_val = d_cval_.get<unsigned long long>();
      t_.c_n = d_cn_.get<unsigned long long>();
      t_.c_cap = coll_cap_;
    }
  }

  static constexpr int64_t AGG2_MIN_CHUNK = 4 << 20;   // below: single-phase
  // partition-buffer rows per two-phase chunk: bigger chunks amortize launch
  // tails and per-chunk syncs (24 B/row + 24 B/row leftover scratch; 256M
  // rows = 12 GB of the 288 GB HBM). Runtime-tunable for A/B sweeps.
  int64_t agg2_chunk_max_ = 256 << 20;
  static constexpr int AGG2_NBUCK_LOG2 = 10;
  static constexpr int AGG3_NBUCK_LOG2 = 9;   // v3: 512 buckets
  static constexpr int AGG3_NBUCK = 1 << AGG3_NBUCK_LOG2;
  static constexpr int AGG3_GRID_LOG2 = 8;    // v3 scatter: 256 x 1024-thr           // 1024 buckets: scatter write-line footprint ~L2-sized (512 buckets overflow the 2048-slot LDS window: 2.3% leftovers, 3x slower)

  // ---- multi-argument chunk (kernels_maarg.hip) --------------------------
  void ma_chunk(const DevBatch& b, int64_t done, int64_t chunk) {
    const DevColumn& key = b.cols.at(key_col_);
    const int64_t* keys = (const int64_t*)key.values + done;
    const uint8_t* kv = key.validity ? key.validity + done / 8 : nullptr;
    if (d_ma_slots_.size() < (size_t)chunk * 4) d_ma_slots_.alloc(chunk * 4);
    MaArgs args;
    if (!merge_mode_) {
      for (int j = 0; j < ma_desc_.n; j++) {
        if (ma_arg_cols_[j] < 0) continue;
        const DevColumn& c = b.cols.at(ma_arg_cols_[j]);
        size_t w = dtype_width(c.dt);
        args.vals[j] = (const uint8_t*)c.values + (size_t)done * w;
        args.valid[j] = c.validity ? c.validity + done / 8 : nullptr;
      }
    }
    hipEvent_t e0, e1;
    AURON_HIP(hipEventCreate(&e0));
    AURON_HIP(hipEventCreate(&e1));
    AURON_HIP(hipEventRecord(e0, stream_));
    launch_slots_upsert(t_, keys, kv, chunk, row_cursor_,
                        d_ma_slots_.get<uint32_t>(), stream_);
    bool any_first = false;
    for (int j = 0; j < ma_desc_.n; j++)
      any_first |= (ma_desc_.a[j].kind == AGGL_FIRST ||
                    ma_desc_.a[j].kind == AGGL_FIRSTIN);
    if (merge_mode_) {
      const DevColumn& buf = b.cols.at(1);
      if (buf.dt != DType::Binary) FAIL("agg-buf column must be Binary");
      launch_ma_merge_frozen(t_, ma_desc_, ma_, mp_, keys, kv,
                             d_ma_slots_.get<uint32_t>(),
                             (const uint8_t*)buf.values, buf.offsets + done,
                             chunk, row_cursor_, stream_);
      if (any_first)
        launch_ma_first_capture_frozen(t_, ma_desc_, ma_,
                                       d_ma_slots_.get<uint32_t>(),
                                       (const uint8_t*)buf.values,
                                       buf.offsets + done, chunk,
                                       row_cursor_, stream_);
    } else {
      launch_ma_update(t_, ma_desc_, ma_, mp_, args, keys, kv,
                       d_ma_slots_.get<uint32_t>(), chunk, row_cursor_,
                       stream_);
      if (any_first)
        launch_ma_first_capture(t_, ma_desc_, ma_, args,
                                d_ma_slots_.get<uint32_t>(), chunk,
                                row_cursor_, stream_);
      update_rows_ += chunk;
    }
    AURON_HIP(hipEventRecord(e1, stream_));
    ev_pairs_.push_back({e0, e1});
  }

  // ---- generalized-key machinery (kernels_gkey.hip) ----------------------
  void init_gkey() {
    gk_pool_cap_ = 64 << 20;
    d_gkpool_.alloc(gk_pool_cap_);
    d_gkpn_.alloc(8);
    AURON_HIP(hipMemsetAsync(d_gkpn_.get(), 0, 8, stream_));
    d_gkoff_.alloc(t_.cap * 8);
    AURON_HIP(hipMemsetAsync(d_gkoff_.get(), 0, t_.cap * 8, stream_));
    g_.off = d_gkoff_.get<unsigned long long>();
    g_.pool = d_gkpool_.get<uint8_t>();
    g_.pool_n = d_gkpn_.get<unsigned long long>();
    g_.pool_cap = gk_pool_cap_;
    uint8_t dts[GKEY_MAX_COLS] = {};
    for (size_t k = 0; k < key_dts_.size(); k++) {
      switch (key_dts_[k]) {
        case DType::Int64: dts[k] = GK_I64; break;
        case DType::Int32: dts[k] = GK_I32; break;
        case DType::Float64: dts[k] = GK_F64; break;
        default: dts[k] = GK_UTF8; break;  // Utf8/Binary
      }
    }
    d_gkdts_.alloc(GKEY_MAX_COLS);
    AURON_HIP(hipMemcpyAsync(d_gkdts_.get(), dts, GKEY_MAX_COLS,
                             hipMemcpyHostToDevice, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
  }

  void gk_ensure_pool(int64_t need_more) {
    if (gk_used_bound_ + need_more <= gk_pool_cap_) return;
    // refresh the true cursor (the bound counts every row as a new group)
    if (pinned_meta_.size() < 16) pinned_meta_.alloc(16);
    AURON_HIP(hipMemcpyAsync(pinned_meta_.get(), d_gkpn_.get(), 8,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    gk_used_bound_ = (int64_t)*pinned_meta_.get<unsigned long long>();
    if (gk_used_bound_ + need_more <= gk_pool_cap_) return;
    int64_t new_cap = gk_pool_cap_;
    while (new_cap < gk_used_bound_ + need_more) new_cap *= 2;
    DevBuf np(new_cap);
    AURON_HIP(hipMemcpyAsync(np.get(), d_gkpool_.get(), gk_pool_cap_,
                             hipMemcpyDeviceToDevice, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    d_gkpool_ = std::move(np);
    gk_pool_cap_ = new_cap;
    g_.pool = d_gkpool_.get<uint8_t>();
    g_.pool_cap = gk_pool_cap_;
  }

  GKeyCols gk_cols(const DevBatch& b, int64_t done) const {
    GKeyCols gc;
    gc.ncols = (int)key_cols_.size();
    for (size_t k = 0; k < key_cols_.size(); k++) {
      const DevColumn& c = b.cols.at(key_cols_[k]);
      switch (key_dts_[k]) {
        case DType::Int64:
        case DType::Float64:
          gc.dt[k] = key_dts_[k] == DType::Int64 ? GK_I64 : GK_F64;
          gc.values[k] = (const uint8_t*)c.values + done * 8;
          break;
        case DType::Int32:
          gc.dt[k] = GK_I32;
          gc.values[k] = (const uint8_t*)c.values + done * 4;
          break;
        default:  // Utf8/Binary: offsets are absolute into the data buffer
          gc.dt[k] = GK_UTF8;
          gc.values[k] = c.values;
          gc.offsets[k] = c.offsets + done;
          break;
      }
      gc.validity[k] = c.validity ? c.validity + done / 8 : nullptr;
    }
    return gc;
  }

  // one bounded-size chunk of the generalized-key path: encode the key
  // tuples -> scan -> upsert (slot per row) -> slot-indexed accumulate or
  // frozen merge
  void gkey_chunk(const DevBatch& b, int64_t done, int64_t chunk) {
    GKeyCols gc = gk_cols(b, done);
    if (d_enclens_.size() < (size_t)(chunk + 1) * 4) {
      d_enclens_.alloc((chunk + 1) * 4);
      d_encoffs_.alloc((chunk + 1) * 4);
      d_gkslots_.alloc(chunk * 4);
    }
    launch_gkey_enc_lens(gc, chunk, d_enclens_.get<uint32_t>(), stream_);
    size_t tb = 0;
    scan_counts_matrix(d_enclens_.get<uint32_t>(), d_encoffs_.get<uint32_t>(),
                       chunk + 1, nullptr, &tb, stream_);
    if (d_gkscan_tmp_.size() < tb) d_gkscan_tmp_.alloc(tb);
    scan_counts_matrix(d_enclens_.get<uint32_t>(), d_encoffs_.get<uint32_t>(),
                       chunk + 1, d_gkscan_tmp_.get(), &tb, stream_);
    if (pinned_meta_.size() < 16) pinned_meta_.alloc(16);
    AURON_HIP(hipMemcpyAsync(pinned_meta_.get(),
                             d_encoffs_.get<uint32_t>() + chunk, 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    int64_t total = (int64_t)*pinned_meta_.get<uint32_t>();
    if (d_encbytes_.size() < (size_t)total + 8)
      d_encbytes_.alloc(total + 8);
    launch_gkey_enc_write(gc, d_encoffs_.get<uint32_t>(),
                          d_encbytes_.get<uint8_t>(), chunk, stream_);
    gk_ensure_pool(total + 4 * chunk);  // worst case: every row a new group
    hipEvent_t e0, e1;
    AURON_HIP(hipEventCreate(&e0));
    AURON_HIP(hipEventCreate(&e1));
    AURON_HIP(hipEventRecord(e0, stream_));
    launch_gkey_upsert(t_, g_, d_encoffs_.get<uint32_t>(),
                       d_encbytes_.get<uint8_t>(), chunk, row_cursor_,
                       d_gkslots_.get<uint32_t>(), stream_);
    if (merge_mode_) {
      const DevColumn& buf = b.cols.back();
      if (buf.dt != DType::Binary) FAIL("agg-buf column must be Binary");
      launch_gkey_merge_frozen_idx(t_, d_gkslots_.get<uint32_t>(),
                                   (const uint8_t*)buf.values,
                                   buf.offsets + done, chunk, layout_,
                                   stream_);
    } else {
      const DevColumn& val = b.cols.at(val_col_);
      launch_gkey_update_idx(t_, d_gkslots_.get<uint32_t>(),
                             (const double*)val.values + done,
                             val.validity ? val.validity + done / 8 : nullptr,
                             chunk, stream_);
      update_rows_ += chunk;
    }
    AURON_HIP(hipEventRecord(e1, stream_));
    ev_pairs_.push_back({e0, e1});
    gk_used_bound_ += total + 4 * chunk;
  }

  // Read the two-phase tuning knobs once, BEFORE the first chunk size is
  // computed (the scratch buffers are sized from agg2_chunk_max_, so the
  // bound must be final by then).
  void init_agg2_conf() {
    if (agg2_conf_read_) return;
    agg2_conf_read_ = true;
    // A/B toggle for the partition-record layout: the 24B AoS record
    // (default) vs 16B kv + 4B rowv split streams — measured a wash within
    // box noise; AURON_AGG2_SPLIT=1 selects the split for experiments.
    const char* e = getenv("AURON_AGG2_SPLIT");
    agg2_split_ = (e && e[0] == '1');
    // hist/scatter grid (blocks = 1<<log2): 512 default (same-box sweeps:
    // occupancy beats write locality — grid 256/128 measured slower)
    const char* g = getenv("AURON_AGG2_GRID_LOG2");
    if (g && g[0]) agg2_grid_log2_ = atoi(g);
    if (agg2_grid_log2_ < 6 || agg2_grid_log2_ > AGG2_GRID_LOG2_MAX)
      agg2_grid_log2_ = 9;
    // hist/scatter/bucket workgroup size: 1024 threads = 16 waves/block
    // fill the chip's 32-wave/CU slots (the 256-thread launch left the
    // bucket kernel at 12 waves/CU = 38% occupancy; microarch guide
    // "Chip-level parameters"/"Register files")
    const char* bl = getenv("AURON_AGG2_BLOCK");
    if (bl && bl[0]) agg2_block_ = atoi(bl);
    if (agg2_block_ != 256 && agg2_block_ != 512 && agg2_block_ != 1024)
      agg2_block_ = 1024;
    const char* cm = getenv("AURON_AGG2_CHUNK_M");  // millions of rows
    if (cm && cm[0]) {
      int64_t m = atoll(cm);
      if (m >= 4 && m <= 1024) agg2_chunk_max_ = m << 20;
    }
    // v3 pipeline (default): LDS-staged packet scatter, 512 buckets,
    // 4096-slot bucket window — kernels_agg3.hip header comment has the
    // evidence trail. AURON_AGG2_V3=0 falls back to the v2 kernels.
    const char* v3 = getenv("AURON_AGG2_V3");
    agg2_v3_ = !(v3 && v3[0] == '0');
    // cross-chunk pipeline: OFF by default — the scatter (151 KB LDS/WG)
    // and bucket (100 KB) kernels cannot co-reside on a CU, so overlapping
    // them only splits the chip and adds sync bubbles (measured 60.5 vs
    // 51 ms/step); kept behind AURON_AGG2_PIPE=1 for experiments
    const char* pe = getenv("AURON_AGG2_PIPE");
    agg2_pipe_ = (pe && pe[0] == '1');
    // v4 scatter: barrier-free producer/flusher rings (kernels_agg3.hip) —
    // experimental until measured; AURON_AGG2_V4=1 selects it
    const char* v4 = getenv("AURON_AGG2_V4");
    agg2_v4_ = (v4 && v4[0] == '1');
    const char* p16 = getenv("AURON_AGG2_P16");
    agg2_p16_en_ = !(p16 && p16[0] == '0');
  }

  // Cross-chunk pipelined two-phase aggregation (v3 kernels): the scatter
  // chain of chunk k+1 runs on the aux stream while the bucket/merge of
  // chunk k runs on the engine stream. Per-chunk scratch is
  // double-buffered; the counter readback of chunk k completes while
  // scatter(k+1) is in flight, so the host enqueues merges without stalling
  // the device. MERGES ENQUEUE BEFORE THE NEXT BUCKET so the shared staged
  // list is consumed in engine-stream order. Partial skipping is disabled
  // on this path (checked by the caller).
  int64_t two_phase_pipelined(const DevBatch& b, int64_t done) {
    const DevColumn& keyc = b.cols.at(key_col_);
    const DevColumn& valc = b.cols.at(val_col_);
    init_pipeline_scratch();
    struct Chunk {
      int64_t beg = 0, n = 0;
      uint64_t row0 = 0;
    };
    std::vector<Chunk> chunks;
    int64_t total = b.num_rows;
    uint64_t rc = row_cursor_;
    while (done < total && total - done >= AGG2_MIN_CHUNK) {
      Chunk c;
      c.beg = done;
      c.n = std::min(total - done, agg2_chunk_max_);
      if (done + c.n < total) c.n &= ~(int64_t)7;
      c.row0 = rc;
      chunks.push_back(c);
      done += c.n;
      rc += (uint64_t)c.n;
    }
    hipEvent_t e0, e1;
    AURON_HIP(hipEventCreate(&e0));
    AURON_HIP(hipEventCreate(&e1));
    AURON_HIP(hipEventRecord(e0, stream_));
    // the aux stream must not run ahead of prior work on the engine stream
    AURON_HIP(hipEventRecord(ev_pipe_[0], stream_));
    AURON_HIP(hipStreamWaitEvent(s_aux_, ev_pipe_[0], 0));
    const int nbuck = AGG3_NBUCK;
    int64_t mat3 = (int64_t)nbuck << AGG3_GRID_LOG2;
    for (size_t k = 0; k <= chunks.size(); k++) {
      if (k < chunks.size()) {
        // aux stream: hist -> line scan -> scatter for chunk k
        const Chunk& c = chunks[k];
        int pb = (int)(k & 1);
        const int64_t* keys = (const int64_t*)keyc.values + c.beg;
        const double* vals = (const double*)valc.values + c.beg;
        const uint8_t* kv =
            keyc.validity ? keyc.validity + c.beg / 8 : nullptr;
        const uint8_t* vv =
            valc.validity ? valc.validity + c.beg / 8 : nullptr;
        // partbuf/leftover[pb] are free once chunk k-2's MERGES completed
        // (the merge event implies its bucket read finished too)
        if (k >= 2)
          AURON_HIP(hipStreamWaitEvent(s_aux_, ev_merge_[pb], 0));
        AURON_HIP(hipMemsetAsync(d_pctr_[pb].get(), 0, 24, s_aux_));
        launch_agg2_hist(keys, kv, c.n, AGG3_NBUCK_LOG2, AGG3_GRID_LOG2,
                         d_pcounts_[pb].get<uint32_t>(),
                         (uint32_t*)(d_pctr_[pb].get<uint8_t>() + 16), 1024,
                         s_aux_);
        launch_agg3_line_sizes(d_pcounts_[pb].get<uint32_t>(), mat3 + 1,
                               d_plinesz_[pb].get<uint32_t>(), 24, s_aux_);
        size_t tb = d_pscan_tmp_.size();
        scan_counts_matrix(d_plinesz_[pb].get<uint32_t>(),
                           d_pscanned_[pb].get<uint32_t>(), mat3 + 1,
                           d_pscan_tmp_.get(), &tb, s_aux_);
        launch_agg3_scatter(keys, kv, vals, vv, c.n, AGG3_NBUCK_LOG2,
                            AGG3_GRID_LOG2, d_pscanned_[pb].get<uint32_t>(),
                            d_ppart_[pb].get<uint8_t>(),
                            d_pleft_[pb].get<PartRow>(),
                            d_pctr_[pb].get<unsigned long long>() + 1,
                            d_plinesz_[pb].get<uint32_t>(),
                            t_.error_flag, 24, 0, s_aux_);
        AURON_HIP(hipEventRecord(ev_scatter_[pb], s_aux_));
      }
      if (k > 0) {
        // merges for chunk k-1; the readback completes while scatter(k)
        // runs on the aux stream
        const Chunk& c = chunks[k - 1];
        int pb = (int)((k - 1) & 1);
        AURON_HIP(hipEventSynchronize(ev_read_[pb]));
        unsigned long long* h = pin_pctr_[pb].get<unsigned long long>();
        int64_t staged_n = (int64_t)h[0];
        int64_t lo_n = (int64_t)h[1];
        uint32_t special_rows = (uint32_t)h[2];
        ng_true_ = h[4];
        ng_bound_ = ng_true_ + (special_rows ? 2u : 0u);
        if (special_rows) {
          const int64_t* keys = (const int64_t*)keyc.values + c.beg;
          const double* vals = (const double*)valc.values + c.beg;
          const uint8_t* kv =
              keyc.validity ? keyc.validity + c.beg / 8 : nullptr;
          const uint8_t* vv =
              valc.validity ? valc.validity + c.beg / 8 : nullptr;
          launch_agg2_specials(t_, keys, kv, vals, vv, c.n, c.row0, stream_);
        }
        // capacity: exact counted bounds; growth forces a full pipeline sync
        if (t_.cap * 3 / 4 < (int64_t)ng_bound_ + staged_n + lo_n) {
          AURON_HIP(hipStreamSynchronize(s_aux_));
          ensure_capacity((int64_t)ng_bound_ + staged_n + lo_n);
        }
        launch_agg2_merge_groups(t_, d_staged_.get<StagedGroup>(), staged_n,
                                 c.row0, stream_);
        if (lo_n)
          launch_agg2_leftovers(t_, d_pleft_[pb].get<PartRow>(), lo_n,
                                c.row0, stream_);
        AURON_HIP(hipEventRecord(ev_merge_[pb], stream_));
        ng_bound_ += (uint64_t)(staged_n + lo_n);
        update_rows_ += c.n;
        row_cursor_ += (uint64_t)c.n;
      }
      if (k < chunks.size()) {
        // engine stream: bucket(k) after scatter(k); stream order places it
        // after merge(k-1) as well
        int pb = (int)(k & 1);
        AURON_HIP(hipStreamWaitEvent(stream_, ev_scatter_[pb], 0));
        launch_agg3_bucket(d_ppart_[pb].get<uint8_t>(),
                           d_pcounts_[pb].get<uint32_t>(),
                           d_plinesz_[pb].get<uint32_t>(),
                           d_pscanned_[pb].get<uint32_t>(), AGG3_GRID_LOG2,
                           val_is_int_ ? 1 : 0, nbuck,
                           d_staged_.get<StagedGroup>(),
                           d_pctr_[pb].get<unsigned long long>(),
                           (int64_t)nbuck * 4096, d_pleft_[pb].get<PartRow>(),
                           d_pctr_[pb].get<unsigned long long>() + 1,
                           t_.error_flag, 24, 0, stream_);
        AURON_HIP(hipMemcpyAsync(pin_pctr_[pb].get(), d_pctr_[pb].get(), 24,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipMemcpyAsync(pin_pctr_[pb].get<uint8_t>() + 32,
                                 t_.num_groups, 8, hipMemcpyDeviceToHost,
                                 stream_));
        AURON_HIP(hipEventRecord(ev_read_[pb], stream_));
      }
    }
    AURON_HIP(hipEventRecord(e1, stream_));
    ev_pairs_.push_back({e0, e1});
    return chunks.empty() ? done : chunks.back().beg + chunks.back().n;
  }

  void init_pipeline_scratch() {
    if (d_ppart_[0]) return;
    init_agg2_conf();
    const int nbuck = AGG3_NBUCK;
    int64_t mat3 = (int64_t)nbuck << AGG3_GRID_LOG2;
    int64_t part_bytes =
        agg2_chunk_max_ * 24 + ((int64_t)nbuck << AGG3_GRID_LOG2) * 64;
    for (int i = 0; i < 2; i++) {
      d_ppart_[i].alloc(part_bytes);
      d_pleft_[i].alloc(agg2_chunk_max_ * sizeof(PartRow));
      d_pcounts_[i].alloc((mat3 + 1) * 4);
      d_plinesz_[i].alloc((mat3 + 1) * 4);
      d_pscanned_[i].alloc((mat3 + 1) * 4);
      d_pctr_[i].alloc(24);
      pin_pctr_[i].alloc(64);
      AURON_HIP(hipEventCreate(&ev_scatter_[i]));
      AURON_HIP(hipEventCreate(&ev_merge_[i]));
      AURON_HIP(hipEventCreate(&ev_read_[i]));
    }
    AURON_HIP(hipEventCreate(&ev_pipe_[0]));
    size_t tb = 0;
    scan_counts_matrix(d_plinesz_[0].get<uint32_t>(),
                       d_pscanned_[0].get<uint32_t>(), mat3 + 1, nullptr,
                       &tb, stream_);
    d_pscan_tmp_.alloc(tb);
    if (!d_staged_)
      d_staged_.alloc((int64_t)nbuck * 4096 * sizeof(StagedGroup));
    if (!d_leftover_) d_leftover_.alloc(agg2_chunk_max_ * sizeof(PartRow));
    AURON_HIP(hipStreamCreateWithFlags(&s_aux_, hipStreamNonBlocking));
  }

  // Two-phase aggregation of rows [done, done+chunk) of batch b:
  // histogram -> scatter to bucket-major SoA -> per-bucket LDS aggregate ->
  // merge counted staged groups into the slot table (kernels_agg2.hip).
  void two_phase_chunk(const DevBatch& b, int64_t done, int64_t chunk) {
    const int nbuck = 1 << AGG2_NBUCK_LOG2;
    const DevColumn& key = b.cols.at(key_col_);
    const DevColumn& val = b.cols.at(val_col_);
    const int64_t* keys = (const int64_t*)key.values + done;
    const double* vals = (const double*)val.values + done;
    const uint8_t* kv = key.validity ? key.validity + done / 8 : nullptr;
    const uint8_t* vv = val.validity ? val.validity + done / 8 : nullptr;

    if (!d_partkv_) {
      if (agg2_v3_) {
        // 24B records + one line of alignment padding per (block,bucket)
        d_partkv_.alloc(agg2_chunk_max_ * 24 +
                        ((int64_t)AGG3_NBUCK << AGG3_GRID_LOG2) * 64);
      } else if (agg2_split_) {
        d_partkv_.alloc(agg2_chunk_max_ * sizeof(PartKV));
        d_rowv_.alloc(agg2_chunk_max_ * 4);
      } else {
        d_partkv_.alloc(agg2_chunk_max_ * sizeof(PartRow));
      }
      d_leftover_.alloc(agg2_chunk_max_ * sizeof(PartRow));
      int64_t mat = (int64_t)nbuck << AGG2_GRID_LOG2_MAX;
      d_counts_.alloc((mat + 1) * 4);   // +1: scan total slot
      d_scanned_.alloc((mat + 1) * 4);
      d_offsets_.alloc((nbuck + 1) * 4);
      if (agg2_v3_) d_linesz_.alloc((mat + 1) * 4);
      size_t tb = 0;
      scan_counts_matrix(d_counts_.get<uint32_t>(), d_scanned_.get<uint32_t>(),
                         mat + 1, nullptr, &tb, stream_);
      d_scan_tmp_.alloc(tb);
      d_staged_.alloc(std::max<int64_t>((int64_t)nbuck * AGG2_LSLOTS,
                                        (int64_t)AGG3_NBUCK * 4096) *
                      sizeof(StagedGroup));
      d_counters_.alloc(24);  // staged_n, lo_n, special_rows
      // layout: counts[nbuck+1] | offs[nbuck+1] | counters[2] (8-aligned)
      pinned_agg2_.alloc(2 * (size_t)(nbuck + 1) * 4 + 64);
    }
    hipEvent_t e0, e1;
    AURON_HIP(hipEventCreate(&e0));
    AURON_HIP(hipEventCreate(&e1));
    AURON_HIP(hipEventRecord(e0, stream_));
    // P1: per-block histogram matrix (+ special-row count). ONE host sync
    // per chunk: the scan total feeds the offsets kernel on-device, and the
    // special/staged/leftover counters are read back together after phase A.
    AURON_HIP(hipMemsetAsync(d_counters_.get(), 0, 24, stream_));
    if (agg2_v3_) {
      // v3: hist -> line-padded scan -> LDS-staged packet scatter -> 4096-
      // slot bucket aggregation (kernels_agg3.hip)
      int64_t mat3 = (int64_t)AGG3_NBUCK << AGG3_GRID_LOG2;
      // adaptive packed-16B records: chunk 1 runs the 24B layout while the
      // hist probes the normal-key min/max; if the observed range fits u32,
      // later chunks store (key - base) u32 — 33% less partition traffic.
      // Out-of-range keys in later chunks take the counted leftover bypass,
      // so the choice is a pure optimization, never a correctness bet.
      if (agg2_p16_en_ && !agg2_v4_ && !p16_decided_) {
        // one-time key-range probe (8 B/row) so even the FIRST chunk runs
        // the packed layout; ~0.35 ms vs ~2.8 ms saved on a 256M chunk
        if (!d_kminmax_) d_kminmax_.alloc(16);
        AURON_HIP(hipMemsetAsync(d_kminmax_.get(), 0xFF, 8, stream_));
        AURON_HIP(hipMemsetAsync(d_kminmax_.get<uint8_t>() + 8, 0, 8,
                                 stream_));
        launch_keys_minmax(keys, kv, chunk,
                           d_kminmax_.get<unsigned long long>(), stream_);
        uint64_t h_kmm[2];
        AURON_HIP(hipMemcpyAsync(h_kmm, d_kminmax_.get(), 16,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        p16_decided_ = true;
        if (h_kmm[0] <= h_kmm[1] && h_kmm[1] - h_kmm[0] <= 0xFFFFFFFFull) {
          packed16_ = true;
          key_base16_ = (int64_t)(h_kmm[0] ^ 0x8000000000000000ull);
          DBG("agg.p16 enabled: key range %llu",
              (unsigned long long)(h_kmm[1] - h_kmm[0]));
        }
      }
      const int rec3 = (packed16_ && !agg2_v4_) ? 16 : 24;
      // the v4 workers cover the same 1024-wide virtual-lane tiles as the
      // 1024-thread hist, so the hist launch is identical for v3 and v4
      launch_agg2_hist(keys, kv, chunk, AGG3_NBUCK_LOG2, AGG3_GRID_LOG2,
                       d_counts_.get<uint32_t>(),
                       (uint32_t*)(d_counters_.get<uint8_t>() + 16), 1024,
                       stream_);
      launch_agg3_line_sizes(d_counts_.get<uint32_t>(), mat3 + 1,
                             d_linesz_.get<uint32_t>(), rec3, stream_);
      size_t tb3 = d_scan_tmp_.size();
      scan_counts_matrix(d_linesz_.get<uint32_t>(),
                         d_scanned_.get<uint32_t>(), mat3 + 1,
                         d_scan_tmp_.get(), &tb3, stream_);
      if (agg2_v4_)
        launch_agg4_scatter(keys, kv, vals, vv, chunk, AGG3_NBUCK_LOG2,
                            AGG3_GRID_LOG2, d_scanned_.get<uint32_t>(),
                            d_partkv_.get<uint8_t>(),
                            d_leftover_.get<PartRow>(),
                            d_counters_.get<unsigned long long>() + 1,
                            d_linesz_.get<uint32_t>(), t_.error_flag, stream_);
      else
        launch_agg3_scatter(keys, kv, vals, vv, chunk, AGG3_NBUCK_LOG2,
                            AGG3_GRID_LOG2, d_scanned_.get<uint32_t>(),
                            d_partkv_.get<uint8_t>(),
                            d_leftover_.get<PartRow>(),
                            d_counters_.get<unsigned long long>() + 1,
                            d_linesz_.get<uint32_t>(),  // free after the scan
                            t_.error_flag, rec3, key_base16_, stream_);
      launch_agg3_bucket(d_partkv_.get<uint8_t>(), d_counts_.get<uint32_t>(),
                         d_linesz_.get<uint32_t>(),
                         d_scanned_.get<uint32_t>(), AGG3_GRID_LOG2,
                         val_is_int_ ? 1 : 0, AGG3_NBUCK,
                         d_staged_.get<StagedGroup>(),
                         d_counters_.get<unsigned long long>(),
                         (int64_t)AGG3_NBUCK * 4096,
                         d_leftover_.get<PartRow>(),
                         d_counters_.get<unsigned long long>() + 1,
                         t_.error_flag, rec3, key_base16_, stream_);
    } else {
    int64_t mat = (int64_t)nbuck << agg2_grid_log2_;
    launch_agg2_hist(keys, kv, chunk, AGG2_NBUCK_LOG2, agg2_grid_log2_,
                     d_counts_.get<uint32_t>(),
                     (uint32_t*)(d_counters_.get<uint8_t>() + 16), agg2_block_,
                     stream_);
    size_t tb = d_scan_tmp_.size();
    scan_counts_matrix(d_counts_.get<uint32_t>(), d_scanned_.get<uint32_t>(),
                       mat + 1, d_scan_tmp_.get(), &tb, stream_);
    launch_agg2_offsets(d_scanned_.get<uint32_t>(), AGG2_NBUCK_LOG2,
                        agg2_grid_log2_, d_offsets_.get<uint32_t>(), stream_);
    // P2: scatter
    if (agg2_split_)
      launch_agg2_scatter(keys, kv, vals, vv, chunk, AGG2_NBUCK_LOG2,
                          agg2_grid_log2_, d_scanned_.get<uint32_t>(),
                          d_partkv_.get<PartKV>(), d_rowv_.get<uint32_t>(),
                          agg2_block_, stream_);
    else
      launch_agg2_scatter24(keys, kv, vals, vv, chunk, AGG2_NBUCK_LOG2,
                            agg2_grid_log2_, d_scanned_.get<uint32_t>(),
                            d_partkv_.get<PartRow>(), agg2_block_, stream_);
    // A: per-bucket LDS aggregation (counters zeroed above; staged_n at +0,
    // lo_n at +1, special count at byte offset 8 via the +2 uint32 slot)
    if (agg2_split_)
      launch_agg2_bucket(d_partkv_.get<PartKV>(), d_rowv_.get<uint32_t>(),
                         d_offsets_.get<uint32_t>(), val_is_int_ ? 1 : 0,
                         nbuck, d_staged_.get<StagedGroup>(),
                         d_counters_.get<unsigned long long>(),
                         (int64_t)nbuck * AGG2_LSLOTS,
                         d_leftover_.get<PartRow>(),
                         d_counters_.get<unsigned long long>() + 1,
                         t_.error_flag, agg2_block_, stream_);
    else
      launch_agg2_bucket24(d_partkv_.get<PartRow>(),
                           d_offsets_.get<uint32_t>(), val_is_int_ ? 1 : 0,
                           nbuck, d_staged_.get<StagedGroup>(),
                           d_counters_.get<unsigned long long>(),
                           (int64_t)nbuck * AGG2_LSLOTS,
                           d_leftover_.get<PartRow>(),
                           d_counters_.get<unsigned long long>() + 1,
                           t_.error_flag, agg2_block_, stream_);
    }
    unsigned long long* h_ctr =
        (unsigned long long*)(pinned_agg2_.get<uint8_t>() +
                              2 * (size_t)(nbuck + 1) * 4);
    AURON_HIP(hipMemcpyAsync(h_ctr, d_counters_.get(), 24,
                             hipMemcpyDeviceToHost, stream_));
    uint64_t* h_ng = (uint64_t*)(h_ctr + 3);
    AURON_HIP(hipMemcpyAsync(h_ng, t_.num_groups, 8, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    int64_t staged_n = (int64_t)h_ctr[0];
    int64_t lo_n = (int64_t)h_ctr[1];
    uint32_t special_rows = (uint32_t)h_ctr[2];
    if (special_rows)
      launch_agg2_specials(t_, keys, kv, vals, vv, chunk, row_cursor_, stream_);
    // true cardinality from the piggybacked pre-merge read (no extra sync);
    // the piecewise loops below grow the bound as they merge
    ng_true_ = *h_ng;
    ng_bound_ = *h_ng + (special_rows ? 2u : 0u);
    // merge: table inserts bounded by the COUNTED lists; piecewise so a
    // VRAM budget (spill) smaller than the staged list still works
    for (int64_t done2 = 0; done2 < staged_n;) {
      ensure_capacity((int64_t)ng_bound_ + std::min<int64_t>(
                                               staged_n - done2, 1 << 20));
      int64_t free2 = t_.cap * 3 / 4 - (int64_t)ng_bound_;
      if (free2 < (1 << 14)) {
        refresh_ng();
        free2 = t_.cap * 3 / 4 - (int64_t)ng_true_;
        if (free2 < (1 << 14)) {
          spill_table();
          continue;
        }
      }
      int64_t piece = std::min(staged_n - done2, free2);
      launch_agg2_merge_groups(t_, d_staged_.get<StagedGroup>() + done2, piece,
                               row_cursor_, stream_);
      done2 += piece;
      ng_bound_ += (uint64_t)piece;
    }
    for (int64_t done2 = 0; done2 < lo_n;) {
      ensure_capacity((int64_t)ng_bound_ +
                      std::min<int64_t>(lo_n - done2, 1 << 20));
      int64_t free2 = t_.cap * 3 / 4 - (int64_t)ng_bound_;
      if (free2 < (1 << 14)) {
        refresh_ng();
        free2 = t_.cap * 3 / 4 - (int64_t)ng_true_;
        if (free2 < (1 << 14)) {
          spill_table();
          continue;
        }
      }
      int64_t piece = std::min(lo_n - done2, free2);
      launch_agg2_leftovers(t_, d_leftover_.get<PartRow>() + done2, piece,
                            row_cursor_, stream_);
      done2 += piece;
      ng_bound_ += (uint64_t)piece;
    }
    AURON_HIP(hipEventRecord(e1, stream_));
    ev_pairs_.push_back({e0, e1});
    update_rows_ += chunk;
    DBG("agg.2phase chunk=%lld staged=%lld leftover=%lld special=%u",
        (long long)chunk, (long long)staged_n, (long long)lo_n, special_rows);
  }

  static uint64_t host_mix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
  }

  struct SpillBucket {
    std::vector<int64_t> keys;
    std::vector<unsigned long long> first_rows;
    std::vector<int32_t> lens;
    std::vector<uint8_t> data;
  };

  // compact the table and return (order_slots sorted by first_row, ng);
  // device buffers owned by the out-params
  int64_t table_order(DevBuf* slots_sorted, DevBuf* first_sorted) {
    uint64_t ng = num_groups_host();
    if (ng == 0) return 0;
    DevBuf slots_u(ng * 4), first(ng * 8), dcount(8);
    slots_sorted->alloc(ng * 4);
    first_sorted->alloc(ng * 8);
    AURON_HIP(hipMemsetAsync(dcount.get(), 0, 8, stream_));
    launch_agg_compact(t_, slots_u.get<uint32_t>(),
                       first.get<unsigned long long>(),
                       dcount.get<unsigned long long>(), stream_);
    size_t tmp_bytes = 0;
    sort_pairs_u64_u32(first.get<unsigned long long>(), slots_u.get<uint32_t>(),
                       first_sorted->get<unsigned long long>(),
                       slots_sorted->get<uint32_t>(), (int64_t)ng, nullptr,
                       &tmp_bytes, stream_);
    DevBuf tmp(tmp_bytes);
    sort_pairs_u64_u32(first.get<unsigned long long>(), slots_u.get<uint32_t>(),
                       first_sorted->get<unsigned long long>(),
                       slots_sorted->get<uint32_t>(), (int64_t)ng, tmp.get(),
                       &tmp_bytes, stream_);
    // pooled buffers (slots_u/first/tmp) must not be recycled while the sort
    // still reads them — the pool, unlike hipFree, does not synchronize
    AURON_HIP(hipStreamSynchronize(stream_));
    return (int64_t)ng;
  }

  // freeze the main-region groups to host spill buckets and reset the main
  // region (the two special groups stay resident across spills)
  void spill_table() {
    drain_timing();
    // COLLECT pool drain (collect.rs spill analog): sort the pool so the
    // freeze kernels can binary-search each group's item run — the frozen
    // records then CARRY the items into the spill buckets; afterwards the
    // pool is reset to the resident specials' items only
    prepare_collect();
    DevBuf order, first;
    int64_t ng = table_order(&order, &first);
    if (ng == 0) return;
    std::vector<uint32_t> h_order(ng);
    std::vector<unsigned long long> h_first(ng);
    AURON_HIP(hipMemcpyAsync(h_order.data(), order.get(), ng * 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipMemcpyAsync(h_first.data(), first.get(), ng * 8,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    // filter out the special slots (they stay in the table)
    std::vector<uint32_t> main_order;
    std::vector<unsigned long long> main_first;
    for (int64_t i = 0; i < ng; i++) {
      if (h_order[i] < (uint32_t)t_.cap) {
        main_order.push_back(h_order[i]);
        main_first.push_back(h_first[i]);
      }
    }
    int64_t n = (int64_t)main_order.size();
    specials_count_ = ng - n;
    if (n > 0) {
      DevBuf d_order(n * 4), keys(n * 8), lens(n * 4), offs((n + 1) * 4);
      AURON_HIP(hipMemcpyAsync(d_order.get(), main_order.data(), n * 4,
                               hipMemcpyHostToDevice, stream_));
      launch_agg_gather_out(t_, d_order.get<uint32_t>(), n,
                            keys.get<int64_t>(), nullptr, nullptr, nullptr,
                            nullptr, nullptr, nullptr, nullptr, nullptr,
                            stream_);
      launch_agg_freeze_len(t_, d_order.get<uint32_t>(), n, lens.get<int32_t>(),
                            layout_, stream_);
      std::vector<int32_t> h_lens(n);
      std::vector<int64_t> h_keys(n);
      AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipMemcpyAsync(h_keys.data(), keys.get(), n * 8,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      std::vector<int32_t> h_offs(n + 1, 0);
      for (int64_t i = 0; i < n; i++) h_offs[i + 1] = h_offs[i] + h_lens[i];
      AURON_HIP(hipMemcpyAsync(offs.get(), h_offs.data(), (n + 1) * 4,
                               hipMemcpyHostToDevice, stream_));
      DevBuf data(h_offs[n] ? h_offs[n] : 1);
      launch_agg_freeze_write(t_, d_order.get<uint32_t>(), n,
                              offs.get<int32_t>(), data.get<uint8_t>(), layout_,
                              stream_);
      std::vector<uint8_t> h_data(h_offs[n]);
      if (h_offs[n])
        AURON_HIP(hipMemcpyAsync(h_data.data(), data.get(), h_offs[n],
                                 hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      if (spill_.empty()) spill_.resize(num_spill_buckets_);
      for (int64_t i = 0; i < n; i++) {
        SpillBucket& b =
            spill_[host_mix64((uint64_t)h_keys[i]) % num_spill_buckets_];
        b.keys.push_back(h_keys[i]);
        b.first_rows.push_back(main_first[i]);
        b.lens.push_back(h_lens[i]);
        b.data.insert(b.data.end(), h_data.begin() + h_offs[i],
                      h_data.begin() + h_offs[i + 1]);
      }
      spill_count_++;
    }
    reset_main();
    if (has_coll_) reset_collect_pool();
    DBG("agg.spill n=%lld buckets=%d", (long long)n, num_spill_buckets_);
  }

  // Rebuild the pool to hold exactly the two RESIDENT special groups' items
  // (taken from the sorted views, order preserved; re-assigned prios 0..m
  // stay below every future row/record priority) and restore the original
  // pool arrays as the append target.
  void reset_collect_pool() {
    if (!has_coll_ || !coll_sorted_) return;
    uint32_t spec[2] = {(uint32_t)t_.cap, (uint32_t)(t_.cap + 1)};
    DevBuf d_spec(8), cnts(8), items;
    AURON_HIP(hipMemcpyAsync(d_spec.get(), spec, 8, hipMemcpyHostToDevice,
                             stream_));
    launch_coll_counts(t_, d_spec.get<uint32_t>(), 2, cnts.get<int32_t>(),
                       stream_);
    int32_t h_cnts[2];
    AURON_HIP(hipMemcpyAsync(h_cnts, cnts.get(), 8, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    int64_t m0 = h_cnts[0], m1 = h_cnts[1];
    int32_t h_offs[3] = {0, (int32_t)m0, (int32_t)(m0 + m1)};
    DevBuf d_offs(12);
    AURON_HIP(hipMemcpyAsync(d_offs.get(), h_offs, 12, hipMemcpyHostToDevice,
                             stream_));
    items.alloc((m0 + m1) * 8 + 8);
    launch_coll_gather(t_, d_spec.get<uint32_t>(), 2, d_offs.get<int32_t>(),
                       items.get<unsigned long long>(), stream_);
    // restore the ORIGINAL pool arrays as the live pool, then refill
    t_.c_key = d_ckey_.get<long long>();
    t_.c_prio = d_cprio_.get<unsigned long long>();
    t_.c_val = d_cval_.get<unsigned long long>();
    t_.c_cap = coll_cap_;
    launch_coll_refill(t_, items.get<unsigned long long>(), m0, m1, stream_);
    unsigned long long nn[2] = {(unsigned long long)m0,
                                (unsigned long long)m1};
    AURON_HIP(hipMemcpyAsync(d_cn_.get(), nn, 16, hipMemcpyHostToDevice,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    coll_sorted_ = false;
    coll_n0_ = coll_n1_ = 0;
    d_cskey_ = DevBuf();
    d_csval_ = DevBuf();
  }

  void reset_main() {
    launch_slots_init(t_.slots, t_.cap, stream_);  // specials untouched
    if (has_mm_) launch_mm_init(t_.mm, t_.cap, stream_);
    if (has_first_) launch_first_init(t_.f_row, t_.f_val, t_.f_st, t_.cap,
                                      stream_);
    uint64_t ng0 = (uint64_t)specials_count_;
    AURON_HIP(hipMemcpyAsync(t_.num_groups, &ng0, 8, hipMemcpyHostToDevice,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    ng_true_ = ng0;
    ng_bound_ = ng0;
  }

  void refresh_ng() {
    ng_true_ = num_groups_host();
    ng_bound_ = ng_true_;
  }

  void drain_timing() {
    if (ev_pairs_.empty()) return;
    AURON_HIP(hipStreamSynchronize(stream_));
    for (auto& [e0, e1] : ev_pairs_) {
      float ms = 0.f;
      AURON_HIP(hipEventElapsedTime(&ms, e0, e1));
      update_ns_ += (int64_t)(ms * 1e6);
      (void)hipEventDestroy(e0);
      (void)hipEventDestroy(e1);
    }
    ev_pairs_.clear();
  }

  void ensure_capacity(int64_t need) {
    while (t_.cap * 3 / 4 < need && t_.cap < max_cap_) grow(t_.cap * 4);
    if (t_.cap * 3 / 4 < need &&
        (int64_t)num_groups_host() > specials_count_) {
      if (gkey_)
        FAIL("spill with Utf8/multi-column grouping keys unsupported "
             "(raise AURON_HIP_MEM_BUDGET)");
      if (ma_mode_)
        FAIL("spill with multi-argument aggregates unsupported "
             "(raise AURON_HIP_MEM_BUDGET)");
      spill_table();  // frees the main region; caller re-checks free space
    }
  }

  void grow(int64_t new_cap) {
    AggTable old = t_;
    GKeyTable oldg = g_;
    DevBuf oslots = std::move(d_slots_), os = std::move(d_special_),
           ong = std::move(d_ng_), oerr = std::move(d_err_),
           omm = std::move(d_mm_), ofrow = std::move(d_frow_),
           ofval = std::move(d_fval_), ofst = std::move(d_fst_),
           ogkoff = std::move(d_gkoff_);
    MaAcc oldma = ma_;
    DevBuf oma = std::move(d_ma_acc_), omam = std::move(d_ma_meta_),
           omas = std::move(d_ma_st_);
    init_table(new_cap);
    if (ma_mode_)
      launch_ma_rebuild(t_, ma_, old, oldma, ma_desc_.n, stream_);
    else if (gkey_)
      launch_gkey_rebuild(t_, g_, old, oldg, stream_);
    else
      launch_agg_rebuild(t_, old, stream_);
    AURON_HIP(hipStreamSynchronize(stream_));
  }

  // emit every group in the table, first-occurrence-ordered, in
  // batch_size_-row output chunks
  void emit_table(std::vector<std::pair<int64_t, std::vector<HostOutCol>>>* out,
                  bool exclude_specials) {
    prepare_collect();  // one-shot per pool state (coll_sorted_ guard)
    prepare_ma_pools();
    DevBuf order, first;
    int64_t ng = table_order(&order, &first);
    DBG("agg.finish ng=%lld excl=%d", (long long)ng, (int)exclude_specials);
    if (ng == 0) return;
    DevBuf filtered;
    const uint32_t* order_ptr = order.get<uint32_t>();
    if (exclude_specials) {
      std::vector<uint32_t> h_order(ng);
      AURON_HIP(hipMemcpyAsync(h_order.data(), order.get(), ng * 4,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      std::vector<uint32_t> keep;
      for (int64_t i = 0; i < ng; i++)
        if (h_order[i] < (uint32_t)t_.cap) keep.push_back(h_order[i]);
      ng = (int64_t)keep.size();
      if (ng == 0) return;
      filtered.alloc(ng * 4);
      AURON_HIP(hipMemcpyAsync(filtered.get(), keep.data(), ng * 4,
                               hipMemcpyHostToDevice, stream_));
      order_ptr = filtered.get<uint32_t>();
    }
    for (int64_t beg = 0; beg < ng; beg += batch_size_) {
      int64_t len = std::min(batch_size_, ng - beg);
      out->emplace_back(emit_groups(order_ptr + beg, len));
    }
  }

  // D2H through a reusable pinned staging buffer (pageable D2H costs ms on
  // this stack); dst vector is filled from the pinned copy
  void d2h_pinned(void* dev, std::vector<uint8_t>* dst, size_t len) {
    if (pinned_emit_.size() < len) pinned_emit_.alloc(len);
    AURON_HIP(hipMemcpyAsync(pinned_emit_.get(), dev, len,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    dst->resize(len);
    memcpy(dst->data(), pinned_emit_.get(), len);
  }

  // decode the generalized-key grouping columns for the ordered groups
  std::vector<HostOutCol> emit_gkey_cols(const uint32_t* order_slots,
                                         int64_t n) {
    std::vector<HostOutCol> out;
    size_t bm = (n + 7) / 8;
    for (size_t k = 0; k < key_cols_.size(); k++) {
      HostOutCol c;
      c.dt = key_dts_[k];
      DevBuf bmbuf(bm);
      if (c.dt == DType::Utf8 || c.dt == DType::Binary) {
        DevBuf lens(n * 4);
        launch_gkey_out_lens(g_, order_slots, n, d_gkdts_.get<uint8_t>(),
                             (int)key_cols_.size(), (int)k,
                             lens.get<uint32_t>(), bmbuf.get<uint8_t>(),
                             stream_);
        std::vector<uint32_t> h_lens(n);
        AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        c.offsets.assign(n + 1, 0);
        for (int64_t i = 0; i < n; i++)
          c.offsets[i + 1] = c.offsets[i] + (int32_t)h_lens[i];
        DevBuf d_off((n + 1) * 4), data(c.offsets[n] ? c.offsets[n] : 1);
        AURON_HIP(hipMemcpyAsync(d_off.get(), c.offsets.data(), (n + 1) * 4,
                                 hipMemcpyHostToDevice, stream_));
        launch_gkey_out_bytes(g_, order_slots, n, d_gkdts_.get<uint8_t>(),
                              (int)key_cols_.size(), (int)k,
                              d_off.get<int32_t>(), data.get<uint8_t>(),
                              stream_);
        d2h_pinned(data.get(), &c.values, c.offsets[n]);
      } else {
        size_t w = dtype_width(c.dt);
        DevBuf vals(n * w);
        launch_gkey_out_fixed(g_, order_slots, n, d_gkdts_.get<uint8_t>(),
                              (int)key_cols_.size(), (int)k,
                              vals.get<uint8_t>(), bmbuf.get<uint8_t>(),
                              stream_);
        d2h_pinned(vals.get(), &c.values, n * w);
      }
      std::vector<uint8_t> h_bm(bm);
      AURON_HIP(hipMemcpyAsync(h_bm.data(), bmbuf.get(), bm,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      attach_validity(&c, h_bm, n);
      out.push_back(std::move(c));
    }
    return out;
  }

  // multi-arg emit: per-agg columns from the accumulator bank; partial mode
  // freezes the per-agg wire instead (ma_freeze_*). Collect aggs read their
  // sorted pool views through the legacy coll kernels (the AggTable's c_*
  // pointers are temporarily re-aimed per pool).
  void emit_ma_aggs(const uint32_t* order_slots, int64_t n,
                    std::vector<HostOutCol>* cols) {
    size_t bm = (n + 7) / 8;
    if (!final_output_) {
      // partial: ONE Binary agg-buf column with the per-agg wire
      DevBuf lens(n * 4), offs((n + 1) * 4);
      launch_ma_freeze_len(t_, ma_desc_, ma_, mp_, order_slots, n,
                           lens.get<int32_t>(), stream_);
      std::vector<int32_t> h_lens(n);
      AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      std::vector<int32_t> h_offs(n + 1, 0);
      for (int64_t i = 0; i < n; i++) h_offs[i + 1] = h_offs[i] + h_lens[i];
      AURON_HIP(hipMemcpyAsync(offs.get(), h_offs.data(), (n + 1) * 4,
                               hipMemcpyHostToDevice, stream_));
      DevBuf data(h_offs[n] ? h_offs[n] : 1);
      launch_ma_freeze_write(t_, ma_desc_, ma_, mp_, order_slots, n,
                             offs.get<int32_t>(), data.get<uint8_t>(),
                             stream_);
      HostOutCol buf_col;
      buf_col.dt = DType::Binary;
      buf_col.offsets = std::move(h_offs);
      d2h_pinned(data.get(), &buf_col.values, (size_t)buf_col.offsets[n]);
      cols->push_back(std::move(buf_col));
      return;
    }
    for (int j = 0; j < ma_desc_.n; j++) {
      const MaAgg& ag = ma_desc_.a[j];
      HostOutCol ac;
      if (ag.kind == AGGL_CLIST || ag.kind == AGGL_CSET) {
        // re-aim the legacy collect view at this pool's sorted arrays
        AggTable tv = t_;
        uint8_t p = ag.pool;
        tv.c_key = mp_.key[p];
        tv.c_val = mp_.val[p];
        tv.c_n = d_mp_n_.get<unsigned long long>() + 2 * p;
        tv.c_cap = mp_counts_[p][0] + mp_counts_[p][1];
        DevBuf ccnt(n * 4);
        launch_coll_counts(tv, order_slots, n, ccnt.get<int32_t>(), stream_);
        std::vector<int32_t> h_cnts(n);
        AURON_HIP(hipMemcpyAsync(h_cnts.data(), ccnt.get(), n * 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        ac.offsets.assign(n + 1, 0);
        for (int64_t i = 0; i < n; i++)
          ac.offsets[i + 1] = ac.offsets[i] + h_cnts[i];
        DevBuf d_off((n + 1) * 4),
            items((int64_t)ac.offsets[n] * 8 + 8);
        AURON_HIP(hipMemcpyAsync(d_off.get(), ac.offsets.data(),
                                 (n + 1) * 4, hipMemcpyHostToDevice,
                                 stream_));
        launch_coll_gather(tv, order_slots, n, d_off.get<int32_t>(),
                           items.get<unsigned long long>(), stream_);
        ac.dt = ma_out_dt(j);
        ac.is_list = true;
        if (ac.dt == DType::Int32) {
          DevBuf narrow((int64_t)ac.offsets[n] * 4 + 4);
          launch_narrow_i64_i32(items.get<int64_t>(), ac.offsets[n],
                                narrow.get<int32_t>(), stream_);
          d2h_pinned(narrow.get(), &ac.values, (size_t)ac.offsets[n] * 4);
        } else {
          d2h_pinned(items.get(), &ac.values, (size_t)ac.offsets[n] * 8);
        }
        cols->push_back(std::move(ac));
        continue;
      }
      int w = (ag.kind == AGGL_CNT) ? 8 : (ag.acc_t == 2 ? 4 : 8);
      DevBuf vals(n * w), bmbuf(bm);
      launch_ma_gather_out(ma_desc_, ma_, j, order_slots, n,
                           vals.get<uint8_t>(), bmbuf.get<uint8_t>(),
                           stream_);
      ac.dt = (ag.kind == AGGL_CNT) ? DType::Int64
              : (ag.kind == AGGL_AVG) ? DType::Float64
                                      : ma_out_dt(j);
      d2h_pinned(vals.get(), &ac.values, (size_t)n * w);
      std::vector<uint8_t> h_bm(bm);
      AURON_HIP(hipMemcpyAsync(h_bm.data(), bmbuf.get(), bm,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      if (ag.kind != AGGL_CNT) attach_validity(&ac, h_bm, n);
      cols->push_back(std::move(ac));
    }
  }

  std::pair<int64_t, std::vector<HostOutCol>> emit_groups(
      const uint32_t* order_slots, int64_t n) {
    DBG("agg.emit n=%lld final=%d", (long long)n, (int)final_output_);
    std::vector<HostOutCol> cols;
    size_t bm = (n + 7) / 8;
    DevBuf keys(n * 8), kvalid(bm), sums(n * 8), svalid(bm), cnts(n * 8);
    DevBuf mins, mvalid, maxs, xvalid;
    if (has_mm_ && final_output_) {
      mins.alloc(n * 8);
      mvalid.alloc(bm);
      maxs.alloc(n * 8);
      xvalid.alloc(bm);
    }
    launch_agg_gather_out(t_, order_slots, n, keys.get<int64_t>(),
                          kvalid.get<uint8_t>(), sums.get<double>(),
                          svalid.get<uint8_t>(), cnts.get<long long>(),
                          mins.get<double>(), mvalid.get<uint8_t>(),
                          maxs.get<double>(), xvalid.get<uint8_t>(), stream_);
    HostOutCol key_col;
    key_col.dt = key_dt_ == DType::Unsupported ? DType::Int64 : key_dt_;
    if (key_col.dt == DType::Int32) {
      DevBuf k32(n * 4);
      launch_narrow_i64_i32(keys.get<int64_t>(), n, k32.get<int32_t>(),
                            stream_);
      d2h_pinned(k32.get(), &key_col.values, n * 4);
    } else {
      d2h_pinned(keys.get(), &key_col.values, n * 8);
    }
    if (ma_mode_) {
      std::vector<uint8_t> kvh(bm);
      AURON_HIP(hipMemcpyAsync(kvh.data(), kvalid.get(), bm,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      attach_validity(&key_col, kvh, n);
      cols.push_back(std::move(key_col));
      emit_ma_aggs(order_slots, n, &cols);
      return {n, std::move(cols)};
    }
    std::vector<uint8_t> kv(bm), sv(bm);
    AURON_HIP(hipMemcpyAsync(kv.data(), kvalid.get(), bm, hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipMemcpyAsync(sv.data(), svalid.get(), bm, hipMemcpyDeviceToHost,
                             stream_));
    if (final_output_) {
      std::vector<uint8_t> h_sums, h_cnts, h_avgs, h_mins, h_maxs;
      std::vector<uint8_t> mv(bm), xv(bm);
      d2h_pinned(sums.get(), &h_sums, n * 8);
      d2h_pinned(cnts.get(), &h_cnts, n * 8);
      bool need_avg = false;
      for (uint32_t k : agg_kinds_) need_avg |= (k == AGGL_AVG);
      DevBuf avgs;
      if (need_avg) {
        avgs.alloc(n * 8);
        launch_avg_div(sums.get<double>(), cnts.get<long long>(), n,
                       avgs.get<double>(), stream_);
        d2h_pinned(avgs.get(), &h_avgs, n * 8);
      }
      if (has_mm_) {
        d2h_pinned(mins.get(), &h_mins, n * 8);
        d2h_pinned(maxs.get(), &h_maxs, n * 8);
        AURON_HIP(hipMemcpyAsync(mv.data(), mvalid.get(), bm,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipMemcpyAsync(xv.data(), xvalid.get(), bm,
                                 hipMemcpyDeviceToHost, stream_));
      }
      std::vector<int32_t> h_coff;
      std::vector<uint8_t> h_citems;
      bool need_coll = false;
      for (uint32_t k : agg_kinds_)
        need_coll |= (k == AGGL_CLIST || k == AGGL_CSET);
      if (need_coll) {
        DevBuf cnts(n * 4);
        launch_coll_counts(t_, order_slots, n, cnts.get<int32_t>(), stream_);
        std::vector<int32_t> h_cnts(n);
        AURON_HIP(hipMemcpyAsync(h_cnts.data(), cnts.get(), n * 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        h_coff.assign(n + 1, 0);
        for (int64_t i = 0; i < n; i++) h_coff[i + 1] = h_coff[i] + h_cnts[i];
        DevBuf d_off((n + 1) * 4), items((int64_t)h_coff[n] * 8 + 8);
        AURON_HIP(hipMemcpyAsync(d_off.get(), h_coff.data(), (n + 1) * 4,
                                 hipMemcpyHostToDevice, stream_));
        launch_coll_gather(t_, order_slots, n, d_off.get<int32_t>(),
                           items.get<unsigned long long>(), stream_);
        d2h_pinned(items.get(), &h_citems, (size_t)h_coff[n] * 8);
      }
      std::vector<uint8_t> h_firsts[2];
      std::vector<uint8_t> fv[2] = {std::vector<uint8_t>(bm),
                                    std::vector<uint8_t>(bm)};
      if (has_first_) {
        DevBuf fvals(n * 8), fvalid(bm);
        for (int w = 0; w < 2; w++) {
          launch_first_gather(t_, order_slots, n, w, fvals.get<double>(),
                              fvalid.get<uint8_t>(), stream_);
          d2h_pinned(fvals.get(), &h_firsts[w], n * 8);
          AURON_HIP(hipMemcpyAsync(fv[w].data(), fvalid.get(), bm,
                                   hipMemcpyDeviceToHost, stream_));
          AURON_HIP(hipStreamSynchronize(stream_));
        }
      }
      AURON_HIP(hipStreamSynchronize(stream_));
      DBG("agg.emit final d2h done");
      if (gkey_) {
        for (auto& kc2 : emit_gkey_cols(order_slots, n))
          cols.push_back(std::move(kc2));
      } else {
        attach_validity(&key_col, kv, n);
        cols.push_back(std::move(key_col));
      }
      const DType vdt = val_is_int_ ? DType::Int64 : DType::Float64;
      for (uint32_t k : agg_kinds_) {
        HostOutCol ac;
        if (k == AGGL_CNT) {
          ac.dt = DType::Int64;
          ac.values = h_cnts;
        } else if (k == AGGL_MIN || k == AGGL_MAX) {
          ac.dt = vdt;
          ac.values = (k == AGGL_MIN) ? h_mins : h_maxs;
          attach_validity(&ac, (k == AGGL_MIN) ? mv : xv, n);
        } else if (k == AGGL_FIRST || k == AGGL_FIRSTIN) {
          int w = (k == AGGL_FIRSTIN) ? 1 : 0;
          ac.dt = vdt;
          ac.values = h_firsts[w];
          attach_validity(&ac, fv[w], n);
        } else if (k == AGGL_CLIST || k == AGGL_CSET) {
          ac.dt = vdt;
          ac.is_list = true;
          ac.offsets = h_coff;
          ac.values = h_citems;
        } else if (k == AGGL_AVG) {
          ac.dt = DType::Float64;
          ac.values = h_avgs;
          attach_validity(&ac, sv, n);
        } else {
          ac.dt = vdt;  // SUM output carries the accumulator type's raw bits
          ac.values = h_sums;
          attach_validity(&ac, sv, n);  // null iff cnt==0 (shared column)
        }
        cols.push_back(std::move(ac));
      }
    } else {
      // freeze (a8): lens → host scan → write
      DevBuf lens(n * 4), offs((n + 1) * 4);
      launch_agg_freeze_len(t_, order_slots, n, lens.get<int32_t>(), layout_,
                            stream_);
      std::vector<int32_t> h_lens(n);
      AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      std::vector<int32_t> h_offs(n + 1, 0);
      for (int64_t i = 0; i < n; i++) h_offs[i + 1] = h_offs[i] + h_lens[i];
      AURON_HIP(hipMemcpyAsync(offs.get(), h_offs.data(), (n + 1) * 4,
                               hipMemcpyHostToDevice, stream_));
      DevBuf data(h_offs[n] ? h_offs[n] : 1);
      launch_agg_freeze_write(t_, order_slots, n, offs.get<int32_t>(),
                              data.get<uint8_t>(), layout_, stream_);
      HostOutCol buf_col;
      buf_col.dt = DType::Binary;
      buf_col.offsets = std::move(h_offs);
      d2h_pinned(data.get(), &buf_col.values, (size_t)buf_col.offsets[n]);
      DBG("agg.emit partial freeze d2h done");
      if (gkey_) {
        cols = emit_gkey_cols(order_slots, n);
        cols.push_back(std::move(buf_col));
      } else {
        attach_validity(&key_col, kv, n);
        cols = {std::move(key_col), std::move(buf_col)};
      }
    }
    return {n, std::move(cols)};
  }

  std::pair<int64_t, std::vector<HostOutCol>> emit_skipped(const DevBatch& b) {
    int64_t n = b.num_rows;
    const DevColumn& key = b.cols.at(key_col_);
    const DevColumn& val = b.cols.at(val_col_);
    DevBuf lens(n * 4), offs((n + 1) * 4);
    launch_skip_freeze_len(val.validity, n, lens.get<int32_t>(), layout_,
                           stream_);
    std::vector<int32_t> h_lens(n);
    AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    std::vector<int32_t> h_offs(n + 1, 0);
    for (int64_t i = 0; i < n; i++) h_offs[i + 1] = h_offs[i] + h_lens[i];
    AURON_HIP(hipMemcpyAsync(offs.get(), h_offs.data(), (n + 1) * 4,
                             hipMemcpyHostToDevice, stream_));
    DevBuf data(h_offs[n] ? h_offs[n] : 1);
    launch_skip_freeze_write((const double*)val.values, val.validity, n,
                             offs.get<int32_t>(), data.get<uint8_t>(), layout_,
                             val_is_int_ ? 1 : 0, stream_);
    HostOutCol key_col, buf_col;
    key_col.dt = key.dt;
    size_t kw = dtype_width(key.dt);
    key_col.values.resize(n * kw);
    AURON_HIP(hipMemcpyAsync(key_col.values.data(), key.values, n * kw,
                             hipMemcpyDeviceToHost, stream_));
    std::vector<uint8_t> kv;
    if (key.validity) {
      kv.resize((n + 7) / 8);
      AURON_HIP(hipMemcpyAsync(kv.data(), key.validity, kv.size(),
                               hipMemcpyDeviceToHost, stream_));
    }
    buf_col.dt = DType::Binary;
    buf_col.offsets = std::move(h_offs);
    buf_col.values.resize(buf_col.offsets[n]);
    AURON_HIP(hipMemcpyAsync(buf_col.values.data(), data.get(),
                             buf_col.values.size(), hipMemcpyDeviceToHost,
                             stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    if (!kv.empty()) key_col.validity = std::move(kv);
    std::vector<HostOutCol> cols;
    cols.push_back(std::move(key_col));
    cols.push_back(std::move(buf_col));
    return {n, std::move(cols)};
  }

  static void attach_validity(HostOutCol* col, const std::vector<uint8_t>& bm,
                              int64_t n) {
    // attach only when nulls exist (Arrow null_count>0 convention, matching
    // batch_serde has_null header semantics)
    bool any_null = false;
    for (int64_t i = 0; i < n; i++)
      if (!((bm[i >> 3] >> (i & 7)) & 1)) {
        any_null = true;
        break;
      }
    if (any_null) col->validity = bm;
  }

 public:
  hipStream_t stream_;
  AggMode mode_ = AggMode::Partial;
  bool merge_mode_ = false, final_output_ = false;
  uint32_t key_col_ = 0, val_col_ = 0;
  DType key_dt_ = DType::Unsupported;
  uint32_t layout_ = 0;
  bool has_mm_ = false;  // agg list contains MIN/MAX: side mm array active
  bool has_coll_ = false, coll_sorted_ = false;  // COLLECT pool active
  bool coll_set_ = false;  // COLLECT_SET: dedup at prepare
  int64_t coll_cap_ = 0, conf_coll_cap_ = 16 << 20;
  int64_t coll_n0_ = 0, coll_n1_ = 0;
  bool val_is_int_ = false, val_typed_seen_ = false;  // i64 accumulator mode
  bool has_first_ = false;  // FIRST family: f_row/f_val/f_st arrays active
  std::vector<uint32_t> agg_kinds_;
  std::vector<std::string> agg_names_;
  std::string key_name_;
  int64_t batch_size_ = 10000;
  bool skip_enabled_ = false, skipping_ = false;
  bool device_out_ = false;
  // generalized grouping keys (Utf8 / multi-column; kernels_gkey.hip)
  bool key_mode_set_ = false, gkey_ = false;
  // multi-argument aggregation (kernels_maarg.hip)
  bool ma_mode_ = false, ma_args_checked_ = false;
  MaDesc ma_desc_;
  std::vector<int> ma_arg_cols_;   // per agg: input column or -1 (NULL lit)
  MaAcc ma_;
  MaPools mp_;
  DevBuf d_ma_acc_, d_ma_meta_, d_ma_st_, d_ma_slots_, d_mp_n_;
  DevBuf d_mp_key_[MA_MAX_POOLS], d_mp_prio_[MA_MAX_POOLS],
      d_mp_val_[MA_MAX_POOLS], d_mp_skey_[MA_MAX_POOLS],
      d_mp_sval_[MA_MAX_POOLS];
  bool ma_pools_sorted_ = false;
  int64_t mp_counts_[MA_MAX_POOLS][2] = {};
  std::vector<uint32_t> key_cols_;
  std::vector<std::string> key_names_;
  std::vector<DType> key_dts_;
  GKeyTable g_;
  int64_t gk_pool_cap_ = 0, gk_used_bound_ = 0;
  DevBuf d_gkpool_, d_gkpn_, d_gkoff_, d_gkdts_, d_enclens_, d_encoffs_,
      d_encbytes_, d_gkslots_, d_gkscan_tmp_;
  double skip_ratio_ = 0.999;
  int64_t skip_min_rows_ = 20000;
  uint64_t row_cursor_ = 0;
  uint64_t ng_bound_ = 0, ng_true_ = 0;  // conservative bound / last readback
  int64_t mem_budget_ = 8LL << 30, max_cap_ = 1 << 23;
  int num_spill_buckets_ = 16;
  int64_t specials_count_ = 0, spill_count_ = 0;
  std::vector<SpillBucket> spill_;
  int64_t update_ns_ = 0, update_rows_ = 0;
  std::vector<std::pair<hipEvent_t, hipEvent_t>> ev_pairs_;
  hipEvent_t ev_start_ = nullptr, ev_stop_ = nullptr;
  AggTable t_;
  DevBuf d_slots_, d_special_, d_ng_, d_err_, d_mm_, d_frow_, d_fval_, d_fst_;
  DevBuf d_ckey_, d_cprio_, d_cval_, d_cn_, d_cskey_, d_csval_;
  PinnedBuf pinned_meta_, pinned_emit_;
  // two-phase scratch (allocated on first large chunk)
  bool agg2_split_ = false;
  bool agg2_v3_ = true;
  bool agg2_v4_ = false;
  bool agg2_p16_en_ = true;   // AURON_AGG2_P16=0 disables the adaptive path
  bool p16_decided_ = false;  // one hist range probe (chunk 1)
  bool packed16_ = false;     // later chunks use 16B partition records
  int64_t key_base16_ = 0;
  DevBuf d_kminmax_;
  bool agg2_pipe_ = false;
  bool agg2_conf_read_ = false;
  int agg2_grid_log2_ = 9;
  int agg2_block_ = 1024;
  DevBuf d_partkv_, d_rowv_, d_leftover_, d_counts_, d_scanned_, d_scan_tmp_, d_offsets_,
      d_staged_, d_counters_, d_linesz_;
  // cross-chunk pipeline scratch (double-buffered) + aux stream
  DevBuf d_ppart_[2], d_pleft_[2], d_pcounts_[2], d_plinesz_[2],
      d_pscanned_[2], d_pctr_[2], d_pscan_tmp_;
  PinnedBuf pin_pctr_[2];
  hipStream_t s_aux_ = nullptr;
  hipEvent_t ev_scatter_[2] = {}, ev_merge_[2] = {}, ev_read_[2] = {},
             ev_pipe_[1] = {};
  PinnedBuf pinned_agg2_;
  std::vector<DevBatch> held_, skipped_;
};

// -------------------------------------------------------------- ShuffleOp --
class ShuffleOp {
 public:
  ShuffleOp(const ShuffleWriterNode& node, const Conf& conf, hipStream_t stream,
            uint32_t task_partition_id)
      : stream_(stream), node_(node), task_partition_id_(task_partition_id) {
    if (node.partitioning.kind == Repartition::Hash) {
      if (node.partitioning.hash_exprs.empty())
        FAIL("hash partitioning without exprs");
      for (const Expr& e : node.partitioning.hash_exprs) {
        if (e.kind != Expr::Column) FAIL("hash expr must be a Column");
        hash_cols_.push_back(e.col_index);
      }
    } else if (node.partitioning.kind != Repartition::Single &&
               node.partitioning.kind != Repartition::RoundRobin) {
      FAIL("unsupported partitioning kind");
    }
    if (node.partitioning.partition_count <= 0 ||
        node.partitioning.partition_count > (1 << 24))
      FAIL("ShuffleWriter: partition_count out of range");
    P_ = (uint32_t)node.partitioning.partition_count;
    batch_size_ = conf.get_i("BATCH_SIZE", 10000);
    if (batch_size_ <= 0) FAIL("invalid BATCH_SIZE conf");
    target_block_ = (size_t)conf.get_i("SHUFFLE_COMPRESSION_TARGET_BUF_SIZE",
                                       4194304);
    std::string codec = conf.get("SPARK_IO_COMPRESSION_CODEC", "lz4");
    if (codec == "lz4") codec_ = 0;
    else if (codec == "zstd") codec_ = 1;  // ipc_compression.rs:189-196
    else FAIL("unsupported shuffle codec (lz4/zstd on this path)");
    zstd_level_ = (int)conf.get_i("SPARK_IO_COMPRESSION_ZSTD_LEVEL", 1);
  }

  void consume(DevBatch&& b) {
    if (b.num_rows > 0) staged_.push_back(std::move(b));
  }

  // partition + write files; returns nothing (shuffle output goes to disk)
  void finish() {
    int64_t n = 0;
    for (const DevBatch& b : staged_) n += b.num_rows;
    std::vector<int64_t> part_offsets(P_ + 1, 0);
    if (n == 0) {
      write_files({}, part_offsets, 0);
      return;
    }
    size_t ncols = staged_[0].cols.size();
    // concat columns (device)
    DevBatch all = concat();
    std::vector<int64_t> h_offsets;
    if (node_.partitioning.kind == Repartition::Single) {
      part_offsets[1] = n;
      for (uint32_t p = 1; p <= P_; p++) part_offsets[p] = n;
      // identity permutation
      DevBuf perm(n * 4);
      launch_iota_u32(perm.get<uint32_t>(), n, stream_);
      emit(all, perm, part_offsets, ncols, n);
      return;
    }
    DevBuf part_ids(n * 4);
    if (node_.partitioning.kind == Repartition::RoundRobin) {
      // evaluate_robin_partition_ids (shuffle/mod.rs:190-202) with
      // start_rows = partition_id * 1000193 (buffered_data.rs:291-293); one
      // continuous counter over all rows == the reference's per-flush chain
      uint32_t start = (uint32_t)(((uint64_t)task_partition_id_ * 1000193ull) %
                                  (uint64_t)P_);
      launch_robin_ids(n, start, P_, part_ids.get<uint32_t>(), stream_);
    } else {
      // murmur3 seed 42 fold over hash cols (shuffle/mod.rs:163-176)
      DevBuf hashes(n * 4);
      launch_hash_init(hashes.get<int32_t>(), 42, n, stream_);
      for (uint32_t ci : hash_cols_) {
        const DevColumn& c = all.cols.at(ci);
        if (c.dt == DType::Int64)
          launch_hash_fold_i64((const int64_t*)c.values, c.validity, n,
                               hashes.get<int32_t>(), stream_);
        else if (c.dt == DType::Int32)
          launch_hash_fold_i32((const int32_t*)c.values, c.validity, n,
                               hashes.get<int32_t>(), stream_);
        else if (c.dt == DType::Utf8 || c.dt == DType::Binary)
          launch_hash_fold_bytes(c.offsets, (const uint8_t*)c.values,
                                 c.validity, n, hashes.get<int32_t>(),
                                 stream_);
        else
          FAIL("hash expr column must be Int64/Int32/Utf8 (hot-path scope)");
      }
      launch_pmod(hashes.get<int32_t>(), n, (int32_t)P_,
                  part_ids.get<uint32_t>(), stream_);
      AURON_HIP(hipStreamSynchronize(stream_));  // hashes is pooled
    }
    // histogram → host scan
    DevBuf counts(P_ * 4);
    AURON_HIP(hipMemsetAsync(counts.get(), 0, P_ * 4, stream_));
    launch_histogram(part_ids.get<uint32_t>(), n, P_, counts.get<uint32_t>(),
                     stream_);
    std::vector<uint32_t> h_counts(P_);
    AURON_HIP(hipMemcpyAsync(h_counts.data(), counts.get(), P_ * 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    for (uint32_t p = 0; p < P_; p++)
      part_offsets[p + 1] = part_offsets[p] + h_counts[p];
    // stable permutation: rocprim stable radix sort of (part_id, row)
    DevBuf iota(n * 4), keys_out(n * 4), perm(n * 4);
    launch_iota_u32(iota.get<uint32_t>(), n, stream_);
    int end_bit = 1;
    while ((1u << end_bit) < P_) end_bit++;
    size_t tmp_bytes = 0;
    sort_pairs_u32_u32(part_ids.get<uint32_t>(), iota.get<uint32_t>(),
                       keys_out.get<uint32_t>(), perm.get<uint32_t>(), n,
                       end_bit, nullptr, &tmp_bytes, stream_);
    DevBuf tmp(tmp_bytes);
    sort_pairs_u32_u32(part_ids.get<uint32_t>(), iota.get<uint32_t>(),
                       keys_out.get<uint32_t>(), perm.get<uint32_t>(), n,
                       end_bit, tmp.get(), &tmp_bytes, stream_);
    emit(all, perm, part_offsets, ncols, n);
  }

 private:
  DevBatch concat() {
    DevBatch all;
    int64_t n = 0;
    for (const DevBatch& b : staged_) n += b.num_rows;
    all.num_rows = n;
    size_t ncols = staged_[0].cols.size();
    for (size_t c = 0; c < ncols; c++) {
      const DevColumn& proto = staged_[0].cols[c];
      DevColumn col;
      col.dt = proto.dt;
      col.len = n;
      bool any_valid = false;
      for (const DevBatch& b : staged_)
        if (b.cols[c].validity) any_valid = true;
      if (col.dt == DType::Binary || col.dt == DType::Utf8) {
        int64_t total_data = 0;
        for (const DevBatch& b : staged_) total_data += b.cols[c].data_len;
        col.data_len = total_data;
        col.own_values.alloc(total_data ? total_data : 1);
        col.own_offsets.alloc((n + 1) * 4);
        // offsets rebased on host (few batches; lens path keeps it simple)
        std::vector<int32_t> h_off(n + 1, 0);
        int64_t row = 0, dpos = 0;
        for (const DevBatch& b : staged_) {
          const DevColumn& src = b.cols[c];
          std::vector<int32_t> so(src.len + 1);
          AURON_HIP(hipMemcpyAsync(so.data(), src.offsets, (src.len + 1) * 4,
                                   hipMemcpyDeviceToHost, stream_));
          AURON_HIP(hipStreamSynchronize(stream_));
          for (int64_t i = 0; i < src.len; i++)
            h_off[row + i + 1] = (int32_t)(dpos + so[i + 1]);
          if (src.data_len)
            AURON_HIP(hipMemcpyAsync(col.own_values.get<uint8_t>() + dpos,
                                     src.values, src.data_len,
                                     hipMemcpyDeviceToDevice, stream_));
          row += src.len;
          dpos += src.data_len;
        }
        AURON_HIP(hipMemcpyAsync(col.own_offsets.get(), h_off.data(),
                                 (n + 1) * 4, hipMemcpyHostToDevice, stream_));
        col.values = col.own_values.get();
        col.offsets = col.own_offsets.get<int32_t>();
      } else {
        size_t w = dtype_width(col.dt);
        col.own_values.alloc(n * w);
        int64_t row = 0;
        for (const DevBatch& b : staged_) {
          AURON_HIP(hipMemcpyAsync(col.own_values.get<uint8_t>() + row * w,
                                   b.cols[c].values, b.cols[c].len * w,
                                   hipMemcpyDeviceToDevice, stream_));
          row += b.cols[c].len;
        }
        col.values = col.own_values.get();
      }
      if (any_valid) {
        // merge validity bitmaps on host (bit shifts across batch boundaries)
        std::vector<uint8_t> bm((n + 7) / 8, 0xFF);
        int64_t row = 0;
        for (const DevBatch& b : staged_) {
          const DevColumn& src = b.cols[c];
          if (src.validity) {
            std::vector<uint8_t> sb((src.len + 7) / 8);
            AURON_HIP(hipMemcpyAsync(sb.data(), src.validity, sb.size(),
                                     hipMemcpyDeviceToHost, stream_));
            AURON_HIP(hipStreamSynchronize(stream_));
            for (int64_t i = 0; i < src.len; i++)
              if (!((sb[i >> 3] >> (i & 7)) & 1))
                bm[(row + i) >> 3] &= (uint8_t)~(1u << ((row + i) & 7));
          }
          row += src.len;
        }
        col.own_validity.alloc(bm.size());
        AURON_HIP(hipMemcpyAsync(col.own_validity.get(), bm.data(), bm.size(),
                                 hipMemcpyHostToDevice, stream_));
        col.validity = col.own_validity.get<uint8_t>();
      }
      all.cols.push_back(std::move(col));
    }
    return all;
  }

  void emit(const DevBatch& all, const DevBuf& perm,
            const std::vector<int64_t>& part_offsets, size_t ncols, int64_t n) {
    // gather columns into partition-contiguous order, D2H, serialize
    std::vector<std::vector<uint8_t>> h_values(ncols), h_validity(ncols);
    std::vector<std::vector<int32_t>> h_offsets(ncols);
    std::vector<HostCol> cols(ncols);
    for (size_t c = 0; c < ncols; c++) {
      const DevColumn& src = all.cols[c];
      if (src.dt == DType::Binary || src.dt == DType::Utf8) {
        DevBuf lens(n * 4);
        launch_gather_lens(src.offsets, perm.get<uint32_t>(), n,
                           lens.get<int32_t>(), stream_);
        std::vector<int32_t> h_lens(n);
        AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), n * 4,
                                 hipMemcpyDeviceToHost, stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        h_offsets[c].assign(n + 1, 0);
        for (int64_t i = 0; i < n; i++)
          h_offsets[c][i + 1] = h_offsets[c][i] + h_lens[i];
        DevBuf d_off((n + 1) * 4), d_data(h_offsets[c][n] ? h_offsets[c][n] : 1);
        AURON_HIP(hipMemcpyAsync(d_off.get(), h_offsets[c].data(), (n + 1) * 4,
                                 hipMemcpyHostToDevice, stream_));
        launch_gather_bytes((const uint8_t*)src.values, src.offsets,
                            perm.get<uint32_t>(), d_off.get<int32_t>(), n,
                            d_data.get<uint8_t>(), stream_);
        h_values[c].resize(h_offsets[c][n]);
        AURON_HIP(hipMemcpyAsync(h_values[c].data(), d_data.get(),
                                 h_values[c].size(), hipMemcpyDeviceToHost,
                                 stream_));
        AURON_HIP(hipStreamSynchronize(stream_));
        cols[c].byte_width = 0;
        cols[c].values = h_values[c].data();
        cols[c].offsets = h_offsets[c].data();
      } else {
        size_t w = dtype_width(src.dt);
        DevBuf d_out(n * w);
        if (w == 8)
          launch_gather_8((const uint8_t*)src.values, perm.get<uint32_t>(), n,
                          d_out.get<uint8_t>(), stream_);
        else if (w == 4)
          launch_gather_4((const uint8_t*)src.values, perm.get<uint32_t>(), n,
                          d_out.get<uint8_t>(), stream_);
        else
          FAIL("shuffle gather supports 4/8-byte primitives (hot path)");
        h_values[c].resize(n * w);
        AURON_HIP(hipMemcpyAsync(h_values[c].data(), d_out.get(), n * w,
                                 hipMemcpyDeviceToHost, stream_));
        cols[c].byte_width = (int)w;
        cols[c].values = h_values[c].data();
      }
      if (src.validity) {
        DevBuf d_bits((n + 7) / 8);
        launch_gather_bits(src.validity, perm.get<uint32_t>(), n,
                           d_bits.get<uint8_t>(), stream_);
        h_validity[c].resize((n + 7) / 8);
        AURON_HIP(hipMemcpyAsync(h_validity[c].data(), d_bits.get(),
                                 h_validity[c].size(), hipMemcpyDeviceToHost,
                                 stream_));
        cols[c].validity = h_validity[c].data();
      }
    }
    AURON_HIP(hipStreamSynchronize(stream_));
    write_files(cols, part_offsets, n);
  }

  void write_files(const std::vector<HostCol>& cols,
                   const std::vector<int64_t>& part_offsets, int64_t n) {
    (void)n;
    std::string err;
    if (!write_shuffle_files(cols, part_offsets, batch_size_,
                             node_.output_data_file, node_.output_index_file,
                             &err, codec_, zstd_level_))
      FAIL(err);
  }

  hipStream_t stream_;
  const ShuffleWriterNode& node_;
  uint32_t task_partition_id_ = 0;
  std::vector<uint32_t> hash_cols_;
  uint32_t P_ = 1;
  int64_t batch_size_ = 10000;
  size_t target_block_ = 4194304;
  int codec_ = 0, zstd_level_ = 1;
  std::vector<DevBatch> staged_;
};

// --------------------------------------------------------------- FilterOp --
// FilterExec (filter_exec.rs:174-198): ANDed predicates -> selection mask ->
// stable compaction. Predicate forms on the hot path: Column <op> Literal,
// IsNotNull/IsNull(Column).
class FilterOp {
 public:
  FilterOp(const FilterNode& node, hipStream_t stream) : stream_(stream) {
    for (const Expr& e : node.predicates) compile(e);
    if (preds_.empty()) FAIL("FilterExec without usable predicates");
  }

  DevBatch apply(DevBatch&& in) {
    int64_t n = in.num_rows;
    if (n == 0) return std::move(in);
    DevBuf mask(n);
    for (size_t pi = 0; pi < preds_.size(); pi++) {
      const Pred& p = preds_[pi];
      const DevColumn& c = in.cols.at(p.col);
      bool first = pi == 0;
      if (p.kind == Pred::NotNull || p.kind == Pred::Null) {
        launch_is_not_null(c.validity, n, mask.get<uint8_t>(), first,
                           p.kind == Pred::Null, stream_);
      } else {
        if (c.dt != p.dtype) FAIL("filter literal/column dtype mismatch");
        launch_cmp_lit(c.dt, c.values, c.validity, n, p.op, p.lit_i, p.lit_f,
                       mask.get<uint8_t>(), first, stream_);
      }
    }
    // stable compaction: positions = exclusive scan of mask
    DevBuf positions((n + 1) * 4);
    size_t tb = 0;
    scan_mask_u8(mask.get<uint8_t>(), positions.get<uint32_t>(), n, nullptr,
                 &tb, stream_);
    if (tb > scan_tmp_.size()) scan_tmp_.alloc(tb);
    scan_mask_u8(mask.get<uint8_t>(), positions.get<uint32_t>(), n,
                 scan_tmp_.get(), &tb, stream_);
    if (!pinned_.get()) pinned_.alloc(8);
    uint32_t* h_cnt = pinned_.get<uint32_t>();
    AURON_HIP(hipMemcpyAsync(h_cnt, positions.get<uint32_t>() + n, 4,
                             hipMemcpyDeviceToHost, stream_));
    AURON_HIP(hipStreamSynchronize(stream_));
    int64_t m = *h_cnt;
    DevBatch out;
    out.num_rows = m;
    if (m == 0) {
      for (auto& c : in.cols) {
        DevColumn oc;
        oc.dt = c.dt;
        oc.len = 0;
        out.cols.push_back(std::move(oc));
      }
      return out;
    }
    DevBuf sel(m * 4);
    launch_sel_rows(mask.get<uint8_t>(), positions.get<uint32_t>(), n,
                    sel.get<uint32_t>(), stream_);
    for (auto& c : in.cols) {
      out.cols.push_back(gather_col(c, sel.get<uint32_t>(), m));
    }
    // sel/mask/positions are pooled: drain the gathers before recycling
    AURON_HIP(hipStreamSynchronize(stream_));
    held_.push_back(std::move(in));  // borrowed inputs stay alive
    return out;
  }

 private:
  struct Pred {
    enum Kind { CmpLit, NotNull, Null } kind = CmpLit;
    uint32_t col = 0;
    CmpOp op = CMP_EQ;
    DType dtype = DType::Unsupported;
    int64_t lit_i = 0;
    double lit_f = 0;
  };

  void compile(const Expr& e) {
    if (e.kind == Expr::IsNotNull || e.kind == Expr::IsNull) {
      const Expr& c = e.children.at(0);
      if (c.kind != Expr::Column) FAIL("null-check operand must be a Column");
      Pred p;
      p.kind = e.kind == Expr::IsNull ? Pred::Null : Pred::NotNull;
      p.col = c.col_index;
      preds_.push_back(p);
      return;
    }
    if (e.kind != Expr::BinaryExpr || e.children.size() != 2)
      FAIL("unsupported filter predicate");
    const Expr& l = e.children[0];
    const Expr& r = e.children[1];
    if (e.op == "And") {  // nested AND: flatten
      compile(l);
      compile(r);
      return;
    }
    CmpOp op;
    if (e.op == "Eq") op = CMP_EQ;
    else if (e.op == "NotEq") op = CMP_NE;
    else if (e.op == "Lt") op = CMP_LT;
    else if (e.op == "LtEq") op = CMP_LE;
    else if (e.op == "Gt") op = CMP_GT;
    else if (e.op == "GtEq") op = CMP_GE;
    else FAIL("unsupported filter operator: " + e.op);
    const Expr* col = nullptr;
    const Expr* lit = nullptr;
    if (l.kind == Expr::Column && r.kind == Expr::Literal) {
      col = &l;
      lit = &r;
    } else if (l.kind == Expr::Literal && r.kind == Expr::Column) {
      col = &r;
      lit = &l;
      // flip: lit OP col == col FLIP(OP) lit
      op = op == CMP_LT ? CMP_GT : op == CMP_LE ? CMP_GE
           : op == CMP_GT ? CMP_LT : op == CMP_GE ? CMP_LE : op;
    } else {
      FAIL("filter predicate must be Column-vs-Literal on this path");
    }
    ScalarLit s;
    std::string err;
    if (!decode_ipc_scalar(lit->literal_ipc.data(), lit->literal_ipc.size(),
                           &s, &err))
      FAIL("literal decode: " + err);
    Pred p;
    p.kind = Pred::CmpLit;
    p.col = col->col_index;
    p.op = op;
    p.dtype = s.dtype;
    p.lit_i = s.i64;
    p.lit_f = s.f64;
    if (s.is_null) FAIL("null literal comparison unsupported");
    preds_.push_back(p);
  }

  DevColumn gather_col(const DevColumn& c, const uint32_t* sel, int64_t m) {
    DevColumn out;
    out.dt = c.dt;
    out.len = m;
    size_t w = dtype_width(c.dt);
    if (c.dt == DType::Binary || c.dt == DType::Utf8) {
      DevBuf lens(m * 4);
      launch_gather_lens(c.offsets, sel, m, lens.get<int32_t>(), stream_);
      std::vector<int32_t> h_lens(m);
      AURON_HIP(hipMemcpyAsync(h_lens.data(), lens.get(), m * 4,
                               hipMemcpyDeviceToHost, stream_));
      AURON_HIP(hipStreamSynchronize(stream_));
      std::vector<int32_t> h_offs(m + 1, 0);
      for (int64_t i = 0; i < m; i++) h_offs[i + 1] = h_offs[i] + h_lens[i];
      out.own_offsets.alloc((m + 1) * 4);
      AURON_HIP(hipMemcpyAsync(out.own_offsets.get(), h_offs.data(),
                               (m + 1) * 4, hipMemcpyHostToDevice, stream_));
      out.offsets = out.own_offsets.get<int32_t>();
      out.data_len = h_offs[m];
      out.own_values.alloc(out.data_len ? out.data_len : 1);
      launch_gather_bytes((const uint8_t*)c.values, c.offsets, sel,
                          out.own_offsets.get<int32_t>(), m,
                          out.own_values.get<uint8_t>(), stream_);
      out.values = out.own_values.get();
    } else if (w == 8) {
      out.own_values.alloc(m * 8);
      launch_gather_8((const uint8_t*)c.values, sel, m,
                      out.own_values.get<uint8_t>(), stream_);
      out.values = out.own_values.get();
    } else if (w == 4) {
      out.own_values.alloc(m * 4);
      launch_gather_4((const uint8_t*)c.values, sel, m,
                      out.own_values.get<uint8_t>(), stream_);
      out.values = out.own_values.get();
    } else {
      FAIL("filter gather: unsupported column width");
    }
    if (c.validity) {
      out.own_validity.alloc((m + 7) / 8);
      launch_gather_bits(c.validity, sel, m, out.own_validity.get<uint8_t>(),
                         stream_);
      out.validity = out.own_validity.get<uint8_t>();
    }
    return out;
  }

  hipStream_t stream_;
  std::vector<Pred> preds_;
  DevBuf scan_tmp_;
  PinnedBuf pinned_;
  std::vector<DevBatch> held_;
};

// -------------------------------------------------------------- ProjectOp --
// ProjectionExec (project_exec.rs): Column-only projection on this path
// (expression evaluation breadth is out of hot-path scope and fails loudly).
class ProjectOp {
 public:
  explicit ProjectOp(const ProjectionNode& node) {
    for (const Expr& e : node.exprs) {
      if (e.kind != Expr::Column)
        FAIL("ProjectionExec: only Column exprs on the hot path");
      cols_.push_back(e.col_index);
    }
  }

  DevBatch apply(DevBatch&& in) {
    DevBatch out;
    out.num_rows = in.num_rows;
    for (uint32_t ci : cols_) {
      DevColumn& src = in.cols.at(ci);
      DevColumn view;
      view.dt = src.dt;
      view.len = src.len;
      view.values = src.values;
      view.validity = src.validity;
      view.offsets = src.offsets;
      view.data_len = src.data_len;
      out.cols.push_back(std::move(view));
    }
    held_.push_back(std::move(in));  // owner of the viewed buffers
    return out;
  }

 private:
  std::vector<uint32_t> cols_;
  std::vector<DevBatch> held_;
};

// ---------------------------------------------------------------- Runtime --
struct Runtime {
  std::unique_ptr<TaskDefinition> td;
  AuronCallbacks cb{};
  hipStream_t stream = nullptr;
  bool started = false;
  bool schema_sent = false;
  std::string error;
  struct Stage {
    enum Kind { AggS, ShuffleS, FilterS, ProjectS } kind = AggS;
    std::unique_ptr<AggOp> agg;
    std::unique_ptr<ShuffleOp> shuffle;
    std::unique_ptr<FilterOp> filter;
    std::unique_ptr<ProjectOp> project;
  };
  std::vector<Stage> stages_;
  std::vector<OutField> out_fields;
  int ipc_codec_ = 0;  // IpcReader block codec (SPARK_IO_COMPRESSION_CODEC)
  std::vector<std::pair<int64_t, std::vector<HostOutCol>>> outputs;
  size_t emit_idx = 0;
  // device-resident outputs (AURON_HIP_DEVICE_OUTPUT): exported through
  // import_device_batch; DevBufs stay owned here until finalize
  std::vector<std::pair<int64_t, std::vector<DevOutCol>>> outputs_dev;
  size_t emit_dev_idx = 0;
  std::map<std::string, int64_t> metrics;

  ~Runtime() {
    if (stream) (void)hipStreamDestroy(stream);
  }

  void set_error(const std::string& msg) {
    error = msg;
    if (cb.set_error) cb.set_error(cb.user, msg.c_str());
  }

  // run the whole pipeline (Agg is a pipeline breaker; outputs are staged)
  void run() {
    auto t0 = std::chrono::steady_clock::now();
    Conf conf{&cb};
    {
      std::string codec = conf.get("SPARK_IO_COMPRESSION_CODEC", "lz4");
      if (codec == "zstd") ipc_codec_ = 1;
      else if (codec != "lz4")
        FAIL("unsupported shuffle codec (lz4/zstd on this path)");
    }
    // collect plan chain root→leaf
    std::vector<const PlanNode*> chain;
    const PlanNode* p = td->plan.get();
    while (p) {
      chain.push_back(p);
      switch (p->kind) {
        case PlanNode::ShuffleWriter: p = p->shuffle_writer->input.get(); break;
        case PlanNode::Agg: p = p->agg->input.get(); break;
        case PlanNode::Filter: p = p->filter->input.get(); break;
        case PlanNode::Projection: p = p->projection->input.get(); break;
        case PlanNode::IpcReader:
        case PlanNode::ParquetScan:
        case PlanNode::FFIReader: p = nullptr; break;
      }
    }
    // leaf must be a source: FFIReader (Arrow in) or IpcReader (shuffle bytes)
    const PlanNode* leaf = chain.back();
    if (leaf->kind != PlanNode::FFIReader &&
        leaf->kind != PlanNode::IpcReader &&
        leaf->kind != PlanNode::ParquetScan)
      FAIL("plan leaf must be FFIReader/IpcReader/ParquetScan on this path");

    // middle ops (leaf-1 ... root), in execution order
    stages_.clear();
    for (auto it = chain.rbegin() + 1; it != chain.rend(); ++it) {
      const PlanNode* node = *it;
      Stage st;
      switch (node->kind) {
        case PlanNode::Agg:
          st.kind = Stage::AggS;
          st.agg = std::make_unique<AggOp>(*node->agg, conf, stream);
          break;
        case PlanNode::ShuffleWriter:
          st.kind = Stage::ShuffleS;
          st.shuffle = std::make_unique<ShuffleOp>(*node->shuffle_writer, conf,
                                                   stream, td->partition_id);
          break;
        case PlanNode::Filter:
          st.kind = Stage::FilterS;
          st.filter = std::make_unique<FilterOp>(*node->filter, stream);
          break;
        case PlanNode::Projection:
          st.kind = Stage::ProjectS;
          st.project = std::make_unique<ProjectOp>(*node->projection);
          break;
        default:
          FAIL("unsupported operator in plan chain");
      }
      if (!stages_.empty() && stages_.back().kind == Stage::ShuffleS)
        FAIL("operator above ShuffleWriter unsupported");
      stages_.push_back(std::move(st));
    }

    // pump input through the chain
    int64_t input_rows = 0;
    if (leaf->kind == PlanNode::FFIReader) {
      const FFIReaderNode& reader = *leaf->ffi_reader;
      while (true) {
        ArrowArray arr;
        ArrowSchema sch;
        ArrowDeviceArray dev;
        memset(&arr, 0, sizeof(arr));
        memset(&sch, 0, sizeof(sch));
        memset(&dev, 0, sizeof(dev));
        if (!cb.next_input_batch) break;
        int rc = cb.next_input_batch(cb.user, reader.resource_id.c_str(), &arr,
                                     &sch, &dev);
        DBG("reader rc=%d", rc);
        if (rc == 0) break;
        DevBatch b;
        if (rc == 2) {
          b = import_batch(&dev.array, sch.release ? &sch : nullptr,
                           reader.schema, true, stream);
        } else {
          b = import_batch(&arr, sch.release ? &sch : nullptr, reader.schema,
                           false, stream);
          if (arr.release) arr.release(&arr);
        }
        if (sch.release) sch.release(&sch);
        input_rows += b.num_rows;
        feed(0, std::move(b));
      }
    } else if (leaf->kind == PlanNode::IpcReader) {
      input_rows = pump_ipc_reader(*leaf->ipc_reader);
    } else {
      input_rows = pump_parquet(*leaf->parquet);
    }
    // drain chain: pipeline breakers emit, transforms pass through
    AggOp* last_agg = nullptr;
    AggOp* first_agg = nullptr;
    bool has_shuffle = false;
    for (size_t i = 0; i < stages_.size(); i++) {
      Stage& st = stages_[i];
      if (st.kind == Stage::AggS) {
        if (!first_agg) first_agg = st.agg.get();
        last_agg = st.agg.get();
        bool terminal = i + 1 == stages_.size();
        if (st.agg->device_out_) {
          if (!terminal)
            FAIL("AURON_HIP_DEVICE_OUTPUT on a non-terminal Agg unsupported");
          if (!cb.import_device_batch)
            FAIL("AURON_HIP_DEVICE_OUTPUT set but import_device_batch is "
                 "NULL");
          outputs_dev = st.agg->finish_dev();
          out_fields = st.agg->output_fields();
        } else {
          auto outs = st.agg->finish();
          for (auto& ob : outs) {
            if (terminal) {
              outputs.push_back(std::move(ob));
            } else {
              DevBatch b = host_out_to_dev(ob, st.agg->output_fields());
              feed(i + 1, std::move(b));
            }
          }
          if (terminal) out_fields = st.agg->output_fields();
        }
      } else if (st.kind == Stage::ShuffleS) {
        st.shuffle->finish();
        has_shuffle = true;
        out_fields.clear();
      }
    }
    if (!has_shuffle && !last_agg)
      FAIL("plan without Agg/ShuffleWriter sink unsupported");
    if (last_agg) {
      metrics["num_groups"] = (int64_t)last_agg->num_groups_host();
      metrics["agg_update_ns"] = first_agg->update_ns_;
      metrics["agg_update_rows"] = first_agg->update_rows_;
      metrics["spill_count"] = first_agg->spill_count_ + last_agg->spill_count_;
    }
    metrics["input_rows"] = input_rows;
    int64_t out_rows = 0;
    for (auto& o : outputs) out_rows += o.first;
    for (auto& o : outputs_dev) out_rows += o.first;
    metrics["output_rows"] = out_rows;
    metrics["elapsed_compute_ns"] =
        std::chrono::duration_cast<std::chrono::nanoseconds>(
            std::chrono::steady_clock::now() - t0)
            .count();
  }

  // Evaluate a Column-vs-Literal pruning predicate against row-group stats:
  // returns false only when the predicate PROVABLY excludes every row
  // (parquet_exec.rs row-group stats pruning analog; conservative on any
  // unsupported shape).
  static bool rg_may_match(const Expr& e, const ParquetFile& pf, int rg) {
    if (e.kind != Expr::BinaryExpr || e.children.size() != 2) return true;
    if (e.op == "And")
      return rg_may_match(e.children[0], pf, rg) &&
             rg_may_match(e.children[1], pf, rg);
    const Expr* col = nullptr;
    const Expr* lit = nullptr;
    bool flipped = false;
    if (e.children[0].kind == Expr::Column &&
        e.children[1].kind == Expr::Literal) {
      col = &e.children[0];
      lit = &e.children[1];
    } else if (e.children[1].kind == Expr::Column &&
               e.children[0].kind == Expr::Literal) {
      col = &e.children[1];
      lit = &e.children[0];
      flipped = true;
    } else {
      return true;
    }
    if (col->col_index >= pf.columns().size()) return true;
    PqColStats s = pf.column_stats(rg, (int)col->col_index);
    if (!s.has_minmax) return true;
    ScalarLit sl;
    std::string err2;
    if (!decode_ipc_scalar(lit->literal_ipc.data(), lit->literal_ipc.size(),
                           &sl, &err2) ||
        sl.is_null)
      return true;
    std::string op = e.op;
    if (flipped) {
      op = op == "Lt" ? "Gt" : op == "LtEq" ? "GtEq"
           : op == "Gt" ? "Lt" : op == "GtEq" ? "LtEq" : op;
    }
    int pt = pf.columns()[col->col_index].physical_type;
    if (pt == 1 || pt == 2) {
      // compare in the int64 domain: a round-trip through double loses
      // precision past 2^53 and could prune a row group that matches
      int64_t lo = s.min_i, hi = s.max_i, v = sl.i64;
      if (op == "Lt") return lo < v;
      if (op == "LtEq") return lo <= v;
      if (op == "Gt") return hi > v;
      if (op == "GtEq") return hi >= v;
      if (op == "Eq") return lo <= v && v <= hi;
      return true;
    }
    if (pt != 4 && pt != 5) return true;
    double lo = s.min_f, hi = s.max_f, v = sl.f64;
    if (op == "Lt") return lo < v;
    if (op == "LtEq") return lo <= v;
    if (op == "Gt") return hi > v;
    if (op == "GtEq") return hi >= v;
    if (op == "Eq") return lo <= v && v <= hi;
    return true;
  }

  // ParquetScanExec source (parquet_exec.rs:150-203; decode scope per
  // parquet.h): footer/pages on host, validity/dictionary expansion on GPU.
  int64_t pump_parquet(const ParquetScanNode& node) {
    int64_t input_rows = 0;
    DevBuf scan_tmp;
    for (const std::string& path : node.files) {
      ParquetFile pf(path);
      const auto& fcols = pf.columns();
      std::vector<uint32_t> proj = node.projection;
      if (proj.empty())
        for (uint32_t i = 0; i < fcols.size(); i++) proj.push_back(i);
      for (uint32_t ci : proj)
        if (ci >= fcols.size()) FAIL("parquet: projection out of range");
      // host decode runs in ROW-GROUP WINDOWS overlapped with the device
      // work: while window k's uploads/kernels/consume run on the stream,
      // the host cores decode window k+1 (the host page-header walk +
      // dict-prefix decode and the GPU page decompression are comparable
      // in wall time — serializing them was ~40% of the config-3 step)
      const int nrg = pf.num_row_groups();
      const size_t width = proj.size();
      std::vector<PqColumnChunkData> decoded((size_t)nrg * width);
      std::string decode_err;
      std::mutex err_mu;
      const int RG_WIN = [&] {
        const char* e = getenv("AURON_PARQUET_WIN");
        if (e && e[0] == '0') return nrg > 0 ? nrg : 1;  // one window
        if (e && atoi(e) > 0) return atoi(e);
        // ~128M rows per window measured best at 1B rows (same-box A/B:
        // 16-rg windows 1341 ms/step vs whole-file 1619, 8-rg 1679)
        return std::max(1, (int)((512u << 20) /
                                 std::max<int64_t>(
                                     1, pf.row_group_rows(0) * 8)));
      }();
      auto decode_window = [&](int rg_lo, int rg_hi) {
        size_t jlo = (size_t)rg_lo * width, jhi = (size_t)rg_hi * width;
        unsigned nw = std::min<unsigned>(
            std::min(48u, std::max(1u, std::thread::hardware_concurrency())),
            (unsigned)(jhi - jlo));
        std::atomic<size_t> next{jlo};
        std::vector<std::thread> ws;
        for (unsigned w = 0; w < nw; w++) {
          ws.emplace_back([&]() {
            for (;;) {
              size_t i = next.fetch_add(1);
              if (i >= jhi) break;
              try {
                decoded[i] =
                    pf.read_chunk((int)(i / width), (int)proj[i % width]);
              } catch (const std::exception& ex) {
                std::lock_guard<std::mutex> lk(err_mu);
                if (decode_err.empty()) decode_err = ex.what();
              }
            }
          });
        }
        for (auto& w : ws) w.join();
      };
      std::future<void> pending = std::async(
          std::launch::async, decode_window, 0, std::min(RG_WIN, nrg));
      // ---- GPU page decompression batch (kernels_pq.hip) ----
      // every gpu_comp chunk's PLAIN-suffix pages across ALL row groups go
      // into ONE wave-per-page decompress+decode launch set (per-chunk
      // launches would leave the chip nearly idle: ~30 pages per chunk).
      // Results live in two blobs (validity bitmaps / dense values) that
      // stay resident for the row-group assembly loop below.
      struct GcResult {
        uint64_t valid_base = 0, dense_base = 0;
        int64_t total_nn = 0, null_total = 0;
        bool ready = false;
      };
      std::vector<GcResult> gc_res((size_t)nrg * width);
      for (int win_lo = 0; win_lo < nrg; win_lo += RG_WIN) {
      const int win_hi = std::min(win_lo + RG_WIN, nrg);
      pending.get();  // this window's host decode is complete
      if (!decode_err.empty()) FAIL(decode_err);
      if (win_hi < nrg)  // overlap: decode the NEXT window on host cores
        pending = std::async(std::launch::async, decode_window, win_hi,
                             std::min(win_hi + RG_WIN, nrg));
      DevBuf d_gc_valid, d_gc_dense;
      {
        std::vector<PqGpuPage> pages;
        std::vector<PqGpuChunk> chmeta;
        std::vector<size_t> ch_di;
        uint64_t comp_total = 0, scratch_total = 0, valid_total = 0,
                 dense_total = 0;
        auto al8 = [](uint64_t v) { return (v + 7) & ~7ull; };
        for (size_t di = (size_t)win_lo * width;
             di < (size_t)win_hi * width; di++) {
          PqColumnChunkData& cd = decoded[di];
          if (!cd.gpu_comp) continue;
          const int vw = cd.value_width;
          int64_t values = cd.prefix_values + cd.suffix_values;
          PqGpuChunk ch;
          ch.page0 = (uint32_t)pages.size();
          ch.npages = (uint32_t)cd.comp_pages.size();
          ch.valid_base = valid_total;
          ch.dense_base = dense_total;
          ch.prefix_nn = (uint32_t)(cd.plain.size() / (size_t)vw);
          ch.vw = (uint32_t)vw;
          uint32_t vbase = (uint32_t)cd.prefix_values;
          for (const auto& pg : cd.comp_pages) {
            PqGpuPage p;
            p.comp_off = comp_total + (uint64_t)(pg.src - cd.comp_pages[0].src);
            p.uncomp_off = scratch_total;
            p.comp_len = pg.comp_len;
            p.uncomp_len = pg.uncomp_len;
            p.num_values = pg.num_values;
            p.value_base = vbase;
            p.has_def = cd.nullable ? 1u : 0u;
            p.chunk_id = (uint32_t)chmeta.size();
            pages.push_back(p);
            vbase += pg.num_values;
            scratch_total += al8(pg.uncomp_len);
          }
          const auto& lastp = cd.comp_pages.back();
          comp_total += al8((uint64_t)(lastp.src - cd.comp_pages[0].src) +
                            lastp.comp_len);
          valid_total += al8(((uint64_t)values + 7) / 8);
          dense_total += al8((uint64_t)values * vw);
          chmeta.push_back(ch);
          ch_di.push_back(di);
        }
        if (!pages.empty()) {
          DevBuf d_comp(comp_total), d_scratch(scratch_total);
          d_gc_valid.alloc(valid_total ? valid_total : 8);
          d_gc_dense.alloc(dense_total);
          DevBuf d_pages(pages.size() * sizeof(PqGpuPage));
          DevBuf d_chunks(chmeta.size() * sizeof(PqGpuChunk));
          DevBuf d_nn(pages.size() * 4), d_voff(pages.size() * 4);
          DevBuf d_pdense(pages.size() * 4), d_chunknn(chmeta.size() * 4);
          DevBuf d_gcerr(4);
          AURON_HIP(hipMemsetAsync(d_gc_valid.get(), 0, valid_total, stream));
          AURON_HIP(hipMemsetAsync(d_gcerr.get(), 0, 4, stream));
          PinnedUploader::inst().copy(d_pages.get(), pages.data(),
                                      pages.size() * sizeof(PqGpuPage),
                                      stream);
          PinnedUploader::inst().copy(d_chunks.get(), chmeta.data(),
                                      chmeta.size() * sizeof(PqGpuChunk),
                                      stream);
          std::vector<std::vector<uint8_t>> pvbits(chmeta.size());
          for (size_t c = 0; c < chmeta.size(); c++) {
            PqColumnChunkData& cd = decoded[ch_di[c]];
            // compressed span (pages of a chunk are contiguous in the file)
            const auto& p0 = cd.comp_pages[0];
            const auto& pl = cd.comp_pages.back();
            size_t span = (size_t)(pl.src - p0.src) + pl.comp_len;
            PinnedUploader::inst().copy(
                d_comp.get<uint8_t>() + pages[chmeta[c].page0].comp_off -
                    (uint64_t)(p0.src - cd.comp_pages[0].src),
                p0.src, span, stream);
            // host-decoded prefix: dense values + validity bits
            if (!cd.plain.empty())
              PinnedUploader::inst().copy(
                  d_gc_dense.get<uint8_t>() + chmeta[c].dense_base,
                  cd.plain.data(), cd.plain.size(), stream);
            if (cd.nullable && cd.prefix_values > 0) {
              size_t pb = ((size_t)cd.prefix_values + 7) / 8;
              auto& vb = pvbits[c];
              if (!cd.validity.empty()) {
                vb.assign(cd.validity.begin(), cd.validity.begin() + pb);
              } else {
                vb.assign(pb, 0xFF);
              }
              if (cd.prefix_values & 7)  // clear bits beyond the prefix:
                vb[pb - 1] &= (uint8_t)((1u << (cd.prefix_values & 7)) - 1);
              PinnedUploader::inst().copy(
                  d_gc_valid.get<uint8_t>() + chmeta[c].valid_base, vb.data(),
                  pb, stream);
            }
          }
          launch_pq_pages_decode(d_comp.get<uint8_t>(),
                                 d_pages.get<PqGpuPage>(), (int)pages.size(),
                                 d_scratch.get<uint8_t>(),
                                 d_gc_valid.get<uint8_t>(),
                                 d_chunks.get<PqGpuChunk>(),
                                 d_nn.get<uint32_t>(), d_voff.get<uint32_t>(),
                                 d_gcerr.get<uint32_t>(), stream);
          launch_pq_page_offsets(d_chunks.get<PqGpuChunk>(),
                                 (int)chmeta.size(), d_nn.get<uint32_t>(),
                                 d_pdense.get<uint32_t>(),
                                 d_chunknn.get<uint32_t>(), stream);
          launch_pq_pages_compact(d_pages.get<PqGpuPage>(),
                                  (int)pages.size(),
                                  d_scratch.get<uint8_t>(),
                                  d_chunks.get<PqGpuChunk>(),
                                  d_nn.get<uint32_t>(),
                                  d_voff.get<uint32_t>(),
                                  d_pdense.get<uint32_t>(),
                                  d_gc_dense.get<uint8_t>(), stream);
          std::vector<uint32_t> h_chunknn(chmeta.size());
          uint32_t h_err = 0;
          AURON_HIP(hipMemcpyAsync(h_chunknn.data(), d_chunknn.get(),
                                   chmeta.size() * 4, hipMemcpyDeviceToHost,
                                   stream));
          AURON_HIP(hipMemcpyAsync(&h_err, d_gcerr.get(), 4,
                                   hipMemcpyDeviceToHost, stream));
          AURON_HIP(hipStreamSynchronize(stream));
          if (h_err) FAIL("parquet: GPU page decode failed (flags " +
                          std::to_string(h_err) + ")");
          for (size_t c = 0; c < chmeta.size(); c++) {
            PqColumnChunkData& cd = decoded[ch_di[c]];
            GcResult& r = gc_res[ch_di[c]];
            r.valid_base = chmeta[c].valid_base;
            r.dense_base = chmeta[c].dense_base;
            r.total_nn = (int64_t)chmeta[c].prefix_nn + h_chunknn[c];
            r.null_total =
                cd.prefix_values + cd.suffix_values - r.total_nn;
            r.ready = true;
          }
        }
      }
      for (int rg = win_lo; rg < win_hi; rg++) {
        bool keep = true;
        for (const Expr& pr : node.pruning)
          keep = keep && rg_may_match(pr, pf, rg);
        if (!keep) {
          DBG("parquet: pruned row group %d", rg);
          continue;
        }
        int64_t rows = pf.row_group_rows(rg);
        DevBatch b;
        b.num_rows = rows;
        for (size_t pi = 0; pi < width; pi++) {
          uint32_t ci = proj[pi];
          PqColumnChunkData cd = std::move(decoded[(size_t)rg * width + pi]);
          DevColumn c;
          c.dt = fcols[ci].dtype();
          c.len = rows;
          bool has_nulls = cd.null_count > 0;
          if (cd.gpu_comp) {
            // device-decompressed chunk: validity bitmap + dense values are
            // already resident in the batch blobs (pre-pass above)
            const GcResult& r = gc_res[(size_t)rg * width + pi];
            if (!r.ready) FAIL("parquet: gpu page results missing");
            if (cd.prefix_values + cd.suffix_values != rows)
              FAIL("parquet: gpu chunk row count mismatch");
            const int w2 = (int)dtype_width(fcols[ci].dtype());
            c.own_values.alloc((size_t)rows * w2);
            const uint8_t* dense = d_gc_dense.get<uint8_t>() + r.dense_base;
            if (r.null_total == 0) {
              AURON_HIP(hipMemcpyAsync(c.own_values.get(), dense,
                                       (size_t)rows * w2,
                                       hipMemcpyDeviceToDevice, stream));
            } else {
              c.own_validity.alloc((rows + 7) / 8);
              AURON_HIP(hipMemcpyAsync(c.own_validity.get(),
                                       d_gc_valid.get<uint8_t>() +
                                           r.valid_base,
                                       (size_t)(rows + 7) / 8,
                                       hipMemcpyDeviceToDevice, stream));
              c.validity = c.own_validity.get<uint8_t>();
              DevBuf d_mask2(rows), d_pos2((rows + 1) * 4);
              launch_bits_to_mask(c.validity, rows, d_mask2.get<uint8_t>(),
                                  stream);
              size_t tb = 0;
              scan_mask_u8(d_mask2.get<uint8_t>(), d_pos2.get<uint32_t>(),
                           rows, nullptr, &tb, stream);
              if (tb > scan_tmp.size()) scan_tmp.alloc(tb);
              scan_mask_u8(d_mask2.get<uint8_t>(), d_pos2.get<uint32_t>(),
                           rows, scan_tmp.get(), &tb, stream);
              launch_scatter_packed(w2, dense, d_pos2.get<uint32_t>(),
                                    d_mask2.get<uint8_t>(), rows,
                                    c.own_values.get<uint8_t>(), stream);
              AURON_HIP(hipStreamSynchronize(stream));  // temps die here
            }
            c.values = c.own_values.get();
            b.cols.push_back(std::move(c));
            continue;
          }
          if (c.dt == DType::Utf8 || c.dt == DType::Binary) {
            // row-aligned offsets/data assembled on host (parquet.cpp)
            c.own_offsets.alloc((rows + 1) * 4);
            AURON_HIP(hipMemcpyAsync(c.own_offsets.get(),
                                     cd.bin_offsets.data(), (rows + 1) * 4,
                                     hipMemcpyHostToDevice, stream));
            c.offsets = c.own_offsets.get<int32_t>();
            c.data_len = (int64_t)cd.bin_data.size();
            c.own_values.alloc(cd.bin_data.empty() ? 1 : cd.bin_data.size());
            if (!cd.bin_data.empty())
              PinnedUploader::inst().copy(c.own_values.get(),
                                          cd.bin_data.data(),
                                          cd.bin_data.size(), stream);
            c.values = c.own_values.get();
            if (has_nulls) {
              c.own_validity.alloc(cd.validity.size());
              AURON_HIP(hipMemcpyAsync(c.own_validity.get(),
                                       cd.validity.data(), cd.validity.size(),
                                       hipMemcpyHostToDevice, stream));
              c.validity = c.own_validity.get<uint8_t>();
            }
            AURON_HIP(hipStreamSynchronize(stream));
            b.cols.push_back(std::move(c));
            continue;
          }
          const int w = (int)dtype_width(fcols[ci].dtype());
          c.own_values.alloc((size_t)rows * w);
          DevBuf d_valid, d_mask, d_positions, d_packed, d_indices;
          DevBuf d_runs, d_runbytes;
          if (cd.gpu_def) {
            // def levels expand to the validity bitmap ON DEVICE (the host
            // only walked the run headers — parquet.cpp rle_bp_runs)
            d_runs.alloc(cd.def_runs.size() * sizeof(PqRun));
            PinnedUploader::inst().copy(d_runs.get(), cd.def_runs.data(),
                                        cd.def_runs.size() * sizeof(PqRun),
                                        stream);
            d_runbytes.alloc(cd.def_bytes.empty() ? 8 : cd.def_bytes.size());
            if (!cd.def_bytes.empty())
              PinnedUploader::inst().copy(d_runbytes.get(),
                                          cd.def_bytes.data(),
                                          cd.def_bytes.size(), stream);
            c.own_validity.alloc((rows + 7) / 8);
            launch_def_expand_validity(d_runs.get(), (int)cd.def_runs.size(),
                                       d_runbytes.get<uint8_t>(), rows,
                                       c.own_validity.get<uint8_t>(), stream);
            c.validity = c.own_validity.get<uint8_t>();
            d_mask.alloc(rows);
            launch_bits_to_mask(c.validity, rows, d_mask.get<uint8_t>(),
                                stream);
            d_positions.alloc((rows + 1) * 4);
            size_t tb = 0;
            scan_mask_u8(d_mask.get<uint8_t>(), d_positions.get<uint32_t>(),
                         rows, nullptr, &tb, stream);
            if (tb > scan_tmp.size()) scan_tmp.alloc(tb);
            scan_mask_u8(d_mask.get<uint8_t>(), d_positions.get<uint32_t>(),
                         rows, scan_tmp.get(), &tb, stream);
          } else if (has_nulls) {
            c.own_validity.alloc(cd.validity.size());
            AURON_HIP(hipMemcpyAsync(c.own_validity.get(), cd.validity.data(),
                                     cd.validity.size(), hipMemcpyHostToDevice,
                                     stream));
            c.validity = c.own_validity.get<uint8_t>();
            d_mask.alloc(rows);
            launch_bits_to_mask(c.validity, rows, d_mask.get<uint8_t>(),
                                stream);
            d_positions.alloc((rows + 1) * 4);
            size_t tb = 0;
            scan_mask_u8(d_mask.get<uint8_t>(), d_positions.get<uint32_t>(),
                         rows, nullptr, &tb, stream);
            if (tb > scan_tmp.size()) scan_tmp.alloc(tb);
            scan_mask_u8(d_mask.get<uint8_t>(), d_positions.get<uint32_t>(),
                         rows, scan_tmp.get(), &tb, stream);
          }
          if (cd.gpu_dict) {
            // dictionary indices: host walked run headers only; expand the
            // dense non-null index array on device, then scatter/gather as
            // in the host-materialized path
            DevBuf d_iruns(cd.idx_runs.size() * sizeof(PqRun));
            PinnedUploader::inst().copy(d_iruns.get(), cd.idx_runs.data(),
                                        cd.idx_runs.size() * sizeof(PqRun),
                                        stream);
            DevBuf d_ibytes(cd.idx_bytes.empty() ? 8 : cd.idx_bytes.size());
            if (!cd.idx_bytes.empty())
              PinnedUploader::inst().copy(d_ibytes.get(), cd.idx_bytes.data(),
                                          cd.idx_bytes.size(), stream);
            DevBuf d_idx_packed((cd.nn_count ? cd.nn_count : 1) * 4);
            launch_runs_expand_u32(d_iruns.get(), (int)cd.idx_runs.size(),
                                   d_ibytes.get<uint8_t>(), cd.nn_count,
                                   d_idx_packed.get<uint32_t>(), stream);
            d_indices.alloc((size_t)rows * 4);
            if (has_nulls) {
              launch_scatter_packed(4, d_idx_packed.get<uint8_t>(),
                                    d_positions.get<uint32_t>(),
                                    d_mask.get<uint8_t>(), rows,
                                    d_indices.get<uint8_t>(), stream);
            } else {
              AURON_HIP(hipMemcpyAsync(d_indices.get(), d_idx_packed.get(),
                                       (size_t)rows * 4,
                                       hipMemcpyDeviceToDevice, stream));
            }
            DevBuf d_dict(cd.dict_values.empty() ? 1 : cd.dict_values.size());
            if (!cd.dict_values.empty())
              AURON_HIP(hipMemcpyAsync(d_dict.get(), cd.dict_values.data(),
                                       cd.dict_values.size(),
                                       hipMemcpyHostToDevice, stream));
            if (w == 8)
              launch_gather_8(d_dict.get<uint8_t>(),
                              d_indices.get<uint32_t>(), rows,
                              c.own_values.get<uint8_t>(), stream);
            else
              launch_gather_4(d_dict.get<uint8_t>(),
                              d_indices.get<uint32_t>(), rows,
                              c.own_values.get<uint8_t>(), stream);
            AURON_HIP(hipStreamSynchronize(stream));  // temps die below
          } else if (!cd.uses_dict) {
            // PLAIN: non-null values packed densely
            if (!has_nulls) {
              PinnedUploader::inst().copy(c.own_values.get(), cd.plain.data(),
                                          cd.plain.size(), stream);
            } else {
              d_packed.alloc(cd.plain.empty() ? 1 : cd.plain.size());
              if (!cd.plain.empty())
                PinnedUploader::inst().copy(d_packed.get(), cd.plain.data(),
                                            cd.plain.size(), stream);
              launch_scatter_packed(w, d_packed.get<uint8_t>(),
                                    d_positions.get<uint32_t>(),
                                    d_mask.get<uint8_t>(), rows,
                                    c.own_values.get<uint8_t>(), stream);
            }
          } else {
            // dictionary: expand packed indices to rows, gather dict values
            d_indices.alloc((size_t)rows * 4);
            DevBuf d_idx_packed(cd.dict_indices.empty()
                                    ? 4
                                    : cd.dict_indices.size() * 4);
            if (!cd.dict_indices.empty())
              PinnedUploader::inst().copy(d_idx_packed.get(),
                                          cd.dict_indices.data(),
                                          cd.dict_indices.size() * 4, stream);
            if (has_nulls) {
              launch_scatter_packed(4, d_idx_packed.get<uint8_t>(),
                                    d_positions.get<uint32_t>(),
                                    d_mask.get<uint8_t>(), rows,
                                    d_indices.get<uint8_t>(), stream);
            } else {
              AURON_HIP(hipMemcpyAsync(d_indices.get(), d_idx_packed.get(),
                                       (size_t)rows * 4,
                                       hipMemcpyDeviceToDevice, stream));
            }
            DevBuf d_dict(cd.dict_values.empty() ? 1 : cd.dict_values.size());
            if (!cd.dict_values.empty())
              AURON_HIP(hipMemcpyAsync(d_dict.get(), cd.dict_values.data(),
                                       cd.dict_values.size(),
                                       hipMemcpyHostToDevice, stream));
            if (w == 8)
              launch_gather_8(d_dict.get<uint8_t>(),
                              d_indices.get<uint32_t>(), rows,
                              c.own_values.get<uint8_t>(), stream);
            else
              launch_gather_4(d_dict.get<uint8_t>(),
                              d_indices.get<uint32_t>(), rows,
                              c.own_values.get<uint8_t>(), stream);
          }
          AURON_HIP(hipStreamSynchronize(stream));  // host cd dies here
          c.values = c.own_values.get();
          b.cols.push_back(std::move(c));
        }
        input_rows += rows;
        feed(0, std::move(b));
      }
      }  // row-group window
    }
    return input_rows;
  }

  // IpcReaderExec source (ipc_reader_exec.rs:62-120): raw shuffle block
  // streams in, decoded+deserialized on host, batches up to the device.
  int64_t pump_ipc_reader(const IpcReaderNode& reader) {
    if (!cb.next_ipc_bytes)
      FAIL("plan holds IpcReaderExec but no next_ipc_bytes callback was given");
    std::vector<int> widths;
    for (const Field& f : reader.schema.fields) {
      if (f.dtype == DType::Binary || f.dtype == DType::Utf8)
        widths.push_back(0);
      else if (dtype_width(f.dtype) > 0)
        widths.push_back((int)dtype_width(f.dtype));
      else
        FAIL("IpcReader: unsupported column dtype");
    }
    int64_t input_rows = 0;
    while (true) {
      const uint8_t* data = nullptr;
      size_t len = 0;
      int rc = cb.next_ipc_bytes(cb.user, reader.resource_id.c_str(), &data,
                                 &len);
      DBG("ipc reader rc=%d len=%zu", rc, len);
      if (rc == 0) break;
      std::vector<uint8_t> payload;
      std::string err;
      if (!ipc_decode_blocks(data, len, &payload, &err, ipc_codec_))
        FAIL(err);
      size_t used = 0;
      while (used < payload.size()) {
        int64_t rows = 0;
        std::vector<OwnedCol> cols;
        if (!serde_read_batch(payload.data(), payload.size(), &used, widths,
                              &rows, &cols, &err))
          FAIL(err);
        DevBatch b;
        b.num_rows = rows;
        for (size_t ci = 0; ci < cols.size(); ci++) {
          OwnedCol& oc = cols[ci];
          DevColumn c;
          c.dt = reader.schema.fields[ci].dtype;
          c.len = rows;
          if (oc.byte_width == 0) {
            c.own_offsets.alloc((rows + 1) * 4);
            AURON_HIP(hipMemcpyAsync(c.own_offsets.get(), oc.offsets.data(),
                                     (rows + 1) * 4, hipMemcpyHostToDevice,
                                     stream));
            c.offsets = c.own_offsets.get<int32_t>();
            c.data_len = (int64_t)oc.values.size();
          }
          c.own_values.alloc(oc.values.empty() ? 1 : oc.values.size());
          if (!oc.values.empty())
            AURON_HIP(hipMemcpyAsync(c.own_values.get(), oc.values.data(),
                                     oc.values.size(), hipMemcpyHostToDevice,
                                     stream));
          c.values = c.own_values.get();
          if (!oc.validity.empty()) {
            c.own_validity.alloc(oc.validity.size());
            AURON_HIP(hipMemcpyAsync(c.own_validity.get(), oc.validity.data(),
                                     oc.validity.size(), hipMemcpyHostToDevice,
                                     stream));
            c.validity = c.own_validity.get<uint8_t>();
          }
          AURON_HIP(hipStreamSynchronize(stream));  // host vectors die below
          b.cols.push_back(std::move(c));
        }
        input_rows += rows;
        feed(0, std::move(b));
      }
    }
    return input_rows;
  }

  void feed(size_t idx, DevBatch&& b) {
    while (idx < stages_.size()) {
      Stage& st = stages_[idx];
      switch (st.kind) {
        case Stage::AggS:
          st.agg->consume(std::move(b));
          return;
        case Stage::ShuffleS:
          st.shuffle->consume(std::move(b));
          return;
        case Stage::FilterS:
          b = st.filter->apply(std::move(b));
          idx++;
          break;
        case Stage::ProjectS:
          b = st.project->apply(std::move(b));
          idx++;
          break;
      }
    }
    FAIL("batch fed past end of chain");
  }

  DevBatch host_out_to_dev(const std::pair<int64_t, std::vector<HostOutCol>>& ob,
                           const std::vector<OutField>& fields) {
    DevBatch b;
    b.num_rows = ob.first;
    for (size_t i = 0; i < ob.second.size(); i++) {
      const HostOutCol& h = ob.second[i];
      DevColumn c;
      c.dt = h.dt;
      c.len = ob.first;
      if (h.dt == DType::Binary || h.dt == DType::Utf8) {
        c.own_offsets.alloc(h.offsets.size() * 4);
        AURON_HIP(hipMemcpyAsync(c.own_offsets.get(), h.offsets.data(),
                                 h.offsets.size() * 4, hipMemcpyHostToDevice,
                                 stream));
        c.offsets = c.own_offsets.get<int32_t>();
        c.data_len = (int64_t)h.values.size();
        c.own_values.alloc(h.values.empty() ? 1 : h.values.size());
        if (!h.values.empty())
          AURON_HIP(hipMemcpyAsync(c.own_values.get(), h.values.data(),
                                   h.values.size(), hipMemcpyHostToDevice,
                                   stream));
        c.values = c.own_values.get();
      } else {
        c.own_values.alloc(h.values.size());
        PinnedUploader::inst().copy(c.own_values.get(), h.values.data(),
                                    h.values.size(), stream);
        c.values = c.own_values.get();
      }
      if (!h.validity.empty()) {
        c.own_validity.alloc(h.validity.size());
        AURON_HIP(hipMemcpyAsync(c.own_validity.get(), h.validity.data(),
                                 h.validity.size(), hipMemcpyHostToDevice,
                                 stream));
        c.validity = c.own_validity.get<uint8_t>();
      }
      (void)fields;
      b.cols.push_back(std::move(c));
    }
    return b;
  }

  int32_t next_batch() {
    try {
      if (!started) {
        started = true;
        run();
      }
      if (!schema_sent && !out_fields.empty() && cb.import_schema) {
        ArrowSchema s;
        export_schema(out_fields, &s);
        cb.import_schema(cb.user, &s);
        schema_sent = true;
      }
      if (emit_dev_idx < outputs_dev.size()) {
        auto& ob = outputs_dev[emit_dev_idx++];
        // per-call export scaffolding: child arrays point at the DevBufs
        // (owned by outputs_dev until finalize); the struct itself is only
        // valid during the callback — the consumer copies the pointers out.
        std::vector<ArrowArray> children(ob.second.size());
        std::vector<ArrowArray*> child_ptrs(ob.second.size());
        std::vector<std::array<const void*, 3>> bufs(ob.second.size());
        for (size_t ci = 0; ci < ob.second.size(); ci++) {
          DevOutCol& oc = ob.second[ci];
          ArrowArray& ch = children[ci];
          memset(&ch, 0, sizeof(ch));
          ch.length = ob.first;
          ch.null_count = -1;  // unknown; validity bitmap always attached
          bufs[ci][0] = oc.validity.size() ? oc.validity.get() : nullptr;
          if (oc.dt == DType::Binary || oc.dt == DType::Utf8) {
            bufs[ci][1] = oc.offsets.get();
            bufs[ci][2] = oc.values.get();
            ch.n_buffers = 3;
            // Arrow C has no data-length field; stash it for the consumer
            ch.private_data = (void*)(intptr_t)oc.data_len;
          } else {
            bufs[ci][1] = oc.values.get();
            ch.n_buffers = 2;
          }
          ch.buffers = (const void**)bufs[ci].data();
          child_ptrs[ci] = &ch;
        }
        ArrowDeviceArray dev;
        memset(&dev, 0, sizeof(dev));
        dev.array.length = ob.first;
        dev.array.n_children = (int64_t)children.size();
        dev.array.children = child_ptrs.data();
        int did = 0;
        (void)hipGetDevice(&did);
        dev.device_id = did;
        dev.device_type = 10;  // ARROW_DEVICE_ROCM
        cb.import_device_batch(cb.user, &dev, nullptr);
        return 1;
      }
      if (emit_idx >= outputs.size()) return 0;
      auto& ob = outputs[emit_idx++];
      ArrowArray a;
      export_batch(ob.first, std::move(ob.second), &a);
      if (cb.import_batch) cb.import_batch(cb.user, &a);
      return 1;
    } catch (const std::exception& e) {
      set_error(e.what());
      return 0;
    }
  }
};

std::mutex g_mu;
std::map<int64_t, std::unique_ptr<Runtime>> g_runtimes;
int64_t g_next_handle = 1;

}  // namespace
}  // namespace auron

using namespace auron;

extern "C" {

int64_t auron_call_native(const uint8_t* task_definition, size_t len,
                          AuronCallbacks* callbacks) {
  auto rt = std::make_unique<Runtime>();
  if (callbacks) rt->cb = *callbacks;
  try {
    DBG("call_native len=%zu", len);
    int ndev = 0;
    hip_check(hipGetDeviceCount(&ndev), "hipGetDeviceCount");
    if (ndev == 0) throw EngineError("no HIP device visible");
    AURON_HIP(hipStreamCreate(&rt->stream));
    std::string err;
    rt->td = decode_task_definition(task_definition, len, &err);
    if (!rt->td) throw EngineError("plan decode failed: " + err);
    DBG("call_native ready");
  } catch (const std::exception& e) {
    rt->set_error(e.what());
    return 0;
  }
  std::lock_guard<std::mutex> lk(g_mu);
  int64_t h = g_next_handle++;
  g_runtimes[h] = std::move(rt);
  return h;
}

int32_t auron_next_batch(int64_t handle) {
  Runtime* rt;
  {
    std::lock_guard<std::mutex> lk(g_mu);
    auto it = g_runtimes.find(handle);
    if (it == g_runtimes.end()) return 0;
    rt = it->second.get();
  }
  return rt->next_batch();
}

void auron_finalize(int64_t handle) {
  std::lock_guard<std::mutex> lk(g_mu);
  g_runtimes.erase(handle);
}

void auron_on_exit(void) {
  std::lock_guard<std::mutex> lk(g_mu);
  g_runtimes.clear();
  DevPool::inst().clear([](void* p) { (void)hipFree(p); });
}

const char* auron_version(void) { return "auron-hip 0.1 gfx950"; }

// test-only: host-side batch_serde + IPC block codec roundtrip over a
// 3-column batch (i64 key, f64 val, binary) — CPU-checkable byte parity
// against the oracle writer plus write->read self-consistency, covering the
// engine's IpcReader decode path without a GPU. Returns the block-stream
// length (copied into out up to cap), or -1 with the error text in out.
int64_t auron_debug_serde_roundtrip(const int64_t* k, const uint8_t* kvalid,
                                    const double* v, const uint8_t* vvalid,
                                    const int32_t* boffs, const uint8_t* bdata,
                                    int64_t n, int64_t batch_size, uint8_t* out,
                                    size_t cap) {
  auto fail = [&](const std::string& m) -> int64_t {
    snprintf((char*)out, cap, "%s", m.c_str());
    return -1;
  };
  std::string err;
  std::vector<HostCol> cols(3);
  cols[0] = {8, (const uint8_t*)k, kvalid, nullptr};
  cols[1] = {8, (const uint8_t*)v, vvalid, nullptr};
  cols[2] = {0, bdata, nullptr, boffs};
  IpcBlockWriter w;
  for (int64_t beg = 0; beg < n; beg += batch_size) {
    int64_t end = std::min(n, beg + batch_size);
    std::vector<uint8_t> payload;
    serde_write_batch(cols, beg, end, &payload);
    if (!w.write_payload(payload.data(), payload.size(), &err))
      return fail(err);
  }
  if (!w.finish_block(&err)) return fail(err);
  std::vector<uint8_t> stream = w.take();

  // decode side: blocks -> payload -> batches; verify every row round-trips
  std::vector<uint8_t> payload;
  if (!ipc_decode_blocks(stream.data(), stream.size(), &payload, &err))
    return fail(err);
  size_t pos = 0;
  int64_t row = 0;
  while (pos < payload.size()) {
    size_t used = 0;
    int64_t rows = 0;
    std::vector<OwnedCol> rc;
    if (!serde_read_batch(payload.data() + pos, payload.size() - pos, &used,
                          {8, 8, 0}, &rows, &rc, &err))
      return fail(err);
    pos += used;
    for (int64_t i = 0; i < rows; i++, row++) {
      bool kv2 = rc[0].validity.empty() ||
                 ((rc[0].validity[i >> 3] >> (i & 7)) & 1);
      bool kv1 = !kvalid || ((kvalid[row >> 3] >> (row & 7)) & 1);
      if (kv1 != kv2) return fail("key validity mismatch");
      if (kv1 && memcmp(rc[0].values.data() + i * 8, (const uint8_t*)&k[row],
                        8) != 0)
        return fail("key value mismatch");
      bool vv2 = rc[1].validity.empty() ||
                 ((rc[1].validity[i >> 3] >> (i & 7)) & 1);
      bool vv1 = !vvalid || ((vvalid[row >> 3] >> (row & 7)) & 1);
      if (vv1 != vv2) return fail("val validity mismatch");
      if (vv1 && memcmp(rc[1].values.data() + i * 8, (const uint8_t*)&v[row],
                        8) != 0)
        return fail("val value mismatch");
      int32_t l1 = boffs[row + 1] - boffs[row];
      int32_t l2 = rc[2].offsets[i + 1] - rc[2].offsets[i];
      if (l1 != l2) return fail("binary length mismatch");
      if (l1 && memcmp(rc[2].values.data() + rc[2].offsets[i],
                       bdata + boffs[row], l1) != 0)
        return fail("binary bytes mismatch");
    }
  }
  if (row != n) return fail("row count mismatch");
  if (stream.size() <= cap) memcpy(out, stream.data(), stream.size());
  return (int64_t)stream.size();
}

// test-only: parse a parquet file on the host and summarize chunk decode
// results (CPU-checkable against pyarrow metadata — no GPU needed)
int32_t auron_debug_parquet_summary(const char* path, char* out, size_t cap) {
  std::string r;
  try {
    ParquetFile pf(path);
    r = "cols=";
    for (const auto& c : pf.columns())
      r += c.name + ":" + std::to_string(c.physical_type) +
           (c.nullable ? "?" : "") + ",";
    r += " rgs=" + std::to_string(pf.num_row_groups());
    for (int rg = 0; rg < pf.num_row_groups(); rg++) {
      r += " [rg" + std::to_string(rg) + " rows=" +
           std::to_string(pf.row_group_rows(rg));
      for (size_t c = 0; c < pf.columns().size(); c++) {
        PqColumnChunkData cd = pf.read_chunk(rg, (int)c);
        pq_materialize_gpu_staging(&cd);  // host-expand staged GPU runs
        // value checksum: wrapping i64 sum of the raw fixed-width values
        // (sign-extended for narrow ints) over the dense non-null stream —
        // lets CPU tests verify DECODED VALUES against numpy ground truth,
        // not just counts
        int64_t csum = 0;
        const size_t w = dtype_width(pf.columns()[c].dtype());
        auto add_vals = [&](const uint8_t* v, size_t cnt) {
          for (size_t i = 0; i < cnt; i++) {
            int64_t x = 0;
            if (w == 8) {
              memcpy(&x, v + i * 8, 8);
            } else if (w == 4) {
              int32_t x32;
              memcpy(&x32, v + i * 4, 4);
              x = x32;
            }
            csum = (int64_t)((uint64_t)csum + (uint64_t)x);
          }
        };
        if (w > 0) {
          if (cd.uses_dict) {
            for (uint32_t ix : cd.dict_indices)
              add_vals(cd.dict_values.data() + (size_t)ix * w, 1);
          } else {
            add_vals(cd.plain.data(), cd.plain.size() / w);
          }
        } else if (!cd.bin_offsets.empty()) {
          // byte-array columns: position-weighted rolling checksum over the
          // row-aligned lengths and data bytes
          for (size_t i = 0; i + 1 < cd.bin_offsets.size(); i++) {
            int64_t l = cd.bin_offsets[i + 1] - cd.bin_offsets[i];
            csum = (int64_t)((uint64_t)csum + (uint64_t)(l * (int64_t)(i + 1)));
          }
          for (size_t j = 0; j < cd.bin_data.size(); j++)
            csum = (int64_t)((uint64_t)csum +
                             (uint64_t)((int64_t)cd.bin_data[j] *
                                        (int64_t)(j + 1)));
        }
        r += " c" + std::to_string(c) + "{n=" + std::to_string(cd.num_values) +
             ",nulls=" + std::to_string(cd.null_count) +
             ",dict=" + std::to_string(cd.uses_dict ? cd.dict_count : 0) +
             ",csum=" + std::to_string(csum) + "}";
      }
      r += "]";
    }
  } catch (const std::exception& ex) {
    r = std::string("ERROR: ") + ex.what();
  }
  if (r.size() + 1 > cap) return -1;
  memcpy(out, r.c_str(), r.size() + 1);
  return (int32_t)r.size();
}

// test-only: decode a ScalarValue ipc_bytes literal and render it
int32_t auron_debug_decode_scalar(const uint8_t* data, size_t len, char* out,
                                  size_t cap) {
  ScalarLit s;
  std::string err;
  std::string r;
  if (!decode_ipc_scalar(data, len, &s, &err)) {
    r = "ERROR: " + err;
  } else if (s.is_null) {
    r = "null dtype=" + std::to_string((int)s.dtype);
  } else if (s.dtype == DType::Utf8 || s.dtype == DType::Binary) {
    r = "str:" + s.utf8;
  } else if (s.dtype == DType::Float64 || s.dtype == DType::Float32) {
    char buf[64];
    snprintf(buf, sizeof(buf), "f:%.17g", s.f64);
    r = buf;
  } else {
    r = "i:" + std::to_string(s.i64) + " dtype=" + std::to_string((int)s.dtype);
  }
  if (r.size() + 1 > cap) return -1;
  memcpy(out, r.c_str(), r.size() + 1);
  return (int32_t)r.size();
}

// test-only: exercise the get_conf callback plumbing from the C side (the
// ctypes out-param contract is easy to get wrong — see GET_CONF note)
int32_t auron_debug_conf_roundtrip(AuronCallbacks* cb, const char* key,
                                   char* out, size_t cap) {
  Conf conf{cb};
  std::string v = conf.get(key, "<default>");
  if (v.size() + 1 > cap) return -1;
  memcpy(out, v.c_str(), v.size() + 1);
  return (int32_t)v.size();
}

// Compute murmur3(seed 42) + pmod partition ids for host-resident i64 keys on
// the GPU (shuffle/mod.rs:163-188). Used by bench.py's RCCL exchange leg to
// route partial-agg records to their owner rank. Returns 0 on success.
int32_t auron_partition_ids(const int64_t* keys, int64_t n, int32_t P,
                            uint32_t* out) {
  try {
    // cached stream + pinned staging: this helper runs once per bench step
    // and pageable H2D/D2H of the ~1M-group arrays costs milliseconds
    static hipStream_t s = nullptr;
    static PinnedBuf pin_in, pin_out;
    if (!s) AURON_HIP(hipStreamCreate(&s));
    if (pin_in.size() < (size_t)n * 8) {
      pin_in.alloc(n * 8);
      pin_out.alloc(n * 4);
    }
    memcpy(pin_in.get(), keys, n * 8);
    DevBuf d_keys(n * 8), d_hash(n * 4), d_out(n * 4);
    AURON_HIP(hipMemcpyAsync(d_keys.get(), pin_in.get(), n * 8,
                             hipMemcpyHostToDevice, s));
    launch_hash_init(d_hash.get<int32_t>(), 42, n, s);
    launch_hash_fold_i64(d_keys.get<int64_t>(), nullptr, n,
                         d_hash.get<int32_t>(), s);
    launch_pmod(d_hash.get<int32_t>(), n, P, d_out.get<uint32_t>(), s);
    AURON_HIP(hipMemcpyAsync(pin_out.get(), d_out.get(), n * 4,
                             hipMemcpyDeviceToHost, s));
    AURON_HIP(hipStreamSynchronize(s));
    memcpy(out, pin_out.get(), n * 4);
    return 0;
  } catch (const std::exception&) {
    return -1;
  }
}

// ---- device repartition (in-memory exchange prep) --------------------------
// The multi-GPU exchange analog of sort_batches_by_partition_id +
// create_batch_interleaver (buffered_data.rs:284-351, selection.rs:65-300):
// murmur3(seed 42) partition ids -> stable sort by (dest rank = pid % world,
// pid) -> gather the (key, a8 accbuf) records into dest-rank-major,
// partition-ordered layout entirely in HBM, returning the per-rank row/byte
// splits the RCCL all-to-all needs. Input/output pointers are device memory;
// output buffers are owned by the returned handle until
// auron_repartition_free.
namespace auron {
namespace {
struct RepartOut {
  DevBuf keys, kvalid, offsets, data;
  hipStream_t stream = nullptr;
};
std::mutex g_rp_mu;
std::map<int64_t, std::unique_ptr<RepartOut>> g_repart;
int64_t g_rp_next = 1;
}  // namespace
}  // namespace auron

extern "C" {

int64_t auron_repartition_device(int64_t n, const void* keys,
                                 const void* key_validity,
                                 const void* offsets, const void* data,
                                 int32_t num_partitions, int32_t world,
                                 const void** out_keys,
                                 const void** out_key_validity,
                                 const void** out_offsets,
                                 const void** out_data,
                                 int64_t* rank_rows, int64_t* rank_bytes) {
  try {
    if (n <= 0 || num_partitions <= 0 || world <= 0 || world > 64) return 0;
    auto rp = std::make_unique<RepartOut>();
    AURON_HIP(hipStreamCreate(&rp->stream));
    hipStream_t s = rp->stream;
    DevBuf hash(n * 4), pids(n * 4), ord(n * 4), ord_sorted(n * 4);
    DevBuf idx(n * 4), perm(n * 4), counts(world * 16);
    launch_hash_init(hash.get<int32_t>(), 42, n, s);
    launch_hash_fold_i64((const int64_t*)keys, nullptr, n, hash.get<int32_t>(),
                         s);
    launch_pmod(hash.get<int32_t>(), n, num_partitions, pids.get<uint32_t>(),
                s);
    launch_exchange_ord(pids.get<uint32_t>(), n, (uint32_t)num_partitions,
                        (uint32_t)world, ord.get<uint32_t>(), s);
    AURON_HIP(hipMemsetAsync(counts.get(), 0, world * 16, s));
    launch_dest_counts(pids.get<uint32_t>(), (const int32_t*)offsets, n,
                       (uint32_t)world, counts.get<unsigned long long>(),
                       counts.get<unsigned long long>() + world, s);
    launch_iota_u32(idx.get<uint32_t>(), n, s);
    // ord < world * P: sort only the live bits
    int end_bit = 1;
    while ((1u << end_bit) < (uint32_t)world * (uint32_t)num_partitions &&
           end_bit < 32)
      end_bit++;
    size_t tb = 0;
    sort_pairs_u32_u32(ord.get<uint32_t>(), idx.get<uint32_t>(),
                       ord_sorted.get<uint32_t>(), perm.get<uint32_t>(), n,
                       end_bit, nullptr, &tb, s);
    DevBuf tmp(tb);
    sort_pairs_u32_u32(ord.get<uint32_t>(), idx.get<uint32_t>(),
                       ord_sorted.get<uint32_t>(), perm.get<uint32_t>(), n,
                       end_bit, tmp.get(), &tb, s);
    rp->keys.alloc(n * 8);
    launch_gather_8((const uint8_t*)keys, perm.get<uint32_t>(), n,
                    rp->keys.get<uint8_t>(), s);
    if (key_validity) {
      rp->kvalid.alloc((n + 7) / 8);
      launch_gather_bits((const uint8_t*)key_validity, perm.get<uint32_t>(), n,
                         rp->kvalid.get<uint8_t>(), s);
    }
    if (offsets && data) {
      DevBuf lens((n + 1) * 4);
      rp->offsets.alloc((n + 1) * 4);
      launch_gather_lens((const int32_t*)offsets, perm.get<uint32_t>(), n,
                         lens.get<int32_t>(), s);
      size_t stb = 0;
      scan_counts_matrix(lens.get<uint32_t>(), rp->offsets.get<uint32_t>(),
                         n + 1, nullptr, &stb, s);
      DevBuf stmp(stb);
      scan_counts_matrix(lens.get<uint32_t>(), rp->offsets.get<uint32_t>(),
                         n + 1, stmp.get(), &stb, s);
      PinnedBuf pin;
      pin.alloc(16);
      AURON_HIP(hipMemcpyAsync(pin.get(), rp->offsets.get<uint32_t>() + n, 4,
                               hipMemcpyDeviceToHost, s));
      AURON_HIP(hipStreamSynchronize(s));
      int64_t total = (int64_t)*pin.get<uint32_t>();
      rp->data.alloc(total ? total : 1);
      launch_gather_bytes((const uint8_t*)data, (const int32_t*)offsets,
                          perm.get<uint32_t>(), rp->offsets.get<int32_t>(), n,
                          rp->data.get<uint8_t>(), s);
    }
    std::vector<unsigned long long> h_counts(world * 2);
    AURON_HIP(hipMemcpyAsync(h_counts.data(), counts.get(), world * 16,
                             hipMemcpyDeviceToHost, s));
    AURON_HIP(hipStreamSynchronize(s));
    for (int32_t w = 0; w < world; w++) {
      if (rank_rows) rank_rows[w] = (int64_t)h_counts[w];
      if (rank_bytes) rank_bytes[w] = (int64_t)h_counts[world + w];
    }
    if (out_keys) *out_keys = rp->keys.get();
    if (out_key_validity)
      *out_key_validity = key_validity ? rp->kvalid.get() : nullptr;
    if (out_offsets) *out_offsets = rp->offsets.get();
    if (out_data) *out_data = rp->data.get();
    std::lock_guard<std::mutex> lk(g_rp_mu);
    int64_t h = g_rp_next++;
    g_repart[h] = std::move(rp);
    return h;
  } catch (const std::exception&) {
    return 0;
  }
}

void auron_repartition_free(int64_t handle) {
  std::lock_guard<std::mutex> lk(g_rp_mu);
  auto it = g_repart.find(handle);
  if (it != g_repart.end()) {
    if (it->second->stream) {
      (void)hipStreamSynchronize(it->second->stream);
      (void)hipStreamDestroy(it->second->stream);
    }
    g_repart.erase(it);
  }
}

}  // extern "C"

// test-only introspection: decode a TaskDefinition and render a one-line
// summary (verifies the hand-rolled proto reader against encoders)
int32_t auron_debug_decode_plan(const uint8_t* data, size_t len, char* out,
                                size_t out_cap) {
  std::string err;
  auto td = decode_task_definition(data, len, &err);
  std::string s;
  if (!td) {
    s = "ERROR: " + err;
  } else {
    s = "task stage=" + std::to_string(td->stage_id) +
        " part=" + std::to_string(td->partition_id) +
        " tid=" + std::to_string(td->task_id) + " plan=";
    const PlanNode* p = td->plan.get();
    while (p) {
      switch (p->kind) {
        case PlanNode::ShuffleWriter: {
          const auto& sw = *p->shuffle_writer;
          s += "ShuffleWriter(kind=" + std::to_string((int)sw.partitioning.kind) +
               ",P=" + std::to_string(sw.partitioning.partition_count) +
               ",nhash=" + std::to_string(sw.partitioning.hash_exprs.size()) +
               ",data=" + sw.output_data_file + ")->";
          p = sw.input.get();
          break;
        }
        case PlanNode::Agg: {
          const auto& ag = *p->agg;
          s += "Agg(mode=" +
               std::to_string(ag.modes.empty() ? -1 : (int)ag.modes[0]) +
               ",ngroup=" + std::to_string(ag.grouping_exprs.size()) +
               ",nagg=" + std::to_string(ag.agg_exprs.size());
          for (const auto& e : ag.agg_exprs)
            s += ",fn" + std::to_string(e.agg_function) + "rt" +
                 std::to_string((int)e.return_type);
          s += ",skip=" + std::to_string((int)ag.supports_partial_skipping) +
               ")->";
          p = ag.input.get();
          break;
        }
        case PlanNode::Filter: {
          s += "Filter(npred=" +
               std::to_string(p->filter->predicates.size()) + ")->";
          p = p->filter->input.get();
          break;
        }
        case PlanNode::Projection: {
          s += "Project(ncols=" +
               std::to_string(p->projection->exprs.size()) + ")->";
          p = p->projection->input.get();
          break;
        }
        case PlanNode::ParquetScan: {
          const auto& pq = *p->parquet;
          s += "ParquetScan(nfiles=" + std::to_string(pq.files.size()) +
               ",nfields=" + std::to_string(pq.schema.fields.size()) +
               ",nproj=" + std::to_string(pq.projection.size()) + ")";
          p = nullptr;
          break;
        }
        case PlanNode::IpcReader: {
          const auto& ir = *p->ipc_reader;
          s += "IpcReader(nfields=" + std::to_string(ir.schema.fields.size()) +
               ",rid=" + ir.resource_id + ")";
          p = nullptr;
          break;
        }
        case PlanNode::FFIReader: {
          const auto& fr = *p->ffi_reader;
          s += "FFIReader(nfields=" + std::to_string(fr.schema.fields.size()) +
               ",rid=" + fr.resource_id + ")";
          p = nullptr;
          break;
        }
      }
    }
  }
  if (s.size() + 1 > out_cap) return -1;
  memcpy(out, s.c_str(), s.size() + 1);
  return (int32_t)s.size();
}

int64_t auron_get_metric(int64_t handle, const char* name) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto it = g_runtimes.find(handle);
  if (it == g_runtimes.end()) return -1;
  auto mit = it->second->metrics.find(name);
  return mit == it->second->metrics.end() ? -1 : mit->second;
}

}  // extern "C"
