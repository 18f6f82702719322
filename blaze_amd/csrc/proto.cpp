// proto.cpp — hand-written proto3 wire-format reader for the plan subset.
// No protoc/protobuf-C++ exists in this image; golden
// encodings for tests are generated with Python google.protobuf.
// Wire format: https://protobuf.dev/programming-guides/encoding/
#include "plan.h"

#include <cstring>

namespace auron {
namespace {

struct Reader {
  const uint8_t* p;
  const uint8_t* end;
  bool ok = true;

  uint64_t varint() {
    uint64_t v = 0;
    int shift = 0;
    while (p < end) {
      uint8_t b = *p++;
      v |= (uint64_t)(b & 0x7f) << shift;
      if (!(b & 0x80)) return v;
      shift += 7;
      if (shift >= 64) break;
    }
    ok = false;
    return 0;
  }

  // returns (field_number, wire_type); field 0 = end
  std::pair<uint32_t, uint32_t> tag() {
    if (p >= end) return {0, 0};
    uint64_t t = varint();
    if (!ok) return {0, 0};
    return {(uint32_t)(t >> 3), (uint32_t)(t & 7)};
  }

  Reader sub() {
    uint64_t len = varint();
    if (!ok || (uint64_t)(end - p) < len) {
      ok = false;
      return {p, p};
    }
    Reader r{p, p + len};
    p += len;
    return r;
  }

  std::string str() {
    Reader r = sub();
    return std::string((const char*)r.p, (size_t)(r.end - r.p));
  }

  void skip(uint32_t wire_type) {
    switch (wire_type) {
      case 0: varint(); break;
      case 1: p = (end - p >= 8) ? p + 8 : (ok = false, end); break;
      case 2: sub(); break;
      case 5: p = (end - p >= 4) ? p + 4 : (ok = false, end); break;
      default: ok = false;
    }
  }
};

DType decode_arrow_type(Reader r) {
  // ArrowType oneof: tag number IS the type id (auron.proto:860-896)
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    if (f >= 1 && f <= 15 && w == 2) {
      r.sub();  // EmptyMessage payload
      return (DType)f;
    }
    r.skip(w);
  }
  return DType::Unsupported;
}

Field decode_field(Reader r) {
  Field f;
  while (true) {
    auto [n, w] = r.tag();
    if (n == 0) break;
    switch (n) {
      case 1: f.name = r.str(); break;
      case 2: f.dtype = decode_arrow_type(r.sub()); break;
      case 3: f.nullable = r.varint() != 0; break;
      default: r.skip(w);
    }
  }
  return f;
}

Schema decode_schema(Reader r) {
  Schema s;
  while (true) {
    auto [n, w] = r.tag();
    if (n == 0) break;
    if (n == 1 && w == 2) s.fields.push_back(decode_field(r.sub()));
    else r.skip(w);
  }
  return s;
}

Expr decode_expr(Reader r, std::string* err);

Expr decode_column(Reader r) {
  Expr e;
  e.kind = Expr::Column;
  while (true) {
    auto [n, w] = r.tag();
    if (n == 0) break;
    switch (n) {
      case 1: e.col_name = r.str(); break;   // PhysicalColumn.name
      case 2: e.col_index = (uint32_t)r.varint(); break;  // .index
      default: r.skip(w);
    }
  }
  return e;
}

Expr decode_agg_expr(Reader r, std::string* err) {
  // PhysicalAggExprNode (auron.proto:143-148)
  Expr e;
  e.kind = Expr::AggExpr;
  while (true) {
    auto [n, w] = r.tag();
    if (n == 0) break;
    switch (n) {
      case 1: e.agg_function = (int32_t)r.varint(); break;
      case 3: e.children.push_back(decode_expr(r.sub(), err)); break;
      case 4: e.return_type = decode_arrow_type(r.sub()); break;
      default: r.skip(w);
    }
  }
  return e;
}

Expr decode_expr(Reader r, std::string* err) {
  // PhysicalExprNode oneof (auron.proto:58-121)
  while (true) {
    auto [n, w] = r.tag();
    if (n == 0) break;
    switch (n) {
      case 1: return decode_column(r.sub());
      case 2: {  // ScalarValue{ipc_bytes = 1} (auron.proto:824-826)
        Expr e;
        e.kind = Expr::Literal;
        Reader s = r.sub();
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1 && sw == 2) {
            Reader bytes = s.sub();
            e.literal_ipc.assign(bytes.p, bytes.end);
          } else {
            s.skip(sw);
          }
        }
        return e;
      }
      case 4: {  // PhysicalBinaryExprNode{l=1, r=2, op=3}
        Expr e;
        e.kind = Expr::BinaryExpr;
        Reader s = r.sub();
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1) e.children.push_back(decode_expr(s.sub(), err));
          else if (sf == 2) e.children.push_back(decode_expr(s.sub(), err));
          else if (sf == 3) e.op = s.str();
          else s.skip(sw);
        }
        return e;
      }
      case 5: return decode_agg_expr(r.sub(), err);
      case 6: case 7: {  // PhysicalIsNull / PhysicalIsNotNull {expr = 1}
        Expr e;
        e.kind = (n == 7) ? Expr::IsNotNull : Expr::IsNull;
        Reader s = r.sub();
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1) e.children.push_back(decode_expr(s.sub(), err));
          else s.skip(sw);
        }
        return e;
      }
      default:
        if (err->empty())
          *err = "unsupported PhysicalExprNode field " + std::to_string(n);
        r.skip(w);
    }
  }
  return Expr{};
}

std::unique_ptr<PlanNode> decode_plan_node(Reader r, std::string* err);

std::unique_ptr<AggNode> decode_agg_node(Reader r, std::string* err) {
  auto n = std::make_unique<AggNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: n->input = decode_plan_node(r.sub(), err); break;
      case 2: n->exec_mode = (int32_t)r.varint(); break;
      case 3: n->grouping_exprs.push_back(decode_expr(r.sub(), err)); break;
      case 4: n->agg_exprs.push_back(decode_expr(r.sub(), err)); break;
      case 5:
        if (w == 2) {  // packed repeated enum
          Reader s = r.sub();
          while (s.p < s.end) n->modes.push_back((AggMode)s.varint());
        } else {
          n->modes.push_back((AggMode)r.varint());
        }
        break;
      case 6: n->grouping_names.push_back(r.str()); break;
      case 7: n->agg_names.push_back(r.str()); break;
      case 8: n->initial_input_buffer_offset = r.varint(); break;
      case 9: n->supports_partial_skipping = r.varint() != 0; break;
      default: r.skip(w);
    }
  }
  return n;
}

Repartition decode_repartition(Reader r, std::string* err) {
  Repartition p;
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: {  // PhysicalSingleRepartition
        Reader s = r.sub();
        p.kind = Repartition::Single;
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1) p.partition_count = s.varint();
          else s.skip(sw);
        }
        break;
      }
      case 2: {  // PhysicalHashRepartition (auron.proto:642-645)
        Reader s = r.sub();
        p.kind = Repartition::Hash;
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1) p.hash_exprs.push_back(decode_expr(s.sub(), err));
          else if (sf == 2) p.partition_count = s.varint();
          else s.skip(sw);
        }
        break;
      }
      case 3: {  // round robin
        Reader s = r.sub();
        p.kind = Repartition::RoundRobin;
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 1) p.partition_count = s.varint();
          else s.skip(sw);
        }
        break;
      }
      default:
        if (err->empty())
          *err = "unsupported PhysicalRepartition kind " + std::to_string(f);
        r.skip(w);
    }
  }
  return p;
}

std::unique_ptr<ShuffleWriterNode> decode_shuffle_writer(Reader r,
                                                         std::string* err) {
  auto n = std::make_unique<ShuffleWriterNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: n->input = decode_plan_node(r.sub(), err); break;
      case 2: n->partitioning = decode_repartition(r.sub(), err); break;
      case 3: n->output_data_file = r.str(); break;
      case 4: n->output_index_file = r.str(); break;
      default: r.skip(w);
    }
  }
  return n;
}

std::unique_ptr<ParquetScanNode> decode_parquet_scan(Reader r,
                                                     std::string* err) {
  auto n = std::make_unique<ParquetScanNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    if (f == 1) {  // FileScanExecConf (auron.proto:404-413)
      Reader s = r.sub();
      while (true) {
        auto [sf, sw] = s.tag();
        if (sf == 0) break;
        switch (sf) {
          case 3: {  // FileGroup{files = 1: PartitionedFile{path = 1}}
            Reader fg = s.sub();
            while (true) {
              auto [ff, fw] = fg.tag();
              if (ff == 0) break;
              if (ff == 1) {
                Reader pf = fg.sub();
                while (true) {
                  auto [pff, pfw] = pf.tag();
                  if (pff == 0) break;
                  if (pff == 1) n->files.push_back(pf.str());
                  else pf.skip(pfw);
                }
              } else {
                fg.skip(fw);
              }
            }
            break;
          }
          case 4: n->schema = decode_schema(s.sub()); break;
          case 6:
            if (sw == 2) {  // packed repeated uint32
              Reader pr = s.sub();
              while (pr.p < pr.end)
                n->projection.push_back((uint32_t)pr.varint());
            } else {
              n->projection.push_back((uint32_t)s.varint());
            }
            break;
          default: s.skip(sw);
        }
      }
    } else if (f == 3) {
      n->fs_resource_id = r.str();
    } else if (f == 2) {
      n->pruning.push_back(decode_expr(r.sub(), err));
    } else {
      r.skip(w);
    }
  }
  return n;
}

std::unique_ptr<IpcReaderNode> decode_ipc_reader(Reader r) {
  auto n = std::make_unique<IpcReaderNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: n->num_partitions = (uint32_t)r.varint(); break;
      case 2: n->schema = decode_schema(r.sub()); break;
      case 3: n->resource_id = r.str(); break;
      default: r.skip(w);
    }
  }
  return n;
}

std::unique_ptr<FFIReaderNode> decode_ffi_reader(Reader r) {
  auto n = std::make_unique<FFIReaderNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: n->num_partitions = (uint32_t)r.varint(); break;
      case 2: n->schema = decode_schema(r.sub()); break;
      case 3: n->resource_id = r.str(); break;
      default: r.skip(w);
    }
  }
  return n;
}

std::unique_ptr<FilterNode> decode_filter(Reader r, std::string* err) {
  auto n = std::make_unique<FilterNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    if (f == 1) n->input = decode_plan_node(r.sub(), err);
    else if (f == 2) n->predicates.push_back(decode_expr(r.sub(), err));
    else r.skip(w);
  }
  return n;
}

std::unique_ptr<ProjectionNode> decode_projection(Reader r, std::string* err) {
  auto n = std::make_unique<ProjectionNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    if (f == 1) n->input = decode_plan_node(r.sub(), err);
    else if (f == 2) n->exprs.push_back(decode_expr(r.sub(), err));
    else if (f == 3) n->names.push_back(r.str());
    else r.skip(w);
  }
  return n;
}

std::unique_ptr<PlanNode> decode_plan_node(Reader r, std::string* err) {
  auto node = std::make_unique<PlanNode>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 2:  // ShuffleWriterExecNode
        node->kind = PlanNode::ShuffleWriter;
        node->shuffle_writer = decode_shuffle_writer(r.sub(), err);
        return node;
      case 3:  // IpcReaderExecNode
        node->kind = PlanNode::IpcReader;
        node->ipc_reader = decode_ipc_reader(r.sub());
        return node;
      case 5:  // ParquetScanExecNode
        node->kind = PlanNode::ParquetScan;
        node->parquet = decode_parquet_scan(r.sub(), err);
        return node;
      case 6:  // ProjectionExecNode
        node->kind = PlanNode::Projection;
        node->projection = decode_projection(r.sub(), err);
        return node;
      case 8:  // FilterExecNode
        node->kind = PlanNode::Filter;
        node->filter = decode_filter(r.sub(), err);
        return node;
      case 16:  // AggExecNode
        node->kind = PlanNode::Agg;
        node->agg = decode_agg_node(r.sub(), err);
        return node;
      case 18:  // FFIReaderExecNode
        node->kind = PlanNode::FFIReader;
        node->ffi_reader = decode_ffi_reader(r.sub());
        return node;
      default:
        // Fail loudly: every other operator is outside the hot-path scope
        //  and must not silently no-op.
        if (err->empty())
          *err = "unsupported PhysicalPlanNode kind (proto field " +
                 std::to_string(f) + ")";
        r.skip(w);
        return nullptr;
    }
  }
  if (err->empty()) *err = "empty PhysicalPlanNode";
  return nullptr;
}

}  // namespace

std::unique_ptr<TaskDefinition> decode_task_definition(const uint8_t* data,
                                                       size_t len,
                                                       std::string* err) {
  Reader r{data, data + len};
  auto td = std::make_unique<TaskDefinition>();
  while (true) {
    auto [f, w] = r.tag();
    if (f == 0) break;
    switch (f) {
      case 1: {  // PartitionId (auron.proto:729-733)
        Reader s = r.sub();
        while (true) {
          auto [sf, sw] = s.tag();
          if (sf == 0) break;
          if (sf == 2) td->stage_id = (uint32_t)s.varint();
          else if (sf == 4) td->partition_id = (uint32_t)s.varint();
          else if (sf == 5) td->task_id = s.varint();
          else s.skip(sw);
        }
        break;
      }
      case 2:
        td->plan = decode_plan_node(r.sub(), err);
        break;
      default: r.skip(w);
    }
  }
  if (!r.ok && err->empty()) *err = "malformed protobuf";
  if (!td->plan && err->empty()) *err = "TaskDefinition without plan";
  if (!err->empty()) return nullptr;
  return td;
}

}  // namespace auron
