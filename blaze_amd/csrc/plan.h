// plan.h — decoded physical-plan structs for the hot-path node subset.
// Mirrors auron-serde/proto/auron.proto (field numbers cited inline) and the
// operator-construction contract of from_proto.rs:110-500.
#pragma once

#include <cstdint>
#include <memory>
#include <string>
#include <vector>

namespace auron {

// ArrowType oneof tags (auron.proto:860-896)
enum class DType : int32_t {
  Unsupported = 0,
  Null = 1,
  Bool = 2,
  UInt8 = 3,
  Int8 = 4,
  UInt16 = 5,
  Int16 = 6,
  UInt32 = 7,
  Int32 = 8,
  UInt64 = 9,
  Int64 = 10,
  Float16 = 11,
  Float32 = 12,
  Float64 = 13,
  Utf8 = 14,
  Binary = 15,
};

inline size_t dtype_width(DType t) {
  switch (t) {
    case DType::Bool: return 0;  // bit-packed
    case DType::UInt8: case DType::Int8: return 1;
    case DType::UInt16: case DType::Int16: case DType::Float16: return 2;
    case DType::UInt32: case DType::Int32: case DType::Float32: return 4;
    case DType::UInt64: case DType::Int64: case DType::Float64: return 8;
    default: return 0;  // variable length / unsupported
  }
}

struct Field {
  std::string name;        // Field.name = 1 (auron.proto:750-757)
  DType dtype = DType::Unsupported;
  bool nullable = false;   // Field.nullable = 3
};

struct Schema {
  std::vector<Field> fields;  // Schema.columns = 1 (auron.proto:746-748)
};

// PhysicalExprNode subset (auron.proto:58-121)
struct Expr {
  enum Kind {
    Column,     // PhysicalColumn (auron.proto:489-492)
    AggExpr,    // PhysicalAggExprNode (auron.proto:143-148)
    Literal,    // ScalarValue{ipc_bytes} (auron.proto:824-826)
    BinaryExpr, // PhysicalBinaryExprNode (auron.proto:172-176)
    IsNotNull,  // PhysicalIsNotNull (auron.proto:161-163)
    IsNull,     // PhysicalIsNull (auron.proto:155-157)
  } kind = Column;
  std::string col_name;
  uint32_t col_index = 0;
  // AggFunction enum: MIN=0 MAX=1 SUM=2 AVG=3 COUNT=4. Default 0 (MIN):
  // proto3 omits zero-valued enums on the wire, so an absent field IS MIN.
  int32_t agg_function = 0;
  std::vector<Expr> children;  // agg args / binary l,r / null-check operand
  DType return_type = DType::Unsupported;
  std::string op;              // binary op name (auron-serde/src/lib.rs:70-96)
  std::vector<uint8_t> literal_ipc;  // raw ScalarValue.ipc_bytes
};

// AggFunction enum values (auron.proto:128-141)
enum AggFunction : int32_t {
  AGG_MIN = 0,
  AGG_MAX = 1,
  AGG_SUM = 2,
  AGG_AVG = 3,
  AGG_COUNT = 4,
  AGG_COLLECT_LIST = 5,
  AGG_COLLECT_SET = 6,
  AGG_FIRST = 7,
  AGG_FIRST_IGNORES_NULL = 8,
};

// AggMode enum (auron.proto:692-696)
enum class AggMode : int32_t { Partial = 0, PartialMerge = 1, Final = 2 };

struct PlanNode;

// AggExecNode (auron.proto:675-685)
struct AggNode {
  std::unique_ptr<PlanNode> input;                 // = 1
  int32_t exec_mode = 0;                           // = 2 (HASH_AGG=0)
  std::vector<Expr> grouping_exprs;                // = 3
  std::vector<Expr> agg_exprs;                     // = 4
  std::vector<AggMode> modes;                      // = 5
  std::vector<std::string> grouping_names;         // = 6
  std::vector<std::string> agg_names;              // = 7
  uint64_t initial_input_buffer_offset = 0;        // = 8
  bool supports_partial_skipping = false;          // = 9
};

// PhysicalRepartition (auron.proto:629-645)
struct Repartition {
  enum Kind { Single, Hash, RoundRobin } kind = Single;
  std::vector<Expr> hash_exprs;  // PhysicalHashRepartition.hash_expr = 1
  uint64_t partition_count = 1;  // .partition_count = 2
};

// ShuffleWriterExecNode (auron.proto:524-529)
struct ShuffleWriterNode {
  std::unique_ptr<PlanNode> input;  // = 1
  Repartition partitioning;         // = 2
  std::string output_data_file;     // = 3
  std::string output_index_file;    // = 4
};

// FFIReaderExecNode (auron.proto:703-707)
struct FFIReaderNode {
  uint32_t num_partitions = 1;   // = 1
  Schema schema;                 // = 2
  std::string resource_id;       // = 3
};

// ParquetScanExecNode (auron.proto:415-419) + FileScanExecConf (:404-413)
struct ParquetScanNode {
  std::vector<std::string> files;     // FileGroup.files[].path
  Schema schema;                      // FileScanExecConf.schema = 4
  std::vector<uint32_t> projection;   // FileScanExecConf.projection = 6
  std::vector<Expr> pruning;          // pruning_predicates = 2 (rg stats)
  std::string fs_resource_id;         // = 3 (unused: POSIX paths read directly)
};

// IpcReaderExecNode (auron.proto:607-611)
struct IpcReaderNode {
  uint32_t num_partitions = 1;   // = 1
  Schema schema;                 // = 2
  std::string resource_id;       // = 3 (ipc_provider_resource_id)
};

// FilterExecNode (auron.proto:363-366): predicates are ANDed
struct FilterNode {
  std::unique_ptr<PlanNode> input;  // = 1
  std::vector<Expr> predicates;     // = 2
};

// ProjectionExecNode (auron.proto:505-510)
struct ProjectionNode {
  std::unique_ptr<PlanNode> input;    // = 1
  std::vector<Expr> exprs;            // = 2
  std::vector<std::string> names;     // = 3
};

// PhysicalPlanNode oneof (auron.proto:27-55)
struct PlanNode {
  enum Kind {
    ShuffleWriter = 2,
    IpcReader = 3,
    ParquetScan = 5,
    Projection = 6,
    Filter = 8,
    Agg = 16,
    FFIReader = 18,
  } kind;
  std::unique_ptr<ShuffleWriterNode> shuffle_writer;
  std::unique_ptr<AggNode> agg;
  std::unique_ptr<FFIReaderNode> ffi_reader;
  std::unique_ptr<FilterNode> filter;
  std::unique_ptr<ProjectionNode> projection;
  std::unique_ptr<IpcReaderNode> ipc_reader;
  std::unique_ptr<ParquetScanNode> parquet;
};

// TaskDefinition (auron.proto:735-740) + PartitionId (:729-733)
struct TaskDefinition {
  uint32_t stage_id = 0;
  uint32_t partition_id = 0;
  uint64_t task_id = 0;
  std::unique_ptr<PlanNode> plan;
};

// Decode a serialized TaskDefinition. Returns nullptr and fills `err` on
// failure (unknown node kinds and expressions fail loudly — no silent
// fallback paths).
std::unique_ptr<TaskDefinition> decode_task_definition(const uint8_t* data,
                                                       size_t len,
                                                       std::string* err);

}  // namespace auron
