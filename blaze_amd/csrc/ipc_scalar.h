// ipc_scalar.h — minimal Arrow IPC stream reader for ScalarValue literals.
// The reference encodes expression literals as an Arrow IPC stream holding a
// single-row, single-column RecordBatch (auron.proto:824-826 ScalarValue;
// auron-serde/src/lib.rs:447-456 reads it back with arrow-ipc). This decodes
// exactly that shape for the primitive types on the hot path.
#pragma once

#include <cstddef>
#include <cstdint>
#include <string>

#include "plan.h"

namespace auron {

struct ScalarLit {
  DType dtype = DType::Unsupported;
  bool is_null = true;
  int64_t i64 = 0;    // Int8..Int64/UInt*/Date32 widened
  double f64 = 0.0;   // Float32/Float64
  std::string utf8;   // Utf8/Binary
};

// returns false and fills err on malformed/unsupported input
bool decode_ipc_scalar(const uint8_t* data, size_t len, ScalarLit* out,
                       std::string* err);

}  // namespace auron
