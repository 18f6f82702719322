/* auron_hip.h — C-ABI of the MI355X-native Auron hot-path engine.
 *
 * This is the drop-in boundary replacing the reference's JNI surface
 * (native-engine/auron/src/exec.rs:42-143):
 *   auron_call_native  <-> Java_org_apache_spark_sql_auron_JniBridge_callNative
 *                          (exec.rs:42-113): plan in, opaque runtime handle out
 *   auron_next_batch   <-> ..._nextBatch (exec.rs:116-123): pull-model, one
 *                          Arrow batch pushed through import_batch per call,
 *                          returns false on stream end
 *   auron_finalize     <-> ..._finalizeNative (exec.rs:127-135)
 *   auron_on_exit      <-> ..._onExit (exec.rs:138-143)
 *
 * Plan input is the reference's own serialized protobuf
 * TaskDefinition{task_id, plan} (auron-serde/proto/auron.proto:735-740),
 * decoded by a built-in proto3 wire-format reader — the operator surface is
 * identical to from_proto.rs:110-500 for the hot-path node subset
 * (ShuffleWriterExecNode, AggExecNode, FFIReaderExecNode).
 *
 * Batches cross the boundary as Arrow C Data Interface structs
 * (rt.rs:157-160,229-259 importSchema/importBatch), extended with the Arrow
 * C *Device* Data Interface so MI355X-resident batches pass zero-copy.
 *
 * A JNI binding for Spark loads this library and forwards the four symbols —
 * see INTEGRATION.md for the stub a maintainer would add.
 */
#ifndef AURON_HIP_H
#define AURON_HIP_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- Arrow C Data Interface (stable ABI, arrow.apache.org/docs/format/CDataInterface) ---- */
#ifndef ARROW_C_DATA_INTERFACE
#define ARROW_C_DATA_INTERFACE

#define ARROW_FLAG_DICTIONARY_ORDERED 1
#define ARROW_FLAG_NULLABLE 2
#define ARROW_FLAG_MAP_KEYS_SORTED 4

struct ArrowSchema {
  const char* format;
  const char* name;
  const char* metadata;
  int64_t flags;
  int64_t n_children;
  struct ArrowSchema** children;
  struct ArrowSchema* dictionary;
  void (*release)(struct ArrowSchema*);
  void* private_data;
};

struct ArrowArray {
  int64_t length;
  int64_t null_count;
  int64_t offset;
  int64_t n_buffers;
  int64_t n_children;
  const void** buffers;
  struct ArrowArray** children;
  struct ArrowArray* dictionary;
  void (*release)(struct ArrowArray*);
  void* private_data;
};
#endif /* ARROW_C_DATA_INTERFACE */

/* ---- Arrow C Device Data Interface (CDeviceDataInterface spec) ---- */
#ifndef ARROW_C_DEVICE_DATA_INTERFACE
#define ARROW_C_DEVICE_DATA_INTERFACE

#define ARROW_DEVICE_CPU 1
#define ARROW_DEVICE_ROCM 10

struct ArrowDeviceArray {
  struct ArrowArray array;   /* buffers are DEVICE pointers */
  int64_t device_id;
  int32_t device_type;       /* ARROW_DEVICE_ROCM for MI355X HBM */
  void* sync_event;          /* hipEvent_t or NULL */
  int64_t reserved[3];
};
#endif /* ARROW_C_DEVICE_DATA_INTERFACE */

/* ---- callbacks (replaces the JNI wrapper object + JniBridge.getResource) ---- */
typedef struct AuronCallbacks {
  void* user;

  /* Conf lookup by the reference's key names (auron-jni-bridge/src/conf.rs:
   * 32-111), e.g. "spark.auron.batchSize". Return 0 and fill `value` when the
   * key is set; nonzero = use the built-in default. May be NULL. */
  int (*get_conf)(void* user, const char* key, char* value, size_t value_cap);

  /* Input side — FFIReaderExecNode's export_iter_provider_resource_id
   * (auron.proto:703-707, parquet_exec.rs:160-163 getResource pattern).
   * Pull the next batch: return 1 and fill host `array`+`schema`, or return 2
   * and fill `dev` (device-resident, zero-copy), or 0 on exhaustion.
   * Ownership of filled structs moves to the engine (release() honored). */
  int (*next_input_batch)(void* user, const char* resource_id,
                          struct ArrowArray* array, struct ArrowSchema* schema,
                          struct ArrowDeviceArray* dev);

  /* Output side — mirrors wrapper.importSchema/importBatch
   * (rt.rs:157-160,236-241). Called with ownership moving to the callee. */
  void (*import_schema)(void* user, struct ArrowSchema* schema);
  void (*import_batch)(void* user, struct ArrowArray* array);
  /* Optional device export: if non-NULL, batches stay in HBM and are handed
   * over as ArrowDeviceArray instead of import_batch. */
  void (*import_device_batch)(void* user, struct ArrowDeviceArray* dev,
                              struct ArrowSchema* schema);

  /* Error channel — mirrors wrapper.setError (rt.rs:284-293). */
  void (*set_error)(void* user, const char* message);

  /* IpcReaderExecNode input (auron.proto:607-611; ipc_reader_exec.rs:62-120):
   * pull the next raw shuffle block-stream segment (repeated [u32-LE
   * len][lz4 frame] records, the format ShuffleWriter emits). Return 1 and
   * set *data/*len
   * (caller-owned, valid until the next call or finalize), 0 on exhaustion.
   * May be NULL when the plan holds no IpcReaderExec. */
  int (*next_ipc_bytes)(void* user, const char* resource_id,
                        const uint8_t** data, size_t* len);
} AuronCallbacks;

/* Create a runtime for one task. Returns a handle (>0) or 0 on error (error
 * text delivered via set_error). The plan starts executing on the first
 * auron_next_batch call. */
int64_t auron_call_native(const uint8_t* task_definition, size_t len,
                          AuronCallbacks* callbacks);

/* Pull one output batch (pushed through import_batch / import_device_batch
 * before returning). Returns 1 while batches remain, 0 at end-of-stream or
 * error. Mirrors exec.rs:116-123. */
int32_t auron_next_batch(int64_t handle);

/* Destroy the runtime. Mirrors exec.rs:127-135. */
void auron_finalize(int64_t handle);

/* Global teardown. Mirrors exec.rs:138-143. */
void auron_on_exit(void);

/* ---- device exchange prep (multi-GPU shuffle leg) ----
 * In-memory analog of sort_batches_by_partition_id + create_batch_interleaver
 * (buffered_data.rs:284-351, selection.rs:65-300) for the RCCL all-to-all
 * exchange: murmur3(seed 42) partition ids -> stable sort by
 * (dest rank = pid % world, pid) -> gather (key, accbuf) records into
 * dest-rank-major partition order, all in HBM. All data pointers are DEVICE
 * memory; out_* buffers are owned by the returned handle (>0) until
 * auron_repartition_free; rank_rows/rank_bytes are host arrays[world] = the
 * all-to-all split sizes. offsets/data may be NULL (keys-only). Returns 0 on
 * error. */
int64_t auron_repartition_device(int64_t n, const void* keys,
                                 const void* key_validity,
                                 const void* offsets, const void* data,
                                 int32_t num_partitions, int32_t world,
                                 const void** out_keys,
                                 const void** out_key_validity,
                                 const void** out_offsets,
                                 const void** out_data,
                                 int64_t* rank_rows, int64_t* rank_bytes);
void auron_repartition_free(int64_t handle);

/* ---- introspection for tests/bench ---- */
/* Version / build info, e.g. "auron-hip 0.1 gfx950". */
const char* auron_version(void);
/* Returns per-op metric value by name for the last finished runtime
 * (hashing_time_ns, output_rows, ... — metric names mirror
 * NativeAggBase.scala:74-94). Returns -1 if unknown. */
int64_t auron_get_metric(int64_t handle, const char* name);

#ifdef __cplusplus
}
#endif
#endif /* AURON_HIP_H */
