"""plan.py — builds serialized protobuf TaskDefinition plans.

Pure-python proto3 wire-format encoder for the operator surface the engine
accepts (auron-serde/proto/auron.proto; field numbers cited per message).
There is no protoc in this image; the encoding is verified against the
engine's independent C++ decoder (auron_debug_decode_plan) in tests.
"""
import struct

# AggFunction enum (auron.proto:128-141)
AGG_MIN, AGG_MAX, AGG_SUM, AGG_AVG, AGG_COUNT = 0, 1, 2, 3, 4
AGG_FIRST, AGG_FIRST_IGNORES_NULL = 7, 8
# AggMode enum (auron.proto:692-696)
MODE_PARTIAL, MODE_PARTIAL_MERGE, MODE_FINAL = 0, 1, 2

# ArrowType oneof tags (auron.proto:860-896)
DT_INT32, DT_INT64, DT_FLOAT64, DT_UTF8, DT_BINARY = 8, 10, 13, 14, 15


def _varint(v):
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(field, wire):
    return _varint((field << 3) | wire)


def _len_field(field, payload):
    return _tag(field, 2) + _varint(len(payload)) + payload


def _varint_field(field, v):
    return _tag(field, 0) + _varint(v)


def arrow_type(dt_tag):
    return _len_field(dt_tag, b"")  # EmptyMessage payload


def field(name, dt_tag, nullable):
    # Field (auron.proto:750-757)
    out = _len_field(1, name.encode())
    out += _len_field(2, arrow_type(dt_tag))
    if nullable:
        out += _varint_field(3, 1)
    return out


def schema(fields):
    # Schema (auron.proto:746-748)
    return b"".join(_len_field(1, f) for f in fields)


def column(name, index):
    # PhysicalExprNode{column = 1} -> PhysicalColumn (auron.proto:489-492)
    col = _len_field(1, name.encode()) + (_varint_field(2, index) if index else b"")
    return _len_field(1, col)


def agg_expr(fn, children, return_type_tag):
    # PhysicalExprNode{agg_expr = 5} -> PhysicalAggExprNode (auron.proto:143-148)
    node = _varint_field(1, fn) if fn else b""
    for c in children:
        node += _len_field(3, c)
    node += _len_field(4, arrow_type(return_type_tag))
    return _len_field(5, node)


def ffi_reader(fields, resource_id, num_partitions=1):
    # PhysicalPlanNode{ffi_reader = 18} (auron.proto:703-707)
    node = _varint_field(1, num_partitions)
    node += _len_field(2, schema(fields))
    node += _len_field(3, resource_id.encode())
    return _len_field(18, node)


def agg(input_plan, grouping, aggs, modes, grouping_names, agg_names,
        supports_partial_skipping=False, exec_mode=0):
    # PhysicalPlanNode{agg = 16} -> AggExecNode (auron.proto:675-685)
    node = _len_field(1, input_plan)
    node += _varint_field(2, exec_mode)
    for g in grouping:
        node += _len_field(3, g)
    for a in aggs:
        node += _len_field(4, a)
    node += _len_field(5, b"".join(_varint(m) for m in modes))  # packed enums
    for n in grouping_names:
        node += _len_field(6, n.encode())
    for n in agg_names:
        node += _len_field(7, n.encode())
    if supports_partial_skipping:
        node += _varint_field(9, 1)
    return _len_field(16, node)


def hash_repartition(hash_exprs, partition_count):
    # PhysicalRepartition{hash_repartition = 2} (auron.proto:629-645)
    inner = b"".join(_len_field(1, e) for e in hash_exprs)
    inner += _varint_field(2, partition_count)
    return _len_field(2, inner)


def single_repartition():
    return _len_field(1, _varint_field(1, 1))


def shuffle_writer(input_plan, repartition, data_file, index_file):
    # PhysicalPlanNode{shuffle_writer = 2} (auron.proto:524-529)
    node = _len_field(1, input_plan)
    node += _len_field(2, repartition)
    node += _len_field(3, data_file.encode())
    node += _len_field(4, index_file.encode())
    return _len_field(2, node)


def task_definition(plan, stage_id=0, partition_id=0, task_id=0):
    # TaskDefinition (auron.proto:735-740) + PartitionId (:729-733)
    pid = _varint_field(2, stage_id) + _varint_field(4, partition_id) + \
        _varint_field(5, task_id)
    return _len_field(1, pid) + _len_field(2, plan)


# ---- canned plans for the north-star configs -------------------------------

def northstar_input_fields(key_nullable=False):
    return [field("key", DT_INT64, key_nullable), field("val", DT_FLOAT64, True)]


def sum_count_aggs(val_index=1):
    return [
        agg_expr(AGG_SUM, [column("val", val_index)], DT_FLOAT64),
        agg_expr(AGG_COUNT, [column("val", val_index)], DT_INT64),
    ]


AGG_COLLECT_LIST, AGG_COLLECT_SET = 5, 6
_AGG_FN = {"min": AGG_MIN, "max": AGG_MAX, "sum": AGG_SUM,
           "avg": AGG_AVG, "count": AGG_COUNT, "first": AGG_FIRST,
           "first_ignores_null": AGG_FIRST_IGNORES_NULL,
           "collect_list": AGG_COLLECT_LIST, "collect_set": AGG_COLLECT_SET}


def named_aggs(names, val_index=1, val_dt=DT_FLOAT64):
    """Agg exprs for a list of function names over the shared val column.
    val_dt is the declared accumulator/output type (sum.rs:78-88 casts the
    input to it; MIN/MAX preserve it — maxmin.rs:81-83)."""
    out = []
    for nm in names:
        fn = _AGG_FN[nm]
        dt = DT_INT64 if fn == AGG_COUNT else val_dt
        out.append(agg_expr(fn, [column("val", val_index)], dt))
    return out


def plan_partial_final_named(agg_fns, resource_id="input0", key_dt=DT_INT64,
                             val_dt=DT_FLOAT64):
    """FFIReader -> Agg(Partial) -> Agg(Final) with an arbitrary agg list
    (e.g. ["min", "max", "sum", "count", "avg"]) over the shared val column."""
    reader = ffi_reader([field("key", key_dt, False),
                         field("val", val_dt, True)], resource_id)
    names = list(agg_fns)
    partial = agg(reader, [column("key", 0)],
                  named_aggs(names, val_dt=val_dt),
                  [MODE_PARTIAL] * len(names), ["key"], names)
    final = agg(partial, [column("key", 0)],
                named_aggs(names, val_dt=val_dt),
                [MODE_FINAL] * len(names), ["key"], names)
    return task_definition(final)


def plan_partial_only_named(agg_fns, resource_id="input0", skipping=False):
    reader = ffi_reader(northstar_input_fields(), resource_id)
    names = list(agg_fns)
    partial = agg(reader, [column("key", 0)], named_aggs(names),
                  [MODE_PARTIAL] * len(names), ["key"], names,
                  supports_partial_skipping=skipping)
    return task_definition(partial)


def plan_final_only_named(agg_fns, resource_id="input0"):
    names = list(agg_fns)
    reader = ffi_reader([field("key", DT_INT64, True),
                         field("agg_buf", DT_BINARY, False)], resource_id)
    final = agg(reader, [column("key", 0)], named_aggs(names),
                [MODE_FINAL] * len(names), ["key"], names)
    return task_definition(final)


def plan_partial_final(resource_id="input0", skipping=False, key_dt=DT_INT64):
    """FFIReader -> Agg(Partial) -> Agg(Final): config 1/2 shape
    (mirrors agg_exec.rs fuzztest:714-843 topology)."""
    reader = ffi_reader([field("key", key_dt, False),
                         field("val", DT_FLOAT64, True)], resource_id)
    partial = agg(reader, [column("key", 0)], sum_count_aggs(1),
                  [MODE_PARTIAL, MODE_PARTIAL], ["key"], ["sum", "cnt"],
                  supports_partial_skipping=skipping)
    final = agg(partial, [column("key", 0)], sum_count_aggs(1),
                [MODE_FINAL, MODE_FINAL], ["key"], ["sum", "cnt"])
    return task_definition(final)


def plan_partial_only(resource_id="input0", skipping=False):
    reader = ffi_reader(northstar_input_fields(), resource_id)
    partial = agg(reader, [column("key", 0)], sum_count_aggs(1),
                  [MODE_PARTIAL, MODE_PARTIAL], ["key"], ["sum", "cnt"],
                  supports_partial_skipping=skipping)
    return task_definition(partial)


def plan_final_only(resource_id="input0"):
    """FFIReader(partial output schema) -> Agg(Final): stage-2 shape."""
    fields = [field("key", DT_INT64, True),
              field("#9223372036854775807", DT_BINARY, False)]
    reader = ffi_reader(fields, resource_id)
    final = agg(reader, [column("key", 0)], sum_count_aggs(1),
                [MODE_FINAL, MODE_FINAL], ["key"], ["sum", "cnt"])
    return task_definition(final)


def plan_partial_final_gkey(key_fields, agg_fns=("sum", "count"),
                            val_dt=DT_FLOAT64, resource_id="input0"):
    """FFIReader -> Agg(Partial) -> Agg(Final) over GENERALIZED grouping
    keys: key_fields = [(name, dt_tag, nullable), ...] (1..4 columns, Utf8 /
    Int64 / Int32 / Float64); the val column follows the key columns."""
    fields = [field(n, dt, nul) for (n, dt, nul) in key_fields]
    vi = len(fields)
    fields.append(field("val", val_dt, True))
    reader = ffi_reader(fields, resource_id)
    fn_tags = {"sum": AGG_SUM, "count": AGG_COUNT, "avg": AGG_AVG,
               "min": AGG_MIN, "max": AGG_MAX}
    aggs = []
    for fn in agg_fns:
        tag = fn_tags[fn]
        dt = DT_INT64 if tag == AGG_COUNT else             (DT_FLOAT64 if tag == AGG_AVG else val_dt)
        aggs.append(agg_expr(tag, [column("val", vi)], dt))
    grouping = [column(n, i) for i, (n, _, _) in enumerate(key_fields)]
    gnames = [n for (n, _, _) in key_fields]
    partial = agg(reader, grouping, aggs, [MODE_PARTIAL] * len(aggs),
                  gnames, list(agg_fns))
    final = agg(partial, [column(n, i) for i, (n, _, _)
                          in enumerate(key_fields)], aggs,
                [MODE_FINAL] * len(aggs), gnames, list(agg_fns))
    return task_definition(final)


def plan_partial_only_gkey(key_fields, agg_fns=("sum", "count"),
                           val_dt=DT_FLOAT64, resource_id="input0"):
    fields = [field(n, dt, nul) for (n, dt, nul) in key_fields]
    vi = len(fields)
    fields.append(field("val", val_dt, True))
    reader = ffi_reader(fields, resource_id)
    fn_tags = {"sum": AGG_SUM, "count": AGG_COUNT, "avg": AGG_AVG,
               "min": AGG_MIN, "max": AGG_MAX}
    aggs = []
    for fn in agg_fns:
        tag = fn_tags[fn]
        dt = DT_INT64 if tag == AGG_COUNT else             (DT_FLOAT64 if tag == AGG_AVG else val_dt)
        aggs.append(agg_expr(tag, [column("val", vi)], dt))
    grouping = [column(n, i) for i, (n, _, _) in enumerate(key_fields)]
    gnames = [n for (n, _, _) in key_fields]
    partial = agg(reader, grouping, aggs, [MODE_PARTIAL] * len(aggs),
                  gnames, list(agg_fns))
    return task_definition(partial)


def plan_agg_shuffle(data_file, index_file, num_partitions=200,
                     resource_id="input0", partition_id=0, skipping=False):
    """FFIReader -> Agg(Partial) -> ShuffleWriter(hash(key), P): config 4
    stage-1 shape (NativeShuffleExchangeBase.scala:246-293)."""
    reader = ffi_reader(northstar_input_fields(), resource_id)
    partial = agg(reader, [column("key", 0)], sum_count_aggs(1),
                  [MODE_PARTIAL, MODE_PARTIAL], ["key"], ["sum", "cnt"],
                  supports_partial_skipping=skipping)
    rep = hash_repartition([column("key", 0)], num_partitions)
    sw = shuffle_writer(partial, rep, data_file, index_file)
    return task_definition(sw, partition_id=partition_id)


def plan_shuffle_only(data_file, index_file, num_partitions=200,
                      resource_id="input0", fields=None, hash_col=("key", 0),
                      partition_id=0):
    if fields is None:
        fields = [field("key", DT_INT64, True),
                  field("#9223372036854775807", DT_BINARY, False)]
    reader = ffi_reader(fields, resource_id)
    rep = hash_repartition([column(*hash_col)], num_partitions)
    sw = shuffle_writer(reader, rep, data_file, index_file)
    return task_definition(sw, partition_id=partition_id)


# ---- filter / projection (a15 in-engine half) ------------------------------

def literal_ipc(value, dtype):
    """ScalarValue.ipc_bytes: an Arrow IPC stream of a 1-row single-column
    batch (auron-serde/src/lib.rs:447-456). Built with pyarrow, exactly like
    the JVM side serializes literals."""
    import io

    import pyarrow as pa
    import pyarrow.ipc as ipc

    arr = pa.array([value], type=dtype)
    batch = pa.record_batch([arr], names=["lit"])
    sink = io.BytesIO()
    with ipc.new_stream(sink, batch.schema) as w:
        w.write_batch(batch)
    return sink.getvalue()


def literal(value, dtype="int32"):
    import pyarrow as pa

    pa_type = {"int32": pa.int32(), "int64": pa.int64(),
               "float64": pa.float64(), "utf8": pa.utf8()}[dtype]
    # PhysicalExprNode{literal = 2} -> ScalarValue{ipc_bytes = 1}
    return _len_field(2, _len_field(1, literal_ipc(value, pa_type)))


def binary_expr(l, r, op):
    # PhysicalExprNode{binary_expr = 4} -> PhysicalBinaryExprNode{l=1,r=2,op=3}
    node = _len_field(1, l) + _len_field(2, r) + _len_field(3, op.encode())
    return _len_field(4, node)


def is_not_null(child):
    return _len_field(7, _len_field(1, child))


def filter_node(input_plan, predicates):
    # PhysicalPlanNode{filter = 8} -> FilterExecNode (auron.proto:363-366)
    node = _len_field(1, input_plan)
    for p in predicates:
        node += _len_field(2, p)
    return _len_field(8, node)


def projection(input_plan, exprs, names):
    # PhysicalPlanNode{projection = 6} -> ProjectionExecNode (:505-510)
    node = _len_field(1, input_plan)
    for e in exprs:
        node += _len_field(2, e)
    for n in names:
        node += _len_field(3, n.encode())
    return _len_field(6, node)


def plan_filter_project_agg(resource_id="input0", cutoff=500_000,
                            cutoff_dtype="int64"):
    """FFIReader(key,val) -> Filter(key < c) -> Project(key,val) ->
    Agg(Partial) -> Agg(Final): the config-3 shape minus Parquet decode."""
    reader = ffi_reader(northstar_input_fields(), resource_id)
    filt = filter_node(reader, [
        binary_expr(column("key", 0), literal(cutoff, cutoff_dtype), "Lt")])
    proj = projection(filt, [column("key", 0), column("val", 1)],
                      ["key", "val"])
    partial = agg(proj, [column("key", 0)], sum_count_aggs(1),
                  [MODE_PARTIAL, MODE_PARTIAL], ["key"], ["sum", "cnt"])
    final = agg(partial, [column("key", 0)], sum_count_aggs(1),
                [MODE_FINAL, MODE_FINAL], ["key"], ["sum", "cnt"])
    return task_definition(final)


# ---- IpcReaderExec (f.3: native shuffle read-back) -------------------------

def ipc_reader(fields, resource_id, num_partitions=1):
    # PhysicalPlanNode{ipc_reader = 3} -> IpcReaderExecNode (auron.proto:607-611)
    node = _varint_field(1, num_partitions)
    node += _len_field(2, schema(fields))
    node += _len_field(3, resource_id.encode())
    return _len_field(3, node)


def plan_agg_shuffle_named(data_file, index_file, agg_fns,
                           num_partitions=200, resource_id="input0",
                           val_dt=DT_FLOAT64, partition_id=0):
    """FFIReader -> Agg(Partial, arbitrary agg list) -> ShuffleWriter."""
    names = list(agg_fns)
    reader = ffi_reader([field("key", DT_INT64, False),
                         field("val", val_dt, True)], resource_id)
    partial = agg(reader, [column("key", 0)],
                  named_aggs(names, val_dt=val_dt),
                  [MODE_PARTIAL] * len(names), ["key"], names)
    rep = hash_repartition([column("key", 0)], num_partitions)
    sw = shuffle_writer(partial, rep, data_file, index_file)
    return task_definition(sw, partition_id=partition_id)


def plan_ipc_final_named(agg_fns, resource_id="ipc0", val_dt=DT_FLOAT64):
    names = list(agg_fns)
    fields = [field("key", DT_INT64, True),
              field("#9223372036854775807", DT_BINARY, False)]
    reader = ipc_reader(fields, resource_id)
    final = agg(reader, [column("key", 0)],
                named_aggs(names, val_dt=val_dt),
                [MODE_FINAL] * len(names), ["key"], names)
    return task_definition(final)


def plan_ipc_final(resource_id="ipc0"):
    """IpcReader(partial output schema) -> Agg(Final): the reference's
    stage-2 topology (NativeShuffleExchangeBase.scala:147-179) fully native."""
    fields = [field("key", DT_INT64, True),
              field("#9223372036854775807", DT_BINARY, False)]
    reader = ipc_reader(fields, resource_id)
    final = agg(reader, [column("key", 0)], sum_count_aggs(1),
                [MODE_FINAL, MODE_FINAL], ["key"], ["sum", "cnt"])
    return task_definition(final)


# ---- ParquetScanExec (a15 / config 3) --------------------------------------

def partitioned_file(path, size):
    # PartitionedFile (auron.proto:373-379)
    return _len_field(1, path.encode()) + _varint_field(2, size)


def parquet_scan(files, fields, projection=(), pruning=(), resource_id="fs0"):
    """PhysicalPlanNode{parquet_scan = 5} -> ParquetScanExecNode
    (auron.proto:415-419) with FileScanExecConf (:404-413) and optional
    pruning_predicates (= 2, row-group stats pruning).
    files: list of (path, size)."""
    fg = b"".join(_len_field(1, partitioned_file(p, sz)) for p, sz in files)
    conf = _varint_field(1, 1) + _varint_field(2, 0)
    conf += _len_field(3, fg)
    conf += _len_field(4, schema(fields))
    if projection:
        conf += _len_field(6, b"".join(_varint(i) for i in projection))
    node = _len_field(1, conf)
    for pr in pruning:
        node += _len_field(2, pr)
    node += _len_field(3, resource_id.encode())
    return _len_field(5, node)


def plan_parquet_filter_agg(files, cutoff=None, key_field="key",
                            val_field="val", key_dt=DT_INT64):
    """Config-3 topology: ParquetScan -> [Filter(key < c)] -> Agg(Partial) ->
    Agg(Final) over (key, val f64)."""
    fields = [field(key_field, key_dt, True), field(val_field, DT_FLOAT64, True)]
    node = parquet_scan(files, fields)
    if cutoff is not None:
        dt = {DT_INT64: "int64", DT_INT32: "int32"}[key_dt]
        node = filter_node(node, [
            binary_expr(column(key_field, 0), literal(cutoff, dt), "Lt")])
    partial = agg(node, [column(key_field, 0)], sum_count_aggs(1),
                  [MODE_PARTIAL, MODE_PARTIAL], [key_field], ["sum", "cnt"])
    final = agg(partial, [column(key_field, 0)], sum_count_aggs(1),
                [MODE_FINAL, MODE_FINAL], [key_field], ["sum", "cnt"])
    return task_definition(final)


def plan_shuffle_robin(data_file, index_file, num_partitions, partition_id,
                       fields=None):
    """FFIReader -> ShuffleWriter(RoundRobin(P)) (shuffle/mod.rs:190-202)."""
    if fields is None:
        fields = northstar_input_fields()
    reader = ffi_reader(fields, "input0")
    rep = _len_field(3, _varint_field(1, num_partitions))  # round_robin = 3
    sw = shuffle_writer(reader, rep, data_file, index_file)
    return task_definition(sw, partition_id=partition_id)
