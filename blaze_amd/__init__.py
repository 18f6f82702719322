"""blaze_amd — MI355X-native Auron hot-path engine (Python harness).

Loads the in-tree libauron_hip.so and drives its C ABI (include/auron_hip.h),
which mirrors the reference JNI surface (exec.rs:42-143). The compute path is
HIP on gfx950 — if the library or a GPU is missing, calls FAIL LOUDLY; there
is no CPU fallback.
"""
import ctypes
import os
import subprocess

import numpy as np

from . import plan  # noqa: F401

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libauron_hip.so")

c = ctypes


class ArrowSchema(c.Structure):
    pass


ArrowSchema._fields_ = [
    ("format", c.c_char_p),
    ("name", c.c_char_p),
    ("metadata", c.c_char_p),
    ("flags", c.c_int64),
    ("n_children", c.c_int64),
    ("children", c.POINTER(c.POINTER(ArrowSchema))),
    ("dictionary", c.POINTER(ArrowSchema)),
    ("release", c.c_void_p),
    ("private_data", c.c_void_p),
]


class ArrowArray(c.Structure):
    pass


ArrowArray._fields_ = [
    ("length", c.c_int64),
    ("null_count", c.c_int64),
    ("offset", c.c_int64),
    ("n_buffers", c.c_int64),
    ("n_children", c.c_int64),
    ("buffers", c.POINTER(c.c_void_p)),
    ("children", c.POINTER(c.POINTER(ArrowArray))),
    ("dictionary", c.POINTER(ArrowArray)),
    ("release", c.c_void_p),
    ("private_data", c.c_void_p),
]


class ArrowDeviceArray(c.Structure):
    _fields_ = [
        ("array", ArrowArray),
        ("device_id", c.c_int64),
        ("device_type", c.c_int32),
        ("sync_event", c.c_void_p),
        ("reserved", c.c_int64 * 3),
    ]


# NOTE: the out-param must be POINTER(c_char), not c_char_p — ctypes passes
# c_char_p callback args as immutable python bytes (a copy), so writes would
# never reach the caller's buffer.
GET_CONF = c.CFUNCTYPE(c.c_int, c.c_void_p, c.c_char_p, c.POINTER(c.c_char),
                       c.c_size_t)
NEXT_INPUT = c.CFUNCTYPE(c.c_int, c.c_void_p, c.c_char_p, c.POINTER(ArrowArray),
                         c.POINTER(ArrowSchema), c.POINTER(ArrowDeviceArray))
IMPORT_SCHEMA = c.CFUNCTYPE(None, c.c_void_p, c.POINTER(ArrowSchema))
IMPORT_BATCH = c.CFUNCTYPE(None, c.c_void_p, c.POINTER(ArrowArray))
IMPORT_DEV = c.CFUNCTYPE(None, c.c_void_p, c.POINTER(ArrowDeviceArray),
                         c.POINTER(ArrowSchema))
SET_ERROR = c.CFUNCTYPE(None, c.c_void_p, c.c_char_p)
NEXT_IPC = c.CFUNCTYPE(c.c_int, c.c_void_p, c.c_char_p,
                       c.POINTER(c.POINTER(c.c_uint8)),
                       c.POINTER(c.c_size_t))


class AuronCallbacks(c.Structure):
    _fields_ = [
        ("user", c.c_void_p),
        ("get_conf", GET_CONF),
        ("next_input_batch", NEXT_INPUT),
        ("import_schema", IMPORT_SCHEMA),
        ("import_batch", IMPORT_BATCH),
        ("import_device_batch", IMPORT_DEV),
        ("set_error", SET_ERROR),
        ("next_ipc_bytes", NEXT_IPC),
    ]


def build(force=False):
    """Compile libauron_hip.so for gfx950 in-tree (hipcc cross-compiles)."""
    if force or not os.path.exists(_SO):
        subprocess.run(["make", "-C", os.path.join(_DIR, "csrc")], check=True,
                       capture_output=True, text=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            build()
        _lib = c.CDLL(_SO)
        _lib.auron_call_native.restype = c.c_int64
        _lib.auron_call_native.argtypes = [c.c_char_p, c.c_size_t,
                                           c.POINTER(AuronCallbacks)]
        _lib.auron_next_batch.restype = c.c_int32
        _lib.auron_next_batch.argtypes = [c.c_int64]
        _lib.auron_finalize.argtypes = [c.c_int64]
        _lib.auron_version.restype = c.c_char_p
        _lib.auron_get_metric.restype = c.c_int64
        _lib.auron_get_metric.argtypes = [c.c_int64, c.c_char_p]
        _lib.auron_repartition_device.restype = c.c_int64
        _lib.auron_repartition_device.argtypes = [
            c.c_int64, c.c_void_p, c.c_void_p, c.c_void_p, c.c_void_p,
            c.c_int32, c.c_int32, c.POINTER(c.c_void_p),
            c.POINTER(c.c_void_p), c.POINTER(c.c_void_p),
            c.POINTER(c.c_void_p), c.POINTER(c.c_int64),
            c.POINTER(c.c_int64)]
        _lib.auron_repartition_free.argtypes = [c.c_int64]
        _lib.auron_debug_decode_plan.restype = c.c_int32
        _lib.auron_debug_decode_plan.argtypes = [c.c_char_p, c.c_size_t,
                                                 c.c_char_p, c.c_size_t]
    return _lib


def debug_decode_plan(task_bytes: bytes) -> str:
    buf = c.create_string_buffer(1 << 16)
    rc = lib().auron_debug_decode_plan(task_bytes, len(task_bytes), buf,
                                       len(buf))
    assert rc >= 0
    return buf.value.decode()


def _bitmap(valid, n):
    if valid is None:
        return None
    v = np.asarray(valid, dtype=bool)
    assert v.shape == (n,)
    if v.all():
        return None
    return np.packbits(v, bitorder="little")


class _HostBatch:
    """Builds an ArrowArray struct batch over numpy columns; keeps buffers
    alive while the engine imports."""

    def __init__(self, cols):
        # cols: list of (values ndarray, valid bool-array-or-None) or
        # ('binary', data u8 ndarray, offsets i32 ndarray, valid)
        self.keep = []
        self.children = []
        n = None
        for col in cols:
            ch = ArrowArray()
            if len(col) > 0 and isinstance(col[0], str) and col[0] == "binary":
                _, data, offsets, valid = col
                data = np.ascontiguousarray(data, dtype=np.uint8)
                offsets = np.ascontiguousarray(offsets, dtype=np.int32)
                rows = len(offsets) - 1
                bm = _bitmap(valid, rows)
                bufs = (c.c_void_p * 3)(
                    c.c_void_p(bm.ctypes.data) if bm is not None else None,
                    c.c_void_p(offsets.ctypes.data),
                    c.c_void_p(data.ctypes.data))
                ch.n_buffers = 3
                nulls = int((~np.asarray(valid, bool)).sum()) if valid is not None else 0
                self.keep += [data, offsets, bm]
            else:
                values, valid = col
                values = np.ascontiguousarray(values)
                rows = len(values)
                bm = _bitmap(valid, rows)
                bufs = (c.c_void_p * 2)(
                    c.c_void_p(bm.ctypes.data) if bm is not None else None,
                    c.c_void_p(values.ctypes.data))
                ch.n_buffers = 2
                nulls = int((~np.asarray(valid, bool)).sum()) if valid is not None else 0
                self.keep += [values, bm]
            n = rows if n is None else n
            assert rows == n
            ch.length = rows
            ch.null_count = nulls if nulls else 0
            ch.offset = 0
            ch.buffers = bufs
            ch.n_children = 0
            ch.release = None
            self.keep.append(bufs)
            self.children.append(ch)
        self.child_ptrs = (c.POINTER(ArrowArray) * len(self.children))(
            *[c.pointer(ch) for ch in self.children])
        self.top = ArrowArray()
        self.top.length = n or 0
        self.top.null_count = 0
        self.top.offset = 0
        self.top.n_buffers = 0
        self.top.buffers = None
        self.top.n_children = len(self.children)
        self.top.children = self.child_ptrs
        self.top.release = None


class DeviceBatch:
    """ArrowDeviceArray over raw HBM pointers (e.g. torch cuda tensors) —
    zero-copy input for the engine. cols: list of dicts
    {"ptr": data_ptr, "len": n, "validity_ptr": p-or-None, "null_count": k}.
    The caller must keep the backing tensors alive for the task lifetime."""

    def __init__(self, cols, device_id=0):
        self.keep = []
        self.children = []
        n = None
        for col in cols:
            ch = ArrowArray()
            rows = col["len"]
            n = rows if n is None else n
            assert rows == n
            if "offsets_ptr" in col:  # binary column: validity/offsets/data
                bufs = (c.c_void_p * 3)(
                    c.c_void_p(col.get("validity_ptr") or None),
                    c.c_void_p(col["offsets_ptr"]),
                    c.c_void_p(col["ptr"]))
                ch.n_buffers = 3
            else:
                bufs = (c.c_void_p * 2)(
                    c.c_void_p(col.get("validity_ptr") or None),
                    c.c_void_p(col["ptr"]))
                ch.n_buffers = 2
            ch.length = rows
            ch.null_count = col.get("null_count", 0)
            ch.offset = 0
            ch.buffers = bufs
            ch.n_children = 0
            ch.release = None
            self.keep.append(bufs)
            self.children.append(ch)
        self.child_ptrs = (c.POINTER(ArrowArray) * len(self.children))(
            *[c.pointer(ch) for ch in self.children])
        self.struct = ArrowDeviceArray()
        top = self.struct.array
        top.length = n or 0
        top.n_children = len(self.children)
        top.children = self.child_ptrs
        top.n_buffers = 0
        top.release = None
        self.struct.device_id = device_id
        self.struct.device_type = 10  # ARROW_DEVICE_ROCM

    def as_input(self):
        return {"struct": self.struct, "keep": self}


def partition_ids(keys, num_partitions):
    """murmur3(seed 42) + pmod on the GPU (engine kernels) for host i64 keys."""
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    out = np.empty(len(keys), dtype=np.uint32)
    L = lib()
    L.auron_partition_ids.restype = c.c_int32
    L.auron_partition_ids.argtypes = [c.c_void_p, c.c_int64, c.c_int32,
                                      c.c_void_p]
    rc = L.auron_partition_ids(keys.ctypes.data, len(keys), num_partitions,
                               out.ctypes.data)
    if rc != 0:
        raise RuntimeError("auron_partition_ids failed (GPU required)")
    return out


class DeviceRepartition:
    """Device-side exchange prep (auron_repartition_device): partition-id +
    stable sort + gather of (key, accbuf) records into dest-rank-major
    partition order, all in HBM. Pointers stay valid until .free()."""

    def __init__(self, n, keys_ptr, offsets_ptr, data_ptr, num_partitions,
                 world, key_validity_ptr=None):
        L = lib()
        ok = c.c_void_p()
        okv = c.c_void_p()
        oo = c.c_void_p()
        od = c.c_void_p()
        rows = (c.c_int64 * world)()
        bts = (c.c_int64 * world)()
        self.handle = L.auron_repartition_device(
            n, c.c_void_p(keys_ptr), c.c_void_p(key_validity_ptr or None),
            c.c_void_p(offsets_ptr or None), c.c_void_p(data_ptr or None),
            num_partitions, world, c.byref(ok), c.byref(okv), c.byref(oo),
            c.byref(od), rows, bts)
        if self.handle == 0:
            raise RuntimeError("auron_repartition_device failed "
                               "(GPU required)")
        self.n = n
        self.keys_ptr = ok.value
        self.key_validity_ptr = okv.value
        self.offsets_ptr = oo.value
        self.data_ptr = od.value
        self.rank_rows = list(rows)
        self.rank_bytes = list(bts)

    def free(self):
        if self.handle:
            lib().auron_repartition_free(self.handle)
            self.handle = 0

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass


def _read_bitmap(ptr, n):
    if not ptr:
        return None
    raw = np.ctypeslib.as_array(c.cast(ptr, c.POINTER(c.c_uint8)),
                                shape=((n + 7) // 8,))
    return np.unpackbits(raw, bitorder="little")[:n].astype(bool)


def _import_output(array_ptr, schema_fields):
    """Convert an engine-exported ArrowArray (host) into numpy columns."""
    a = array_ptr.contents
    out = []
    for i in range(a.n_children):
        ch = a.children[i].contents
        dt = schema_fields[i][1]
        n = ch.length
        valid = _read_bitmap(ch.buffers[0], n) if ch.null_count else None
        if dt.startswith("+l:"):
            # list<prim>: offsets on the parent, values on the child array
            offsets = np.ctypeslib.as_array(
                c.cast(ch.buffers[1], c.POINTER(c.c_int32)),
                shape=(n + 1,)).copy()
            item = ch.children[0].contents
            npdt = {"l": np.int64, "g": np.float64,
                    "i": np.int32}[dt[3:]]
            w = np.dtype(npdt).itemsize
            ni = int(item.length)
            vals = np.ctypeslib.as_array(
                c.cast(item.buffers[1], c.POINTER(c.c_uint8)),
                shape=(max(ni, 1) * w,))[:ni * w].view(npdt).copy()
            out.append(dict(dtype="list", offsets=offsets, values=vals,
                            valid=valid))
            continue
        if dt in ("z", "u"):
            offsets = np.ctypeslib.as_array(
                c.cast(ch.buffers[1], c.POINTER(c.c_int32)), shape=(n + 1,)).copy()
            nbytes = int(offsets[-1])
            data = np.ctypeslib.as_array(
                c.cast(ch.buffers[2], c.POINTER(c.c_uint8)),
                shape=(max(nbytes, 1),))[:nbytes].copy()
            out.append(dict(dtype="binary", offsets=offsets, data=data,
                            valid=valid))
        else:
            npdt = {"l": np.int64, "g": np.float64, "i": np.int32}[dt]
            vals = np.ctypeslib.as_array(
                c.cast(ch.buffers[1], c.POINTER(c.c_uint8)),
                shape=(n * np.dtype(npdt).itemsize,)).view(npdt).copy()
            out.append(dict(dtype=np.dtype(npdt).name, values=vals, valid=valid))
    return out


class Task:
    """One native task execution (mirrors JniBridge callNative/nextBatch)."""

    def __init__(self, task_bytes, batches=None, conf=None, device_batches=None,
                 ipc_segments=None):
        """batches: list of _HostBatch col-spec lists (host numpy input).
        device_batches: list of prebuilt ArrowDeviceArray + keepalive (see
        bench.py) for zero-copy HBM input.
        ipc_segments: list of bytes (shuffle block streams) for IpcReaderExec
        plans — plays the JVM fetch iterator's role."""
        self._conf = dict(conf or {})
        self._in_host = [_HostBatch(cols) for cols in (batches or [])]
        self._in_dev = list(device_batches or [])
        self._ipc = [np.frombuffer(seg, dtype=np.uint8).copy()
                     for seg in (ipc_segments or [])]
        self._ipc_cursor = 0
        self._cursor = 0
        self.schema_fields = []   # (name, fmt)
        self.outputs = []
        self.error = None
        self._keep = []

        @GET_CONF
        def get_conf(user, key, value, cap):
            k = key.decode()
            if k in self._conf:
                v = str(self._conf[k]).encode()[: cap - 1]
                c.memmove(value, v + b"\x00", len(v) + 1)
                return 0
            return 1

        @NEXT_INPUT
        def next_input(user, rid, arr, sch, dev):
            i = self._cursor
            if i >= len(self._in_host) + len(self._in_dev):
                return 0
            self._cursor += 1
            if i < len(self._in_host):
                hb = self._in_host[i]
                c.memmove(arr, c.byref(hb.top), c.sizeof(ArrowArray))
                return 1
            db = self._in_dev[i - len(self._in_host)]
            c.memmove(dev, c.byref(db["struct"]), c.sizeof(ArrowDeviceArray))
            return 2

        @IMPORT_SCHEMA
        def import_schema(user, sp):
            s = sp.contents
            for i in range(s.n_children):
                chs = s.children[i].contents
                fmt = chs.format.decode()
                if fmt == "+l":  # list<item>: record the item format too
                    item = chs.children[0].contents.format.decode()
                    fmt = "+l:" + item
                self.schema_fields.append(
                    (chs.name.decode() if chs.name else "", fmt))
            rel = s.release
            if rel:
                c.CFUNCTYPE(None, c.POINTER(ArrowSchema))(rel)(sp)

        @IMPORT_BATCH
        def import_batch(user, ap):
            self.outputs.append(_import_output(ap, self.schema_fields))
            rel = ap.contents.release
            if rel:
                c.CFUNCTYPE(None, c.POINTER(ArrowArray))(rel)(ap)

        @SET_ERROR
        def set_error(user, msg):
            self.error = msg.decode()

        @NEXT_IPC
        def next_ipc(user, rid, data_out, len_out):
            i = self._ipc_cursor
            if i >= len(self._ipc):
                return 0
            self._ipc_cursor += 1
            seg = self._ipc[i]
            data_out[0] = c.cast(c.c_void_p(seg.ctypes.data),
                                 c.POINTER(c.c_uint8))
            len_out[0] = seg.size
            return 1

        @IMPORT_DEV
        def import_device_batch(user, devp, schp):
            # pointers stay valid until finalize (engine contract); binary
            # data length rides in child.private_data (Arrow C has no
            # data-length field)
            d = devp.contents
            a = d.array
            cols = []
            for i in range(a.n_children):
                ch = a.children[i].contents
                fmt = self.schema_fields[i][1] if \
                    i < len(self.schema_fields) else ""
                col = {"len": int(ch.length), "fmt": fmt,
                       "validity_ptr": ch.buffers[0]}
                if ch.n_buffers == 3:
                    col["offsets_ptr"] = ch.buffers[1]
                    col["ptr"] = ch.buffers[2]
                    col["data_len"] = int(
                        c.cast(ch.private_data, c.c_void_p).value or 0)
                else:
                    col["ptr"] = ch.buffers[1]
                cols.append(col)
            self.device_outputs.append(
                {"num_rows": int(a.length), "device_id": int(d.device_id),
                 "cols": cols})

        self.device_outputs = []
        use_dev_out = str(self._conf.get("AURON_HIP_DEVICE_OUTPUT", "")) == "1"
        self._cb = AuronCallbacks(
            user=None, get_conf=get_conf, next_input_batch=next_input,
            import_schema=import_schema, import_batch=import_batch,
            import_device_batch=import_device_batch if use_dev_out
            else c.cast(None, IMPORT_DEV), set_error=set_error,
            next_ipc_bytes=next_ipc)
        self._keep += [get_conf, next_input, import_schema, import_batch,
                       set_error, next_ipc, import_device_batch]
        self.handle = lib().auron_call_native(task_bytes, len(task_bytes),
                                              c.byref(self._cb))
        if self.handle == 0:
            raise RuntimeError(f"auron_call_native failed: {self.error}")

    def run(self):
        """Pump all output batches (mirrors the JVM nextBatch loop)."""
        while lib().auron_next_batch(self.handle):
            pass
        if self.error:
            raise RuntimeError(f"native task failed: {self.error}")
        return self.outputs

    def metric(self, name):
        return lib().auron_get_metric(self.handle, name.encode())

    def finalize(self):
        if self.handle:
            lib().auron_finalize(self.handle)
            self.handle = 0

    def __del__(self):
        try:
            self.finalize()
        except Exception:
            pass
