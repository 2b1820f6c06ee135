"""api.py — ctypes binding to libdd_shuffle.so (the C ABI, include/dd_shuffle.h).

Harness layer only: uploads/downloads numpy-described Arrow columns (the column-dict
convention of oracle/pyref.py) to device memory and drives the C ABI. The compute path is
entirely inside libdd_shuffle.so; there is NO CPU fallback here — calls on a GPU-less box
raise DDError(DD_ERR_NO_DEVICE), by design.
"""

import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libdd_shuffle.so")

DD_OK = 0
STATUS_NAMES = {
    0: "DD_OK", 1: "DD_ERR_INVALID", 2: "DD_ERR_NO_DEVICE", 3: "DD_ERR_HIP",
    4: "DD_ERR_RCCL", 5: "DD_ERR_NOT_FOUND", 6: "DD_ERR_UNSUPPORTED",
}

DTYPE_CODE = {"u8": 1, "i16": 2, "i32": 3, "i64": 4, "f32": 5, "f64": 6,
              "bool": 7, "utf8": 8, "dict32": 9}
FIXED_NP = {"u8": np.uint8, "bool": np.uint8, "i16": np.int16, "i32": np.int32,
            "f32": np.float32, "i64": np.int64, "f64": np.float64, "dict32": np.int32}
ELEM_SIZE = {"u8": 1, "bool": 1, "i16": 2, "i32": 4, "f32": 4, "i64": 8, "f64": 8,
             "dict32": 4}

DD_MAX_COLS = 24
UNIQUE_ID_BYTES = 128


class DDError(RuntimeError):
    def __init__(self, status, msg):
        super().__init__(f"{STATUS_NAMES.get(status, status)}: {msg}")
        self.status = status


class ColDesc(ctypes.Structure):
    _fields_ = [
        ("dtype", ctypes.c_int32),
        ("data", ctypes.c_void_p),
        ("validity", ctypes.c_void_p),
        ("offsets", ctypes.c_void_p),
        ("data_len", ctypes.c_int64),
        ("dict_bytes", ctypes.c_void_p),
        ("dict_offsets", ctypes.c_void_p),
        ("dict_n", ctypes.c_int64),
    ]


class BatchDesc(ctypes.Structure):
    _fields_ = [
        ("n_rows", ctypes.c_int64),
        ("n_cols", ctypes.c_int32),
        ("cols", ColDesc * DD_MAX_COLS),
    ]


class TaskKeyC(ctypes.Structure):
    _fields_ = [
        ("query_id_hi", ctypes.c_uint64),
        ("query_id_lo", ctypes.c_uint64),
        ("stage_id", ctypes.c_uint64),
        ("task_number", ctypes.c_uint64),
    ]


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO):
            raise FileNotFoundError(
                f"{_SO} missing — build it with __graft_entry__.build(); the product path "
                "has no fallback")
        L = ctypes.CDLL(_SO)
        L.dd_last_error.restype = ctypes.c_char_p
        L.dd_version.restype = ctypes.c_char_p
        L.dd_device_count.restype = ctypes.c_int
        L.dd_partitioner_pids.restype = ctypes.c_void_p
        L.dd_partitioner_col_data.restype = ctypes.c_void_p
        L.dd_partitioner_col_validity.restype = ctypes.c_void_p
        L.dd_partitioner_col_lengths.restype = ctypes.c_void_p
        L.dd_partitioner_var_offsets64.restype = ctypes.c_void_p
        L.dd_exchanged_col_data.restype = ctypes.c_void_p
        L.dd_exchanged_col_validity.restype = ctypes.c_void_p
        L.dd_exchanged_col_lengths.restype = ctypes.c_void_p
        L.dd_exchanged_total_rows.restype = ctypes.c_int64
        L.dd_bcast_n_rows.restype = ctypes.c_int64
        L.dd_bcast_n_cols.restype = ctypes.c_int32
        L.dd_bcast_col_data.restype = ctypes.c_void_p
        L.dd_bcast_col_validity.restype = ctypes.c_void_p
        L.dd_bcast_col_offsets.restype = ctypes.c_void_p
        L.dd_reducer_n_rows.restype = ctypes.c_int64
        L.dd_reducer_kernel_ms.restype = ctypes.c_float
        _lib = L
    return _lib


def _check(st):
    if st != DD_OK:
        raise DDError(st, lib().dd_last_error().decode())


def device_count():
    return lib().dd_device_count()


def _stream_arg(stream):
    """Accept None, a raw int hipStream_t, or a torch.cuda.Stream."""
    if stream is None:
        return None
    h = getattr(stream, "cuda_stream", stream)
    return ctypes.c_void_p(h)


def _dev_alloc(nbytes):
    p = ctypes.c_void_p()
    _check(lib().dd_dev_alloc(ctypes.c_int64(max(nbytes, 1)), ctypes.byref(p)))
    return p


def _h2d(dst, arr):
    arr = np.ascontiguousarray(arr)
    _check(lib().dd_memcpy_h2d(dst, arr.ctypes.data_as(ctypes.c_void_p),
                               ctypes.c_int64(arr.nbytes)))


def _d2h(src_ptr, nbytes, dtype):
    out = np.empty(nbytes // np.dtype(dtype).itemsize, dtype=dtype)
    _check(lib().dd_memcpy_d2h(out.ctypes.data_as(ctypes.c_void_p),
                               ctypes.c_void_p(src_ptr), ctypes.c_int64(nbytes)))
    return out


class DeviceBatch:
    """Device-resident batch uploaded from the column-dict convention."""

    def __init__(self, cols):
        self.cols = cols
        self.n_rows = (len(cols[0]["offsets"]) - 1 if cols[0]["dtype"] == "utf8"
                       else len(cols[0]["data"]))
        self._bufs = []
        self.desc = BatchDesc()
        self.desc.n_rows = self.n_rows
        self.desc.n_cols = len(cols)
        for i, col in enumerate(cols):
            cd = self.desc.cols[i]
            cd.dtype = DTYPE_CODE[col["dtype"]]
            if col["dtype"] == "utf8":
                data = np.ascontiguousarray(col["data"], dtype=np.uint8)
                off = np.ascontiguousarray(col["offsets"], dtype=np.int32)
                b = self._up(data)
                cd.data = b
                cd.offsets = self._up(off)
                cd.data_len = int(data.nbytes)
            else:
                npdt = FIXED_NP[col["dtype"]]
                data = np.ascontiguousarray(col["data"], dtype=npdt)
                cd.data = self._up(data)
                cd.data_len = 0
            if col.get("valid") is not None:
                cd.validity = self._up(np.ascontiguousarray(col["valid"], np.uint8))
            if col["dtype"] == "dict32":
                cd.dict_bytes = self._up(np.ascontiguousarray(col["dict_bytes"], np.uint8))
                cd.dict_offsets = self._up(np.ascontiguousarray(col["dict_offsets"], np.int32))
                cd.dict_n = len(col["dict_offsets"]) - 1

    @classmethod
    def from_device(cls, view_cols, n_rows):
        """Non-owning device-view batch: each col gives raw device pointers
        ({"dtype", "data_ptr", "valid_ptr", "offsets_ptr", "data_len", dict...}).
        Used by the two-level scatter to re-consume pass-A bucket slices without
        copies."""
        self = cls.__new__(cls)
        self.cols = view_cols
        self.n_rows = n_rows
        self._bufs = []
        self.desc = BatchDesc()
        self.desc.n_rows = n_rows
        self.desc.n_cols = len(view_cols)
        for i, col in enumerate(view_cols):
            cd = self.desc.cols[i]
            cd.dtype = DTYPE_CODE[col["dtype"]]
            cd.data = col["data_ptr"]
            cd.data_len = int(col.get("data_len", 0))
            if col.get("valid_ptr"):
                cd.validity = col["valid_ptr"]
            if col.get("offsets_ptr"):
                cd.offsets = col["offsets_ptr"]
            if col.get("dict_bytes_ptr"):
                cd.dict_bytes = col["dict_bytes_ptr"]
                cd.dict_offsets = col["dict_offsets_ptr"]
                cd.dict_n = col["dict_n"]
        return self

    def _up(self, arr):
        p = _dev_alloc(arr.nbytes)
        _h2d(p, arr)
        self._bufs.append(p)
        return p.value

    def free(self):
        for b in self._bufs:
            lib().dd_dev_free(b)
        self._bufs = []


class Partitioner:
    """Wraps dd_partitioner_*: the producer-head replacement."""

    def __init__(self, batch: DeviceBatch, key_idx, nparts, ranged=None):
        """ranged=(pid_total, coarse_div): coarse-bucket mode — nparts is then
        pid_total // coarse_div contiguous ranges (dd_partitioner_create_ranged)."""
        self.batch = batch
        self.key_idx = list(key_idx)
        keys = (ctypes.c_int32 * len(key_idx))(*key_idx)
        self.h = ctypes.c_void_p()
        if ranged is None:
            self.nparts = nparts
            _check(lib().dd_partitioner_create(ctypes.byref(batch.desc), keys,
                                               len(key_idx), ctypes.c_uint32(nparts),
                                               ctypes.byref(self.h)))
        else:
            total, div = ranged
            self.nparts = total // div
            _check(lib().dd_partitioner_create_ranged(
                ctypes.byref(batch.desc), keys, len(key_idx), ctypes.c_uint32(total),
                ctypes.c_uint32(div), ctypes.byref(self.h)))

    def run(self, stream=None):
        _check(lib().dd_partitioner_run(self.h, _stream_arg(stream)))

    def run_phase1(self, stream=None):
        """K1 hash+count + K2 scans (batch-pipelining split; see dd_shuffle.h)."""
        _check(lib().dd_partitioner_run_phase1(self.h, _stream_arg(stream)))

    def run_phase2(self, stream=None):
        """K3 scatter (+K4 var bytes)."""
        _check(lib().dd_partitioner_run_phase2(self.h, _stream_arg(stream)))

    def wait_phase1(self, stream):
        _check(lib().dd_partitioner_wait_phase1(self.h, _stream_arg(stream)))

    def wait_phase2(self, stream):
        _check(lib().dd_partitioner_wait_phase2(self.h, _stream_arg(stream)))

    def sync(self):
        _check(lib().dd_device_sync())

    def has_pid_array(self):
        """Fast null-check of the device pid pointer (no download): False when the spec
        path recomputes hashes in-kernel (rhash) and no pid array exists."""
        return bool(lib().dd_partitioner_pids(self.h))

    def pids(self):
        """Per-row partition ids (u32; widened from u8 when the precomputed-layout path
        stores byte pids), or None when the spec path recomputed hashes in-kernel (no
        pid array exists; outputs + row_offsets fully define the partitioning)."""
        p = lib().dd_partitioner_pids(self.h)
        if not p:
            return None
        if lib().dd_partitioner_pid_elem(self.h) == 1:
            return _d2h(p, self.batch.n_rows, np.uint8).astype(np.uint32)
        return _d2h(p, self.batch.n_rows * 4, np.uint32)

    def row_offsets(self):
        out = np.empty(self.nparts + 1, dtype=np.int64)
        _check(lib().dd_partitioner_row_offsets(
            self.h, out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))))
        return out

    def byte_offsets(self, col):
        out = np.empty(self.nparts + 1, dtype=np.int64)
        _check(lib().dd_partitioner_byte_offsets(
            self.h, col, out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))))
        return out

    def kernel_ms(self):
        out = (ctypes.c_float * 3)()
        _check(lib().dd_partitioner_kernel_ms(self.h, out))
        return list(out)

    def col_out(self, i):
        """Download partitioned output of column i (tests/smoke only)."""
        col = self.batch.cols[i]
        n = self.batch.n_rows
        res = {"dtype": col["dtype"]}
        if col["dtype"] == "utf8":
            dl = int(np.asarray(col["data"], dtype=np.uint8).nbytes)
            res["data"] = _d2h(lib().dd_partitioner_col_data(self.h, i), dl, np.uint8)
            res["lengths"] = _d2h(lib().dd_partitioner_col_lengths(self.h, i), n * 4,
                                  np.uint32)
        else:
            npdt = FIXED_NP[col["dtype"]]
            res["data"] = _d2h(lib().dd_partitioner_col_data(self.h, i),
                               n * ELEM_SIZE[col["dtype"]], npdt)
        vp = lib().dd_partitioner_col_validity(self.h, i)
        if vp:
            res["valid"] = _d2h(vp, n, np.uint8)
        return res

    def destroy(self):
        if self.h:
            lib().dd_partitioner_destroy(self.h)
            self.h = None


class Comm:
    """RCCL communicator (dd_comm_*)."""

    def __init__(self, unique_id: bytes, rank: int, nranks: int):
        assert len(unique_id) == UNIQUE_ID_BYTES
        self.rank, self.nranks = rank, nranks
        self.h = ctypes.c_void_p()
        buf = ctypes.create_string_buffer(unique_id, UNIQUE_ID_BYTES)
        _check(lib().dd_comm_init(buf, rank, nranks, ctypes.byref(self.h)))

    @staticmethod
    def unique_id() -> bytes:
        buf = ctypes.create_string_buffer(UNIQUE_ID_BYTES)
        _check(lib().dd_comm_unique_id(buf))
        return buf.raw

    def exchange(self, part: Partitioner, stream=None):
        h = ctypes.c_void_p()
        _check(lib().dd_exchange_run(self.h, part.h, _stream_arg(stream),
                                     ctypes.byref(h)))
        return Exchanged(h, part, self)

    def coalesce(self, part: Partitioner, consumer_tasks: int, stream=None):
        """NetworkCoalesceExec data plane (dd_coalesce_run): this consumer rank receives
        the whole partitioned output of every producer rank in its contiguous group."""
        h = ctypes.c_void_p()
        _check(lib().dd_coalesce_run(self.h, part.h, consumer_tasks,
                                     _stream_arg(stream), ctypes.byref(h)))
        ex = Exchanged(h, part, self)
        return ex

    def broadcast(self, batch, root=0, stream=None):
        """Replicate the root's device batch on every rank (dd_broadcast_run: the
        BroadcastExec/NetworkBroadcastExec data plane)."""
        h = ctypes.c_void_p()
        desc = ctypes.byref(batch.desc) if batch is not None else None
        _check(lib().dd_broadcast_run(self.h, desc, root, _stream_arg(stream),
                                      ctypes.byref(h)))
        return Broadcasted(h)

    def destroy(self):
        if self.h:
            lib().dd_comm_destroy(self.h)
            self.h = None


class Broadcasted:
    def __init__(self, h):
        self.h = h

    @property
    def n_rows(self):
        return lib().dd_bcast_n_rows(self.h)

    @property
    def n_cols(self):
        return lib().dd_bcast_n_cols(self.h)

    def col(self, i):
        dtype = ctypes.c_int32()
        dlen = ctypes.c_int64()
        _check(lib().dd_bcast_col_meta(self.h, i, ctypes.byref(dtype),
                                       ctypes.byref(dlen)))
        name = {v: k for k, v in DTYPE_CODE.items()}[dtype.value]
        n = self.n_rows
        out = {"dtype": name}
        if name == "utf8":
            out["data"] = _d2h(lib().dd_bcast_col_data(self.h, i), int(dlen.value), np.uint8)
            out["offsets"] = _d2h(lib().dd_bcast_col_offsets(self.h, i), (n + 1) * 4,
                                  np.int32)
        else:
            out["data"] = _d2h(lib().dd_bcast_col_data(self.h, i), int(dlen.value),
                               FIXED_NP[name])
        vp = lib().dd_bcast_col_validity(self.h, i)
        if vp:
            out["valid"] = _d2h(vp, n, np.uint8)
        return out

    def destroy(self):
        if self.h:
            lib().dd_bcast_destroy(self.h)
            self.h = None


class Exchanged:
    def __init__(self, h, part, comm):
        self.h = h
        self.part = part
        self.comm = comm

    @property
    def total_rows(self):
        return lib().dd_exchanged_total_rows(self.h)

    def row_counts(self, producers=None, parts=None):
        """Per-(producer, partition) row counts. Defaults fit dd_exchange_run's window
        shape; pass producers/parts explicitly for dd_coalesce_run results (group size,
        full P)."""
        producers = self.comm.nranks if producers is None else producers
        parts = (self.part.nparts // self.comm.nranks) if parts is None else parts
        out = np.empty(max(producers * parts, 1), dtype=np.int64)
        _check(lib().dd_exchanged_row_counts(
            self.h, out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))))
        return out.reshape(max(producers, 1), -1)

    def byte_counts(self, col):
        out = np.empty(self.comm.nranks * (self.part.nparts // self.comm.nranks),
                       dtype=np.int64)
        _check(lib().dd_exchanged_byte_counts(
            self.h, col, out.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))))
        return out.reshape(self.comm.nranks, -1)

    def col_data(self, i):
        col = self.part.batch.cols[i]
        n = self.total_rows
        if col["dtype"] == "utf8":
            total_b = int(self.byte_counts(i).sum())
            data = _d2h(lib().dd_exchanged_col_data(self.h, i), total_b, np.uint8)
            lens = _d2h(lib().dd_exchanged_col_lengths(self.h, i), n * 4, np.uint32)
            return {"dtype": "utf8", "data": data, "lengths": lens}
        npdt = FIXED_NP[col["dtype"]]
        return {"dtype": col["dtype"],
                "data": _d2h(lib().dd_exchanged_col_data(self.h, i),
                             n * ELEM_SIZE[col["dtype"]], npdt)}

    def col_validity(self, i):
        vp = lib().dd_exchanged_col_validity(self.h, i)
        return _d2h(vp, self.total_rows, np.uint8) if vp else None

    def stats(self):
        ms = ctypes.c_float()
        eg = ctypes.c_int64()
        _check(lib().dd_exchanged_stats(self.h, ctypes.byref(ms), ctypes.byref(eg)))
        return float(ms.value), int(eg.value)

    def destroy(self):
        if self.h:
            lib().dd_exchanged_destroy(self.h)
            self.h = None


AGG_OPS = {"sum_f64": 0, "count": 1, "sum_i64": 2, "min_f64": 3, "max_f64": 4,
           "min_i64": 5, "max_i64": 6}


def partial_reduce(batch: DeviceBatch, key_idx, aggs):
    """GPU partial aggregation below the shuffle (dd_partial_reduce_run).

    aggs: list of (col_index_or_None, op) with op in {"sum_f64", "count", "sum_i64"}.
    Returns {"keys": u64[n, nk] canonical bits, "keynull": u32[n] per-key null bitmask,
    "aggs": f64[n, na], "nn": u64[n, na] non-null input counts} — integer aggregates
    (count / sum_i64) are bit-cast in the f64 slots (view with .view(np.int64)). May
    contain duplicate groups (partial semantics). The final merge sums "nn" per group and
    emits NULL for an aggregate whose total non-null count is 0 (DataFusion semantics).
    """
    nk = len(key_idx)
    na = len(aggs)
    keysc = (ctypes.c_int32 * nk)(*key_idx)
    aggc = (ctypes.c_int32 * na)(*[c if c is not None else 0 for c, _ in aggs])
    opsc = (ctypes.c_int32 * na)(*[AGG_OPS[o] for _, o in aggs])
    h = ctypes.c_void_p()
    _check(lib().dd_partial_reduce_run(ctypes.byref(batch.desc), keysc, nk, aggc, opsc,
                                       na, None, ctypes.byref(h)))
    try:
        kernel_ms = float(lib().dd_reducer_kernel_ms(h))
        n = lib().dd_reducer_n_rows(h)
        keys = np.empty((n, nk), dtype=np.uint64)
        keynull = np.empty(n, dtype=np.uint32)
        vals = np.empty((n, na), dtype=np.float64)
        nn = np.zeros((n, na), dtype=np.uint64)
        if n:
            _check(lib().dd_reducer_fetch(
                h, keys.ctypes.data_as(ctypes.c_void_p),
                keynull.ctypes.data_as(ctypes.c_void_p),
                vals.ctypes.data_as(ctypes.c_void_p)))
            _check(lib().dd_reducer_fetch_nn(h, nn.ctypes.data_as(ctypes.c_void_p)))
        return {"keys": keys, "keynull": keynull, "aggs": vals, "nn": nn,
                "kernel_ms": kernel_ms}
    finally:
        lib().dd_reducer_destroy(h)
