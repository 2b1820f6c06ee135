"""wire.py — pyarrow RecordBatch <-> the C library's Arrow IPC + lz4 stream (dd_wire.cpp).

The cross-node hop of the exchange boundary: partition batches materialized by
arrow_boundary (or any pyarrow RecordBatches of the supported types) are encoded into
the reference's on-wire format — Arrow IPC streaming + per-buffer lz4-frame compression
(src/protocol/grpc/worker_service.rs:363-433) — by the C implementation, and decoded
back. This is a thin marshalling layer over dd_ipc_writer_* / dd_ipc_reader_*; the
format logic lives in C. Types: fixed widths, bool, utf8 (dictionary columns are
materialized by arrow_boundary before they reach the wire)."""

import ctypes

import numpy as np

from . import api


class IpcField(ctypes.Structure):
    _fields_ = [("dtype", ctypes.c_int32), ("name", ctypes.c_char_p),
                ("nullable", ctypes.c_int32)]


class IpcArray(ctypes.Structure):
    _fields_ = [("data", ctypes.c_void_p), ("data_len", ctypes.c_int64),
                ("validity", ctypes.c_void_p), ("null_count", ctypes.c_int64),
                ("offsets", ctypes.c_void_p)]


_PA_DT = None


def _pa_dtypes(pa):
    global _PA_DT
    if _PA_DT is None:
        _PA_DT = {
            pa.uint8(): "u8", pa.int8(): "u8", pa.int16(): "i16", pa.int32(): "i32",
            pa.date32(): "i32", pa.int64(): "i64", pa.float32(): "f32",
            pa.float64(): "f64", pa.bool_(): "bool", pa.string(): "utf8",
        }
    return _PA_DT


def encode_batches(batches, schema=None, use_lz4=True):
    """Encode pyarrow RecordBatches into one IPC stream (bytes). `schema` defaults to
    the first batch's; zero-column batches need it passed explicitly or carry it."""
    import pyarrow as pa

    if schema is None:
        schema = batches[0].schema
    dtmap = _pa_dtypes(pa)
    names = schema.names
    dts = []
    for f in schema:
        if f.type not in dtmap:
            raise ValueError(f"unsupported wire type {f.type}")
        dts.append(dtmap[f.type])
    L = api.lib()
    n = len(names)
    fields = (IpcField * max(n, 1))()
    keep = []
    for i, (name, dt) in enumerate(zip(names, dts)):
        fields[i].dtype = api.DTYPE_CODE[dt]
        nm = name.encode()
        keep.append(nm)
        fields[i].name = nm
        fields[i].nullable = 1
    w = ctypes.c_void_p()
    api._check(L.dd_ipc_writer_create(fields, n, 1 if use_lz4 else 0, ctypes.byref(w)))
    try:
        for b in batches:
            arrs = (IpcArray * max(n, 1))()
            for i, dt in enumerate(dts):
                col = b.column(i)
                if col.offset != 0:
                    col = col.combine_chunks() if hasattr(col, "combine_chunks") else col
                    col = pa.concat_arrays([col])  # re-base to offset 0
                nc = col.null_count
                if dt == "utf8":
                    off = np.frombuffer(col.buffers()[1], dtype=np.int32,
                                        count=len(col) + 1)
                    base = off[0]
                    if base != 0:
                        off = off - base
                    data = np.frombuffer(col.buffers()[2], dtype=np.uint8)[
                        base:base + int(off[-1])]
                    keep += [off, data]
                    arrs[i].offsets = off.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].data = data.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].data_len = int(off[-1])
                elif dt == "bool":
                    u8 = np.asarray(col.cast(pa.uint8()).fill_null(0)).astype(np.uint8)
                    keep.append(u8)
                    arrs[i].data = u8.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].data_len = int(u8.nbytes)
                else:
                    npdt = api.FIXED_NP[dt]
                    vals = np.frombuffer(col.buffers()[1], dtype=npdt, count=len(col))
                    keep.append(vals)
                    arrs[i].data = vals.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].data_len = int(vals.nbytes)
                if nc > 0:
                    vu8 = np.asarray(col.is_valid()).astype(np.uint8)
                    keep.append(vu8)
                    arrs[i].validity = vu8.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].null_count = int(nc)
            api._check(L.dd_ipc_writer_batch(w, ctypes.c_int64(b.num_rows), arrs))
        p = ctypes.c_void_p()
        ln = ctypes.c_int64()
        api._check(L.dd_ipc_writer_finish(w, ctypes.byref(p), ctypes.byref(ln)))
        return ctypes.string_at(p.value, ln.value) if ln.value else b""
    finally:
        L.dd_ipc_writer_destroy(w)


def decode_batches(blob):
    """Decode an IPC stream (ours or pyarrow's) into pyarrow RecordBatches."""
    import pyarrow as pa

    L = api.lib()
    L.dd_ipc_reader_field_name.restype = ctypes.c_char_p
    L.dd_ipc_reader_batch_rows.restype = ctypes.c_int64
    r = ctypes.c_void_p()
    api._check(L.dd_ipc_reader_create(blob, ctypes.c_int64(len(blob)), ctypes.byref(r)))
    try:
        nf = L.dd_ipc_reader_n_fields(r)
        nb = L.dd_ipc_reader_n_batches(r)
        rev = {v: k for k, v in api.DTYPE_CODE.items()}
        fields = [(L.dd_ipc_reader_field_name(r, i).decode(),
                   rev[L.dd_ipc_reader_field_dtype(r, i)]) for i in range(nf)]
        out = []
        for b in range(nb):
            rows = L.dd_ipc_reader_batch_rows(r, b)
            arrays = []
            for c, (name, dt) in enumerate(fields):
                a = IpcArray()
                api._check(L.dd_ipc_reader_batch_col(r, b, c, ctypes.byref(a)))
                mask = None
                if a.validity:
                    v = np.frombuffer(ctypes.string_at(a.validity, rows), dtype=np.uint8)
                    mask = v == 0
                if dt == "utf8":
                    off = np.frombuffer(ctypes.string_at(a.offsets, (rows + 1) * 4),
                                        dtype=np.int32)
                    data = (np.frombuffer(ctypes.string_at(a.data, a.data_len),
                                          dtype=np.uint8)
                            if a.data_len else np.zeros(0, np.uint8))
                    arr = pa.StringArray.from_buffers(
                        rows, pa.py_buffer(off.tobytes()), pa.py_buffer(data.tobytes()),
                        pa.py_buffer(np.packbits(~mask, bitorder="little").tobytes())
                        if mask is not None else None)
                elif dt == "bool":
                    u8 = np.frombuffer(ctypes.string_at(a.data, rows), dtype=np.uint8)
                    arr = pa.array(u8.astype(bool), mask=mask)
                else:
                    npdt = api.FIXED_NP[dt]
                    vals = np.frombuffer(ctypes.string_at(a.data, a.data_len), dtype=npdt)
                    arr = pa.array(vals, mask=mask)
                arrays.append(arr)
            if nf == 0:
                # zero-column batch with a row count (empty_columns_between_workers)
                out.append(pa.RecordBatch.from_struct_array(
                    pa.array([{}] * rows, type=pa.struct([]))))
            else:
                out.append(pa.RecordBatch.from_arrays(arrays,
                                                      names=[f[0] for f in fields]))
        return out
    finally:
        L.dd_ipc_reader_destroy(r)
