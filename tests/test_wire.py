"""Arrow IPC + lz4 wire format (dd_wire.cpp) — CPU suite, SURVEY §8(f) row 2.

Cross-implementation parity, both directions: streams WE write are read by pyarrow.ipc
(independent Arrow implementation) and streams pyarrow writes (compression="lz4") are
read by US — fixed/bool/utf8 columns, nulls, lz4 on/off, multiple batches, and the
ZERO-COLUMN batches the reference pins in tests/empty_columns_between_workers.rs:11-31
(row count survives with no fields)."""

import ctypes
import io

import numpy as np
import pyarrow as pa
import pytest

from datafusion_distributed_amd import api


class IpcField(ctypes.Structure):
    _fields_ = [("dtype", ctypes.c_int32), ("name", ctypes.c_char_p),
                ("nullable", ctypes.c_int32)]


class IpcArray(ctypes.Structure):
    _fields_ = [("data", ctypes.c_void_p), ("data_len", ctypes.c_int64),
                ("validity", ctypes.c_void_p), ("null_count", ctypes.c_int64),
                ("offsets", ctypes.c_void_p)]


DT = api.DTYPE_CODE
NP_OF = {"u8": np.uint8, "i16": np.int16, "i32": np.int32, "i64": np.int64,
         "f32": np.float32, "f64": np.float64, "bool": np.uint8}
PA_OF = {"u8": pa.uint8(), "i16": pa.int16(), "i32": pa.int32(), "i64": pa.int64(),
         "f32": pa.float32(), "f64": pa.float64(), "bool": pa.bool_(),
         "utf8": pa.string()}


def write_stream(cols_schema, batches, use_lz4):
    """cols_schema: [(name, dtype)]; batches: list of dicts name->
    {data, valid(None|u8), offsets(utf8)} with n_rows."""
    L = api.lib()
    n = len(cols_schema)
    fields = (IpcField * max(n, 1))()
    for i, (name, dt) in enumerate(cols_schema):
        fields[i].dtype = DT[dt]
        fields[i].name = name.encode()
        fields[i].nullable = 1
    w = ctypes.c_void_p()
    api._check(L.dd_ipc_writer_create(fields, n, 1 if use_lz4 else 0, ctypes.byref(w)))
    keep = []
    try:
        for n_rows, batch in batches:
            arrs = (IpcArray * max(n, 1))()
            for i, (name, dt) in enumerate(cols_schema):
                col = batch[name]
                data = np.ascontiguousarray(col["data"],
                                            dtype=np.uint8 if dt == "utf8" else NP_OF[dt])
                keep.append(data)
                arrs[i].data = data.ctypes.data_as(ctypes.c_void_p).value
                arrs[i].data_len = int(data.nbytes)
                if col.get("valid") is not None:
                    v = np.ascontiguousarray(col["valid"], dtype=np.uint8)
                    keep.append(v)
                    arrs[i].validity = v.ctypes.data_as(ctypes.c_void_p).value
                    arrs[i].null_count = int((v == 0).sum())
                if dt == "utf8":
                    off = np.ascontiguousarray(col["offsets"], dtype=np.int32)
                    keep.append(off)
                    arrs[i].offsets = off.ctypes.data_as(ctypes.c_void_p).value
            api._check(L.dd_ipc_writer_batch(w, ctypes.c_int64(n_rows), arrs))
        data_p = ctypes.c_void_p()
        dlen = ctypes.c_int64()
        api._check(L.dd_ipc_writer_finish(w, ctypes.byref(data_p), ctypes.byref(dlen)))
        return ctypes.string_at(data_p.value, dlen.value) if dlen.value else b""
    finally:
        L.dd_ipc_writer_destroy(w)


def read_stream(blob):
    L = api.lib()
    r = ctypes.c_void_p()
    api._check(L.dd_ipc_reader_create(blob, ctypes.c_int64(len(blob)), ctypes.byref(r)))
    try:
        nf = L.dd_ipc_reader_n_fields(r)
        nb = L.dd_ipc_reader_n_batches(r)
        L.dd_ipc_reader_field_name.restype = ctypes.c_char_p
        L.dd_ipc_reader_batch_rows.restype = ctypes.c_int64
        fields = [(L.dd_ipc_reader_field_name(r, i).decode(),
                   L.dd_ipc_reader_field_dtype(r, i)) for i in range(nf)]
        out = []
        for b in range(nb):
            rows = L.dd_ipc_reader_batch_rows(r, b)
            cols = []
            for c in range(nf):
                a = IpcArray()
                api._check(L.dd_ipc_reader_batch_col(r, b, c, ctypes.byref(a)))
                entry = {"null_count": a.null_count}
                entry["data"] = ctypes.string_at(a.data, a.data_len) if a.data_len else b""
                entry["valid"] = (np.frombuffer(ctypes.string_at(a.validity, rows),
                                                dtype=np.uint8)
                                  if a.validity else None)
                if a.offsets:
                    entry["offsets"] = np.frombuffer(
                        ctypes.string_at(a.offsets, (rows + 1) * 4), dtype=np.int32)
                cols.append(entry)
            out.append((rows, cols))
        return fields, out
    finally:
        L.dd_ipc_reader_destroy(r)


def make_pa_batch(rng, n):
    """One batch in both representations (ours + pyarrow)."""
    i64 = rng.integers(-(2**60), 2**60, n, dtype=np.int64)
    f64 = rng.normal(size=n)
    bl = (rng.random(n) > 0.5).astype(np.uint8)
    valid = (rng.random(n) > 0.2).astype(np.uint8)
    lens = rng.integers(0, 20, n)
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    sbytes = rng.integers(97, 123, int(off[-1]), dtype=np.int64).astype(np.uint8)
    ours = {"a": {"data": i64, "valid": None},
            "b": {"data": f64, "valid": valid},
            "c": {"data": bl, "valid": None},
            "d": {"data": sbytes, "valid": None, "offsets": off}}
    pa_batch = pa.record_batch([
        pa.array(i64),
        pa.array([float(x) if v else None for x, v in zip(f64, valid)]),
        pa.array(bl.astype(bool)),
        pa.StringArray.from_buffers(n, pa.py_buffer(off.tobytes()),
                                    pa.py_buffer(sbytes.tobytes())),
    ], names=["a", "b", "c", "d"])
    return ours, pa_batch


SCHEMA = [("a", "i64"), ("b", "f64"), ("c", "bool"), ("d", "utf8")]


@pytest.mark.parametrize("use_lz4", [False, True])
def test_our_writer_read_by_pyarrow(use_lz4):
    rng = np.random.default_rng(5)
    b1, pab1 = make_pa_batch(rng, 1000)
    b2, pab2 = make_pa_batch(rng, 777)
    blob = write_stream(SCHEMA, [(1000, b1), (777, b2)], use_lz4)
    got = pa.ipc.open_stream(io.BytesIO(blob)).read_all()
    want = pa.Table.from_batches([pab1, pab2])
    assert got.schema.names == ["a", "b", "c", "d"]
    assert got.num_rows == want.num_rows
    assert got.equals(want), "pyarrow read different values than we wrote"


@pytest.mark.parametrize("use_lz4", [False, True])
def test_pyarrow_writer_read_by_us(use_lz4):
    rng = np.random.default_rng(6)
    ours, pab = make_pa_batch(rng, 1500)
    sink = io.BytesIO()
    opts = pa.ipc.IpcWriteOptions(compression="lz4" if use_lz4 else None)
    with pa.ipc.new_stream(sink, pab.schema, options=opts) as w:
        w.write_batch(pab)
    fields, batches = read_stream(sink.getvalue())
    assert [f[0] for f in fields] == ["a", "b", "c", "d"]
    assert len(batches) == 1
    rows, cols = batches[0]
    assert rows == 1500
    assert cols[0]["data"] == np.asarray(ours["a"]["data"]).tobytes()
    f64 = np.frombuffer(cols[1]["data"], dtype=np.float64)
    valid = cols[1]["valid"]
    assert valid is not None and (valid == ours["b"]["valid"]).all()
    w_ok = ours["b"]["valid"].astype(bool)
    assert (f64[w_ok] == np.asarray(ours["b"]["data"])[w_ok]).all()
    assert (np.frombuffer(cols[2]["data"], dtype=np.uint8) == ours["c"]["data"]).all()
    assert (cols[3]["offsets"] == ours["d"]["offsets"]).all()
    assert cols[3]["data"] == ours["d"]["data"].tobytes()


def test_roundtrip_ourselves_lz4():
    rng = np.random.default_rng(7)
    ours, _ = make_pa_batch(rng, 3000)
    blob = write_stream(SCHEMA, [(3000, ours)], True)
    fields, batches = read_stream(blob)
    rows, cols = batches[0]
    assert rows == 3000
    assert cols[0]["data"] == np.asarray(ours["a"]["data"]).tobytes()
    assert cols[3]["data"] == ours["d"]["data"].tobytes()


def test_zero_column_batches_roundtrip():
    """The reference wire edge case (tests/empty_columns_between_workers.rs:11-31):
    batches with a ROW COUNT but no columns must cross the wire intact."""
    blob = write_stream([], [(123, {}), (0, {}), (45, {})], True)
    fields, batches = read_stream(blob)
    assert fields == []
    assert [r for r, _ in batches] == [123, 0, 45]
    # and pyarrow agrees on the row counts
    got = list(pa.ipc.open_stream(io.BytesIO(blob)))
    assert [b.num_rows for b in got] == [123, 0, 45]
    # pyarrow-written zero-column stream read by us
    sink = io.BytesIO()
    schema = pa.schema([])
    with pa.ipc.new_stream(sink, schema) as w:
        w.write_batch(pa.record_batch([], schema=schema))
    fields, batches = read_stream(sink.getvalue())
    assert fields == [] and [r for r, _ in batches] == [0]


def test_incompressible_passthrough():
    """Random bytes defeat lz4: the writer must fall back to the spec's -1 raw
    passthrough per buffer, and pyarrow must still read it."""
    rng = np.random.default_rng(8)
    n = 4096
    data = rng.integers(0, 2**63, n, dtype=np.int64)
    blob = write_stream([("x", "i64")], [(n, {"x": {"data": data, "valid": None}})], True)
    got = pa.ipc.open_stream(io.BytesIO(blob)).read_all()
    assert got.column("x").to_pylist() == data.tolist()


def test_dictionary_fields_rejected():
    L = api.lib()
    fields = (IpcField * 1)()
    fields[0].dtype = DT["dict32"]
    fields[0].name = b"d"
    fields[0].nullable = 0
    w = ctypes.c_void_p()
    st = L.dd_ipc_writer_create(fields, 1, 0, ctypes.byref(w))
    assert st == 6  # DD_ERR_UNSUPPORTED, stated not silent


def test_wire_py_roundtrips():
    """The productized binding (datafusion_distributed_amd.wire): RecordBatches ->
    our C encoder -> pyarrow reader, and pyarrow writer -> our C decoder, value-equal."""
    from datafusion_distributed_amd import wire

    rng = np.random.default_rng(9)
    _, pab = make_pa_batch(rng, 2048)
    blob = wire.encode_batches([pab], use_lz4=True)
    got = pa.ipc.open_stream(io.BytesIO(blob)).read_all()
    assert got.equals(pa.Table.from_batches([pab]))

    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, pab.schema,
                           options=pa.ipc.IpcWriteOptions(compression="lz4")) as w:
        w.write_batch(pab)
    back = wire.decode_batches(sink.getvalue())
    assert len(back) == 1
    assert pa.Table.from_batches(back).equals(pa.Table.from_batches([pab]))

    # zero-column batches keep their row counts through our decoder too
    blob0 = write_stream([], [(7, {}), (3, {})], True)
    back0 = wire.decode_batches(blob0)
    assert [b.num_rows for b in back0] == [7, 3]


def test_reader_survives_hostile_bytes():
    """The wire may carry corrupt/hostile bytes: every mutation of a valid stream (and
    pure noise) must return an error or a truncated-but-safe result — never crash or
    read out of bounds (the reader bounds-checks all flatbuffer/body accesses)."""
    rng = np.random.default_rng(11)
    ours, _ = make_pa_batch(rng, 200)
    blob = bytearray(write_stream(SCHEMA, [(200, ours)], True))
    L = api.lib()
    for trial in range(300):
        mutated = bytearray(blob)
        for _ in range(rng.integers(1, 8)):
            mutated[rng.integers(0, len(mutated))] = rng.integers(0, 256)
        r = ctypes.c_void_p(0)
        st = L.dd_ipc_reader_create(bytes(mutated), ctypes.c_int64(len(mutated)),
                                    ctypes.byref(r))
        if st == 0 and r.value:
            # decodable-enough stream: accessors must stay in bounds too
            nb = L.dd_ipc_reader_n_batches(r)
            nf = L.dd_ipc_reader_n_fields(r)
            for b in range(min(nb, 4)):
                for c in range(min(nf, 8)):
                    a = IpcArray()
                    L.dd_ipc_reader_batch_col(r, b, c, ctypes.byref(a))
            L.dd_ipc_reader_destroy(r)
    for trial in range(100):  # pure noise
        noise = bytes(rng.integers(0, 256, rng.integers(8, 400), dtype=np.int64)
                      .astype(np.uint8))
        r = ctypes.c_void_p(0)
        st = L.dd_ipc_reader_create(noise, ctypes.c_int64(len(noise)), ctypes.byref(r))
        if st == 0 and r.value:
            L.dd_ipc_reader_destroy(r)


def test_zstd_compression_rejected_cleanly():
    """The reference supports lz4|zstd|none (distributed_config.rs:36-38); this tier
    implements lz4 (the default) — a zstd stream must fail loudly, not misdecode."""
    if not pa.Codec.is_available("zstd"):
        pytest.skip("no zstd in this pyarrow")
    rng = np.random.default_rng(12)
    _, pab = make_pa_batch(rng, 100)
    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, pab.schema,
                           options=pa.ipc.IpcWriteOptions(compression="zstd")) as w:
        w.write_batch(pab)
    L = api.lib()
    r = ctypes.c_void_p(0)
    blob = sink.getvalue()
    st = L.dd_ipc_reader_create(blob, ctypes.c_int64(len(blob)), ctypes.byref(r))
    assert st == 6  # DD_ERR_UNSUPPORTED


def test_wire_py_sliced_and_null_bool():
    """Sliced RecordBatches (array offset != 0) are rebased before encoding, and
    null-bearing bool columns survive — both read back by pyarrow value-equal."""
    from datafusion_distributed_amd import wire

    b = pa.record_batch([pa.array(np.arange(100, dtype=np.int64)),
                         pa.array([f"s{i}" for i in range(100)])], names=["x", "s"])
    sl = b.slice(10, 50)
    got = pa.ipc.open_stream(io.BytesIO(wire.encode_batches([sl], use_lz4=True))).read_all()
    assert got.equals(pa.Table.from_batches([sl]))

    bb = pa.record_batch([pa.array([True, None, False] * 30)], names=["b"])
    got2 = pa.ipc.open_stream(io.BytesIO(wire.encode_batches([bb]))).read_all()
    assert got2.equals(pa.Table.from_batches([bb]))
