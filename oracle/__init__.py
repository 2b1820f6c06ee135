"""oracle — ctypes binding to the C restatement (libdd_oracle.so).

TEST INFRASTRUCTURE ONLY: only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline
leg may import this package (see dd_oracle.c header). The product path never routes here.
"""

import ctypes
import os
import subprocess

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_DIR, "libdd_oracle.so")

_DTYPE_CODE = {
    "u8": 1, "i16": 2, "i32": 3, "i64": 4, "f32": 5, "f64": 6,
    "bool": 7, "utf8": 8, "dict32": 9,
}
FIXED_SIZE = {"u8": 1, "bool": 1, "i16": 2, "i32": 4, "f32": 4, "i64": 8, "f64": 8}


class _OCol(ctypes.Structure):
    _fields_ = [
        ("dtype", ctypes.c_int32),
        ("data", ctypes.c_void_p),
        ("valid", ctypes.c_void_p),
        ("offsets", ctypes.c_void_p),
        ("dict_bytes", ctypes.c_void_p),
        ("dict_offsets", ctypes.c_void_p),
        ("dict_n", ctypes.c_int64),
    ]


def _build():
    subprocess.run(["make", "-s", "-C", _DIR], check=True)


_lib = None


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(_SO) or os.path.getmtime(_SO) < os.path.getmtime(
            os.path.join(_DIR, "dd_oracle.c")
        ):
            _build()
        _lib = ctypes.CDLL(_SO)
        _lib.dd_oracle_mix64.restype = ctypes.c_uint64
        _lib.dd_oracle_mix64.argtypes = [ctypes.c_uint64]
        _lib.dd_oracle_hash_bytes.restype = ctypes.c_uint64
        _lib.dd_oracle_hash_bytes.argtypes = [ctypes.c_char_p, ctypes.c_int64]
        _lib.dd_oracle_num_threads.restype = ctypes.c_int
    return _lib


def _ptr(a):
    return a.ctypes.data_as(ctypes.c_void_p) if a is not None else None


def _make_ocol(col):
    oc = _OCol()
    oc.dtype = _DTYPE_CODE[col["dtype"]]
    # keep refs alive via attributes until the ctypes call returns
    oc._keep = [np.ascontiguousarray(col["data"])]
    oc.data = ctypes.cast(oc._keep[0].ctypes.data, ctypes.c_void_p)
    if col.get("valid") is not None:
        v = np.ascontiguousarray(col["valid"], dtype=np.uint8)
        oc._keep.append(v)
        oc.valid = ctypes.cast(v.ctypes.data, ctypes.c_void_p)
    if col["dtype"] == "utf8":
        o = np.ascontiguousarray(col["offsets"], dtype=np.int32)
        oc._keep.append(o)
        oc.offsets = ctypes.cast(o.ctypes.data, ctypes.c_void_p)
    if col["dtype"] == "dict32":
        db = np.ascontiguousarray(col["dict_bytes"], dtype=np.uint8)
        do = np.ascontiguousarray(col["dict_offsets"], dtype=np.int32)
        oc._keep += [db, do]
        oc.dict_bytes = ctypes.cast(db.ctypes.data, ctypes.c_void_p)
        oc.dict_offsets = ctypes.cast(do.ctypes.data, ctypes.c_void_p)
        oc.dict_n = len(do) - 1
    return oc


def num_rows(col):
    return len(col["offsets"]) - 1 if col["dtype"] == "utf8" else len(col["data"])


def hash_cols(key_cols, n=None):
    L = lib()
    if n is None:
        n = num_rows(key_cols[0])
    ocols = (_OCol * len(key_cols))()
    keeps = []
    for i, c in enumerate(key_cols):
        oc = _make_ocol(c)
        keeps.append(oc)
        ocols[i] = oc
    h = np.empty(n, dtype=np.uint64)
    L.dd_oracle_hash_cols(len(key_cols), ocols, ctypes.c_int64(n), _ptr(h))
    return h


def pids(h, nparts):
    L = lib()
    out = np.empty(len(h), dtype=np.uint32)
    L.dd_oracle_pids(_ptr(h), ctypes.c_int64(len(h)), ctypes.c_uint32(nparts), _ptr(out))
    return out


def order(pid, nparts):
    L = lib()
    n = len(pid)
    nt = L.dd_oracle_num_threads()
    out = np.empty(n, dtype=np.int64)
    poff = np.empty(nparts + 1, dtype=np.int64)
    scratch = np.empty(max(nt * nparts, nparts), dtype=np.int64)
    L.dd_oracle_order(
        _ptr(np.ascontiguousarray(pid, dtype=np.uint32)),
        ctypes.c_int64(n),
        ctypes.c_uint32(nparts),
        _ptr(out),
        _ptr(poff),
        _ptr(scratch),
        ctypes.c_int64(len(scratch)),
    )
    return out, poff


def gather_fixed(src, order_idx):
    L = lib()
    src = np.ascontiguousarray(src)
    out = np.empty_like(src)
    L.dd_oracle_gather_fixed(
        _ptr(src), ctypes.c_int32(src.itemsize), _ptr(order_idx),
        ctypes.c_int64(len(order_idx)), _ptr(out),
    )
    return out


def gather_var(offsets, data, order_idx):
    L = lib()
    offsets = np.ascontiguousarray(offsets, dtype=np.int32)
    data = np.ascontiguousarray(data, dtype=np.uint8)
    n = len(order_idx)
    out_len = np.empty(n, dtype=np.uint32)
    total = int(offsets[-1] - offsets[0]) if n else 0
    out_bytes = np.empty(max(total, 1), dtype=np.uint8)
    L.dd_oracle_gather_var(
        _ptr(offsets), _ptr(data), _ptr(order_idx), ctypes.c_int64(n),
        _ptr(out_len), _ptr(out_bytes),
    )
    return out_len, out_bytes[:total]


def repartition(cols, key_idx, nparts):
    """Full C-oracle pipeline; same result structure as pyref.repartition."""
    n = num_rows(cols[0])
    h = hash_cols([cols[k] for k in key_idx], n)
    pid = pids(h, nparts)
    ordr, poff = order(pid, nparts)
    out_cols = []
    for col in cols:
        oc = {"dtype": col["dtype"]}
        if col["dtype"] in FIXED_SIZE or col["dtype"] == "dict32":
            oc["data"] = gather_fixed(col["data"], ordr)
        elif col["dtype"] == "utf8":
            lens, by = gather_var(col["offsets"], col["data"], ordr)
            oc["lengths"], oc["data"] = lens, by
        if col.get("valid") is not None:
            oc["valid"] = gather_fixed(np.ascontiguousarray(col["valid"], np.uint8), ordr)
        if col["dtype"] == "dict32":
            oc["dict_offsets"] = col["dict_offsets"]
            oc["dict_bytes"] = col["dict_bytes"]
        out_cols.append(oc)
    return {"hash": h, "pid": pid, "order": ordr, "part_offsets": poff, "cols": out_cols}
