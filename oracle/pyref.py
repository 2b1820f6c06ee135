"""oracle/pyref.py — independent numpy restatement of the normative shuffle semantics.

TEST INFRASTRUCTURE ONLY (see dd_oracle.c header): used to cross-check the C oracle and to
generate the committed golden vectors under tests/golden/. Pure-Python loops appear only on
small (golden-sized) inputs; fixed-width hashing is vectorized.

Spec: DESIGN.md §3 (normative — partition-assignment parity vs the reference is unpinned,
SURVEY.md §8c). Reference semantics restated: datafusion-physical-plan 55.0.0 RepartitionExec
hash partitioning as constructed at /root/reference/src/execution_plans/network_shuffle.rs:121-127
and /root/reference/src/distributed_planner/network_boundary.rs:100-103.

Column convention (shared with tests): a column is a dict
  {"dtype": one of u8,i16,i32,i64,f32,f64,bool,utf8,dict32,
   "data": np.ndarray            # fixed-width values | utf8: uint8 byte buffer | dict32: int32 indices
   "valid": np.ndarray(u8)|None, # unpacked validity, 1=valid
   "offsets": np.ndarray(i32),   # utf8 only, len n+1
   "dict_offsets", "dict_bytes"} # dict32 only (the value list as a utf8 column)
"""

import numpy as np

U64 = np.uint64
M64 = (1 << 64) - 1

_FIXED_NP = {
    "u8": np.uint8,
    "i16": np.int16,
    "i32": np.int32,
    "i64": np.int64,
    "f32": np.float32,
    "f64": np.float64,
    "bool": np.uint8,
}


def mix64_np(x):
    x = x.astype(U64, copy=True)
    x ^= x >> U64(30)
    x *= U64(0xBF58476D1CE4E5B9)
    x ^= x >> U64(27)
    x *= U64(0x94D049BB133111EB)
    x ^= x >> U64(31)
    return x


def mix64_scalar(x):
    x &= M64
    x ^= x >> 30
    x = (x * 0xBF58476D1CE4E5B9) & M64
    x ^= x >> 27
    x = (x * 0x94D049BB133111EB) & M64
    x ^= x >> 31
    return x


def hash_bytes_scalar(b: bytes) -> int:
    h = 0x9E3779B97F4A7C15 ^ ((len(b) * 0xFF51AFD7ED558CCD) & M64)
    for i in range(0, len(b) - 7, 8):
        c = int.from_bytes(b[i : i + 8], "little")
        h = mix64_scalar(h ^ c)
    tail = len(b) % 8
    if tail:
        c = int.from_bytes(b[len(b) - tail :] + b"\x00" * (8 - tail), "little")
        h = mix64_scalar(h ^ c)
    return h


def _value_bits(col):
    """64-bit canonical bits of a fixed-width column (zero-extended raw LE bits)."""
    dt = col["dtype"]
    data = col["data"]
    if dt in ("u8", "bool"):
        return data.view(np.uint8).astype(U64)
    if dt == "i16":
        return data.view(np.uint16).astype(U64)
    if dt in ("i32",):
        return data.view(np.uint32).astype(U64)
    if dt == "i64":
        return data.view(np.uint64).copy()
    if dt == "f32":
        v = data.astype(np.float32, copy=True)
        v[v == 0.0] = 0.0
        bits = v.view(np.uint32).astype(U64)
        bits[np.isnan(v)] = U64(0x7FC00000)
        return bits
    if dt == "f64":
        v = data.astype(np.float64, copy=True)
        v[v == 0.0] = 0.0
        bits = v.view(np.uint64).copy()
        bits[np.isnan(v)] = U64(0x7FF8000000000000)
        return bits
    raise ValueError(dt)


def value_hashes(col, n):
    dt = col["dtype"]
    if dt in _FIXED_NP:
        return mix64_np(_value_bits(col))
    if dt == "utf8":
        off = col["offsets"]
        buf = col["data"].tobytes()
        return np.array(
            [hash_bytes_scalar(buf[off[i] : off[i + 1]]) for i in range(n)], dtype=U64
        )
    if dt == "dict32":
        doff = col["dict_offsets"]
        dbuf = col["dict_bytes"].tobytes()
        vh = np.array(
            [hash_bytes_scalar(dbuf[doff[k] : doff[k + 1]]) for k in range(len(doff) - 1)],
            dtype=U64,
        )
        idx = col["data"].astype(np.int64)
        safe = np.clip(idx, 0, len(vh) - 1) if len(vh) else idx * 0
        return vh[safe] if len(vh) else np.zeros(n, dtype=U64)
    raise ValueError(dt)


def hash_cols(key_cols, n):
    h = np.zeros(n, dtype=U64)
    for col in key_cols:
        vh = value_hashes(col, n)
        mixed = h ^ ((vh + U64(0x9E3779B97F4A7C15) + (h << U64(6)) + (h >> U64(2))))
        if col.get("valid") is not None:
            valid = col["valid"].astype(bool)
            h = np.where(valid, mixed, h)
        else:
            h = mixed
    return h


def pids(h, nparts):
    return (h % U64(nparts)).astype(np.uint32)


def stable_order(pid, nparts):
    """Partition-major stable row order + part_offsets[P+1] (independent impl: stable argsort)."""
    order = np.argsort(pid, kind="stable").astype(np.int64)
    counts = np.bincount(pid, minlength=nparts).astype(np.int64)
    part_offsets = np.zeros(nparts + 1, dtype=np.int64)
    np.cumsum(counts, out=part_offsets[1:])
    return order, part_offsets


def repartition(cols, key_idx, nparts):
    """Full restatement: returns dict with pid, order, part_offsets, and gathered columns."""
    n = _num_rows(cols[0])
    h = hash_cols([cols[k] for k in key_idx], n)
    pid = pids(h, nparts)
    order, part_offsets = stable_order(pid, nparts)
    out_cols = []
    for col in cols:
        oc = {"dtype": col["dtype"]}
        if col["dtype"] in _FIXED_NP or col["dtype"] == "dict32":
            oc["data"] = np.ascontiguousarray(col["data"][order])
        elif col["dtype"] == "utf8":
            off = col["offsets"].astype(np.int64)
            lens = (off[1:] - off[:-1])[order]
            oc["lengths"] = lens.astype(np.uint32)
            buf = col["data"]
            parts = [buf[off[r] : off[r] + lens[i]] for i, r in enumerate(order)]
            oc["data"] = np.concatenate(parts) if parts else np.zeros(0, dtype=np.uint8)
        if col.get("valid") is not None:
            oc["valid"] = np.ascontiguousarray(col["valid"][order])
        if col["dtype"] == "dict32":
            oc["dict_offsets"] = col["dict_offsets"]
            oc["dict_bytes"] = col["dict_bytes"]
        out_cols.append(oc)
    return {
        "hash": h,
        "pid": pid,
        "order": order,
        "part_offsets": part_offsets,
        "cols": out_cols,
    }


def _num_rows(col):
    if col["dtype"] == "utf8":
        return len(col["offsets"]) - 1
    return len(col["data"])
