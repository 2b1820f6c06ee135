"""listcol.py — List<T> columns through the shuffle, by decomposition.

The reference bench schema carries `tags: List<Utf8>` as a PAYLOAD column
(src/execution_plans/benchmarks/fixture.rs:26-31); TPC-H/ClickBench plans never hash
list keys, so list columns only need to MOVE. Rather than a third nested-offset kernel
family, a list column is decomposed host-side into columns the device engine already
scatters bit-exactly, and recomposed at the consumer boundary:

  List<Utf8> (list_offsets L[n+1], child offsets C[m+1], child bytes B, validity at both
  levels) becomes
    1. a var column of the row's CONCATENATED BYTES   (offsets B_i = C[L[i]], data B)
    2. a var column of the row's CHILD LENGTHS        (offsets 4*L[i], data u32 lengths)
    3. a var column of the row's CHILD VALIDITY       (offsets L[i], data u8) — only
       when the child carries validity
  with the LIST-level validity riding column 1 as ordinary u8 validity.

  List<fixed> is the same with (1)'s offsets = elem_size * L[i] over the raw child
  values and no (3) unless the child is nullable.

Every derived column is a legal Arrow-shaped var column (i32 monotone offsets), so
stability, chunking, the exchange, and the oracle parity guarantees apply unchanged.
Recompose rebuilds a pyarrow ListArray per partition; parity is pinned against
pyarrow's own take() in tests/test_gpu_listcol.py. Caps inherited from the var engine:
total child bytes / 4*m / m each <= INT32_MAX (chunk the batch beyond that). List keys
(hashing) are out of scope — the planner never emits them for this path."""

import numpy as np

FIXED_SIZE = {"u8": 1, "bool": 1, "i16": 2, "i32": 4, "f32": 4, "i64": 8, "f64": 8}
FIXED_NP = {"u8": np.uint8, "bool": np.uint8, "i16": np.int16, "i32": np.int32,
            "f32": np.float32, "i64": np.int64, "f64": np.float64}


def decompose(col):
    """col: {"dtype": "list", "list_offsets": i32[n+1], "child": <utf8-or-fixed col
    dict>, "valid": u8[n]|None}. Returns the derived device-ready column dicts."""
    L = np.ascontiguousarray(col["list_offsets"], dtype=np.int64)
    child = col["child"]
    out = []
    if child["dtype"] == "utf8":
        C = np.ascontiguousarray(child["offsets"], dtype=np.int64)
        bytes_off = C[L]  # B_i = C[L[i]] — monotone, covers all child bytes in order
        if bytes_off[-1] > np.iinfo(np.int32).max:
            raise ValueError("list child bytes exceed int32 offsets: chunk the batch")
        out.append({"dtype": "utf8",
                    "data": np.ascontiguousarray(child["data"], np.uint8),
                    "offsets": bytes_off.astype(np.int32),
                    "valid": col.get("valid")})
        lens = (C[1:] - C[:-1]).astype(np.uint32)
        if 4 * L[-1] > np.iinfo(np.int32).max:
            raise ValueError("4*child_count exceeds int32 offsets: chunk the batch")
        out.append({"dtype": "utf8", "data": lens.view(np.uint8),
                    "offsets": (4 * L).astype(np.int32), "valid": None})
    else:
        es = FIXED_SIZE[child["dtype"]]
        vals = np.ascontiguousarray(child["data"], FIXED_NP[child["dtype"]])
        if es * L[-1] > np.iinfo(np.int32).max:
            raise ValueError("list child values exceed int32 offsets: chunk the batch")
        out.append({"dtype": "utf8", "data": vals.view(np.uint8),
                    "offsets": (es * L).astype(np.int32),
                    "valid": col.get("valid")})
    if child.get("valid") is not None:
        if L[-1] > np.iinfo(np.int32).max:
            raise ValueError("child count exceeds int32 offsets: chunk the batch")
        out.append({"dtype": "utf8",
                    "data": np.ascontiguousarray(child["valid"], np.uint8),
                    "offsets": L.astype(np.int32), "valid": None})
    return out


def recompose_partition(pa, child_dtype, parts):
    """Rebuild a pyarrow ListArray for one partition from the partitioned derived
    columns. parts: dict with per-partition numpy views:
      bytes: {"data": u8[], "lengths": u32[n]}  (column 1's partition slice)
      lens:  {"data": u8[], "lengths": u32[n]}  (column 2; utf8 child only)
      ivalid: {"data": u8[], "lengths": u32[n]} (column 3 / validity column, optional)
      lvalid: u8[n] | None                      (list-level validity of column 1)
    """
    n = len(parts["bytes"]["lengths"])
    if child_dtype == "utf8":
        counts = (parts["lens"]["lengths"] // 4).astype(np.int64)
        child_lens = parts["lens"]["data"].view(np.uint32).astype(np.int64)
    else:
        es = FIXED_SIZE[child_dtype]
        counts = (parts["bytes"]["lengths"] // es).astype(np.int64)
    list_off = np.zeros(n + 1, dtype=np.int32)
    list_off[1:] = np.cumsum(counts)
    ivalid = parts.get("ivalid")
    child_mask = None
    if ivalid is not None:
        child_mask = ivalid["data"].astype(bool)
    if child_dtype == "utf8":
        child_off = np.zeros(int(counts.sum()) + 1, dtype=np.int32)
        child_off[1:] = np.cumsum(child_lens)
        values = pa.StringArray.from_buffers(
            int(counts.sum()), pa.py_buffer(child_off.tobytes()),
            pa.py_buffer(parts["bytes"]["data"].tobytes()),
            pa.py_buffer(np.packbits(child_mask, bitorder="little").tobytes())
            if child_mask is not None else None,
            -1 if child_mask is None else int((~child_mask).sum()))
    else:
        vals = parts["bytes"]["data"].view(FIXED_NP[child_dtype])
        values = pa.array(vals, type=pa.from_numpy_dtype(FIXED_NP[child_dtype]),
                          mask=None if child_mask is None else ~child_mask)
    lvalid = parts.get("lvalid")
    lmask = None if lvalid is None else ~lvalid.astype(bool)
    return pa.ListArray.from_arrays(pa.array(list_off, type=pa.int32()), values,
                                    mask=None if lmask is None else pa.array(lmask))
