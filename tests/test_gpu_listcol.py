"""List<T> payload columns through the GPU shuffle (marked gpu): the host-side
decomposition (listcol.py) into derived var columns is scattered by the device engine
and recomposed per partition; parity is pinned against pyarrow's OWN take() of the
original ListArray on the same rows — closing the bench-schema `tags: List<Utf8>` gap
(src/execution_plans/benchmarks/fixture.rs:26-31)."""

import numpy as np
import pytest
import pyarrow as pa

from datafusion_distributed_amd import api, listcol

pytestmark = pytest.mark.gpu


def make_list_utf8(rng, n, item_nulls=True, list_nulls=True):
    counts = rng.integers(0, 6, n)
    m = int(counts.sum())
    L = np.zeros(n + 1, dtype=np.int32)
    L[1:] = np.cumsum(counts)
    slens = rng.integers(0, 15, m)
    C = np.zeros(m + 1, dtype=np.int32)
    C[1:] = np.cumsum(slens)
    B = rng.integers(97, 123, int(C[-1]), dtype=np.int64).astype(np.uint8)
    ivalid = (rng.random(m) > 0.15).astype(np.uint8) if item_nulls else None
    lvalid = (rng.random(n) > 0.1).astype(np.uint8) if list_nulls else None
    col = {"dtype": "list", "list_offsets": L, "valid": lvalid,
           "child": {"dtype": "utf8", "offsets": C, "data": B, "valid": ivalid}}
    # the same array in pyarrow
    values = pa.StringArray.from_buffers(
        m, pa.py_buffer(C.tobytes()), pa.py_buffer(B.tobytes()),
        pa.py_buffer(np.packbits(ivalid.astype(bool), bitorder="little").tobytes())
        if ivalid is not None else None,
        -1 if ivalid is None else int((ivalid == 0).sum()))
    arr = pa.ListArray.from_arrays(
        pa.array(L, type=pa.int32()), values,
        mask=None if lvalid is None else pa.array(~lvalid.astype(bool)))
    return col, arr


def make_list_i64(rng, n):
    counts = rng.integers(0, 4, n)
    m = int(counts.sum())
    L = np.zeros(n + 1, dtype=np.int32)
    L[1:] = np.cumsum(counts)
    vals = rng.integers(-(2**60), 2**60, m, dtype=np.int64)
    col = {"dtype": "list", "list_offsets": L, "valid": None,
           "child": {"dtype": "i64", "data": vals, "valid": None}}
    arr = pa.ListArray.from_arrays(pa.array(L, type=pa.int32()), pa.array(vals))
    return col, arr


def _col_parts(part, ci, p, off):
    """Partition p's slice of derived var column ci: bytes + per-row lengths."""
    out = part.col_out(ci)
    boff = part.byte_offsets(ci)
    lo, hi = off[p], off[p + 1]
    return {"data": out["data"][boff[p]:boff[p + 1]],
            "lengths": out["lengths"][lo:hi]}, out.get("valid")


@pytest.mark.parametrize("kind", ["utf8_nulls", "utf8_plain", "i64"])
def test_list_column_shuffle_matches_pyarrow_take(kind):
    rng = np.random.default_rng(61)
    n, P = 120_000, 16
    key = rng.integers(0, 10**9, n, dtype=np.int64)
    if kind == "i64":
        lcol, arr = make_list_i64(rng, n)
        child_dtype = "i64"
    else:
        lcol, arr = make_list_utf8(rng, n, item_nulls=(kind == "utf8_nulls"),
                                   list_nulls=(kind == "utf8_nulls"))
        child_dtype = "utf8"

    derived = listcol.decompose(lcol)
    cols = [{"dtype": "i64", "data": key, "valid": None}] + derived
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], P)
    part.run()
    part.sync()

    off = part.row_offsets()
    pid = part.pids()
    assert pid is not None
    nderived = len(derived)
    for p in range(P):
        rows = np.flatnonzero(pid == p)
        want = arr.take(pa.array(rows, type=pa.int64()))
        parts = {}
        bts, lvalid_full = _col_parts(part, 1, p, off)
        parts["bytes"] = bts
        if lvalid_full is not None:
            parts["lvalid"] = lvalid_full[off[p]:off[p + 1]]
        ci = 2
        if child_dtype == "utf8":
            parts["lens"], _ = _col_parts(part, ci, p, off)
            ci += 1
        if nderived == ci:  # item-validity column present
            parts["ivalid"], _ = _col_parts(part, ci, p, off)
        got = listcol.recompose_partition(pa, child_dtype, parts)
        assert got.equals(want), f"partition {p} ({kind}) differs from pyarrow take()"
    batch.free()
    part.destroy()
