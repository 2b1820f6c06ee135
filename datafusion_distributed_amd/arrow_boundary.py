"""arrow_boundary.py — materialize per-partition Arrow RecordBatches at the host boundary.

This is the boundary conversion DESIGN.md §4 specifies: inside the device path validity is
unpacked u8 and var-width data is (lengths, bytes, partition byte offsets); at the consumer
boundary those become real Arrow arrays — validity re-packed to bitmaps, offsets rebuilt per
partition — exactly what the Rust shim's `streams_from_partitioner` does in INTEGRATION.md.
`batch_size` slicing mirrors RepartitionExec's output coalescing
(src/distributed_planner/distributed_config.rs:39-45): partitions are emitted as
<= batch_size row batches.
"""

import numpy as np


def _arrow():
    import pyarrow as pa

    return pa


_PA_TYPE = {
    "u8": "uint8", "i16": "int16", "i32": "int32", "i64": "int64",
    "f32": "float32", "f64": "float64", "bool": "uint8",
}


def _validity_mask(valid_u8):
    return valid_u8.astype(bool) if valid_u8 is not None else None


def partition_batches(part, lo, hi, batch_size=8192):
    """Arrow RecordBatches for partitions [lo, hi) of a run Partitioner (downloads the
    partition-major buffers once and slices views). Yields (partition, RecordBatch)."""
    pa = _arrow()
    roff = part.row_offsets()
    cols_meta = part.batch.cols
    fixed = {}
    lens = {}
    var_bytes = {}
    var_boff = {}
    valid = {}
    for i, c in enumerate(cols_meta):
        out = part.col_out(i)
        if c["dtype"] == "utf8":
            lens[i] = out["lengths"]
            var_bytes[i] = out["data"]
            var_boff[i] = part.byte_offsets(i)
        else:
            fixed[i] = out["data"]
        if "valid" in out:
            valid[i] = out["valid"]

    for p in range(lo, hi):
        r0, r1 = int(roff[p]), int(roff[p + 1])
        # one empty batch for an empty partition (the reference emits empty streams)
        slices = [(r0, r0)] if r1 == r0 else [
            (b0, min(b0 + batch_size, r1)) for b0 in range(r0, r1, batch_size)]
        for b0, b1 in slices:
            arrays, names = [], []
            for i, c in enumerate(cols_meta):
                name = c.get("name", f"c{i}")
                mask = None
                if i in valid:
                    mask = ~_validity_mask(valid[i][b0:b1])
                if c["dtype"] == "utf8":
                    seg_lens = lens[i][b0:b1].astype(np.int64)
                    start = int(var_boff[i][p]) + int(
                        lens[i][r0:b0].astype(np.int64).sum())
                    nbytes = int(seg_lens.sum())
                    offsets = np.zeros(len(seg_lens) + 1, dtype=np.int32)
                    np.cumsum(seg_lens, out=offsets[1:])
                    arr = pa.StringArray.from_buffers(
                        b1 - b0,
                        pa.py_buffer(offsets.tobytes()),
                        pa.py_buffer(var_bytes[i][start:start + nbytes].tobytes()),
                        pa.py_buffer(np.packbits(~mask, bitorder="little").tobytes())
                        if mask is not None else None,
                    )
                elif c["dtype"] == "dict32":
                    idx = pa.array(fixed[i][b0:b1], type=pa.int32(), mask=mask)
                    # Rebuild values from the raw offsets/bytes buffers: the device path
                    # hashes/moves raw bytes, so round-tripping through str would silently
                    # mutate (or mask) non-UTF8 dictionary bytes.
                    doff = np.asarray(c["dict_offsets"], dtype=np.int32)
                    dbuf = np.asarray(c["dict_bytes"], dtype=np.uint8)
                    values = pa.StringArray.from_buffers(
                        len(doff) - 1,
                        pa.py_buffer(doff.tobytes()),
                        pa.py_buffer(dbuf.tobytes()),
                        None,
                    )
                    arr = pa.DictionaryArray.from_arrays(idx, values)
                elif c["dtype"] == "bool":
                    arr = pa.array(fixed[i][b0:b1].astype(bool), type=pa.bool_(), mask=mask)
                else:
                    arr = pa.array(fixed[i][b0:b1], type=getattr(pa, _PA_TYPE[c["dtype"]])(),
                                   mask=mask)
                arrays.append(arr)
                names.append(name)
            yield p, pa.RecordBatch.from_arrays(arrays, names=names)
