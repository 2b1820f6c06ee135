"""Arrow boundary materialization (marked gpu): per-partition pyarrow RecordBatches built
from the partitioner's device buffers must equal a pyarrow-native take() of the same rows
(validity re-packed, offsets rebuilt, batch_size coalescing — DESIGN.md §4 boundary)."""

import numpy as np
import pytest

import oracle
from datafusion_distributed_amd import api
from datafusion_distributed_amd.arrow_boundary import partition_batches

pytestmark = pytest.mark.gpu


def test_partition_batches_match_pyarrow_take():
    import pyarrow as pa

    rng = np.random.default_rng(83)
    n, P = 30000, 8
    rows = [bytes(rng.integers(97, 123, rng.integers(0, 20), dtype=np.int64)
                  .astype(np.uint8)) for _ in range(n)]
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(r) for r in rows])
    cols = [
        {"name": "k", "dtype": "i64",
         "data": rng.integers(0, 10**9, n, dtype=np.int64), "valid": None},
        {"name": "v", "dtype": "f64", "data": rng.normal(size=n),
         "valid": (rng.random(n) > 0.15).astype(np.uint8)},
        {"name": "s", "dtype": "utf8",
         "data": np.frombuffer(b"".join(rows), dtype=np.uint8), "offsets": off,
         "valid": (rng.random(n) > 0.1).astype(np.uint8)},
        {"name": "b", "dtype": "bool", "data": (rng.random(n) > 0.5).astype(np.uint8),
         "valid": None},
    ]
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], P)
    part.run()
    part.sync()
    ref = oracle.repartition(cols, [0], P)

    # pyarrow-native reference table
    tbl = pa.table({
        "k": cols[0]["data"],
        "v": pa.array(cols[1]["data"], mask=~cols[1]["valid"].astype(bool)),
        "s": pa.array([r.decode() if vv else None
                       for r, vv in zip(rows, cols[2]["valid"])]),
        "b": pa.array(cols[3]["data"].astype(bool)),
    })

    seen_rows = 0
    for p, rb in partition_batches(part, 0, P, batch_size=1024):
        # rows of this batch, in partition-major (stable) order
        idx = ref["order"][seen_rows: seen_rows + rb.num_rows]
        expected = tbl.take(pa.array(idx)) if rb.num_rows else tbl.slice(0, 0)
        got = pa.table(rb)
        assert got.equals(pa.table({n: expected[n] for n in got.column_names})), \
            f"partition {p} batch mismatch"
        seen_rows += rb.num_rows
    assert seen_rows == n
    part.destroy()
    batch.free()
