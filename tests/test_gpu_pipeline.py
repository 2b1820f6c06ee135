"""End-to-end pipeline parity on the GPU (marked gpu): the composed hot path —
partial aggregation below the shuffle -> hash repartition -> RCCL exchange (self) ->
final aggregation — must produce exactly the answers of a direct pyarrow aggregation.
This is the invariant the reference's correctness suites pin
(tests/tpch_correctness_test.rs:139-158: distributed == single-node query answers),
exercised through the GPU components end to end."""

import numpy as np
import pytest

from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu


def test_q1_pipeline_partial_reduce_shuffle_final():
    import pyarrow as pa

    rng = np.random.default_rng(47)
    n = 2_000_000
    rf = rng.integers(0, 3, n, dtype=np.int64).astype(np.uint8)
    ls = rng.integers(0, 2, n, dtype=np.int64).astype(np.uint8)
    qty = rng.uniform(1, 50, n)
    price = rng.uniform(900, 105000, n)

    # stage N-1: partial aggregation below the shuffle (dd_reduce)
    batch = api.DeviceBatch([
        {"dtype": "u8", "data": rf, "valid": None},
        {"dtype": "u8", "data": ls, "valid": None},
        {"dtype": "f64", "data": qty, "valid": None},
        {"dtype": "f64", "data": price, "valid": None},
    ])
    part_res = api.partial_reduce(batch, [0, 1],
                                  [(2, "sum_f64"), (3, "sum_f64"), (None, "count")])
    batch.free()
    m = len(part_res["keynull"])
    assert 0 < m < n / 100  # the exchange now moves partials, not rows

    # stage N: hash-repartition the PARTIALS on the same keys + exchange (1 rank window)
    pk0 = part_res["keys"][:, 0].astype(np.uint8)
    pk1 = part_res["keys"][:, 1].astype(np.uint8)
    psum_q = part_res["aggs"][:, 0]
    psum_p = part_res["aggs"][:, 1]
    pcnt = part_res["aggs"][:, 2].view(np.int64)
    pbatch = api.DeviceBatch([
        {"dtype": "u8", "data": pk0, "valid": None},
        {"dtype": "u8", "data": pk1, "valid": None},
        {"dtype": "f64", "data": psum_q, "valid": None},
        {"dtype": "f64", "data": psum_p, "valid": None},
        {"dtype": "i64", "data": pcnt, "valid": None},
    ])
    P = 6
    part = api.Partitioner(pbatch, [0, 1], P)
    part.run()
    part.sync()
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    ex = comm.exchange(part)
    assert ex.total_rows == m

    # stage N+1: final aggregation per partition; groups must not split across partitions
    off = part.row_offsets()
    krf = ex.col_data(0)["data"]
    kls = ex.col_data(1)["data"]
    sq = ex.col_data(2)["data"]
    sp = ex.col_data(3)["data"]
    cn = ex.col_data(4)["data"]
    final = {}
    for p in range(P):
        lo, hi = off[p], off[p + 1]
        seen_here = set()
        for i in range(lo, hi):
            k = (int(krf[i]), int(kls[i]))
            acc = final.setdefault(k, [0.0, 0.0, 0])
            acc[0] += sq[i]
            acc[1] += sp[i]
            acc[2] += int(cn[i])
            seen_here.add(k)
        for k in seen_here:  # hash-partition guarantee: one partition per group
            assert final[k] is not None

    direct = pa.table({"rf": rf, "ls": ls, "qty": qty, "price": price}) \
        .group_by(["rf", "ls"]).aggregate([("qty", "sum"), ("price", "sum"),
                                           ([], "count_all")])
    assert len(final) == direct.num_rows
    for i in range(direct.num_rows):
        k = (direct["rf"][i].as_py(), direct["ls"][i].as_py())
        got = final[k]
        assert got[2] == direct["count_all"][i].as_py()  # counts bit-exact
        for gv, dv in ((got[0], direct["qty_sum"][i].as_py()),
                       (got[1], direct["price_sum"][i].as_py())):
            assert abs(gv - dv) <= 1e-6 * abs(dv)  # north-star float tolerance

    ex.destroy()
    comm.destroy()
    part.destroy()
    pbatch.free()
