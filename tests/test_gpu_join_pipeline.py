"""TPC-H q3-shaped join pipeline end-to-end on the GPU (marked gpu) — BASELINE
config[2]. Composition: broadcast the filtered build side (dd_broadcast_run, the
CollectLeft analog of src/execution_plans/broadcast.rs:22-35) -> GPU hash-shuffle of the
orders-side join result AND lineitem on the join key (co-partitioned, P shared) -> RCCL
exchange -> per-partition hash join + aggregation on the host (pyarrow, the engine above
the seam) -> whole-query answer asserted against a direct pyarrow plan over the full
tables: group keys and counts exact, float revenue <= 1e-6 relative — the invariant the
reference's correctness suites pin (tests/tpch_correctness_test.rs:139-158)."""

import numpy as np
import pytest
import pyarrow.compute as pc

from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu

CUTOFF = 9204  # days: 1995-03-15 in the synthetic date space


def make_tables(rng, n_cust=150_000, n_orders=1_500_000, n_li=6_000_000):
    cust = {
        "c_custkey": np.arange(1, n_cust + 1, dtype=np.int64),
        "c_mktsegment": rng.integers(0, 5, n_cust).astype(np.int32),  # 0 = BUILDING
    }
    orders = {
        "o_orderkey": np.arange(1, n_orders + 1, dtype=np.int64) * 4,
        "o_custkey": rng.integers(1, n_cust + 1, n_orders, dtype=np.int64),
        "o_orderdate": rng.integers(8000, 10500, n_orders, dtype=np.int64)
                          .astype(np.int32),
        "o_shippriority": rng.integers(0, 2, n_orders, dtype=np.int64).astype(np.int32),
    }
    li = {
        "l_orderkey": rng.integers(1, n_orders + 1, n_li, dtype=np.int64) * 4,
        "l_extendedprice": rng.uniform(900, 105000, n_li),
        "l_discount": np.round(rng.uniform(0, 0.1, n_li), 2),
        "l_shipdate": rng.integers(8000, 10500, n_li, dtype=np.int64).astype(np.int32),
    }
    return cust, orders, li


def direct_answer(pa, cust, orders, li):
    """The single-node plan: pure pyarrow over the full tables."""
    tc = pa.table(cust)
    to = pa.table(orders)
    tl = pa.table(li)
    tc = tc.filter(pc.equal(tc["c_mktsegment"], 0))
    to = to.filter(pc.less(to["o_orderdate"], CUTOFF))
    tl = tl.filter(pc.greater(tl["l_shipdate"], CUTOFF))
    oj = to.join(tc.select(["c_custkey"]), keys="o_custkey", right_keys="c_custkey",
                 join_type="inner")
    lj = tl.join(oj.select(["o_orderkey", "o_orderdate", "o_shippriority"]),
                 keys="l_orderkey", right_keys="o_orderkey", join_type="inner")
    rev = pc.multiply(lj["l_extendedprice"],
                              pc.subtract(pa.scalar(1.0), lj["l_discount"]))
    lj = lj.append_column("revenue", rev)
    agg = lj.group_by(["l_orderkey", "o_orderdate", "o_shippriority"]).aggregate(
        [("revenue", "sum")])
    return agg


def test_q3_join_pipeline_matches_direct_plan():
    import pyarrow as pa

    rng = np.random.default_rng(53)
    cust, orders, li = make_tables(rng)
    P = 16

    # ---- stage 1: build side. Filter customer on the segment, broadcast the keys
    # (CollectLeft build replication; nranks=1 exercises the RCCL data plane).
    keep = cust["c_mktsegment"] == 0
    build_keys = cust["c_custkey"][keep]
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    bbatch = api.DeviceBatch([{"dtype": "i64", "data": build_keys, "valid": None}])
    bc = comm.broadcast(bbatch, root=0)
    got_keys = bc.col(0)["data"]
    assert (got_keys == build_keys).all()
    bc.destroy()
    bbatch.free()
    build_set = got_keys  # every rank now holds the build side

    # ---- stage 2a: probe side A — orders filtered + joined with the broadcast build
    # side (the join above BroadcastExec, host-side), then GPU-shuffled on o_orderkey.
    omask = orders["o_orderdate"] < CUTOFF
    of = {k: v[omask] for k, v in orders.items()}
    isin = pc.is_in(pa.array(of["o_custkey"]), value_set=pa.array(build_set))
    omask2 = np.asarray(isin)
    oj = {k: v[omask2] for k, v in of.items()}
    obatch = api.DeviceBatch([
        {"dtype": "i64", "data": oj["o_orderkey"], "valid": None},
        {"dtype": "i32", "data": oj["o_orderdate"], "valid": None},
        {"dtype": "i32", "data": oj["o_shippriority"], "valid": None},
    ])
    opart = api.Partitioner(obatch, [0], P)
    opart.run()
    opart.sync()
    oex = comm.exchange(opart)

    # ---- stage 2b: probe side B — lineitem filtered, GPU-shuffled on l_orderkey
    # (same P: co-partitioned with the orders side).
    lmask = li["l_shipdate"] > CUTOFF
    lf = {k: v[lmask] for k, v in li.items()}
    lbatch = api.DeviceBatch([
        {"dtype": "i64", "data": lf["l_orderkey"], "valid": None},
        {"dtype": "f64", "data": lf["l_extendedprice"], "valid": None},
        {"dtype": "f64", "data": lf["l_discount"], "valid": None},
    ])
    lpart = api.Partitioner(lbatch, [0], P)
    lpart.run()
    lpart.sync()
    lex = comm.exchange(lpart)

    # ---- stage 3: per-partition hash join + partial agg (host; the engine above the
    # seam). Identical hash => both sides of a key land in the same partition.
    ooff = opart.row_offsets()
    loff = lpart.row_offsets()
    ok = oex.col_data(0)["data"]
    od = oex.col_data(1)["data"]
    osp = oex.col_data(2)["data"]
    lk = lex.col_data(0)["data"]
    lp = lex.col_data(1)["data"]
    ld = lex.col_data(2)["data"]
    merged = {}
    total_joined = 0
    for p in range(P):
        olo, ohi = ooff[p], ooff[p + 1]
        llo, lhi = loff[p], loff[p + 1]
        tor = pa.table({"o_orderkey": ok[olo:ohi], "o_orderdate": od[olo:ohi],
                        "o_shippriority": osp[olo:ohi]})
        tli = pa.table({"l_orderkey": lk[llo:lhi], "l_extendedprice": lp[llo:lhi],
                        "l_discount": ld[llo:lhi]})
        j = tli.join(tor, keys="l_orderkey", right_keys="o_orderkey", join_type="inner")
        if j.num_rows == 0:
            continue
        total_joined += j.num_rows
        rev = pc.multiply(j["l_extendedprice"],
                                  pc.subtract(pa.scalar(1.0), j["l_discount"]))
        j = j.append_column("revenue", rev)
        agg = j.group_by(["l_orderkey", "o_orderdate", "o_shippriority"]).aggregate(
            [("revenue", "sum")])
        for row in agg.to_pylist():
            key = (row["l_orderkey"], row["o_orderdate"], row["o_shippriority"])
            assert key not in merged, "group split across partitions"
            merged[key] = row["revenue_sum"]
    obatch.free()
    lbatch.free()
    opart.destroy()
    lpart.destroy()
    oex.destroy()
    lex.destroy()
    comm.destroy()

    # ---- the whole-query answer must equal the direct single-node plan
    want = direct_answer(pa, cust, orders, li)
    assert len(merged) == want.num_rows, "group count differs"
    for row in want.to_pylist():
        key = (row["l_orderkey"], row["o_orderdate"], row["o_shippriority"])
        assert key in merged, f"missing group {key}"
        w = row["revenue_sum"]
        g = merged[key]
        assert abs(g - w) <= 1e-6 * max(abs(w), 1.0), f"revenue differs for {key}"
    # and the q3 presentation layer: top-10 by revenue (stable tie-break on the keys)
    top_w = sorted(want.to_pylist(),
                   key=lambda r: (-r["revenue_sum"], r["l_orderkey"]))[:10]
    top_g = sorted(((k, v) for k, v in merged.items()), key=lambda kv: (-kv[1], kv[0][0]))[:10]
    for wrow, (gkey, grev) in zip(top_w, top_g):
        assert gkey == (wrow["l_orderkey"], wrow["o_orderdate"], wrow["o_shippriority"])
        assert abs(grev - wrow["revenue_sum"]) <= 1e-6 * max(abs(wrow["revenue_sum"]), 1.0)
