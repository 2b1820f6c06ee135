"""GPU partial aggregation (marked gpu): final-merge of the GPU partials must equal direct
aggregation by pyarrow — exact on counts/keys, <=1e-6 relative on float sums (the
reference's correctness bar: tests/tpch_correctness_test.rs:139-158 analog)."""

import numpy as np
import pytest

from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu


def merge_partials(res, nk, na):
    """Final-aggregate step: sum duplicate groups (what the reference's mode=Final does)."""
    out = {}
    for i in range(len(res["keynull"])):
        key = tuple(res["keys"][i]) + (int(res["keynull"][i]),)
        acc = out.setdefault(key, np.zeros(na))
        acc += 0  # ensure array exists
        out[key] = acc + res["aggs"][i]
    return out


def merge_partials_mixed(res, int_ops):
    """Merge with integer slots bit-cast (count/sum_i64). NULL semantics: a non-count
    aggregate whose total non-null input count (res["nn"]) is 0 merges to None — exactly
    what DataFusion's mode=Final emits for SUM/MIN/MAX over an all-null group."""
    out = {}
    nnsum = {}
    for i in range(len(res["keynull"])):
        key = tuple(int(x) for x in res["keys"][i]) + (int(res["keynull"][i]),)
        vals = []
        for g in range(res["aggs"].shape[1]):
            v = res["aggs"][i, g]
            vals.append(int(np.float64(v).view(np.int64)) if g in int_ops else float(v))
        nn = [int(x) for x in res["nn"][i]]
        if key in out:
            out[key] = [a + b for a, b in zip(out[key], vals)]
            nnsum[key] = [a + b for a, b in zip(nnsum[key], nn)]
        else:
            out[key] = vals
            nnsum[key] = nn
    for key, vals in out.items():
        out[key] = [None if nnsum[key][g] == 0 else vals[g] for g in range(len(vals))]
    return out


def test_partial_reduce_q1_low_cardinality():
    import pyarrow as pa

    rng = np.random.default_rng(23)
    n = 1_000_000
    rf = rng.integers(0, 3, n, dtype=np.int64).astype(np.uint8)
    ls = rng.integers(0, 2, n, dtype=np.int64).astype(np.uint8)
    qty = rng.uniform(1, 50, n)
    price = rng.uniform(900, 105000, n)
    cols = [
        {"dtype": "u8", "data": rf, "valid": None},
        {"dtype": "u8", "data": ls, "valid": None},
        {"dtype": "f64", "data": qty, "valid": None},
        {"dtype": "f64", "data": price, "valid": None},
    ]
    batch = api.DeviceBatch(cols)
    res = api.partial_reduce(batch, [0, 1], [(2, "sum_f64"), (3, "sum_f64"),
                                             (None, "count")])
    batch.free()
    # partial output is small: 6 true groups per block plus bounded claim-race
    # duplicates (dd_reduce.hip probe notes); far below the input row count
    assert len(res["keynull"]) <= 64 * 64
    merged = merge_partials_mixed(res, int_ops={2})

    tbl = pa.table({"rf": rf, "ls": ls, "qty": qty, "price": price})
    direct = tbl.group_by(["rf", "ls"]).aggregate([("qty", "sum"), ("price", "sum"),
                                                   ([], "count_all")])
    assert len(merged) == direct.num_rows
    for i in range(direct.num_rows):
        key = (direct["rf"][i].as_py(), direct["ls"][i].as_py(), 0)
        got = merged[key]
        assert got[2] == direct["count_all"][i].as_py()  # counts exact
        assert abs(got[0] - direct["qty_sum"][i].as_py()) <= 1e-6 * abs(
            direct["qty_sum"][i].as_py())
        assert abs(got[1] - direct["price_sum"][i].as_py()) <= 1e-6 * abs(
            direct["price_sum"][i].as_py())


def test_partial_reduce_high_cardinality_spills():
    # 200k distinct i64 keys over 1M rows: tables overflow -> spill path exercised
    rng = np.random.default_rng(29)
    n = 1_000_000
    keys = rng.integers(0, 200_000, n, dtype=np.int64)
    vals = rng.normal(size=n)
    batch = api.DeviceBatch([
        {"dtype": "i64", "data": keys, "valid": None},
        {"dtype": "f64", "data": vals, "valid": None},
    ])
    res = api.partial_reduce(batch, [0], [(1, "sum_f64"), (None, "count")])
    batch.free()
    merged = merge_partials_mixed(res, int_ops={1})
    # counts conserve rows
    assert sum(v[1] for v in merged.values()) == n
    # numpy groupby cross-check on a sample of groups
    order = np.argsort(keys, kind="stable")
    sk, sv = keys[order], vals[order]
    bounds = np.flatnonzero(np.diff(sk)) + 1
    starts = np.concatenate([[0], bounds])
    ends = np.concatenate([bounds, [n]])
    assert len(merged) >= len(starts)  # duplicates allowed, never fewer groups
    uk = sk[starts]
    sums = np.add.reduceat(sv, starts)
    counts = ends - starts
    for j in rng.integers(0, len(uk), 50):
        key = (int(np.uint64(uk[j])), 0)
        got = merged[key]
        assert got[1] == counts[j]
        assert abs(got[0] - sums[j]) <= 1e-6 * max(abs(sums[j]), 1.0)


def test_partial_reduce_null_keys_and_null_aggs():
    import pyarrow as pa

    rng = np.random.default_rng(31)
    n = 100_000
    k = rng.integers(0, 5, n, dtype=np.int64).astype(np.int32)
    kvalid = (rng.random(n) > 0.1).astype(np.uint8)
    v = rng.normal(size=n)
    vvalid = (rng.random(n) > 0.2).astype(np.uint8)
    batch = api.DeviceBatch([
        {"dtype": "i32", "data": k, "valid": kvalid},
        {"dtype": "f64", "data": v, "valid": vvalid},
    ])
    res = api.partial_reduce(batch, [0], [(1, "sum_f64"), (None, "count")])
    batch.free()
    merged = merge_partials_mixed(res, int_ops={1})

    karr = pa.array([int(x) if vv else None for x, vv in zip(k, kvalid)])
    varr = pa.array([float(x) if vv else None for x, vv in zip(v, vvalid)])
    tbl = pa.table({"k": karr, "v": varr})
    direct = tbl.group_by("k").aggregate([("v", "sum"), ([], "count_all")])
    assert len(merged) == direct.num_rows
    for i in range(direct.num_rows):
        kv = direct["k"][i].as_py()
        key = (0, 1) if kv is None else (kv, 0)
        got = merged[key]
        assert got[1] == direct["count_all"][i].as_py()
        want = direct["v_sum"][i].as_py()
        if want is None:
            assert got[0] is None
        else:
            assert abs(got[0] - want) <= 1e-6 * max(abs(want), 1.0)


def test_partial_reduce_all_null_group_is_null():
    """A group whose aggregate inputs are ALL NULL must merge to NULL, not the op
    identity (DataFusion: SUM(v)=NULL, MIN(v)=NULL over an all-null group; COUNT(*)
    still counts rows). Pinned vs pyarrow direct aggregation."""
    import pyarrow as pa

    rng = np.random.default_rng(41)
    n = 50_000
    k = rng.integers(0, 8, n, dtype=np.int64)
    v = rng.normal(size=n)
    vvalid = (k != 3).astype(np.uint8)  # group 3: every aggregate input is NULL
    batch = api.DeviceBatch([
        {"dtype": "i64", "data": k, "valid": None},
        {"dtype": "f64", "data": v, "valid": vvalid},
    ])
    res = api.partial_reduce(batch, [0], [(1, "sum_f64"), (1, "min_f64"),
                                          (None, "count")])
    batch.free()
    # ops-aware final merge (sum / min / count), NULL when total non-null count is 0
    merged = {}
    nnsum = {}
    for i in range(len(res["keynull"])):
        key = (int(res["keys"][i, 0]), int(res["keynull"][i]))
        vals = [float(res["aggs"][i, 0]), float(res["aggs"][i, 1]),
                int(np.float64(res["aggs"][i, 2]).view(np.int64))]
        nn = [int(x) for x in res["nn"][i]]
        if key in merged:
            cur = merged[key]
            # a partial with nn[g]==0 carries the op identity (NaN for min) — skip it
            merged[key] = [cur[0] + vals[0],
                           cur[1] if nn[1] == 0 else
                           (vals[1] if nnsum[key][1] == 0 else min(cur[1], vals[1])),
                           cur[2] + vals[2]]
            nnsum[key] = [a + b for a, b in zip(nnsum[key], nn)]
        else:
            merged[key] = vals
            nnsum[key] = nn
    for key in merged:
        merged[key] = [None if nnsum[key][g] == 0 else merged[key][g] for g in range(3)]

    varr = pa.array([float(x) if vv else None for x, vv in zip(v, vvalid)])
    tbl = pa.table({"k": pa.array(k), "v": varr})
    direct = tbl.group_by("k").aggregate([("v", "sum"), ("v", "min"), ([], "count_all")])
    assert len(merged) == direct.num_rows
    for i in range(direct.num_rows):
        kv = direct["k"][i].as_py()
        got = merged[(kv, 0)]
        for gi, col in ((0, "v_sum"), (1, "v_min")):
            want = direct[col][i].as_py()
            if want is None:
                assert got[gi] is None, f"group {kv} {col}: want NULL, got {got[gi]}"
            else:
                assert abs(got[gi] - want) <= 1e-6 * max(abs(want), 1.0)
        assert got[2] == direct["count_all"][i].as_py()


def test_partial_reduce_min_max():
    import pyarrow as pa

    rng = np.random.default_rng(37)
    n = 500_000
    k = rng.integers(0, 64, n, dtype=np.int64)
    f = rng.normal(size=n) * 1000
    i = rng.integers(-(2**60), 2**60, n, dtype=np.int64)
    batch = api.DeviceBatch([
        {"dtype": "i64", "data": k, "valid": None},
        {"dtype": "f64", "data": f, "valid": None},
        {"dtype": "i64", "data": i, "valid": None},
    ])
    res = api.partial_reduce(batch, [0], [(1, "min_f64"), (1, "max_f64"),
                                          (2, "min_i64"), (2, "max_i64")])
    batch.free()
    # final merge: min of mins, max of maxes
    merged = {}
    for r in range(len(res["keynull"])):
        key = int(res["keys"][r, 0])
        v = res["aggs"][r]
        iv = res["aggs"][r].view(np.int64)
        cur = merged.setdefault(key, [np.inf, -np.inf, 2**63 - 1, -(2**63)])
        cur[0] = min(cur[0], v[0])
        cur[1] = max(cur[1], v[1])
        cur[2] = min(cur[2], int(iv[2]))
        cur[3] = max(cur[3], int(iv[3]))

    tbl = pa.table({"k": k, "f": f, "i": i})
    direct = tbl.group_by("k").aggregate([("f", "min"), ("f", "max"), ("i", "min"),
                                          ("i", "max")])
    assert len(merged) == direct.num_rows
    for r in range(direct.num_rows):
        key = direct["k"][r].as_py()
        got = merged[key]
        assert got[0] == direct["f_min"][r].as_py()
        assert got[1] == direct["f_max"][r].as_py()
        assert got[2] == direct["i_min"][r].as_py()
        assert got[3] == direct["i_max"][r].as_py()


def test_partial_reduce_dict_key():
    import pyarrow as pa

    rng = np.random.default_rng(41)
    n = 200_000
    nvals = 10
    vals = [f"cat{v}".encode() for v in range(nvals)]
    dby = np.frombuffer(b"".join(vals), dtype=np.uint8).copy()
    doff = np.zeros(nvals + 1, dtype=np.int32)
    doff[1:] = np.cumsum([len(v) for v in vals])
    idx = rng.integers(0, nvals, n).astype(np.int32)
    x = rng.normal(size=n)
    batch = api.DeviceBatch([
        {"dtype": "dict32", "data": idx, "dict_bytes": dby, "dict_offsets": doff,
         "valid": None},
        {"dtype": "f64", "data": x, "valid": None},
    ])
    res = api.partial_reduce(batch, [0], [(1, "sum_f64"), (None, "count")])
    batch.free()
    merged = merge_partials_mixed(res, int_ops={1})
    tbl = pa.table({"k": idx, "v": x})
    direct = tbl.group_by("k").aggregate([("v", "sum"), ([], "count_all")])
    assert len(merged) == direct.num_rows
    for r in range(direct.num_rows):
        key = (direct["k"][r].as_py(), 0)
        got = merged[key]
        assert got[1] == direct["count_all"][r].as_py()
        want = direct["v_sum"][r].as_py()
        assert abs(got[0] - want) <= 1e-6 * max(abs(want), 1.0)
