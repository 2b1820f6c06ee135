"""Oracle tests: golden-vector pinning, C vs numpy cross-check, edge cases, and the
query-result parity invariant the reference's own suites pin (partition-then-aggregate ==
direct aggregation; /root/reference/tests/tpch_correctness_test.rs:139-158 analog), with
pyarrow as the independent aggregation implementation."""

import glob
import os

import numpy as np
import pytest

import oracle
from oracle import pyref

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")

import importlib.util

_spec = importlib.util.spec_from_file_location("gen_golden", os.path.join(GOLDEN, "gen_golden.py"))
gen_golden = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(gen_golden)


@pytest.mark.parametrize("fname", sorted(glob.glob(os.path.join(GOLDEN, "*_p*.npz"))))
def test_c_oracle_matches_golden(fname):
    cols, key_idx, nparts, exp = gen_golden.load_case(fname)
    res = oracle.repartition(cols, key_idx, nparts)
    assert (res["hash"] == exp["hash"]).all()
    assert (res["pid"] == exp["pid"]).all()
    assert (res["order"] == exp["order"]).all()
    assert (res["part_offsets"] == exp["part_offsets"]).all()


def test_hash_known_answers():
    z = np.load(os.path.join(GOLDEN, "hash_kat.npz"))
    L = oracle.lib()
    for x, want in zip(z["mix64_in"], z["mix64_out"]):
        assert L.dd_oracle_mix64(int(x)) == int(want)
    buf = bytes(z["bytes_in"])
    for k, want in enumerate(z["bytes_prefix_out"]):
        assert L.dd_oracle_hash_bytes(buf[:k], k) == int(want)
        assert pyref.hash_bytes_scalar(buf[:k]) == int(want)


def test_c_matches_pyref_fresh_random():
    cols = gen_golden.make_bench_batch(3000, seed=987)
    for keys, P in [([0], 7), ([1], 3), ([0, 2, 3, 4], 31), ([5], 256)]:
        rc = oracle.repartition(cols, keys, P)
        rp = pyref.repartition(cols, keys, P)
        assert (rc["hash"] == rp["hash"]).all(), (keys, P)
        assert (rc["order"] == rp["order"]).all(), (keys, P)
        for a, b in zip(rc["cols"], rp["cols"]):
            for k in a:
                if isinstance(a[k], np.ndarray) and k in b:
                    if a[k].dtype.kind == "f":
                        assert np.array_equal(a[k], b[k], equal_nan=True)
                    else:
                        assert (a[k] == b[k]).all()


def test_empty_batch():
    """Zero-row batches must round-trip (the reference tests zero-column/empty batches over
    the wire: tests/empty_columns_between_workers.rs:11-31)."""
    cols = [{"dtype": "i64", "data": np.zeros(0, dtype=np.int64), "valid": None}]
    res = oracle.repartition(cols, [0], 8)
    assert (res["part_offsets"] == 0).all()
    assert len(res["pid"]) == 0


def test_all_null_key():
    n = 100
    cols = [{"dtype": "i64", "data": np.arange(n, dtype=np.int64),
             "valid": np.zeros(n, dtype=np.uint8)}]
    res = oracle.repartition(cols, [0], 4)
    # null rows leave h=0 -> all rows in partition 0 % 4 == 0
    assert (res["pid"] == 0).all()
    assert res["part_offsets"][1] == n
    # and the gather is the identity (stable)
    assert (res["cols"][0]["data"] == np.arange(n)).all()


def test_single_partition():
    cols = [{"dtype": "i64", "data": np.random.default_rng(0).integers(0, 100, 50,
             dtype=np.int64), "valid": None}]
    res = oracle.repartition(cols, [0], 1)
    assert (res["pid"] == 0).all()
    assert (res["order"] == np.arange(50)).all()


def test_float_canonicalization():
    # -0.0 == +0.0 and all NaNs hash equal
    a = {"dtype": "f64", "data": np.array([0.0, -0.0, np.nan, np.float64("nan")]),
         "valid": None}
    h = oracle.hash_cols([a])
    assert h[0] == h[1]
    assert h[2] == h[3]


def test_dict_hash_equals_value_hash():
    # dictionary encoding is invisible: dict32 of values hashes like plain utf8
    vals = [b"x", b"hello", b"", b"longer-string-here"]
    dby = np.frombuffer(b"".join(vals), dtype=np.uint8)
    doff = np.zeros(len(vals) + 1, dtype=np.int32)
    doff[1:] = np.cumsum([len(v) for v in vals])
    idx = np.array([3, 1, 0, 2, 1], dtype=np.int32)
    dcol = {"dtype": "dict32", "data": idx, "dict_bytes": dby, "dict_offsets": doff,
            "valid": None}
    # equivalent plain utf8
    rows = [vals[i] for i in idx]
    off = np.zeros(len(rows) + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(r) for r in rows])
    ucol = {"dtype": "utf8", "data": np.frombuffer(b"".join(rows), dtype=np.uint8),
            "offsets": off, "valid": None}
    assert (oracle.hash_cols([dcol]) == oracle.hash_cols([ucol])).all()


def test_stability_within_partition():
    rng = np.random.default_rng(3)
    n = 5000
    cols = [{"dtype": "i64", "data": rng.integers(0, 8, n, dtype=np.int64), "valid": None}]
    res = oracle.repartition(cols, [0], 4)
    # rows within each partition keep input order
    for p in range(4):
        lo, hi = res["part_offsets"][p], res["part_offsets"][p + 1]
        idx = res["order"][lo:hi]
        assert (np.diff(idx) > 0).all()


# ---------------- query-result parity (the invariant the reference pins) ----------------

def _weather_like(n=366, seed=11):
    """Synthetic stand-in for testdata/weather (the real file is a git-lfs pointer —
    absent). Shape mirrors tests/distributed_aggregation.rs:23: RainToday in
    {Yes, No, null}, count(*) GROUP BY RainToday."""
    rng = np.random.default_rng(seed)
    choice = rng.random(n)
    rows = [b"Yes" if c < 0.18 else b"No" for c in choice]
    valid = (rng.random(n) > 0.02).astype(np.uint8)
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(r) for r in rows])
    return {
        "dtype": "utf8",
        "data": np.frombuffer(b"".join(rows), dtype=np.uint8),
        "offsets": off,
        "valid": valid,
    }


def _utf8_to_pylist(col):
    off = np.asarray(col["offsets"], dtype=np.int64)
    buf = col["data"].tobytes()
    out = []
    valid = col.get("valid")
    n = len(off) - 1
    for i in range(n):
        if valid is not None and not valid[i]:
            out.append(None)
        else:
            out.append(buf[off[i]: off[i + 1]].decode("utf-8", "replace"))
    return out


def test_partition_then_aggregate_weather():
    import pyarrow as pa

    col = _weather_like()
    n = len(col["offsets"]) - 1
    # direct aggregation with pyarrow (independent implementation)
    tbl = pa.table({"RainToday": _utf8_to_pylist(col)})
    direct = tbl.group_by("RainToday").aggregate([([], "count_all")])
    direct_map = dict(zip([v.as_py() for v in direct["RainToday"]],
                          [v.as_py() for v in direct["count_all"]]))

    # partition into 6 (the reference plan's Hash([RainToday], 6),
    # distributed_aggregation.rs plan snapshot), aggregate per partition, merge
    res = oracle.repartition([col], [0], 6)
    merged = {}
    lens = res["cols"][0]["lengths"]
    byts = res["cols"][0]["data"].tobytes()
    valid = res["cols"][0]["valid"]
    starts = np.zeros(n, dtype=np.int64)
    if n:
        starts[1:] = np.cumsum(lens[:-1])
    for p in range(6):
        lo, hi = res["part_offsets"][p], res["part_offsets"][p + 1]
        part_counts = {}
        for i in range(lo, hi):
            key = None if not valid[i] else byts[starts[i]: starts[i] + lens[i]].decode()
            part_counts[key] = part_counts.get(key, 0) + 1
        for k, v in part_counts.items():
            merged[k] = merged.get(k, 0) + v
    assert merged == direct_map
    # every group key lands in exactly one partition (hash-partition guarantee)
    seen = {}
    for p in range(6):
        lo, hi = res["part_offsets"][p], res["part_offsets"][p + 1]
        for i in range(lo, hi):
            key = None if not valid[i] else byts[starts[i]: starts[i] + lens[i]].decode()
            assert seen.setdefault(key, p) == p


def test_partition_then_aggregate_tpch_q1_like():
    """TPC-H q1 shape: GROUP BY (l_returnflag, l_linestatus), SUM aggregates; bit-exact on
    counts/keys, <=1e-6 relative on float sums (the north star's tolerance)."""
    import pyarrow as pa

    rng = np.random.default_rng(5)
    n = 20000
    rf = rng.integers(0, 3, n).astype(np.int32)   # returnflag (3 values)
    ls = rng.integers(0, 2, n).astype(np.int32)   # linestatus (2 values)
    qty = rng.uniform(1, 50, n)
    price = rng.uniform(900, 105000, n)
    cols = [
        {"dtype": "i32", "data": rf, "valid": None},
        {"dtype": "i32", "data": ls, "valid": None},
        {"dtype": "f64", "data": qty, "valid": None},
        {"dtype": "f64", "data": price, "valid": None},
    ]
    tbl = pa.table({"rf": rf, "ls": ls, "qty": qty, "price": price})
    direct = tbl.group_by(["rf", "ls"]).aggregate([("qty", "sum"), ("price", "sum"),
                                                   ([], "count_all")])
    dmap = {}
    for i in range(direct.num_rows):
        k = (direct["rf"][i].as_py(), direct["ls"][i].as_py())
        dmap[k] = (direct["qty_sum"][i].as_py(), direct["price_sum"][i].as_py(),
                   direct["count_all"][i].as_py())

    P = 6
    res = oracle.repartition(cols, [0, 1], P)
    gmap = {}
    rfo, lso = res["cols"][0]["data"], res["cols"][1]["data"]
    qo, po = res["cols"][2]["data"], res["cols"][3]["data"]
    for p in range(P):
        lo, hi = res["part_offsets"][p], res["part_offsets"][p + 1]
        part = {}
        for i in range(lo, hi):
            k = (int(rfo[i]), int(lso[i]))
            s = part.setdefault(k, [0.0, 0.0, 0])
            s[0] += qo[i]
            s[1] += po[i]
            s[2] += 1
        for k, s in part.items():
            assert k not in gmap, "group split across partitions"
            gmap[k] = tuple(s)
    assert set(gmap) == set(dmap)
    for k in dmap:
        assert gmap[k][2] == dmap[k][2]  # counts bit-exact
        assert abs(gmap[k][0] - dmap[k][0]) <= 1e-6 * abs(dmap[k][0])
        assert abs(gmap[k][1] - dmap[k][1]) <= 1e-6 * abs(dmap[k][1])


@pytest.mark.parametrize("case", range(12))
def test_c_matches_pyref_random_shapes(case):
    """Deeper C-oracle vs independent-numpy cross-check: random dtype mixes, null
    densities, sizes and partition counts (the GPU fuzz distribution, CPU-only legs).
    The two restatements share no code (C loops vs numpy stable argsort), so agreement
    on hash, order and every gathered buffer pins the normative spec itself."""
    import tests.test_gpu_fuzz as gf

    rng = np.random.default_rng(5000 + case)
    n = int(rng.choice([1, 2, 63, 65, 1000, 4096, 20000]))
    ncols = int(rng.integers(1, 7))
    dtypes = list(rng.choice(gf.FIXED + ["utf8", "dict32"], ncols))
    cols = [gf.random_col(rng, n, dt, float(rng.choice([0, 0, 0.1, 0.5])))
            for dt in dtypes]
    nkeys = int(rng.integers(1, min(ncols, 4) + 1))
    keys = [int(k) for k in rng.choice(ncols, nkeys, replace=False)]
    P = int(rng.choice([1, 2, 7, 16, 128, 777]))
    rc = oracle.repartition(cols, keys, P)
    rp = pyref.repartition(cols, keys, P)
    assert (rc["hash"] == rp["hash"]).all(), (case, keys, P)
    assert (rc["order"] == rp["order"]).all(), (case, keys, P)
    assert (rc["part_offsets"] == rp["part_offsets"]).all()
    for a, b in zip(rc["cols"], rp["cols"]):
        for k in a:
            if isinstance(a[k], np.ndarray) and k in b:
                if a[k].dtype.kind == "f":
                    assert np.array_equal(a[k], b[k], equal_nan=True)
                else:
                    assert (a[k] == b[k]).all()
