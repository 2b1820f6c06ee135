"""Two-level var scatter (marked gpu, opt-in path): pass A coarse buckets + pass B final
partitions must be bit-identical to the oracle's single-pass stable partition."""

import numpy as np
import pytest

import oracle
from datafusion_distributed_amd.twolevel import two_level_partition

pytestmark = pytest.mark.gpu


def test_two_level_matches_oracle():
    rng = np.random.default_rng(101)
    n, P, B = 300_000, 128, 8
    lens = rng.integers(0, 32, n)
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64), "valid": None},
        {"dtype": "utf8", "data": rng.integers(32, 127, int(off[-1]), dtype=np.int64)
         .astype(np.uint8), "offsets": off,
         "valid": (rng.random(n) > 0.1).astype(np.uint8)},
        {"dtype": "f64", "data": rng.normal(size=n), "valid": None},
    ]
    ref = oracle.repartition(cols, [0], P)
    res = two_level_partition(cols, [0], P, buckets=B)
    assert (res["part_row_offsets"] == ref["part_offsets"]).all()
    assert (res["cols"][0]["data"] == ref["cols"][0]["data"]).all()
    assert (res["cols"][1]["lengths"] == ref["cols"][1]["lengths"]).all()
    assert res["cols"][1]["data"].tobytes() == ref["cols"][1]["data"].tobytes()
    assert (res["cols"][1]["valid"] == ref["cols"][1]["valid"]).all()
    assert np.array_equal(res["cols"][2]["data"], ref["cols"][2]["data"], equal_nan=True)


def test_two_level_utf8_key():
    # utf8 as the KEY: the bucket view's rebuilt offsets feed the hash in pass B
    rng = np.random.default_rng(103)
    n, P, B = 100_000, 64, 4
    lens = rng.integers(1, 16, n)
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    cols = [
        {"dtype": "utf8", "data": rng.integers(97, 123, int(off[-1]), dtype=np.int64)
         .astype(np.uint8), "offsets": off, "valid": None},
        {"dtype": "i32", "data": rng.integers(0, 100, n, dtype=np.int64).astype(np.int32),
         "valid": None},
    ]
    ref = oracle.repartition(cols, [0], P)
    res = two_level_partition(cols, [0], P, buckets=B)
    assert (res["part_row_offsets"] == ref["part_offsets"]).all()
    assert res["cols"][0]["data"].tobytes() == ref["cols"][0]["data"].tobytes()
    assert (res["cols"][1]["data"] == ref["cols"][1]["data"]).all()
