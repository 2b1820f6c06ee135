"""Chunked partitioning (marked gpu): an input forced through multiple row-chunks (tiny
caps) must produce bit-identical results to the oracle on the WHOLE input — the stability
contract across chunk boundaries (chunk order == input order within every partition).
This is the harness path for inputs beyond the single-launch caps (SF100 / full
ClickBench var sizes; SURVEY §8d 'chunked to fit')."""

import numpy as np
import pytest

import oracle
from datafusion_distributed_amd import api
from datafusion_distributed_amd.chunked import chunked_partition

pytestmark = pytest.mark.gpu


def test_chunked_matches_oracle_whole_input():
    rng = np.random.default_rng(91)
    n, P = 200_000, 32
    lens = rng.integers(0, 30, n)
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
    # force ~7 chunks via a tiny row cap, and exercise the byte cap too
    res = chunked_partition(cols, [0], P, max_chunk_rows=30_000,
                            max_chunk_var_bytes=300_000)
    assert res["n_chunks"] >= 6
    assert (res["part_row_offsets"] == ref["part_offsets"]).all()
    assert (res["cols"][0]["data"] == ref["cols"][0]["data"]).all()
    assert (res["cols"][1]["lengths"] == ref["cols"][1]["lengths"]).all()
    assert res["cols"][1]["data"].tobytes() == ref["cols"][1]["data"].tobytes()
    assert (res["cols"][1]["valid"] == ref["cols"][1]["valid"]).all()
    assert np.array_equal(res["cols"][2]["data"], ref["cols"][2]["data"], equal_nan=True)


def test_chunked_single_chunk_degenerates():
    rng = np.random.default_rng(93)
    n, P = 50_000, 8
    cols = [{"dtype": "i64", "data": rng.integers(0, 10**9, n, dtype=np.int64),
             "valid": None}]
    ref = oracle.repartition(cols, [0], P)
    res = chunked_partition(cols, [0], P)
    assert res["n_chunks"] == 1
    assert (res["part_row_offsets"] == ref["part_offsets"]).all()
    assert (res["cols"][0]["data"] == ref["cols"][0]["data"]).all()
