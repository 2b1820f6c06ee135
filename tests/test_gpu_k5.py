"""K5 (LDS-staged var-byte scatter) eligibility boundaries and edges (marked gpu):
max string length exactly at / just past the TH=128 gate (K5 vs the K4 gather fallback),
empty strings, all-empty columns, tiny and ragged row counts — all bit-exact vs the
oracle on BOTH engines (DD_K5=0 forces the gather)."""

import numpy as np
import pytest

from tests.test_gpu_parity import check_against_oracle

pytestmark = pytest.mark.gpu


def utf8_col(rng, n, minlen, maxlen):
    lens = rng.integers(minlen, maxlen + 1, n) if n else np.zeros(0, dtype=np.int64)
    if n:
        lens[rng.integers(0, n)] = maxlen  # pin the max so the gate decision is exact
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    data = rng.integers(33, 127, int(off[-1]), dtype=np.int64).astype(np.uint8)
    return {"dtype": "utf8", "data": data, "offsets": off, "valid": None}


@pytest.mark.parametrize("maxlen", [128, 129])  # K5 gate boundary: <=128 in, 129 out
@pytest.mark.parametrize("k5", ["1", "0"])
def test_k5_threshold_boundary(maxlen, k5, monkeypatch):
    monkeypatch.setenv("DD_K5", k5)
    rng = np.random.default_rng(maxlen)
    n = 120_000
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64),
         "valid": None},
        utf8_col(rng, n, 8, maxlen),
        {"dtype": "i32", "data": rng.integers(0, 99, n).astype(np.int32), "valid": None},
    ]
    check_against_oracle(cols, [0], 64)
    check_against_oracle(cols, [0], 100)


def test_k5_edge_shapes(monkeypatch):
    monkeypatch.delenv("DD_K5", raising=False)
    rng = np.random.default_rng(99)
    # all-empty strings (zero byte payload), tiny n, ragged last round (n % 1024 != 0)
    for n in [1, 63, 1024, 1025, 120_001]:
        empty = {"dtype": "utf8", "data": np.zeros(0, np.uint8),
                 "offsets": np.zeros(n + 1, dtype=np.int32), "valid": None}
        key = {"dtype": "i64", "data": rng.integers(0, 10**9, n, dtype=np.int64),
               "valid": None}
        check_against_oracle([key, empty], [0], 16)
        check_against_oracle([key, utf8_col(rng, n, 0, 17)], [0], 16)
