"""RCCL broadcast on the GPU box (marked gpu; nranks=1 self-broadcast — the multi-rank
placement is a single collective with no routing logic)."""

import numpy as np
import pytest

from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu


def test_self_broadcast_roundtrip():
    rng = np.random.default_rng(5)
    n = 50000
    rows = [b"b" * int(l) for l in rng.integers(0, 24, n)]
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(r) for r in rows])
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64), "valid": None},
        {"dtype": "utf8", "data": np.frombuffer(b"".join(rows), dtype=np.uint8),
         "offsets": off, "valid": (rng.random(n) > 0.2).astype(np.uint8)},
    ]
    batch = api.DeviceBatch(cols)
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    b = comm.broadcast(batch, root=0)
    assert b.n_rows == n and b.n_cols == 2
    c0 = b.col(0)
    assert (c0["data"] == cols[0]["data"]).all()
    c1 = b.col(1)
    assert (c1["offsets"] == off).all()
    assert c1["data"].tobytes() == cols[1]["data"].tobytes()
    assert (c1["valid"] == cols[1]["valid"]).all()
    b.destroy()
    comm.destroy()
    batch.free()
