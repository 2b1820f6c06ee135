"""RCCL exchange on the GPU box (marked gpu). gpurun provides one GPU, so the collective is
exercised at nranks=1 (self-exchange through RCCL send/recv): the received window must equal
the partitioner's full output. N>1 placement logic is pinned by tests/test_exchange_gloo.py's
model and by the driver's multi-GPU scale run."""

import numpy as np
import pytest

import oracle
from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu


def test_self_exchange_roundtrip():
    rng = np.random.default_rng(31)
    n, P = 100000, 16
    rows = [b"s" * int(l) for l in rng.integers(0, 20, n)]
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum([len(r) for r in rows])
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64), "valid": None},
        {"dtype": "f64", "data": rng.normal(size=n), "valid": None},
        {"dtype": "utf8", "data": np.frombuffer(b"".join(rows), dtype=np.uint8),
         "offsets": off, "valid": (rng.random(n) > 0.1).astype(np.uint8)},
    ]
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], P)
    part.run()
    part.sync()
    ref = oracle.repartition(cols, [0], P)

    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    ex = comm.exchange(part)
    assert ex.total_rows == n
    rc = ex.row_counts()
    assert rc.shape == (1, P)
    assert (rc[0] == ref["part_offsets"][1:] - ref["part_offsets"][:-1]).all()

    got0 = ex.col_data(0)
    assert (got0["data"] == ref["cols"][0]["data"]).all()
    got1 = ex.col_data(1)
    assert np.array_equal(got1["data"], ref["cols"][1]["data"], equal_nan=True)
    got2 = ex.col_data(2)
    assert (got2["lengths"] == ref["cols"][2]["lengths"]).all()
    assert got2["data"].tobytes() == ref["cols"][2]["data"].tobytes()
    v = ex.col_validity(2)
    assert (v == ref["cols"][2]["valid"]).all()
    ms, egress = ex.stats()
    assert ms >= 0 and egress == 0  # single rank: no xGMI egress

    ex.destroy()
    comm.destroy()
    part.destroy()
    batch.free()


def test_self_exchange_empty_batch():
    """Zero-row batch through partition + exchange (the reference's empty-batch wire edge
    case, tests/empty_columns_between_workers.rs analog at our boundary)."""
    cols = [{"dtype": "i64", "data": np.zeros(0, dtype=np.int64), "valid": None}]
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], 4)
    part.run()
    part.sync()
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    ex = comm.exchange(part)
    assert ex.total_rows == 0
    assert (ex.row_counts() == 0).all()
    ex.destroy()
    comm.destroy()
    part.destroy()
    batch.free()


def test_self_exchange_sparse_partitions():
    """Highly skewed input: most partitions empty (zero-size slices must pair correctly)."""
    n = 10000
    cols = [{"dtype": "i64", "data": np.full(n, 123456789, dtype=np.int64), "valid": None},
            {"dtype": "f64", "data": np.arange(n, dtype=np.float64), "valid": None}]
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], 64)
    part.run()
    part.sync()
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    ex = comm.exchange(part)
    assert ex.total_rows == n
    rc = ex.row_counts()
    assert (rc > 0).sum() == 1  # single key -> single partition
    got = ex.col_data(1)["data"]
    assert (np.sort(got) == np.arange(n, dtype=np.float64)).all()
    ex.destroy()
    comm.destroy()
    part.destroy()
    batch.free()


def test_self_coalesce_roundtrip():
    """dd_coalesce_run at nranks=1, consumer_tasks=1: the consumer's group is {rank 0},
    so the received buffers equal the partitioner's own output (whole-partition fetch,
    no repartition — NetworkCoalesceExec's data plane, network_coalesce.rs:24-70)."""
    rng = np.random.default_rng(71)
    n, P = 50000, 8
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**10, n, dtype=np.int64), "valid": None},
        {"dtype": "i32", "data": rng.integers(0, 100, n, dtype=np.int64).astype(np.int32),
         "valid": (rng.random(n) > 0.1).astype(np.uint8)},
    ]
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], P)
    part.run()
    part.sync()
    ref = oracle.repartition(cols, [0], P)
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    co = comm.coalesce(part, consumer_tasks=1)
    assert co.total_rows == n
    rc = co.row_counts(producers=1, parts=P)
    assert (rc[0] == ref["part_offsets"][1:] - ref["part_offsets"][:-1]).all()
    got = co.col_data(0)["data"]
    assert (got == ref["cols"][0]["data"]).all()
    v = co.col_validity(1)
    assert (v == ref["cols"][1]["valid"]).all()
    co.destroy()
    comm.destroy()
    part.destroy()
    batch.free()


def test_exchange_boundary_to_wire_read_by_pyarrow():
    """SURVEY §8f row 2 end-to-end: the EXCHANGE BOUNDARY's bytes cross the (simulated)
    node hop in the reference's wire format — GPU shuffle -> arrow_boundary partition
    batches -> our C Arrow IPC + lz4 encoder -> read back by pyarrow.ipc, value-equal
    with a direct pyarrow take() of the same partition rows."""
    import io

    import pyarrow as pa

    from datafusion_distributed_amd import arrow_boundary, wire

    rng = np.random.default_rng(83)
    n, P = 200_000, 8
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64),
         "valid": None},
        {"dtype": "f64", "data": rng.normal(size=n),
         "valid": (rng.random(n) > 0.2).astype(np.uint8)},
    ]
    lens = rng.integers(0, 24, n)
    off = np.zeros(n + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    cols.append({"dtype": "utf8",
                 "data": rng.integers(97, 123, int(off[-1]), dtype=np.int64)
                 .astype(np.uint8),
                 "offsets": off, "valid": None})
    batch = api.DeviceBatch(cols)
    part = api.Partitioner(batch, [0], P)
    part.run()
    part.sync()

    names = ["k", "v", "s"]
    direct = pa.table({"k": cols[0]["data"],
                       "v": pa.array([float(x) if vv else None
                                      for x, vv in zip(cols[1]["data"],
                                                       cols[1]["valid"])]),
                       "s": [bytes(cols[2]["data"][off[i]:off[i + 1]]).decode()
                             for i in range(n)]})
    pid = part.pids()
    checked = 0
    for p, pbatch in arrow_boundary.partition_batches(part, 0, P, batch_size=10**9):
        if pbatch.num_rows == 0:
            continue
        pbatch = pbatch.rename_columns(names)
        blob = wire.encode_batches([pbatch], use_lz4=True)
        got = pa.ipc.open_stream(io.BytesIO(blob)).read_all()
        rows = np.flatnonzero(pid == p)
        want = direct.take(pa.array(rows, type=pa.int64())).combine_chunks()
        assert got.num_rows == len(rows)
        assert got.equals(want), f"partition {p} wire round-trip differs"
        checked += 1
        if checked >= 2:  # two partitions through the hop pin the wire path
            break
    assert checked
    batch.free()
    part.destroy()
