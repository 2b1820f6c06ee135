"""GPU end-to-end of the protobuf plan payload (marked gpu): a prost-shaped
SetPlanRequest / ExecuteTaskRequest pair — encoded by google.protobuf, the independent
implementation — drives the real task cache and partition kernels; results bit-exact vs
the oracle. Mirrors the reference worker's SetPlan -> ExecuteTask flow
(src/worker/impl_execute_task.rs:19-59 with the RepartitionExecHead of
worker.proto:134-170)."""

import ctypes
import uuid

import numpy as np
import pytest

from datafusion_distributed_amd import api
from oracle import pyref as oracle
from tests.test_proto import M, make_partitioning, make_task_key

pytestmark = pytest.mark.gpu


def _wrap_partitioner(handle, batch, nparts):
    p = api.Partitioner.__new__(api.Partitioner)
    p.h = handle
    p.batch = batch
    p.nparts = nparts
    return p


def test_proto_set_plan_execute_roundtrip():
    rng = np.random.default_rng(91)
    n, P = 200_000, 32
    cols = [
        {"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64), "valid": None},
        {"dtype": "f64", "data": rng.normal(size=n), "valid": None},
    ]
    batch = api.DeviceBatch(cols)
    qid = uuid.uuid4()

    sp = M["ddtest.SetPlanRequest"]()
    sp.task_key.CopyFrom(make_task_key(qid, 2, 1))
    sp.task_count = 4
    sp.plan_proto = b"opaque datafusion-proto subplan bytes"
    sp_blob = bytes(sp.SerializeToString())
    api._check(api.lib().dd_set_plan_proto(sp_blob, ctypes.c_int64(len(sp_blob)),
                                           ctypes.byref(batch.desc)))

    et = M["ddtest.ExecuteTaskRequest"]()
    et.task_key.CopyFrom(make_task_key(qid, 2, 1))
    et.target_partition_start = 0
    et.target_partition_end = P
    et.repartition.partitioning = make_partitioning([("key", 0)], P)
    et_blob = bytes(et.SerializeToString())

    h = ctypes.c_void_p()
    api._check(api.lib().dd_execute_task_proto(et_blob, ctypes.c_int64(len(et_blob)),
                                               None, ctypes.byref(h)))
    # second execute for another partition window reuses the same cached run
    h2 = ctypes.c_void_p()
    api._check(api.lib().dd_execute_task_proto(et_blob, ctypes.c_int64(len(et_blob)),
                                               None, ctypes.byref(h2)))
    assert h.value == h2.value

    part = _wrap_partitioner(h, batch, P)
    part.sync()
    ref = oracle.repartition(cols, [0], P)
    pd = part.pids()
    assert pd is not None and (pd == ref["pid"]).all()
    assert (part.row_offsets() == ref["part_offsets"]).all()
    got = part.col_out(1)
    assert got["data"].tobytes() == ref["cols"][1]["data"].tobytes()

    # drop via the proto TaskKey; further executes -> NOT_FOUND
    tk_blob = bytes(make_task_key(qid, 2, 1).SerializeToString())
    api._check(api.lib().dd_drop_task_proto(tk_blob, ctypes.c_int64(len(tk_blob))))
    st = api.lib().dd_execute_task_proto(et_blob, ctypes.c_int64(len(et_blob)), None,
                                         ctypes.byref(h))
    assert st == 5  # DD_ERR_NOT_FOUND
    batch.free()


def test_proto_execute_unknown_key_and_bad_range():
    rng = np.random.default_rng(93)
    n, P = 1000, 8
    cols = [{"dtype": "i64", "data": rng.integers(0, 99, n, dtype=np.int64),
             "valid": None}]
    batch = api.DeviceBatch(cols)
    qid = uuid.uuid4()
    et = M["ddtest.ExecuteTaskRequest"]()
    et.task_key.CopyFrom(make_task_key(qid, 0, 0))
    et.target_partition_end = P
    et.repartition.partitioning = make_partitioning([("k", 0)], P)
    blob = bytes(et.SerializeToString())
    h = ctypes.c_void_p()
    st = api.lib().dd_execute_task_proto(blob, ctypes.c_int64(len(blob)), None,
                                         ctypes.byref(h))
    assert st == 5  # no SetPlanRequest seen

    sp = M["ddtest.SetPlanRequest"]()
    sp.task_key.CopyFrom(make_task_key(qid, 0, 0))
    sp_blob = bytes(sp.SerializeToString())
    api._check(api.lib().dd_set_plan_proto(sp_blob, ctypes.c_int64(len(sp_blob)),
                                           ctypes.byref(batch.desc)))
    et.target_partition_end = P + 1  # outside the decoded partitioning
    blob = bytes(et.SerializeToString())
    st = api.lib().dd_execute_task_proto(blob, ctypes.c_int64(len(blob)), None,
                                         ctypes.byref(h))
    assert st == 1  # DD_ERR_INVALID
    tk_blob = bytes(make_task_key(qid, 0, 0).SerializeToString())
    api.lib().dd_drop_task_proto(tk_blob, ctypes.c_int64(len(tk_blob)))
    batch.free()
