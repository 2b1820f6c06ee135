"""C-ABI error paths (marked gpu — validation runs on create, which needs a device):
every invalid request must fail with the right dd_status and never crash (mirrors the
reference's error mapping discipline, src/protocol/grpc/errors/)."""

import ctypes

import numpy as np
import pytest

from datafusion_distributed_amd import api

pytestmark = pytest.mark.gpu


def make_batch(n=64):
    rng = np.random.default_rng(0)
    return api.DeviceBatch([
        {"dtype": "i64", "data": rng.integers(0, 100, n, dtype=np.int64), "valid": None}])


def expect_status(status, fn, *args):
    h = ctypes.c_void_p()
    st = fn(*args, ctypes.byref(h))
    assert st == status, f"expected {status}, got {st}: {api.lib().dd_last_error()}"


def test_partitioner_create_errors():
    b = make_batch()
    keys = (ctypes.c_int32 * 1)(0)
    bad_keys = (ctypes.c_int32 * 1)(5)
    L = api.lib()
    # P out of range -> UNSUPPORTED (6)
    expect_status(6, L.dd_partitioner_create, ctypes.byref(b.desc), keys, 1,
                  ctypes.c_uint32(0))
    expect_status(6, L.dd_partitioner_create, ctypes.byref(b.desc), keys, 1,
                  ctypes.c_uint32(4096))
    # key index out of range -> INVALID (1)
    expect_status(1, L.dd_partitioner_create, ctypes.byref(b.desc), bad_keys, 1,
                  ctypes.c_uint32(8))
    # zero keys -> INVALID
    expect_status(1, L.dd_partitioner_create, ctypes.byref(b.desc), keys, 0,
                  ctypes.c_uint32(8))
    b.free()


def test_partial_reduce_errors():
    b = make_batch()
    keys = (ctypes.c_int32 * 1)(0)
    aggs = (ctypes.c_int32 * 1)(0)
    bad_op = (ctypes.c_int32 * 1)(99)
    ok_op = (ctypes.c_int32 * 1)(1)  # COUNT
    L = api.lib()
    # unknown op -> INVALID
    expect_status(1, L.dd_partial_reduce_run, ctypes.byref(b.desc), keys, 1, aggs,
                  bad_op, 1, None)
    # SUM_F64 on an i64 column -> UNSUPPORTED
    sum_f64 = (ctypes.c_int32 * 1)(0)
    expect_status(6, L.dd_partial_reduce_run, ctypes.byref(b.desc), keys, 1, aggs,
                  sum_f64, 1, None)
    # valid COUNT works
    h = ctypes.c_void_p()
    assert L.dd_partial_reduce_run(ctypes.byref(b.desc), keys, 1, aggs, ok_op, 1, None,
                                   ctypes.byref(h)) == 0
    L.dd_reducer_destroy(h)
    b.free()


def test_exchange_requires_run_and_divisibility():
    b = make_batch()
    part = api.Partitioner(b, [0], 7)  # 7 % nranks(1) == 0, but not yet run
    L = api.lib()
    comm = api.Comm(api.Comm.unique_id(), 0, 1)
    h = ctypes.c_void_p()
    st = L.dd_exchange_run(comm.h, part.h, None, ctypes.byref(h))
    assert st == 1  # partitioner has not run
    comm.destroy()
    part.destroy()
    b.free()
