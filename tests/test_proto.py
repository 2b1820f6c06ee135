"""Protobuf plan/stage wire payload (dd_proto.cpp) — CPU suite.

The blobs are encoded by google.protobuf (an INDEPENDENT protobuf implementation) from a
dynamically-built descriptor of the reference's schema
(/root/reference/src/protocol/grpc/worker.proto:84-179 + datafusion-proto 55.0.0's
Partitioning — restated in dd_proto.cpp's header), then decoded by the C library's
hand-rolled wire reader. This pins our decoder against the wire format itself, the way
the reference's prost encoding would produce it. No GPU needed: decoding is pure host
logic (the execute path is covered by tests/test_gpu_proto.py)."""

import ctypes
import uuid

import pytest
from google.protobuf import descriptor_pb2, message_factory

from datafusion_distributed_amd import api

T = descriptor_pb2.FieldDescriptorProto


def _build_messages():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "dd_test.proto"
    fdp.package = "ddtest"
    fdp.syntax = "proto3"

    def msg(name, fields):
        m = fdp.message_type.add()
        m.name = name
        for fname, num, ftype, type_name, repeated in fields:
            f = m.field.add()
            f.name = fname
            f.number = num
            f.type = ftype
            f.label = T.LABEL_REPEATED if repeated else T.LABEL_OPTIONAL
            if type_name:
                f.type_name = f".ddtest.{type_name}"
        return m

    # reference worker.proto shapes (field numbers per worker.proto:84-179)
    msg("TaskKey", [("query_id", 1, T.TYPE_BYTES, None, False),
                    ("stage_id", 2, T.TYPE_UINT64, None, False),
                    ("task_number", 3, T.TYPE_UINT64, None, False)])
    msg("NoneHead", [])
    msg("BroadcastExecHead", [("output_partitions", 1, T.TYPE_UINT64, None, False)])
    msg("RepartitionExecHead", [("partitioning", 1, T.TYPE_BYTES, None, False)])
    msg("SetPlanRequest", [("task_key", 1, T.TYPE_MESSAGE, "TaskKey", False),
                           ("task_count", 2, T.TYPE_UINT64, None, False),
                           ("plan_proto", 3, T.TYPE_BYTES, None, False),
                           ("target_worker_url", 5, T.TYPE_STRING, None, False),
                           ("query_start_time_ns", 6, T.TYPE_UINT64, None, False)])
    msg("ExecuteTaskRequest",
        [("task_key", 1, T.TYPE_MESSAGE, "TaskKey", False),
         ("target_partition_start", 2, T.TYPE_UINT64, None, False),
         ("target_partition_end", 3, T.TYPE_UINT64, None, False),
         ("none", 6, T.TYPE_MESSAGE, "NoneHead", False),
         ("broadcast", 7, T.TYPE_MESSAGE, "BroadcastExecHead", False),
         ("repartition", 8, T.TYPE_MESSAGE, "RepartitionExecHead", False)])
    # datafusion-proto 55.0.0 Partitioning subset (restated; dd_proto.cpp header)
    msg("PhysicalColumn", [("name", 1, T.TYPE_STRING, None, False),
                           ("index", 2, T.TYPE_UINT32, None, False)])
    msg("PhysicalExprNode", [("column", 1, T.TYPE_MESSAGE, "PhysicalColumn", False)])
    msg("PhysicalHashRepartition",
        [("hash_expr", 1, T.TYPE_MESSAGE, "PhysicalExprNode", True),
         ("partition_count", 2, T.TYPE_UINT64, None, False)])
    msg("Partitioning", [("round_robin", 1, T.TYPE_UINT64, None, False),
                         ("hash", 2, T.TYPE_MESSAGE, "PhysicalHashRepartition", False),
                         ("unknown", 3, T.TYPE_UINT64, None, False)])
    return message_factory.GetMessages([fdp])


M = _build_messages()


def make_partitioning(cols, pcount):
    part = M["ddtest.Partitioning"]()
    for name, idx in cols:
        e = part.hash.hash_expr.add()
        e.column.name = name
        e.column.index = idx
    part.hash.partition_count = pcount
    return part.SerializeToString()


def make_task_key(qid: uuid.UUID, stage, task):
    tk = M["ddtest.TaskKey"]()
    tk.query_id = qid.bytes
    tk.stage_id = stage
    tk.task_number = task
    return tk


class DecodedKey(ctypes.Structure):
    _fields_ = [("hi", ctypes.c_uint64), ("lo", ctypes.c_uint64),
                ("stage", ctypes.c_uint64), ("task", ctypes.c_uint64)]


def decode_partitioning(blob):
    L = api.lib()
    cols = (ctypes.c_int32 * 8)()
    nk = ctypes.c_int32()
    np_ = ctypes.c_uint32()
    st = L.dd_decode_partitioning(bytes(blob), ctypes.c_int64(len(blob)), cols, 8,
                                  ctypes.byref(nk), ctypes.byref(np_))
    return st, [cols[i] for i in range(nk.value)], np_.value


def decode_execute(blob):
    L = api.lib()
    key = DecodedKey()
    lo = ctypes.c_uint64()
    hi = ctypes.c_uint64()
    head = ctypes.c_int32()
    cols = (ctypes.c_int32 * 8)()
    nk = ctypes.c_int32()
    np_ = ctypes.c_uint32()
    st = L.dd_decode_execute_task(bytes(blob), ctypes.c_int64(len(blob)),
                                  ctypes.byref(key), ctypes.byref(lo), ctypes.byref(hi),
                                  ctypes.byref(head), cols, 8, ctypes.byref(nk),
                                  ctypes.byref(np_))
    return st, key, lo.value, hi.value, head.value, [cols[i] for i in range(nk.value)], np_.value


def test_decode_partitioning_hash_columns():
    blob = make_partitioning([("l_orderkey", 0), ("l_suppkey", 3)], 128)
    st, cols, p = decode_partitioning(blob)
    assert st == 0 and cols == [0, 3] and p == 128


def test_decode_partitioning_rejects_round_robin():
    part = M["ddtest.Partitioning"]()
    part.round_robin = 16
    st, _, _ = decode_partitioning(part.SerializeToString())
    assert st == 6  # DD_ERR_UNSUPPORTED


def test_decode_set_plan_roundtrip():
    qid = uuid.UUID("0123456789abcdef0123456789abcdef")
    req = M["ddtest.SetPlanRequest"]()
    req.task_key.CopyFrom(make_task_key(qid, 3, 7))
    req.task_count = 4
    req.plan_proto = b"\x01\x02\x03opaque-datafusion-plan"
    req.target_worker_url = "http://worker-3:8080"  # skipped field (forward-compat)
    req.query_start_time_ns = 1726400000000000000
    blob = bytes(req.SerializeToString())  # held alive: plan_proto is a borrowed view

    L = api.lib()
    key = DecodedKey()
    tc = ctypes.c_uint64()
    plan = ctypes.c_void_p()
    plen = ctypes.c_int64()
    st = L.dd_decode_set_plan(blob, ctypes.c_int64(len(blob)), ctypes.byref(key),
                              ctypes.byref(tc), ctypes.byref(plan), ctypes.byref(plen))
    assert st == 0
    assert key.hi == 0x0123456789ABCDEF and key.lo == 0x0123456789ABCDEF
    assert key.stage == 3 and key.task == 7
    assert tc.value == 4
    got = ctypes.string_at(plan.value, plen.value)
    assert got == req.plan_proto


def test_decode_execute_task_repartition_head():
    qid = uuid.uuid5(uuid.NAMESPACE_DNS, "dd-test")
    req = M["ddtest.ExecuteTaskRequest"]()
    req.task_key.CopyFrom(make_task_key(qid, 1, 2))
    req.target_partition_start = 128
    req.target_partition_end = 256
    req.repartition.partitioning = make_partitioning([("k", 5)], 512)
    st, key, lo, hi, head, cols, p = decode_execute(req.SerializeToString())
    assert st == 0
    assert (key.hi, key.lo) == (int.from_bytes(qid.bytes[:8], "big"),
                                int.from_bytes(qid.bytes[8:], "big"))
    assert (lo, hi) == (128, 256)
    assert head == 2  # DD_HEAD_REPARTITION
    assert cols == [5] and p == 512


def test_decode_execute_task_none_and_broadcast_heads():
    qid = uuid.uuid4()
    req = M["ddtest.ExecuteTaskRequest"]()
    req.task_key.CopyFrom(make_task_key(qid, 0, 0))
    req.none.SetInParent()
    st, _, _, _, head, _, _ = decode_execute(req.SerializeToString())
    assert st == 0 and head == 0  # DD_HEAD_NONE
    req2 = M["ddtest.ExecuteTaskRequest"]()
    req2.task_key.CopyFrom(make_task_key(qid, 0, 1))
    req2.broadcast.output_partitions = 6
    st, _, _, _, head, _, p = decode_execute(req2.SerializeToString())
    assert st == 0 and head == 1 and p == 6  # DD_HEAD_BROADCAST


def test_decode_rejects_malformed():
    st, _, _ = decode_partitioning(b"\xff\xff\xff\xff\xff\xff")
    assert st != 0
    L = api.lib()
    key = DecodedKey()
    tc = ctypes.c_uint64()
    plan = ctypes.c_void_p()
    plen = ctypes.c_int64()
    # truncated SetPlanRequest (length-delimited field runs past the buffer)
    st = L.dd_decode_set_plan(b"\x0a\x7f\x01", ctypes.c_int64(3), ctypes.byref(key),
                              ctypes.byref(tc), ctypes.byref(plan), ctypes.byref(plen))
    assert st != 0


def test_decode_rejects_non_column_hash_expr():
    # a PhysicalExprNode with an unknown ExprType field (e.g. literal = 2): the decoder
    # must refuse rather than mis-hash
    part = M["ddtest.Partitioning"]()
    h = part.hash
    e = h.hash_expr.add()
    e.column.index = 1
    h.partition_count = 8
    blob = bytearray(part.SerializeToString())
    # surgically retag the inner column field (1, wiretype 2 -> tag 0x0a) as field 2
    # (0x12) inside the expr node to fake a non-column expr
    idx = blob.rfind(b"\x0a")
    blob[idx] = 0x12
    st, _, _ = decode_partitioning(bytes(blob))
    assert st == 6  # DD_ERR_UNSUPPORTED


def test_proto_decoder_survives_hostile_bytes():
    """Mutated and random blobs must decode to an error or bounded values — never
    crash or read out of bounds (the wire reader in dd_proto.cpp checks every
    varint/length against the buffer end)."""
    rng = __import__("numpy").random.default_rng(13)
    np = __import__("numpy")
    qid = uuid.uuid4()
    req = M["ddtest.ExecuteTaskRequest"]()
    req.task_key.CopyFrom(make_task_key(qid, 1, 2))
    req.target_partition_start = 5
    req.target_partition_end = 9
    req.repartition.partitioning = make_partitioning([("a", 1), ("b", 2)], 64)
    base = bytearray(req.SerializeToString())
    for _ in range(500):
        m = bytearray(base)
        for _ in range(rng.integers(1, 6)):
            m[rng.integers(0, len(m))] = rng.integers(0, 256)
        decode_execute(bytes(m))  # any status; must not crash
    for _ in range(200):
        noise = bytes(rng.integers(0, 256, rng.integers(0, 120), dtype=np.int64)
                      .astype(np.uint8))
        decode_execute(noise)
        decode_partitioning(noise)
