"""Host-mirror logic tests (CPU): partition-window math, producer-head scaling, connection
pool exactly-once semantics, shuffle fan-in shape — mirroring the reference's own unit-test
targets (network_shuffle.rs window math :221-251; worker_connection_pool.rs:31-32,128-137)."""

import uuid

import pytest

from datafusion_distributed_amd import (
    DistributedTaskContext,
    ExecuteTaskRequest,
    NetworkShuffleExec,
    TaskKey,
    WorkerConnectionPool,
    partition_window,
    scale_partitioning,
)


def test_scale_partitioning():
    # producer_head: P_total = P * consumer_tasks (network_shuffle.rs:158-165)
    assert scale_partitioning(6, 1) == 6
    assert scale_partitioning(6, 2) == 12
    assert scale_partitioning(3, 3) == 9
    with pytest.raises(ValueError):
        scale_partitioning(0, 1)


def test_partition_window():
    # off = P * task_index (network_shuffle.rs:232-244)
    assert partition_window(0, 6) == (0, 6)
    assert partition_window(1, 6) == (6, 12)
    assert partition_window(2, 3) == (6, 9)


def test_windows_tile_the_partition_space():
    # total partitions across consumer tasks == one producer task's P_total
    # (network_shuffle.rs:92-96 invariant)
    P, tasks = 5, 4
    covered = []
    for t in range(tasks):
        lo, hi = partition_window(t, P)
        covered.extend(range(lo, hi))
    assert covered == list(range(scale_partitioning(P, tasks)))


class RecordingChannel:
    def __init__(self, producer):
        self.producer = producer
        self.requests = []

    def execute_task(self, request):
        self.requests.append(request)
        return (self.producer, request.target_partition_start,
                request.target_partition_end)


def test_pool_exactly_once():
    made = {}

    def factory(i):
        made[i] = RecordingChannel(i)
        return made[i]

    pool = WorkerConnectionPool(2, factory)
    key = TaskKey(uuid.uuid4(), 1, 0)
    req = ExecuteTaskRequest(key, 0, 3, 3)
    pool.execute(0, req, 1)
    # same slot again -> exactly-once violation
    with pytest.raises(RuntimeError, match="already consumed"):
        pool.execute(0, req, 1)
    # different partition, same producer: ok, connection reused (lazy, one per producer)
    pool.execute(0, req, 2)
    assert list(made) == [0]
    # out-of-range partition rejected
    with pytest.raises(ValueError):
        pool.execute(1, req, 7)


def test_network_shuffle_fan_in():
    channels = {}

    def factory(i):
        channels[i] = RecordingChannel(i)
        return channels[i]

    P, producers = 4, 3
    ex = NetworkShuffleExec(P, producers, TaskKey(uuid.uuid4(), 2, 0), factory)
    ctx = DistributedTaskContext(task_index=1, task_count=2)
    results = ex.execute(0, ctx)
    # one result per producer task (fan-in, network_shuffle.rs:235-245)
    assert len(results) == producers
    # every producer got the same window request: task 1's window is [4, 8)
    for i, ch in channels.items():
        (req,) = ch.requests
        assert (req.target_partition_start, req.target_partition_end) == (4, 8)
        # producer head scaled by consumer task count: P_total = 4*2
        assert req.producer_partitions == 8
        assert req.task_key.task_number == i
    # partition out of window range rejected
    with pytest.raises(ValueError):
        ex.execute(P, ctx)


def test_task_key_to_c_roundtrip():
    q = uuid.uuid4()
    k = TaskKey(q, 7, 3).to_c()
    assert (int(k.query_id_hi) << 64) | int(k.query_id_lo) == q.int
    assert k.stage_id == 7 and k.task_number == 3
