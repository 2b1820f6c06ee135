"""BroadcastExec / NetworkBroadcastExec mirror tests (CPU), mirroring the reference's own
virtual-partition cache tests (/root/reference/src/execution_plans/broadcast.rs:341,464:
virtual partitions map to real partition i % N and reuse the cached result)."""

import uuid

import pytest

from datafusion_distributed_amd.broadcast import BroadcastExec, NetworkBroadcastExec
from datafusion_distributed_amd.shuffle import DistributedTaskContext, TaskKey


def test_virtual_partition_cache():
    ex = BroadcastExec(input_partition_count=3, consumer_task_count=4,
                       execute_real=lambda p: f"part{p}")
    assert ex.output_partition_count == 12
    # virtual i -> real i % 3, executed exactly once per real partition
    assert ex.execute(0) == "part0"
    assert ex.execute(3) == "part0"   # cache hit
    assert ex.execute(7) == "part1"
    assert ex.execute(11) == "part2"
    assert ex.executions == 3
    with pytest.raises(ValueError):
        ex.execute(12)


class RecordingChannel:
    def __init__(self, producer):
        self.producer = producer
        self.requests = []

    def execute_task(self, request):
        self.requests.append(request)
        return (self.producer, request.target_partition_start,
                request.target_partition_end)


def test_network_broadcast_fans_to_all_producers():
    channels = {}

    def factory(i):
        channels[i] = RecordingChannel(i)
        return channels[i]

    P, producers = 3, 2
    ex = NetworkBroadcastExec(P, producers, TaskKey(uuid.uuid4(), 4, 0), factory)
    ctx = DistributedTaskContext(task_index=2, task_count=4)
    results = ex.execute(1, ctx)
    assert len(results) == producers  # every producer queried (network_broadcast.rs:258-266)
    for i, ch in channels.items():
        (req,) = ch.requests
        # window off = P * task_index
        assert (req.target_partition_start, req.target_partition_end) == (6, 9)
        # producer head: BroadcastExec{output_partitions = P * consumer_tasks}
        assert req.producer_partitions == 12
