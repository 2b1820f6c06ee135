"""NetworkCoalesceExec mirror tests (CPU) — mirroring the reference's own case table
(/root/reference/src/execution_plans/network_coalesce.rs:430-500: group contiguity, full
coverage, no duplicates, padding-slot accounting, partition sizing by max group)."""

import uuid

import pytest

from datafusion_distributed_amd.coalesce import (
    EMPTY,
    NetworkCoalesceExec,
    coalesced_partition_count,
    task_group,
)
from datafusion_distributed_amd.shuffle import DistributedTaskContext, TaskKey


CASES = [  # (input_tasks, consumer_tasks) — the reference's shapes incl. uneven groups
    (1, 1), (2, 1), (3, 1), (4, 2), (5, 2), (6, 4), (7, 3), (8, 8), (9, 4),
]


@pytest.mark.parametrize("input_tasks,consumer_tasks", CASES)
def test_groups_contiguous_cover_all(input_tasks, consumer_tasks):
    seen = [False] * input_tasks
    expected_start = 0
    padding = 0
    max_group = max(-(-input_tasks // consumer_tasks), 1)
    for t in range(consumer_tasks):
        g = task_group(input_tasks, t, consumer_tasks)
        assert g.start_task == expected_start, "groups must be contiguous"
        assert g.start_task + g.len <= input_tasks
        assert g.max_len == max_group
        for k in range(g.start_task, g.start_task + g.len):
            assert not seen[k], "input task appears twice"
            seen[k] = True
        expected_start = g.start_task + g.len
        padding += g.max_len - g.len
    assert expected_start == input_tasks, "groups must cover all input tasks"
    assert all(seen)
    assert padding == consumer_tasks * max_group - input_tasks


@pytest.mark.parametrize("input_tasks,consumer_tasks", CASES)
def test_partition_sizing_by_max_group(input_tasks, consumer_tasks):
    P = 3
    out = coalesced_partition_count(P, input_tasks, consumer_tasks)
    assert out == P * max(-(-input_tasks // consumer_tasks), 1)


class RecordingChannel:
    def __init__(self, producer):
        self.producer = producer
        self.requests = []

    def execute_task(self, request):
        self.requests.append(request)
        return (self.producer, request.target_partition_start,
                request.target_partition_end)


def test_execute_mapping_even():
    channels = {}

    def factory(i):
        channels[i] = RecordingChannel(i)
        return channels[i]

    # 4 producers, 2 consumers, 3 partitions per producer -> out = 3 * 2 = 6 per consumer
    ex = NetworkCoalesceExec(3, 4, TaskKey(uuid.uuid4(), 2, 0), factory, consumer_tasks=2)
    assert ex.out_partitions == 6
    ctx = DistributedTaskContext(task_index=1, task_count=2)
    # consumer 1's group = producers [2, 4); partition 4 -> producer offset 1, partition 1
    r = ex.execute(4, ctx)
    assert r == (3, 0, 3)
    (req,) = channels[3].requests
    assert req.task_key.task_number == 3
    # ProducerHead::None: producer partitions unscaled
    assert req.producer_partitions == 3
    assert (req.target_partition_start, req.target_partition_end) == (0, 3)


def test_execute_padding_returns_empty():
    calls = []

    def factory(i):
        c = RecordingChannel(i)
        calls.append(i)
        return c

    # 3 producers, 2 consumers: groups [2, 1], max group 2 -> out = P * 2 = 4 (P=2)
    ex = NetworkCoalesceExec(2, 3, TaskKey(uuid.uuid4(), 2, 0), factory, consumer_tasks=2)
    assert ex.out_partitions == 4
    ctx1 = DistributedTaskContext(task_index=1, task_count=2)
    # consumer 1's group has len 1; partitions 2..4 map to offset 1 -> PADDING, no call
    assert ex.execute(2, ctx1) is EMPTY
    assert ex.execute(3, ctx1) is EMPTY
    assert calls == []  # padding made no network call (network_coalesce.rs:315-317)
    # real slots still work
    r = ex.execute(0, ctx1)
    assert r == (2, 0, 2)


def test_invalid_task_context():
    ex = NetworkCoalesceExec(2, 3, TaskKey(uuid.uuid4(), 2, 0), lambda i: None,
                             consumer_tasks=2)
    with pytest.raises(ValueError, match="invalid task context"):
        ex.execute(0, DistributedTaskContext(task_index=2, task_count=2))
