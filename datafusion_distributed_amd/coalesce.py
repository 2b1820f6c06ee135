"""coalesce.py — host-side mirror of NetworkCoalesceExec.

Reference: /root/reference/src/execution_plans/network_coalesce.rs — coalesces partitions
from N producer tasks into M consumer tasks WITHOUT repartitioning (ProducerHead::None,
:209-212). Mirrored semantics:

  - output partitions per consumer task are sized by the MAXIMUM producer-group size:
    out_partitions = producer_partitions * ceil(input_tasks / consumer_tasks)   (:95-103)
  - input tasks are split into contiguous groups; the first (input_tasks % consumer_tasks)
    groups get one extra task (task_group, :376-400)
  - partition p maps to (input_task_offset = p // partitions_per_task,
    target_partition = p % partitions_per_task); offsets beyond the group's real length
    are PADDING and yield an empty stream with no network call (:306-317)

The data plane reuses the shuffle transport (WorkerConnectionPool semantics: whole range
0..partitions_per_task requested per producer connection, :329-337). No kernel is involved
— this boundary moves partitions as-is.
"""

from dataclasses import dataclass

from .shuffle import DistributedTaskContext, ExecuteTaskRequest, TaskKey, WorkerConnectionPool


@dataclass(frozen=True)
class TaskGroup:
    start_task: int
    len: int
    max_len: int


def task_group(input_task_count: int, task_index: int, task_count: int) -> TaskGroup:
    """network_coalesce.rs:376-400 (contiguous groups, remainder spread over the first)."""
    if task_count == 0:
        return TaskGroup(0, 0, 0)
    base = input_task_count // task_count
    extra = input_task_count % task_count
    length = base + (1 if task_index < extra else 0)
    start = task_index * base + min(task_index, extra)
    max_len = base + (1 if extra > 0 else 0)
    return TaskGroup(start, length, max_len)


def coalesced_partition_count(producer_partitions: int, input_tasks: int,
                              consumer_tasks: int) -> int:
    """try_from_stage: out = P * ceil(input_tasks / consumer_tasks) (:95-103)."""
    max_group = max(-(-input_tasks // consumer_tasks), 1)
    return producer_partitions * max_group


EMPTY = object()  # sentinel: padding partition -> empty stream, no network call (:315-317)


class NetworkCoalesceExec:
    """Consumer-side mirror of network_coalesce.rs:337-343 execute()."""

    def __init__(self, producer_partitions: int, producer_tasks: int,
                 task_key_base: TaskKey, make_channel, consumer_tasks: int):
        self.producer_partitions = producer_partitions
        self.producer_tasks = producer_tasks
        self.consumer_tasks = consumer_tasks
        self.key_base = task_key_base
        self.pool = WorkerConnectionPool(producer_tasks, make_channel)
        self.out_partitions = coalesced_partition_count(
            producer_partitions, producer_tasks, consumer_tasks)

    def execute(self, partition: int, ctx: DistributedTaskContext):
        if ctx.task_index >= ctx.task_count:
            raise ValueError(
                f"invalid task context: task_index={ctx.task_index} >= "
                f"task_count={ctx.task_count}")
        partitions_per_task = self.out_partitions // max(
            -(-self.producer_tasks // ctx.task_count), 1)
        if partitions_per_task == 0:
            raise ValueError("NetworkCoalesceExec has 0 partitions per input task")
        group = task_group(self.producer_tasks, ctx.task_index, ctx.task_count)
        input_task_offset = partition // partitions_per_task
        target_partition = partition % partitions_per_task
        if input_task_offset >= group.len:
            return EMPTY  # padding slot (uneven grouping)
        if input_task_offset >= group.max_len:
            raise AssertionError("input_task_offset >= group.max_len")
        target_task = group.start_task + input_task_offset
        req = ExecuteTaskRequest(
            task_key=TaskKey(self.key_base.query_id, self.key_base.stage_id, target_task),
            target_partition_start=0,
            target_partition_end=partitions_per_task,
            producer_partitions=self.producer_partitions,  # ProducerHead::None: unscaled
        )
        return self.pool.execute(target_task, req, target_partition)
