"""broadcast.py — host-side mirror of BroadcastExec / NetworkBroadcastExec
(the build side of CollectLeft joins).

Reference:
  - BroadcastExec (/root/reference/src/execution_plans/broadcast.rs): N input partitions
    exposed as N*M virtual partitions (M = consumer tasks); virtual partition i returns the
    CACHED result of real partition i % N (:24-28, :163) — the input is executed exactly
    once per real partition, subsequent virtual partitions are cache hits.
  - NetworkBroadcastExec (/root/reference/src/execution_plans/network_broadcast.rs):
    consumer task t requests window off = P*t (:255-266) from EVERY producer task;
    producer head = BroadcastExec{output_partitions = P * consumer_tasks} (:176-181).

The GPU data plane is dd_broadcast_run (RCCL ncclBroadcast over xGMI, api.Comm.broadcast):
the build-side batch is replicated to every rank in one collective.
"""

from dataclasses import dataclass, field

from .shuffle import DistributedTaskContext, ExecuteTaskRequest, TaskKey, WorkerConnectionPool


@dataclass
class BroadcastExec:
    """Virtual-partition cache mirror (broadcast.rs:84-123,163)."""

    input_partition_count: int
    consumer_task_count: int
    execute_real: object = None  # callable real_partition -> result
    _cache: dict = field(default_factory=dict)
    executions: int = 0

    @property
    def output_partition_count(self):
        return self.input_partition_count * self.consumer_task_count

    def execute(self, virtual_partition: int):
        if not (0 <= virtual_partition < self.output_partition_count):
            raise ValueError("virtual partition out of range")
        real = virtual_partition % self.input_partition_count
        if real not in self._cache:
            self.executions += 1
            self._cache[real] = self.execute_real(real)
        return self._cache[real]


class NetworkBroadcastExec:
    """Consumer-side mirror (network_broadcast.rs:245-266): fan partition requests to ALL
    producer tasks over this consumer task's virtual-partition window."""

    def __init__(self, partitions: int, producer_tasks: int, task_key_base: TaskKey,
                 make_channel):
        self.partitions = partitions  # P: partitions per consumer task
        self.producer_tasks = producer_tasks
        self.key_base = task_key_base
        self.pool = WorkerConnectionPool(producer_tasks, make_channel)

    def producer_head_partitions(self, consumer_task_count: int) -> int:
        """ProducerHead::BroadcastExec{output_partitions = P * consumer_tasks}
        (network_broadcast.rs:176-181)."""
        return self.partitions * consumer_task_count

    def execute(self, partition: int, ctx: DistributedTaskContext):
        if not (0 <= partition < self.partitions):
            raise ValueError("partition out of range")
        off = self.partitions * ctx.task_index
        results = []
        for producer in range(self.producer_tasks):
            req = ExecuteTaskRequest(
                task_key=TaskKey(self.key_base.query_id, self.key_base.stage_id, producer),
                target_partition_start=off,
                target_partition_end=off + self.partitions,
                producer_partitions=self.producer_head_partitions(ctx.task_count),
            )
            results.append(self.pool.execute(producer, req, off + partition))
        return results
