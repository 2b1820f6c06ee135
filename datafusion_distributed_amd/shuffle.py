"""shuffle.py — host-side mirror of the reference's shuffle operator seam.

Mirrors, for the hot path only:
  - TaskKey            src/protocol/worker_channel.rs:57-65
  - DistributedTaskContext (task_index, task_count)   src/stage.rs:209-224
  - scale_partitioning P_total = P * consumer_tasks   src/execution_plans/common.rs:18-30,
                                                      network_shuffle.rs:158-165
  - the consumer partition window off = P*task_index  network_shuffle.rs:221-251
  - WorkerConnectionPool: lazy one-connection-per-producer-task, whole partition range per
    connection, exactly-once stream slots               src/worker/worker_connection_pool.rs:19-44,
                                                        :31-32,128-137
  - NetworkShuffleExec.execute fan-in                  network_shuffle.rs:221-251

The data plane behind the seam is the C ABI (api.py -> libdd_shuffle.so). The in-process
channel mirrors InProcessWorkerClient (src/protocol/in_process/worker_client.rs:16-53):
zero-serde loopback straight into the worker task cache (dd_set_plan / dd_execute_task).
"""

import uuid
from dataclasses import dataclass, field


@dataclass(frozen=True)
class TaskKey:
    """(query_id, stage_id, task_number) — worker_channel.rs:57-65."""
    query_id: uuid.UUID
    stage_id: int
    task_number: int

    def to_c(self):
        from .api import TaskKeyC
        hi = self.query_id.int >> 64
        lo = self.query_id.int & ((1 << 64) - 1)
        return TaskKeyC(hi, lo, self.stage_id, self.task_number)


@dataclass(frozen=True)
class DistributedTaskContext:
    """stage.rs:209-224."""
    task_index: int
    task_count: int


def scale_partitioning(partitions: int, consumer_tasks: int) -> int:
    """producer_head scaling: P_total = P * consumer_tasks (common.rs:18-30)."""
    if partitions < 1 or consumer_tasks < 1:
        raise ValueError("partitions and consumer_tasks must be >= 1")
    return partitions * consumer_tasks


def partition_window(task_index: int, partitions: int):
    """Consumer task's partition range off..off+P (network_shuffle.rs:232-244)."""
    off = partitions * task_index
    return off, off + partitions


@dataclass
class ExecuteTaskRequest:
    """worker_channel.rs:163-176."""
    task_key: TaskKey
    target_partition_start: int
    target_partition_end: int
    producer_partitions: int  # ProducerHead::RepartitionExec partitioning count (P_total)


class WorkerChannel:
    """Transport seam (worker_channel.rs:19-46, execute_task only for this path)."""

    def execute_task(self, request: ExecuteTaskRequest):
        raise NotImplementedError


class InProcessGpuChannel(WorkerChannel):
    """InProcessWorkerClient mirror: loopback into the worker task cache on this GPU."""

    def execute_task(self, request: ExecuteTaskRequest):
        import ctypes

        from .api import Partitioner, _check, lib

        key = request.task_key.to_c()
        h = ctypes.c_void_p()
        _check(lib().dd_execute_task(
            ctypes.byref(key), request.target_partition_start,
            request.target_partition_end, None, ctypes.byref(h)))
        p = Partitioner.__new__(Partitioner)
        p.h = h
        p.nparts = request.producer_partitions
        p.batch = None
        return p


@dataclass
class WorkerConnectionPool:
    """Mirror of worker_connection_pool.rs: one lazy connection per producer task; the
    first caller requests the WHOLE partition range; each (producer, partition) stream
    slot is consumed exactly once (:31-32, 128-137)."""

    producer_tasks: int
    make_channel: object = None  # callable producer_index -> WorkerChannel
    _channels: dict = field(default_factory=dict)
    _consumed: set = field(default_factory=set)

    def execute(self, producer_index: int, request: ExecuteTaskRequest, partition: int):
        if not (request.target_partition_start <= partition < request.target_partition_end):
            raise ValueError("partition outside the requested range")
        slot = (producer_index, partition)
        if slot in self._consumed:
            raise RuntimeError(
                f"stream slot {slot} already consumed (exactly-once: "
                "worker_connection_pool.rs:128-137)")
        self._consumed.add(slot)
        if producer_index not in self._channels:
            if self.make_channel is None:
                raise RuntimeError("no channel factory")
            self._channels[producer_index] = self.make_channel(producer_index)
        return self._channels[producer_index].execute_task(request)


class NetworkShuffleExec:
    """Consumer-side mirror (network_shuffle.rs:102-108, 221-251): fan-in of one partition
    from every producer task, over this consumer task's partition window."""

    def __init__(self, partitions: int, producer_tasks: int, task_key_base: TaskKey,
                 make_channel):
        self.partitions = partitions  # P: partitions per consumer task
        self.producer_tasks = producer_tasks
        self.key_base = task_key_base
        self.pool = WorkerConnectionPool(producer_tasks, make_channel)

    def producer_head_partitions(self, consumer_task_count: int) -> int:
        return scale_partitioning(self.partitions, consumer_task_count)

    def execute(self, partition: int, ctx: DistributedTaskContext):
        """Returns one result handle per producer task for `partition` within this
        task's window (caller merges; producer order is deterministic — DESIGN.md §9)."""
        if not (0 <= partition < self.partitions):
            raise ValueError("partition out of range")
        lo, hi = partition_window(ctx.task_index, self.partitions)
        results = []
        for producer in range(self.producer_tasks):
            req = ExecuteTaskRequest(
                task_key=TaskKey(self.key_base.query_id, self.key_base.stage_id, producer),
                target_partition_start=lo,
                target_partition_end=hi,
                producer_partitions=self.producer_head_partitions(ctx.task_count),
            )
            results.append(self.pool.execute(producer, req, lo + partition))
        return results
