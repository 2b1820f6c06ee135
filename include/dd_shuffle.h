/* include/dd_shuffle.h — C ABI of the MI355X-native shuffle/exchange path.
 *
 * This is the drop-in boundary: the exports mirror, 1:1, the reference's
 * transport/operator seam for the hash-shuffle hot path so that a one-file Rust shim
 * implementing `WorkerChannel` (reference: src/protocol/worker_channel.rs:19-46, resolved via
 * src/protocol/channel_resolver.rs:27-41) can delegate to these functions wherever cargo
 * exists. The reference-side binding a maintainer would add is shown in INTEGRATION.md.
 *
 * No torch types anywhere; plain pointers and sizes. All device pointers are HIP device
 * memory on the current device. All entry points return dd_status; dd_last_error() gives a
 * human-readable message (mirrors the reference's DataFusionError <-> tonic Status mapping,
 * src/protocol/grpc/errors/).
 */

#ifndef DD_SHUFFLE_H
#define DD_SHUFFLE_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------------- status / errors ---------------- */

typedef enum dd_status {
    DD_OK = 0,
    DD_ERR_INVALID = 1,     /* bad arguments / unsupported shape */
    DD_ERR_NO_DEVICE = 2,   /* no HIP device — the product path FAILS here, no CPU fallback */
    DD_ERR_HIP = 3,         /* HIP runtime error */
    DD_ERR_RCCL = 4,        /* RCCL error */
    DD_ERR_NOT_FOUND = 5,   /* unknown TaskKey (mirrors worker "plan not found" timeout,
                               src/worker/impl_execute_task.rs:29-34) */
    DD_ERR_UNSUPPORTED = 6  /* dtype/partition-count combination outside round-1 coverage */
} dd_status;

const char *dd_last_error(void);
int dd_device_count(void); /* number of visible HIP devices (0 on a GPU-less box) */
const char *dd_version(void);

/* ---------------- data model ----------------
 * Device-resident Arrow-layout columns. Validity is UNPACKED u8 (1=valid).
 * dtype codes shared with oracle/dd_oracle.c. */

typedef enum dd_dtype {
    DD_DT_U8 = 1,
    DD_DT_I16 = 2,
    DD_DT_I32 = 3,
    DD_DT_I64 = 4,
    DD_DT_F32 = 5,
    DD_DT_F64 = 6,
    DD_DT_BOOL = 7,  /* unpacked u8 0/1 */
    DD_DT_UTF8 = 8,  /* i32 offsets[n+1] + byte buffer */
    DD_DT_DICT32 = 9 /* i32 indices; dictionary values as utf8 (offsets+bytes) */
} dd_dtype;

#define DD_MAX_COLS 24
#define DD_MAX_KEYS 8
#define DD_MAX_PARTITIONS 2048u /* round-1 cap (LDS budget), DESIGN.md §5 */

typedef struct dd_col_desc {
    int32_t dtype;            /* dd_dtype */
    const void *data;         /* device: values, or utf8 bytes */
    const uint8_t *validity;  /* device, unpacked u8, NULL => all valid */
    const int32_t *offsets;   /* device, UTF8 only: offsets[n_rows+1] */
    int64_t data_len;         /* UTF8: byte length of `data`; else 0 */
    const void *dict_bytes;   /* DICT32: device value bytes */
    const int32_t *dict_offsets; /* DICT32: device offsets[dict_n+1] */
    int64_t dict_n;           /* DICT32: number of dictionary values */
} dd_col_desc;

typedef struct dd_batch_desc {
    int64_t n_rows;
    int32_t n_cols;
    dd_col_desc cols[DD_MAX_COLS];
} dd_batch_desc;

/* ---------------- hash-repartition (the producer head) ----------------
 * Replaces the `RepartitionExec(Hash(keys, P_total))` head the reference inserts on the
 * producer: ProducerHead::insert (src/distributed_planner/network_boundary.rs:86-106),
 * required input shape at src/execution_plans/network_shuffle.rs:121-127, lazily re-created
 * per task at src/worker/task_data.rs:104-116. Semantics: DESIGN.md §3 (normative; stable).
 *
 * Two-phase (prepare/execute) so repeated executions reuse workspace + output buffers —
 * mirroring the worker's lazy plan-head caching. All output buffers are partition-major
 * contiguous. */

typedef struct dd_partitioner dd_partitioner; /* opaque */

dd_status dd_partitioner_create(const dd_batch_desc *batch, const int32_t *key_cols,
                                int32_t n_keys, uint32_t n_partitions,
                                dd_partitioner **out);
/* Runs K1 (hash+count), K2 (scan), K3 (stable scatter) on `stream` (a hipStream_t, or NULL
 * for the default stream). Asynchronous; results valid after stream sync. */
dd_status dd_partitioner_run(dd_partitioner *p, void *stream);
/* coarse-bucket variant: pid = (h % pid_total) / coarse_div — pid_total/coarse_div
 * CONTIGUOUS ranges of the final partition space (pass A of the two-level var scatter,
 * DESIGN.md §11; both arguments must be powers of two) */
dd_status dd_partitioner_create_ranged(const dd_batch_desc *batch, const int32_t *key_cols,
                                       int32_t n_keys, uint32_t pid_total,
                                       uint32_t coarse_div, dd_partitioner **out);
/* Phase split for batch pipelining (the reference streams batches through
 * RepartitionExec continuously; overlapping batch s+1's hash/count with batch s's
 * scatter mirrors that): phase1 = K1+K2 (+byte scans), phase2 = K3 (+K4). phase2 must be
 * ordered after phase1 of the SAME partitioner (same stream, or an event). */
dd_status dd_partitioner_run_phase1(dd_partitioner *p, void *stream);
dd_status dd_partitioner_run_phase2(dd_partitioner *p, void *stream);
/* make `stream` wait for this partitioner's last phase1 / phase2 completion (for
 * cross-stream batch pipelining) */
dd_status dd_partitioner_wait_phase1(dd_partitioner *p, void *stream);
dd_status dd_partitioner_wait_phase2(dd_partitioner *p, void *stream);
void dd_partitioner_destroy(dd_partitioner *p);

/* result accessors (pointers are device memory owned by the partitioner) */
/* u32[n_rows] of per-row partition ids. NULL ONLY when the opt-in DD_RHASH=1 recompute
 * path is active (all-fixed no-validity batches with integer keys, measured slower and
 * OFF by default — DESIGN.md §9): shim authors must handle NULL for robustness but will
 * not see it under default configuration; partition membership is always fully defined
 * by col_data + row_offsets either way. */
const uint32_t *dd_partitioner_pids(const dd_partitioner *p);
/* element size of the pid array: 4 (u32) normally, 1 (u8) when the precomputed-layout
 * path stores byte pids (n_partitions <= 256) — cast the pids pointer accordingly */
int32_t dd_partitioner_pid_elem(const dd_partitioner *p);
const void *dd_partitioner_col_data(const dd_partitioner *p, int32_t col); /* partition-major */
const uint8_t *dd_partitioner_col_validity(const dd_partitioner *p, int32_t col);
const uint32_t *dd_partitioner_col_lengths(const dd_partitioner *p, int32_t col); /* utf8 */
/* copies part_row_offsets[P+1] (rows) to host */
dd_status dd_partitioner_row_offsets(const dd_partitioner *p, int64_t *host_out);
/* staged-var path only: device pointer to the rebuilt 64-bit Arrow offsets [n_rows+1]
 * of a var column (K4c output); NULL on the v1 path */
const uint64_t *dd_partitioner_var_offsets64(const dd_partitioner *p, int32_t col);
/* out32[i] = (int32)(off64[lo_row + i] - off64[lo_row]) for i in [0, n]; device buffers;
 * lets a bucket/chunk of a var column be re-consumed as an Arrow batch view */
dd_status dd_make_offsets32(const uint64_t *off64, int64_t lo_row, int64_t n,
                            int32_t *out32, void *stream);
/* copies per-var-col part_byte_offsets[P+1] to host */
dd_status dd_partitioner_byte_offsets(const dd_partitioner *p, int32_t col, int64_t *host_out);
/* per-kernel last-run durations in ms, measured with hipEvents on the run stream:
 * [0]=K1 hash+count, [1]=K2 scan, [2]=K3 scatter. Valid after a synced run. */
dd_status dd_partitioner_kernel_ms(const dd_partitioner *p, float out_ms[3]);

/* ---------------- exchange (the transport data plane) ----------------
 * Replaces the Arrow-Flight-over-gRPC data plane (client demux src/protocol/grpc/
 * worker_client.rs:89-320; server encode+tagging src/protocol/grpc/worker_service.rs:133-184,
 * 363-433) with an RCCL all-to-all-v over xGMI: one rank per GPU, rank = task
 * (TaskKey.task_number, src/stage.rs:134-139). Consumer rank r owns partition window
 * [P*r, P*(r+1)) of P_total = P*nranks (src/execution_plans/network_shuffle.rs:232-244,
 * scale_partitioning src/execution_plans/common.rs:18-30). Received data is concatenated
 * in producer-rank order (deterministic refinement of the reference's select_all merge). */

typedef struct dd_comm dd_comm; /* opaque: RCCL communicator + streams */

#define DD_UNIQUE_ID_BYTES 128 /* == NCCL_UNIQUE_ID_BYTES */
dd_status dd_comm_unique_id(void *bytes128); /* rank 0 calls; share out-of-band */
dd_status dd_comm_init(const void *bytes128, int rank, int nranks, dd_comm **out);
void dd_comm_destroy(dd_comm *c);

typedef struct dd_exchanged dd_exchanged; /* opaque: received window buffers */

/* Exchange the partitioned result: every rank sends partition window [P*j, P*(j+1)) of its
 * local result to rank j and receives its own window from every rank. P (= partitions per
 * consumer) is inferred: partitioner P_total must equal P*nranks. Synchronous on `stream`. */
dd_status dd_exchange_run(dd_comm *c, const dd_partitioner *p, void *stream,
                          dd_exchanged **out);
void dd_exchanged_destroy(dd_exchanged *e);

/* accessors: my window, producer-major then partition-major per producer */
int64_t dd_exchanged_total_rows(const dd_exchanged *e);
const void *dd_exchanged_col_data(const dd_exchanged *e, int32_t col);
const uint8_t *dd_exchanged_col_validity(const dd_exchanged *e, int32_t col);
const uint32_t *dd_exchanged_col_lengths(const dd_exchanged *e, int32_t col);
/* row counts per (producer, local partition): host_out[nranks*P] */
dd_status dd_exchanged_row_counts(const dd_exchanged *e, int64_t *host_out);
/* bytes per (producer, local partition) for a var col: host_out[nranks*P] */
dd_status dd_exchanged_byte_counts(const dd_exchanged *e, int32_t col, int64_t *host_out);
/* last exchange wall time (ms, hipEvent on the exchange stream) and payload bytes sent by
 * this rank to OTHER ranks (xGMI egress) */
dd_status dd_exchanged_stats(const dd_exchanged *e, float *ms, int64_t *egress_bytes);

/* ---------------- coalesce (N->M task coalesce, no repartition) ----------------
 * Data plane for NetworkCoalesceExec (src/execution_plans/network_coalesce.rs:24-70):
 * consumer rank t receives the WHOLE partitioned output of every producer rank in its
 * contiguous group (task_group, :376-400; groups = ceil-split of nranks over
 * consumer_tasks). No repartition head (ProducerHead::None, :209-212). Reuses the
 * dd_exchanged result shape: producer-major concatenation, per-(producer, partition)
 * row counts. Ranks outside any group's producer set still participate (empty sends). */
dd_status dd_coalesce_run(dd_comm *c, const dd_partitioner *p, int32_t consumer_tasks,
                          void *stream, dd_exchanged **out);

/* ---------------- broadcast (build side of CollectLeft joins) ----------------
 * Replaces BroadcastExec's cache + NetworkBroadcastExec's per-consumer fetch
 * (src/execution_plans/broadcast.rs:24-28,163; network_broadcast.rs:245-266) with one
 * RCCL ncclBroadcast over xGMI: the root rank's device batch is replicated on every rank.
 * A size header is broadcast first so non-root ranks can allocate. */

typedef struct dd_bcast dd_bcast; /* opaque: replicated batch */

/* batch: the root's device batch (ignored on other ranks, may be NULL there) */
dd_status dd_broadcast_run(dd_comm *c, const dd_batch_desc *batch, int root, void *stream,
                           dd_bcast **out);
void dd_bcast_destroy(dd_bcast *b);
int64_t dd_bcast_n_rows(const dd_bcast *b);
int32_t dd_bcast_n_cols(const dd_bcast *b);
const void *dd_bcast_col_data(const dd_bcast *b, int32_t col);
const uint8_t *dd_bcast_col_validity(const dd_bcast *b, int32_t col);
const int32_t *dd_bcast_col_offsets(const dd_bcast *b, int32_t col);
dd_status dd_bcast_col_meta(const dd_bcast *b, int32_t col, int32_t *dtype,
                            int64_t *data_len);

/* ---------------- task cache (worker execute path) ----------------
 * Mirrors SetPlanRequest / ExecuteTaskRequest (src/protocol/worker_channel.rs:74-93,163-176)
 * and the worker's TaskData cache (src/worker/task_data.rs:16-29,104-116;
 * src/worker/worker_service.rs:12,31 — the moka TTI cache becomes an explicit
 * set/execute/drop lifecycle). The "plan" payload here is the shuffle-path plan descriptor:
 * a device batch + key columns + partitioning, i.e. exactly the producer head the reference
 * ships for this path. */

typedef struct dd_task_key {
    uint64_t query_id_hi, query_id_lo; /* Uuid (src/protocol/worker_channel.rs:57-65) */
    uint64_t stage_id;
    uint64_t task_number;
} dd_task_key;

dd_status dd_set_plan(const dd_task_key *key, const dd_batch_desc *batch,
                      const int32_t *key_cols, int32_t n_keys, uint32_t n_partitions);
/* Executes (or reuses) the cached task's partitioner and returns it; partition range
 * [part_lo, part_hi) mirrors ExecuteTaskRequest.target_partition_start/end. The returned
 * partitioner is owned by the cache; do not destroy. Lifetime: concurrent
 * set_plan/execute/drop calls on the same key are safe (entries are refcounted and an
 * in-flight execute pins its entry), but the pointer *out refers to the cache entry as
 * of this call — it is invalidated by a later dd_drop_task or dd_set_plan on the same
 * key, so callers must not use it past either (the reference's worker has the same
 * contract: task state lives until the TTI cache evicts it, task_data.rs:16-29). */
dd_status dd_execute_task(const dd_task_key *key, uint32_t part_lo, uint32_t part_hi,
                          void *stream, dd_partitioner **out);
dd_status dd_drop_task(const dd_task_key *key); /* task cleanup (stateful_data_cleanup) */

/* ---------------- protobuf plan/stage wire payload (dd_proto.cpp) ----------------
 * The reference's own wire shapes at this boundary (field numbers cited in
 * dd_proto.cpp): SetPlanRequest / ExecuteTaskRequest / TaskKey
 * (src/protocol/grpc/worker.proto:84-110,134-155,171-179) with the producer_head's
 * RepartitionExecHead carrying a datafusion-proto `Partitioning` message
 * (worker.proto:167-170; datafusion-proto 55.0.0, Cargo.lock pin). A Rust shim passes
 * the prost-encoded request bytes straight through — no re-encoding. */

enum { DD_HEAD_NONE = 0, DD_HEAD_BROADCAST = 1, DD_HEAD_REPARTITION = 2 };

/* decode a datafusion-proto Partitioning (hash exprs must be columns) */
dd_status dd_decode_partitioning(const uint8_t *buf, int64_t len, int32_t *key_cols,
                                 int32_t max_keys, int32_t *n_keys,
                                 uint32_t *n_partitions);
/* decode an ExecuteTaskRequest: task key, partition range, producer head (repartition
 * heads also fill key_cols/n_keys/n_partitions; broadcast fills n_partitions) */
dd_status dd_decode_execute_task(const uint8_t *buf, int64_t len, dd_task_key *key,
                                 uint64_t *part_start, uint64_t *part_end,
                                 int32_t *head_kind, int32_t *key_cols, int32_t max_keys,
                                 int32_t *n_keys, uint32_t *n_partitions);
/* decode a SetPlanRequest; *plan_proto is a borrowed view into buf */
dd_status dd_decode_set_plan(const uint8_t *buf, int64_t len, dd_task_key *key,
                             uint64_t *task_count, const uint8_t **plan_proto,
                             int64_t *plan_len);
/* register a plan under the TaskKey decoded from a prost-encoded SetPlanRequest; the
 * device batch is what the subplan below the network boundary produces (the shim
 * materializes it — the plan_proto subplan itself stays host-side per the tier) */
dd_status dd_set_plan_proto(const uint8_t *set_plan_pb, int64_t len,
                            const dd_batch_desc *batch);
/* execute from a prost-encoded ExecuteTaskRequest: decodes the RepartitionExecHead,
 * lazily inserts the head (TaskData::plan / ProducerHead::insert semantics) and runs
 * the partition kernels; same returned-pointer lifetime as dd_execute_task */
dd_status dd_execute_task_proto(const uint8_t *execute_task_pb, int64_t len, void *stream,
                                dd_partitioner **out);
dd_status dd_drop_task_proto(const uint8_t *task_key_pb, int64_t len);

/* ---------------- Arrow IPC + lz4 wire format (dd_wire.cpp) ----------------
 * The on-wire batch encoding for cross-node (non-xGMI) hops, reimplementing the
 * reference's Flight encode/decode (src/protocol/grpc/worker_service.rs:363-433 /
 * worker_client.rs:302): Apache Arrow IPC STREAMING format, MetadataVersion V5, with
 * per-buffer lz4-frame BodyCompression (the reference's default,
 * distributed_config.rs:36-38). Zero-column batches (row count, no fields) round-trip —
 * the edge case the reference wire-tests pin (tests/empty_columns_between_workers.rs).
 * Validity and bool data are unpacked u8 at this ABI (packed to Arrow bitmaps on the
 * wire). Host memory only; this is the serialization layer, not the device path. */

typedef struct dd_ipc_field {
    int32_t dtype;    /* dd_dtype (DICT32 unsupported on the wire: indices + values
                         cross via arrow_boundary materialization) */
    const char *name; /* optional */
    int32_t nullable;
} dd_ipc_field;

typedef struct dd_ipc_array {
    const void *data;        /* fixed: values (bool: unpacked u8); utf8: byte buffer */
    int64_t data_len;        /* utf8: byte length; reader fills for fixed too */
    const uint8_t *validity; /* unpacked u8, NULL = all valid */
    int64_t null_count;
    const int32_t *offsets;  /* utf8: [n_rows+1] */
} dd_ipc_array;

typedef struct dd_ipc_writer dd_ipc_writer;
typedef struct dd_ipc_reader dd_ipc_reader;

dd_status dd_ipc_writer_create(const dd_ipc_field *fields, int32_t n_fields,
                               int32_t use_lz4, dd_ipc_writer **out);
dd_status dd_ipc_writer_batch(dd_ipc_writer *w, int64_t n_rows, const dd_ipc_array *cols);
/* appends the EOS marker (idempotent) and exposes the stream; owned by the writer */
dd_status dd_ipc_writer_finish(dd_ipc_writer *w, const uint8_t **data, int64_t *len);
void dd_ipc_writer_destroy(dd_ipc_writer *w);

dd_status dd_ipc_reader_create(const uint8_t *data, int64_t len, dd_ipc_reader **out);
int32_t dd_ipc_reader_n_fields(const dd_ipc_reader *r);
int32_t dd_ipc_reader_n_batches(const dd_ipc_reader *r);
int32_t dd_ipc_reader_field_dtype(const dd_ipc_reader *r, int32_t i);
const char *dd_ipc_reader_field_name(const dd_ipc_reader *r, int32_t i);
int64_t dd_ipc_reader_batch_rows(const dd_ipc_reader *r, int32_t b);
/* pointers into reader-owned memory (decompressed; validity/bool unpacked to u8) */
dd_status dd_ipc_reader_batch_col(const dd_ipc_reader *r, int32_t b, int32_t c,
                                  dd_ipc_array *out);
void dd_ipc_reader_destroy(dd_ipc_reader *r);

/* ---------------- partial aggregation (below the shuffle) ----------------
 * Mirrors the partial-reduce pass (src/distributed_planner/
 * partial_reduce_below_network_shuffles.rs; `distributed.partial_reduce`,
 * distributed_config.rs:50-54): a mode=Partial aggregate run before dd_partition so the
 * exchange moves per-group partials. May emit duplicate groups (one per block); the
 * downstream final aggregate merges them. Fixed-width keys (<=4), aggregate ops below. */

typedef struct dd_reducer dd_reducer;

#define DD_AGG_SUM_F64 0 /* sum of an f64 column (nulls skipped) */
#define DD_AGG_COUNT 1   /* count(*) (agg_col ignored) */
#define DD_AGG_SUM_I64 2 /* sum of an i64 column (nulls skipped) */
#define DD_AGG_MIN_F64 3 /* min/max skip nulls; NaN sorts above +inf (Arrow semantics); */
#define DD_AGG_MAX_F64 4 /* an all-null group yields the identity (0 / +-extreme), not   */
#define DD_AGG_MIN_I64 5 /* SQL NULL — pair with count(col) downstream to reconstruct    */
#define DD_AGG_MAX_I64 6 /* null results */

dd_status dd_partial_reduce_run(const dd_batch_desc *batch, const int32_t *key_cols,
                                int32_t n_keys, const int32_t *agg_cols,
                                const int32_t *agg_ops, int32_t n_aggs, void *stream,
                                dd_reducer **out);
int64_t dd_reducer_n_rows(const dd_reducer *r);
float dd_reducer_kernel_ms(const dd_reducer *r); /* reduce-kernel time of the last run */
/* host_keys[n][n_keys] canonical 64-bit key bits; host_keynull[n] per-key null bitmask;
 * host_aggs[n][n_aggs] (i64 sums / counts bit-cast into the double slot) */
dd_status dd_reducer_fetch(const dd_reducer *r, uint64_t *host_keys, uint32_t *host_keynull,
                           double *host_aggs);
/* host_nn[n][n_aggs] non-null input counts per aggregate: the final merge sums them and
 * emits NULL when the total is 0 (DataFusion: SUM/MIN/MAX over an all-null group is NULL,
 * not the op identity; COUNT counts rows and is never NULL) */
dd_status dd_reducer_fetch_nn(const dd_reducer *r, uint64_t *host_nn);
void dd_reducer_destroy(dd_reducer *r);

/* ---------------- device helpers (harness convenience; not part of the seam) ------- */
dd_status dd_dev_alloc(int64_t bytes, void **out);
dd_status dd_dev_free(void *p);
dd_status dd_memcpy_h2d(void *dst, const void *src, int64_t bytes);
dd_status dd_memcpy_d2h(void *dst, const void *src, int64_t bytes);
dd_status dd_memcpy_d2d(void *dst, const void *src, int64_t bytes);
dd_status dd_device_sync(void);

#ifdef __cplusplus
}
#endif

#endif /* DD_SHUFFLE_H */
