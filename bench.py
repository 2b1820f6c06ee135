"""bench.py — measures the BASELINE.json metric: TPC-H SF10 shuffle GB/s + rows/s on
1/2/4/8 MI355X, with the HBM roofline of the dominant kernel (K3 scatter).

Workload: synthetic TPC-H SF10 lineitem shuffle (q3/q5 shape) —
59,986,052 rows per rank (weak scaling), hash key l_orderkey (int64), payload = q3
projection (l_orderkey i64, l_extendedprice f64, l_discount f64, l_shipdate date32) =
28 B/row. One "step" = one full hash-repartition pass over the HBM-resident input
(+ the RCCL all-to-all-v exchange at N>1). P_total = 128 * n_gpus.

value = whole-job GB/s of input bytes (rows x 28 x n_gpus / time). dtype "int64" is the
arithmetic type of the hash/scatter path, not a precision claim. Inputs are resident in
HBM when the timed region starts; the PCIe-inclusive staging rate is reported in
config.h2d_GBps (never as value).

cpu_baseline: the C oracle (oracle/dd_oracle.c — same algorithm; kind "port"; the
reference itself is Rust and unbuildable here) timed on this box's host
cores on a bounded sample. Rank 0, N=1 only.

--workload selects additional BASELINE config shapes for extra evidence lines
(tpch_sf1_q1_repartition = config[1]; clickbench_userid_shuffle = config[4] stand-in);
the driver's contract lines always use the default tpch_sf10_lineitem_shuffle.
"""

import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

SF10_ROWS = 59_986_052
PARTS_PER_RANK = 128
HBM_PEAK = 8.0e12

FIXED_SIZE = {"u8": 1, "bool": 1, "i16": 2, "i32": 4, "f32": 4, "i64": 8, "f64": 8,
              "dict32": 4}


def make_lineitem_q3(rows, seed):
    """Default workload: TPC-H SF10 lineitem q3/q5 shuffle shape (28 B/row, i64 key)."""
    rng = np.random.default_rng(seed)
    return [
        {"dtype": "i64", "data": rng.integers(1, 60_000_000, rows, dtype=np.int64),
         "valid": None},
        {"dtype": "f64", "data": rng.uniform(900.0, 105000.0, rows), "valid": None},
        {"dtype": "f64", "data": rng.uniform(0.0, 0.1, rows), "valid": None},
        {"dtype": "i32", "data": rng.integers(8000, 11000, rows, dtype=np.int64)
         .astype(np.int32), "valid": None},
    ], [0]


def make_lineitem_q1(rows, seed):
    """BASELINE config[1] shape: TPC-H SF1 q1 — repartition on the low-cardinality
    (l_returnflag, l_linestatus) pair; q1 aggregate projection payload."""
    rng = np.random.default_rng(seed)
    cols = [
        {"dtype": "u8", "data": rng.integers(0, 3, rows, dtype=np.int64).astype(np.uint8),
         "valid": None},  # l_returnflag in {A,N,R}
        {"dtype": "u8", "data": rng.integers(0, 2, rows, dtype=np.int64).astype(np.uint8),
         "valid": None},  # l_linestatus in {F,O}
        {"dtype": "f64", "data": rng.uniform(1.0, 50.0, rows), "valid": None},
        {"dtype": "f64", "data": rng.uniform(900.0, 105000.0, rows), "valid": None},
        {"dtype": "f64", "data": rng.uniform(0.0, 0.1, rows), "valid": None},
        {"dtype": "f64", "data": rng.uniform(0.0, 0.08, rows), "valid": None},
        {"dtype": "i32", "data": rng.integers(8000, 11000, rows, dtype=np.int64)
         .astype(np.int32), "valid": None},
    ]
    return cols, [0, 1]


def make_lineitem_q3_multikey(rows, seed):
    """BASELINE config[2]'s 'multi-key hash shuffle': l_orderkey + l_suppkey composite
    key over the q3/q5 projection (exercises the multi-column combine on the GPU)."""
    cols, _ = make_lineitem_q3(rows, seed)
    rng = np.random.default_rng(seed + 1)
    cols.append({"dtype": "i32", "data": rng.integers(1, 100_001, rows, dtype=np.int64)
                 .astype(np.int32), "valid": None})
    return cols, [0, 4]


def make_lineitem_utf8key(rows, seed):
    """SURVEY §8(d) microbench variant '2-key (int64+utf8(16B avg))': composite i64+utf8
    key over the q3 payload — exercises the chunked byte hash at full scale."""
    rng = np.random.default_rng(seed)
    lens = rng.integers(8, 25, rows)  # ~16 B avg
    off = np.zeros(rows + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    data = rng.integers(48, 122, int(off[-1]), dtype=np.int64).astype(np.uint8)
    return [
        {"dtype": "i64", "data": rng.integers(1, 60_000_000, rows, dtype=np.int64),
         "valid": None},
        {"dtype": "utf8", "data": data, "offsets": off, "valid": None},
        {"dtype": "f64", "data": rng.uniform(900.0, 105000.0, rows), "valid": None},
        {"dtype": "f64", "data": rng.uniform(0.0, 0.1, rows), "valid": None},
    ], [0, 1]


def make_clickbench_userid(rows, seed):
    """BASELINE config[4] stand-in: GROUP BY UserID — Zipf(1.1)-skewed i64 key + wide
    var-width URL column (exercises the v1 var-width path)."""
    rng = np.random.default_rng(seed)
    uid = (rng.zipf(1.1, rows) % 100_000).astype(np.int64)
    lens = rng.integers(16, 112, rows)
    off = np.zeros(rows + 1, dtype=np.int32)
    off[1:] = np.cumsum(lens)
    url = rng.integers(33, 127, int(off[-1]), dtype=np.int64).astype(np.uint8)
    return [
        {"dtype": "i64", "data": uid, "valid": None},
        {"dtype": "utf8", "data": url, "offsets": off, "valid": None},
        {"dtype": "i64", "data": rng.integers(0, 2**41, rows, dtype=np.int64),
         "valid": None},
        {"dtype": "i32", "data": rng.integers(0, 10**6, rows, dtype=np.int64)
         .astype(np.int32), "valid": None},
    ], [0]


def make_dictkey(rows, seed):
    """SURVEY §8(d) microbench variant 'dictionary key': dict32(idx→utf8) key over the
    q3 payload — exercises the precomputed per-distinct-value hash table (K0 dict pass)
    with the scatter on 4-byte indices. 100k distinct values, ≈12 B avg."""
    rng = np.random.default_rng(seed)
    nvals = 100_000
    lens = rng.integers(6, 20, nvals)
    doff = np.zeros(nvals + 1, dtype=np.int32)
    doff[1:] = np.cumsum(lens)
    dbytes = rng.integers(97, 123, int(doff[-1]), dtype=np.int64).astype(np.uint8)
    cols, _ = make_lineitem_q3(rows, seed)
    cols[0] = {"dtype": "dict32", "data": rng.integers(0, nvals, rows).astype(np.int32),
               "dict_bytes": dbytes, "dict_offsets": doff, "valid": None}
    return cols, [0]


WORKLOADS = {
    "tpch_sf10_lineitem_shuffle": (make_lineitem_q3, SF10_ROWS),
    "dictkey_shuffle": (make_dictkey, 30_000_000),
    "tpch_sf10_multikey_shuffle": (make_lineitem_q3_multikey, SF10_ROWS),
    "tpch_sf10_utf8key_shuffle": (make_lineitem_utf8key, 30_000_000),
    "tpch_sf1_q1_repartition": (make_lineitem_q1, 6_001_215),
    "clickbench_userid_shuffle": (make_clickbench_userid, 20_000_000),
}


def workload_bytes(cols):
    """(total input bytes, K3 algorithmic bytes) for the roofline formula.
    K3: read pid (4/row) + read+write every column; var cols add offsets read (4/row) and
    lengths write (4/row)."""
    n = (len(cols[0]["offsets"]) - 1 if cols[0]["dtype"] == "utf8"
         else len(cols[0]["data"]))
    total = 0
    k3 = 4 * n
    for c in cols:
        if c["dtype"] == "utf8":
            b = int(np.asarray(c["data"]).nbytes)
            total += b + 4 * n
            k3 += 2 * b + 4 * n + 4 * n
        else:
            b = FIXED_SIZE[c["dtype"]] * n
            total += b
            k3 += 2 * b
        if c.get("valid") is not None:
            total += n
            k3 += 2 * n
    return total, k3


def cpu_baseline_leg(rows_sample, key, cols_np, min_seconds=8.0, max_seconds=30.0):
    """Time the C oracle on a bounded sample of the same workload (kind: port)."""
    import oracle

    sample = []
    for c in cols_np:
        sc = dict(c)
        if c["dtype"] == "utf8":
            off = c["offsets"][: rows_sample + 1]
            sc["offsets"] = off
            sc["data"] = c["data"][: int(off[-1])]
        else:
            sc["data"] = c["data"][:rows_sample]
        if c.get("valid") is not None:
            sc["valid"] = c["valid"][:rows_sample]
        sample.append(sc)
    reps = 0
    t0 = time.perf_counter()
    while True:
        oracle.repartition(sample, key, PARTS_PER_RANK)
        reps += 1
        el = time.perf_counter() - t0
        if el >= min_seconds or el >= max_seconds:
            break
    el = time.perf_counter() - t0
    sample_bytes, _ = workload_bytes(sample)
    gbps = sample_bytes * reps / el / 1e9
    return {
        "value": round(gbps, 3),
        "unit": "GB/s",
        "cores": oracle.lib().dd_oracle_num_threads(),
        "kind": "port",
        "sample": f"{rows_sample} rows x {reps} reps of the same hash-repartition "
                  f"({sample_bytes // max(rows_sample, 1)} B/row, P={PARTS_PER_RANK}), "
                  f"{el:.1f}s on host cores",
    }


def run_ingest(args):
    """Pinned-host staged ingest (VERDICT r01 item 8 / DESIGN §11.4): the real pipeline's
    input arrives from host-side parquet scans. Double-buffered PINNED staging chunks are
    H2D-copied on a dedicated copy stream while the partition kernels of the PREVIOUS
    chunk run on the compute stream — step time converges to max(PCIe, kernels), not the
    sum, and pinned H2D runs at link speed (pageable measured 7-9 GB/s round 1)."""
    import torch

    from datafusion_distributed_amd import api

    if api.device_count() == 0:
        print(json.dumps({"error": "no HIP device"}))
        sys.exit(1)
    torch.cuda.set_device(0)
    chunk_rows = args.rows or 8_000_000
    nchunks = max(args.steps, 2)
    gen, _ = WORKLOADS["tpch_sf10_lineitem_shuffle"]
    cols_np, key_idx = gen(chunk_rows, seed=42)
    tdt = {"i64": torch.int64, "f64": torch.float64, "i32": torch.int32}

    # measure raw H2D once: pageable vs pinned
    nbytes = sum(int(np.asarray(c["data"]).nbytes) for c in cols_np)
    dev_probe = torch.empty(nbytes, dtype=torch.uint8, device="cuda")
    pageable = torch.from_numpy(np.concatenate(
        [np.asarray(c["data"]).view(np.uint8) for c in cols_np]))
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    dev_probe.copy_(pageable)
    torch.cuda.synchronize()
    h2d_pageable = nbytes / (time.perf_counter() - t0) / 1e9
    pinned_probe = pageable.pin_memory()
    t0 = time.perf_counter()
    dev_probe.copy_(pinned_probe, non_blocking=True)
    torch.cuda.synchronize()
    h2d_pinned = nbytes / (time.perf_counter() - t0) / 1e9
    del dev_probe, pageable, pinned_probe

    # double-buffered staging: 2 pinned host sets, 2 device sets, 2 partitioners
    pinned = []
    dev = []
    parts = []
    for b in range(2):
        hs = [torch.from_numpy(np.ascontiguousarray(c["data"])).pin_memory()
              for c in cols_np]
        ds = [torch.empty_like(h, device="cuda") for h in hs]
        views = []
        for c, t in zip(cols_np, ds):
            views.append({"dtype": c["dtype"], "data_ptr": t.data_ptr()})
        vb = api.DeviceBatch.from_device(views, chunk_rows)
        vb.cols = cols_np  # dtype metadata for accessors
        parts.append(api.Partitioner(vb, key_idx, args.parts_per_rank))
        pinned.append(hs)
        dev.append(ds)
    copy_s = torch.cuda.Stream()
    comp_s = torch.cuda.Stream()
    evs = [torch.cuda.Event() for _ in range(2)]

    def pipeline(k):
        for i in range(k):
            b = i % 2
            with torch.cuda.stream(copy_s):
                parts[b].wait_phase2(copy_s)  # chunk buffers free?
                for h, d in zip(pinned[b], dev[b]):
                    d.copy_(h, non_blocking=True)
                evs[b].record(copy_s)
            comp_s.wait_event(evs[b])
            parts[b].run(stream=comp_s)
        torch.cuda.synchronize()

    pipeline(2)  # warmup
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    pipeline(nchunks)
    dt = time.perf_counter() - t0
    total = nbytes * nchunks
    kms = parts[0].kernel_ms()
    line = {
        "metric": "tpch_sf10_staged_ingest_GBps",
        "value": round(total / dt / 1e9, 2),
        "unit": "GB/s",
        "n_gpus": 1,
        "steps": nchunks,
        "warmup": 2,
        "ms_per_step": round(dt / nchunks * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "int64",
        "data": "synthetic",
        "config": {
            "workload": "tpch_sf10_staged_ingest",
            "chunk_rows": chunk_rows,
            "h2d_pageable_GBps": round(h2d_pageable, 2),
            "h2d_pinned_GBps": round(h2d_pinned, 2),
            "kernel_ms_last": {"k1_hash": round(kms[0], 3), "k2_scan": round(kms[1], 3),
                               "k3_scatter": round(kms[2], 3)},
            "note": "pipeline GB/s ~= min(pinned PCIe, kernel) leg; staging overlapped",
        },
    }
    print(json.dumps(line))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows", type=int, default=None, help="rows per rank")
    ap.add_argument("--workload", choices=sorted(WORKLOADS), 
                    default="tpch_sf10_lineitem_shuffle")
    ap.add_argument("--parts-per-rank", type=int, default=PARTS_PER_RANK)
    ap.add_argument("--traffic-bytes", type=float, default=None,
                    help="measured HBM bytes per K3 launch from a separate rocprofv3 "
                         "--pmc run (profiles/); null if not provided")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--ingest", action="store_true",
                    help="pinned-host staged-ingest pipeline: double-buffered H2D on a "
                         "copy stream overlapped with the partition kernels (the "
                         "'data originates host-side' scenario; PCIe staging hidden). "
                         "Prints its own line; not the contract metric.")
    ap.add_argument("--pipeline", action="store_true",
                    help="two-stream batch pipeline at N=1 (phase1 of batch s+1 "
                         "overlapped with phase2 of batch s). Measured SLOWER than "
                         "serial on this HBM-bound path (2.98 vs 1.65 ms/step: both "
                         "phases already saturate the memory system and co-residency "
                         "thrashes) — kept for measurement honesty and for callers "
                         "whose phases are not bandwidth-bound.")
    args = ap.parse_args()
    if args.ingest:
        run_ingest(args)
        return

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    import torch

    if world > 1:
        import torch.distributed as dist

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        ndev = torch.cuda.device_count()
        # gloo for the control plane when GPUs are oversubscribed (testing); the data
        # plane is always our own RCCL comm (exchange.create_comm)
        backend = "nccl" if ndev >= world else "gloo"
        dist.init_process_group(backend, rank=rank, world_size=world)
        torch.cuda.set_device(local_rank % max(ndev, 1))
    elif torch.cuda.is_available():
        torch.cuda.set_device(0)

    from datafusion_distributed_amd import api
    from datafusion_distributed_amd.exchange import create_comm

    if api.device_count() == 0:
        print(json.dumps({"error": "no HIP device"}))
        sys.exit(1)

    p_total = args.parts_per_rank * world
    gen, default_rows = WORKLOADS[args.workload]
    rows = args.rows or default_rows
    cols_np, key_idx = gen(rows, seed=42 + rank)
    total_bytes_in, k3_bytes = workload_bytes(cols_np)

    th2d0 = time.perf_counter()
    batch = api.DeviceBatch(cols_np)
    api.lib().dd_device_sync()
    th2d = time.perf_counter() - th2d0
    h2d_gbps = total_bytes_in / th2d / 1e9

    part = api.Partitioner(batch, key_idx, p_total)
    if not part.has_pid_array():
        # rhash spec path: K3 recomputes hashes from preloaded registers;
        # the 4 B/row pid read is not part of its algorithmic bytes
        k3_bytes -= 4 * rows
    elif api.lib().dd_partitioner_pid_elem(part.h) == 1:
        # u8 pid array (pre path, P <= 256): K3 reads 1 B/row of pid, not 4
        k3_bytes -= 3 * rows
    comm = create_comm(rank, world) if world > 1 else None
    pipeline = world == 1 and args.pipeline and torch.cuda.is_available()
    if pipeline:
        part2 = api.Partitioner(batch, key_idx, p_total)
        parts2 = [part, part2]
        s1, s2 = torch.cuda.Stream(), torch.cuda.Stream()

    def step():
        part.run()
        if comm is not None:
            ex = comm.exchange(part)
            ex.destroy()
        else:
            part.sync()

    def pipelined_steps(k):
        """k batch passes with phase1(s+1) on stream 1 overlapped with phase2(s) on
        stream 2 (events order phases of the same partitioner across streams)."""
        for i in range(k):
            cur = parts2[i % 2]
            cur.wait_phase2(s1)   # don't overwrite pid/counts its last scatter still reads
            cur.run_phase1(s1)
            if i > 0:
                prv = parts2[(i + 1) % 2]
                prv.wait_phase1(s2)
                prv.run_phase2(s2)
        last = parts2[(k - 1) % 2]
        last.wait_phase1(s2)
        last.run_phase2(s2)
        torch.cuda.synchronize()

    def barrier_sync():
        if world > 1:
            import torch.distributed as dist

            torch.cuda.synchronize()
            dist.barrier()
        torch.cuda.synchronize() if torch.cuda.is_available() else api.lib().dd_device_sync()

    for _ in range(args.warmup):
        if pipeline:
            part2.run()
        step()
    barrier_sync()

    k3_ms_total = 0.0
    t0 = time.perf_counter()
    if pipeline:
        pipelined_steps(args.steps)
        k3_ms_total = part.kernel_ms()[2] * args.steps  # last-run K3 (events, per pass)
    else:
        for _ in range(args.steps):
            step()
            k3_ms_total += part.kernel_ms()[2]
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if world > 1:
        import torch.distributed as dist

        dev = ("cuda" if (torch.cuda.is_available() and
                          dist.get_backend() == "nccl") else "cpu")
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_bytes = total_bytes_in * world * args.steps
    value = total_bytes / elapsed / 1e9
    rows_per_s = rows * world * args.steps / elapsed

    k1_ms, k2_ms, k3_ms = part.kernel_ms()
    k3_mean_ms = k3_ms_total / args.steps
    achieved = k3_bytes / (k3_mean_ms / 1e3)
    roofline = {
        "bound": "hbm",
        "achieved": round(achieved / 1e9, 1),
        "peak": round(HBM_PEAK / 1e9, 1),
        "unit": "GB/s",
        "frac": round(achieved / HBM_PEAK, 4),
        "traffic": args.traffic_bytes,
    }

    cpu_baseline = None
    if rank == 0 and world == 1 and not args.skip_cpu_baseline:
        cpu_baseline = cpu_baseline_leg(min(rows, 8_000_000), key_idx, cols_np)

    config = {
        "workload": args.workload,
        "batch_pipeline": bool(pipeline),
        "rows_per_rank": rows,
        "row_bytes": round(total_bytes_in / max(rows, 1), 1),
        "p_total": p_total,
        "parts_per_rank": args.parts_per_rank,
        "key_cols": key_idx,
        "rows_per_s": round(rows_per_s, 0),
        "kernel_ms_last": {"k1_hash": round(k1_ms, 3), "k2_scan": round(k2_ms, 3),
                           "k3_scatter": round(k3_ms, 3)},
        "h2d_GBps": round(h2d_gbps, 2),
    }
    if comm is not None:
        ex = comm.exchange(part)
        ms, egress = ex.stats()
        config["xgmi_egress_GBps"] = round(egress / (ms / 1e3) / 1e9, 2) if ms > 0 else None
        config["data_plane"] = getattr(comm, "data_plane", "rccl/xgmi")
        ex.destroy()

    line = {
        "metric": ("tpch_sf10_shuffle_GBps" if args.workload == "tpch_sf10_lineitem_shuffle"
                   else args.workload + "_GBps"),
        "value": round(value, 3),
        "unit": "GB/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1e3, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published number for this metric
        "dtype": "int64",
        "data": "synthetic",
        "config": config,
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    if rank == 0:
        print(json.dumps(line))

    part.destroy()
    batch.free()
    if comm is not None:
        comm.destroy()
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
