"""Multi-process CPU tests (gloo, world_size 2) for the N>1 control logic: the unique-id
bootstrap path and the exchange placement model (who sends which partition slice to whom,
and in what order the consumer sees producers) — the same model dd_exchange_run implements
(DESIGN.md §6). The RCCL data plane itself is exercised on the GPU box
(tests/test_gpu_exchange.py, N=1) and by the driver's multi-GPU scale run."""

import os

import numpy as np
import pytest

import oracle


def exchange_model(per_rank_results, P, nranks):
    """Pure-numpy model of dd_exchange_run's placement: consumer j receives, producer-major,
    each producer's partitions [P*j, P*(j+1)). Returns per consumer: (row_counts[R][P],
    per-col concatenated data)."""
    out = []
    for j in range(nranks):
        row_counts = np.zeros((nranks, P), dtype=np.int64)
        col_chunks = None
        for r, res in enumerate(per_rank_results):
            off = res["part_offsets"]
            lo, hi = off[P * j], off[P * (j + 1)]
            for q in range(P):
                row_counts[r, q] = off[P * j + q + 1] - off[P * j + q]
            if col_chunks is None:
                col_chunks = [[] for _ in res["cols"]]
            for ci, col in enumerate(res["cols"]):
                col_chunks[ci].append(col["data"][lo:hi] if col["dtype"] != "utf8" else None)
        out.append({"row_counts": row_counts,
                    "cols": [np.concatenate(c) if c[0] is not None else None
                             for c in col_chunks]})
    return out


def test_exchange_model_conservation():
    """Every row lands at exactly one consumer; totals are conserved; producer order is
    deterministic."""
    rng = np.random.default_rng(21)
    nranks, P = 2, 8
    P_total = P * nranks
    per_rank = []
    inputs = []
    for r in range(nranks):
        n = 5000 + r * 777
        cols = [{"dtype": "i64", "data": rng.integers(0, 10**12, n, dtype=np.int64),
                 "valid": None},
                {"dtype": "f64", "data": rng.normal(size=n), "valid": None}]
        inputs.append(cols)
        per_rank.append(oracle.repartition(cols, [0], P_total))

    consumers = exchange_model(per_rank, P, nranks)
    total_in = sum(len(c[0]["data"]) for c in inputs)
    total_out = sum(int(c["row_counts"].sum()) for c in consumers)
    assert total_in == total_out
    # each consumer's rows hash into its own window
    for j, cons in enumerate(consumers):
        keys = cons["cols"][0]
        h = oracle.hash_cols([{"dtype": "i64", "data": keys, "valid": None}])
        pid = oracle.pids(h, P_total)
        assert ((pid >= P * j) & (pid < P * (j + 1))).all()
    # multiset of key values is conserved
    all_in = np.sort(np.concatenate([c[0]["data"] for c in inputs]))
    all_out = np.sort(np.concatenate([c["cols"][0] for c in consumers]))
    assert (all_in == all_out).all()


def _bootstrap_worker(rank, world, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29531"
    import torch
    import torch.distributed as dist

    dist.init_process_group("gloo", rank=rank, world_size=world)
    # mirror exchange.create_comm's broadcast, with a random token instead of an
    # ncclUniqueId (dd_comm_init needs a GPU; the broadcast path is identical)
    if rank == 0:
        uid = os.urandom(128)
        t = torch.tensor(list(uid), dtype=torch.uint8)
    else:
        t = torch.zeros(128, dtype=torch.uint8)
    dist.broadcast(t, src=0)
    q.put((rank, bytes(t.tolist())))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_unique_id_broadcast_gloo_world2():
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_bootstrap_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    got = {}
    for _ in range(2):
        r, uid = q.get(timeout=90)
        got[r] = uid
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert got[0] == got[1] and len(got[0]) == 128
