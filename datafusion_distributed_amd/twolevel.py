"""twolevel.py — L3-aware two-level var scatter (opt-in).

A random gather of small strings fetches a full 128 B line per string (8x read
amplification at 16 B averages — profiles). Pass A partitions into `buckets` CONTIGUOUS
coarse ranges of the final partition space (pid = (h % P) / (P/buckets), the ranged
partitioner); pass B partitions each bucket to its final partitions with plain h % (P/B).
Pass B's K4 gather then reads from a bucket ~1/buckets the size — Infinity-Cache resident
(256 MB) — instead of HBM-latency-bound random reads. Bucket outputs are contiguous
slices of the final partition-major layout, so the final buffers are a handful of
bucket-sized d2d copies.

Stability: pass A is stable over the input; pass B is stable over each bucket; bucket g's
final partitions are exactly [g*S, (g+1)*S) — the concatenation is bit-identical to the
single-pass partitioner / the oracle.
"""

import ctypes

import numpy as np

from . import api


def two_level_partition(cols, key_idx, nparts, buckets=8):
    """Returns {"part_row_offsets", "cols": downloaded final buffers, "timings_ms"}."""
    assert nparts % buckets == 0 and buckets & (buckets - 1) == 0
    assert nparts & (nparts - 1) == 0
    S = nparts // buckets
    L = api.lib()
    n = (len(cols[0]["offsets"]) - 1 if cols[0]["dtype"] == "utf8" else len(cols[0]["data"]))
    var_idx = [i for i, c in enumerate(cols) if c["dtype"] == "utf8"]

    batch = api.DeviceBatch(cols)
    import os
    import time
    t0 = time.perf_counter()
    # pass A on the v1 direct path: sequential reads, writes into only `buckets` streams
    # (large runs) — avoids running the expensive K4 gather twice. Pass B (staged-var)
    # then gathers within an L3-resident bucket.
    prev = os.environ.get("DD_V2_VAR")
    os.environ["DD_V2_VAR"] = "0"
    try:
        passA = api.Partitioner(batch, key_idx, None, ranged=(nparts, S))
    finally:
        if prev is None:
            os.environ.pop("DD_V2_VAR", None)
        else:
            os.environ["DD_V2_VAR"] = prev
    passA.run()
    passA.sync()
    tA = time.perf_counter() - t0
    roffA = passA.row_offsets()
    boffA = {i: passA.byte_offsets(i) for i in var_idx}
    # v1 pass A has no rebuilt offsets64: build bucket offsets from the lengths instead
    lensA = {i: api._d2h(L.dd_partitioner_col_lengths(passA.h, i), n * 4, np.uint32)
             for i in var_idx}

    # reusable i32 offsets scratch for bucket views
    max_rows = int((roffA[1:] - roffA[:-1]).max()) if n else 0
    off32 = {i: api._dev_alloc((max_rows + 1) * 4) for i in var_idx}

    elem = {i: api.ELEM_SIZE[c["dtype"]] for i, c in enumerate(cols)
            if c["dtype"] != "utf8"}
    t1 = time.perf_counter()
    passB = []
    for g in range(buckets):
        lo, hi = int(roffA[g]), int(roffA[g + 1])
        nrows = hi - lo
        view = []
        for i, c in enumerate(cols):
            v = {"dtype": c["dtype"]}
            if c["dtype"] == "utf8":
                # host-built i32 offsets for the bucket (lengths prefix), uploaded
                ob = np.zeros(nrows + 1, dtype=np.int32)
                np.cumsum(lensA[i][lo:hi], out=ob[1:].view(np.uint32)[:nrows])
                api._check(L.dd_memcpy_h2d(off32[i],
                                           ob.ctypes.data_as(ctypes.c_void_p),
                                           ctypes.c_int64(ob.nbytes)))
                v["data_ptr"] = L.dd_partitioner_col_data(passA.h, i) + int(boffA[i][g])
                v["offsets_ptr"] = off32[i].value
                v["data_len"] = int(boffA[i][g + 1] - boffA[i][g])
            else:
                v["data_ptr"] = L.dd_partitioner_col_data(passA.h, i) + lo * elem[i]
            vp = L.dd_partitioner_col_validity(passA.h, i)
            if vp:
                v["valid_ptr"] = vp + lo
            view.append(v)
        vbatch = api.DeviceBatch.from_device(view, nrows)
        pb = api.Partitioner(vbatch, key_idx, S)
        pb.run()
        pb.sync()  # off32 scratch is reused next bucket: must be consumed first
        passB.append(pb)
    tB = time.perf_counter() - t1

    kernel_ms = sum(passA.kernel_ms()) + sum(sum(pb.kernel_ms()) for pb in passB)

    # global offsets: bucket-contiguous
    grow = np.zeros(nparts + 1, dtype=np.int64)
    gbyte = {i: np.zeros(nparts + 1, dtype=np.int64) for i in var_idx}
    for g in range(buckets):
        roffB = passB[g].row_offsets()
        grow[g * S + 1: (g + 1) * S + 1] = roffA[g] + roffB[1:]
        for i in var_idx:
            boffB = passB[g].byte_offsets(i)
            gbyte[i][g * S + 1: (g + 1) * S + 1] = boffA[i][g] + boffB[1:]
    assert grow[-1] == n

    # final buffers: one d2d per (stream, bucket)
    t2 = time.perf_counter()
    final = {}
    for i, c in enumerate(cols):
        if c["dtype"] == "utf8":
            total_b = int(boffA[i][-1])
            final[i] = {"data": api._dev_alloc(total_b),
                        "lengths": api._dev_alloc(n * 4)}
        else:
            final[i] = {"data": api._dev_alloc(n * elem[i])}
        if c.get("valid") is not None:
            final[i]["valid"] = api._dev_alloc(n)

    def d2d(dst, dst_off, src_ptr, nbytes):
        if nbytes:
            api._check(L.dd_memcpy_d2d(ctypes.c_void_p(dst.value + dst_off),
                                       ctypes.c_void_p(src_ptr), ctypes.c_int64(nbytes)))

    for g in range(buckets):
        lo = int(roffA[g])
        nrows = int(roffA[g + 1]) - lo
        for i, c in enumerate(cols):
            if c["dtype"] == "utf8":
                d2d(final[i]["data"], int(boffA[i][g]),
                    L.dd_partitioner_col_data(passB[g].h, i),
                    int(boffA[i][g + 1] - boffA[i][g]))
                d2d(final[i]["lengths"], lo * 4,
                    L.dd_partitioner_col_lengths(passB[g].h, i), nrows * 4)
            else:
                d2d(final[i]["data"], lo * elem[i],
                    L.dd_partitioner_col_data(passB[g].h, i), nrows * elem[i])
            if c.get("valid") is not None:
                d2d(final[i]["valid"], lo, L.dd_partitioner_col_validity(passB[g].h, i),
                    nrows)
    tC = time.perf_counter() - t2

    out_cols = []
    for i, c in enumerate(cols):
        oc = {"dtype": c["dtype"]}
        if c["dtype"] == "utf8":
            oc["data"] = api._d2h(final[i]["data"].value, int(boffA[i][-1]), np.uint8)
            oc["lengths"] = api._d2h(final[i]["lengths"].value, n * 4, np.uint32)
        else:
            oc["data"] = api._d2h(final[i]["data"].value, n * elem[i],
                                  api.FIXED_NP[c["dtype"]])
        if c.get("valid") is not None:
            oc["valid"] = api._d2h(final[i]["valid"].value, n, np.uint8)
        out_cols.append(oc)

    for g in range(buckets):
        passB[g].destroy()
    passA.destroy()
    batch.free()
    for i in var_idx:
        L.dd_dev_free(off32[i])
    for i in final:
        for b in final[i].values():
            L.dd_dev_free(b)
    return {"part_row_offsets": grow, "cols": out_cols,
            "timings_ms": {"passA": tA * 1e3, "passB": tB * 1e3, "concat": tC * 1e3,
                           "kernels": kernel_ms}}
