"""Generates the committed golden vectors under tests/golden/.

Run from the repo root: python tests/golden/gen_golden.py

The fixtures pin the normative hash/partition semantics so that the C oracle
and the HIP kernels can be checked bit-exactly on any box without this script. Inputs mirror
the reference's bench fixture schema (/root/reference/src/execution_plans/benchmarks/
fixture.rs:12-32: id i64, metric f64, flag bool 10% null, label utf8 10% null, category
dict(i32->utf8), raw u8, ts i64, count i32; the `tags` list column is deferred — DESIGN.md §8)
and its create_random_batch(.., null_density=0.1, str_len factor 0.5) shape. Expected values
are produced by the numpy restatement (oracle/pyref.py) — generation is deliberately through
pyref, and tests re-check the C oracle against these files, so the two restatements pin each
other through the committed artifact.
"""

import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)

from oracle import pyref  # noqa: E402


def make_bench_batch(n, seed, null_density=0.1):
    rng = np.random.default_rng(seed)

    def nulls():
        return (rng.random(n) >= null_density).astype(np.uint8)

    def utf8_col(max_len):
        lens = rng.integers(0, max_len + 1, n)
        off = np.zeros(n + 1, dtype=np.int32)
        off[1:] = np.cumsum(lens)
        data = rng.integers(32, 127, int(off[-1]), dtype=np.int64).astype(np.uint8)
        return off, data

    cols = []
    cols.append({"name": "id", "dtype": "i64",
                 "data": rng.integers(-(2**62), 2**62, n, dtype=np.int64), "valid": None})
    metric = rng.normal(size=n)
    metric[:: max(n // 37, 1)] = 0.0
    metric[1:: max(n // 41, 1)] = -0.0
    metric[2:: max(n // 43, 1)] = np.nan
    cols.append({"name": "metric", "dtype": "f64", "data": metric, "valid": None})
    cols.append({"name": "flag", "dtype": "bool",
                 "data": (rng.random(n) > 0.5).astype(np.uint8), "valid": nulls()})
    off, data = utf8_col(16)
    cols.append({"name": "label", "dtype": "utf8", "data": data, "offsets": off,
                 "valid": nulls()})
    dict_values = [b"", b"alpha", b"beta", b"gamma", b"delta-longer-value", b"\xc3\xa9clair"]
    dby = np.frombuffer(b"".join(dict_values), dtype=np.uint8).copy()
    doff = np.zeros(len(dict_values) + 1, dtype=np.int32)
    doff[1:] = np.cumsum([len(v) for v in dict_values])
    cols.append({"name": "category", "dtype": "dict32",
                 "data": rng.integers(0, len(dict_values), n).astype(np.int32),
                 "dict_bytes": dby, "dict_offsets": doff, "valid": nulls()})
    cols.append({"name": "raw", "dtype": "u8",
                 "data": rng.integers(0, 256, n, dtype=np.int64).astype(np.uint8),
                 "valid": None})
    cols.append({"name": "ts", "dtype": "i64",
                 "data": rng.integers(0, 2**48, n, dtype=np.int64), "valid": None})
    cols.append({"name": "count", "dtype": "i32",
                 "data": rng.integers(-(2**31), 2**31, n, dtype=np.int64).astype(np.int32),
                 "valid": None})
    return cols


def save_case(fname, cols, key_idx, nparts):
    res = pyref.repartition(cols, key_idx, nparts)
    out = {
        "key_idx": np.array(key_idx, dtype=np.int32),
        "nparts": np.array([nparts], dtype=np.int32),
        "exp_hash": res["hash"],
        "exp_pid": res["pid"],
        "exp_order": res["order"],
        "exp_part_offsets": res["part_offsets"],
    }
    for i, c in enumerate(cols):
        out[f"col{i}_dtype"] = np.frombuffer(c["dtype"].encode(), dtype=np.uint8)
        out[f"col{i}_data"] = c["data"]
        if c.get("valid") is not None:
            out[f"col{i}_valid"] = c["valid"]
        if c["dtype"] == "utf8":
            out[f"col{i}_offsets"] = c["offsets"]
        if c["dtype"] == "dict32":
            out[f"col{i}_dict_bytes"] = c["dict_bytes"]
            out[f"col{i}_dict_offsets"] = c["dict_offsets"]
    np.savez_compressed(fname, **out)
    sizes = res["part_offsets"][1:] - res["part_offsets"][:-1]
    print(f"{os.path.basename(fname)}: n={len(res['pid'])} P={nparts} "
          f"min/max part {sizes.min()}/{sizes.max()}")


def load_case(fname):
    z = np.load(fname)
    cols = []
    i = 0
    while f"col{i}_dtype" in z:
        dt = bytes(z[f"col{i}_dtype"]).decode()
        c = {"dtype": dt, "data": z[f"col{i}_data"], "valid": None}
        if f"col{i}_valid" in z:
            c["valid"] = z[f"col{i}_valid"]
        if dt == "utf8":
            c["offsets"] = z[f"col{i}_offsets"]
        if dt == "dict32":
            c["dict_bytes"] = z[f"col{i}_dict_bytes"]
            c["dict_offsets"] = z[f"col{i}_dict_offsets"]
        cols.append(c)
        i += 1
    exp = {
        "hash": z["exp_hash"], "pid": z["exp_pid"], "order": z["exp_order"],
        "part_offsets": z["exp_part_offsets"],
    }
    return cols, list(z["key_idx"]), int(z["nparts"][0]), exp


def main():
    d = os.path.dirname(os.path.abspath(__file__))
    # bench-schema batch, single i64 key, P=16 (mirrors local_repartition hash scenarios)
    cols = make_bench_batch(8192, seed=42)
    save_case(os.path.join(d, "bench_i64key_p16.npz"), cols, [0], 16)
    # multi-key incl. utf8 + dict + f64, P=13 (prime, exercises % path)
    save_case(os.path.join(d, "bench_multikey_p13.npz"), cols, [0, 3, 4, 1], 13)
    # utf8-only key with nulls, P=8
    save_case(os.path.join(d, "bench_utf8key_p8.npz"), cols, [3], 8)
    # tiny edge batch: 1 row, P=5
    tiny = make_bench_batch(1, seed=7)
    save_case(os.path.join(d, "tiny_1row_p5.npz"), tiny, [0], 5)
    # P=2048 (round-1 max partition count)
    cols2 = make_bench_batch(16384, seed=123)
    save_case(os.path.join(d, "bench_i64key_p2048.npz"), cols2, [0], 2048)
    # bool + narrow-int keys (a missing BOOL case in the device hash escaped the
    # original fixtures — pin it)
    save_case(os.path.join(d, "bench_boolkey_p7.npz"), cols, [2, 5, 7], 7)
    # scalar hash known-answer vectors (pin mix64 / hash_bytes directly)
    from oracle.pyref import hash_bytes_scalar, mix64_scalar
    vec_in = [0, 1, 0x9E3779B97F4A7C15, 2**64 - 1, 42]
    np.savez_compressed(
        os.path.join(d, "hash_kat.npz"),
        mix64_in=np.array(vec_in, dtype=np.uint64),
        mix64_out=np.array([mix64_scalar(v) for v in vec_in], dtype=np.uint64),
        bytes_in=np.frombuffer(b"hello world, datafusion shuffle", dtype=np.uint8),
        bytes_prefix_out=np.array(
            [hash_bytes_scalar(b"hello world, datafusion shuffle"[:k]) for k in range(32)],
            dtype=np.uint64,
        ),
    )
    print("hash_kat.npz written")


if __name__ == "__main__":
    main()
