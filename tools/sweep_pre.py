"""Interleaved one-box A/B of the K3 scatter variants on the bench shape.

Sequential per-config blocks measured ±4-5% within-box drift (clock/thermal), swamping
the ~3-6% deltas under test. This sweep creates every config's partitioner up-front
(env knobs are read at create time; outputs stay resident — 288 GB HBM), then runs the
configs ROUND-ROBIN and reports per-kernel MEDIANS, so drift hits all configs equally.
Usage: python tools/sweep_pre.py [rows] [reps]
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from datafusion_distributed_amd import api

N = int(sys.argv[1]) if len(sys.argv) > 1 else 59_986_052
REPS = int(sys.argv[2]) if len(sys.argv) > 2 else 21
P = 128

CONFIGS = [
    ("pre-rpb1", {"DD_PRE_RPB": "1"}),
    ("pre-rpb2", {"DD_PRE_RPB": "2"}),
    ("pre-rpb4", {"DD_PRE_RPB": "4"}),
    ("hl", {"DD_K3_PRE": "0"}),
    ("plain", {"DD_K3_PRE": "0", "DD_K3_HL": "0"}),
]

KNOBS = ["DD_K3_PRE", "DD_K3_HL", "DD_PID8", "DD_PRE_NT", "DD_PRE_RPB", "DD_PRE_GMAX",
         "DD_PRE_WPB"]


def main():
    rng = np.random.default_rng(42)
    cols = [
        {"dtype": "i64", "data": rng.integers(1, 15_000_000 * 4, N), "valid": None},
        {"dtype": "f64", "data": rng.normal(size=N) * 1000, "valid": None},
        {"dtype": "f64", "data": rng.random(N) * 0.1, "valid": None},
        {"dtype": "i32", "data": rng.integers(8000, 12000, N).astype(np.int32),
         "valid": None},
    ]
    batch = api.DeviceBatch(cols)
    row_bytes = 28.0
    parts = []
    for name, env in CONFIGS:
        for k in KNOBS:
            os.environ.pop(k, None)
        os.environ.update(env)
        parts.append((name, api.Partitioner(batch, [0], P)))
    samples = {name: [] for name, _ in CONFIGS}
    for r in range(REPS + 2):
        for name, part in parts:
            part.run()
            part.sync()
            if r >= 2:
                samples[name].append(part.kernel_ms())
    results = {}
    for name, part in parts:
        ks = np.median(np.array(samples[name]), axis=0)
        step_ms = float(ks.sum())
        results[name] = {"k1": round(float(ks[0]), 4), "k2": round(float(ks[1]), 4),
                         "k3": round(float(ks[2]), 4), "step_ms": round(step_ms, 4),
                         "GBps": round(N * row_bytes / step_ms / 1e6, 1)}
        print(name, json.dumps(results[name]), flush=True)
        part.destroy()
    batch.free()
    print("SWEEP", json.dumps(results))


if __name__ == "__main__":
    main()
