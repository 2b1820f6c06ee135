"""One-box A/B sweep of the K3 scatter variants (pre / hl / knobs) on the bench shape.

Runs every configuration back-to-back on the SAME box (box-to-box spread measured ~20%,
far above the deltas under test), repeating the first config at the end to bound drift.
Prints one line per config: mean kernel ms over reps (k1/k2/k3) + whole-step GB/s.
Environment knobs are read at partitioner-create time, so each config rebuilds its
partitioner. Usage: python tools/sweep_pre.py [rows] [reps]
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from datafusion_distributed_amd import api

N = int(sys.argv[1]) if len(sys.argv) > 1 else 59_986_052
REPS = int(sys.argv[2]) if len(sys.argv) > 2 else 15
P = 128

CONFIGS = [
    ("pre", {}),
    ("pre-pid32", {"DD_PID8": "0"}),
    ("pre-nt0", {"DD_PRE_NT": "0"}),
    ("pre-rpb2", {"DD_PRE_RPB": "2"}),
    ("pre-rpb4", {"DD_PRE_RPB": "4"}),
    ("pre-wpb8", {"DD_PRE_WPB": "8"}),
    ("pre-wpb8-rpb2", {"DD_PRE_WPB": "8", "DD_PRE_RPB": "2"}),
    ("hl", {"DD_K3_PRE": "0"}),
    ("plain", {"DD_K3_PRE": "0", "DD_K3_HL": "0"}),
    ("pre/again", {}),  # drift check
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
    results = {}
    for name, env in CONFIGS:
        for k in KNOBS:
            os.environ.pop(k, None)
        os.environ.update(env)
        part = api.Partitioner(batch, [0], P)
        ks = np.zeros(3)
        for r in range(REPS + 3):
            part.run()
            part.sync()
            if r >= 3:
                ks += np.array(part.kernel_ms())
        part.destroy()
        ks /= REPS
        step_ms = ks.sum()
        gbps = N * row_bytes / step_ms / 1e6
        results[name] = {"k1": round(ks[0], 4), "k2": round(ks[1], 4),
                         "k3": round(ks[2], 4), "step_ms": round(step_ms, 4),
                         "GBps": round(gbps, 1)}
        print(name, json.dumps(results[name]), flush=True)
    batch.free()
    print("SWEEP", json.dumps(results))


if __name__ == "__main__":
    main()
