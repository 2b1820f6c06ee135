"""One-shot extended parity sweep (run manually on a GPU box; not part of the pytest
suite so the round-end GPU run stays fast). Reuses the fuzz generator and the oracle
checker from tests/: 40 extra random seeds plus adversarial shapes the uniform fuzz
rarely produces — heavy key skew (every row in one partition stresses the staged
flush's longest runs), two-value keys, constant keys with P=2048, empty strings.

Usage: python tools/fuzz_sweep.py  (prints one line per case; exits nonzero on mismatch)
"""

import sys
import numpy as np

sys.path.insert(0, ".")
from tests.test_gpu_fuzz import random_col  # noqa: E402
from tests.test_gpu_parity import check_against_oracle  # noqa: E402


def skew_col(n, nvals, dtype=np.int64):
    """keys drawn from nvals distinct values (nvals=1 -> single partition)."""
    rng = np.random.default_rng(42 + nvals)
    vals = rng.integers(-(2**62), 2**62, nvals, dtype=np.int64)
    return {"dtype": "i64", "data": vals[rng.integers(0, nvals, n)].astype(dtype),
            "valid": None}


def main():
    failures = 0

    # 40 extra random seeds over the same distribution as tests/test_gpu_fuzz.py
    for case in range(40):
        rng = np.random.default_rng(2000 + case)
        n = int(rng.choice([1, 2, 63, 64, 65, 1000, 4096, 30000, 250000, 1 << 20]))
        ncols = int(rng.integers(1, 9))
        dtypes = list(rng.choice(["u8", "i16", "i32", "i64", "f32", "f64", "bool",
                                  "utf8", "dict32"], ncols))
        cols = [random_col(rng, n, dt, float(rng.choice([0, 0, 0.1, 0.5])))
                for dt in dtypes]
        nkeys = int(rng.integers(1, min(ncols, 4) + 1))
        key_idx = [int(k) for k in rng.choice(ncols, nkeys, replace=False)]
        nparts = int(rng.choice([1, 2, 3, 7, 8, 16, 100, 128, 777, 2048]))
        try:
            check_against_oracle(cols, key_idx, nparts)
            print(f"seed {2000 + case}: n={n} ncols={ncols} P={nparts} OK")
        except AssertionError as e:
            failures += 1
            print(f"seed {2000 + case}: FAIL {e}")

    # adversarial skew: nvals distinct keys, large n
    rng = np.random.default_rng(7)
    for nvals, n, nparts in [(1, 250000, 128), (1, 1 << 20, 2048), (2, 500000, 128),
                             (3, 250000, 7), (16, 1 << 20, 128)]:
        cols = [skew_col(n, nvals), random_col(rng, n, "f64", 0.1),
                random_col(rng, n, "utf8", 0)]
        try:
            check_against_oracle(cols, [0], nparts)
            print(f"skew nvals={nvals} n={n} P={nparts} OK")
        except AssertionError as e:
            failures += 1
            print(f"skew nvals={nvals} n={n} P={nparts}: FAIL {e}")

    # all-empty strings + constant key at max P
    n = 100000
    cols = [skew_col(n, 1),
            {"dtype": "utf8", "data": np.zeros(0, dtype=np.uint8),
             "offsets": np.zeros(n + 1, dtype=np.int32), "valid": None}]
    try:
        check_against_oracle(cols, [0], 2048)
        print("empty-strings constant-key P=2048 OK")
    except AssertionError as e:
        failures += 1
        print(f"empty-strings constant-key: FAIL {e}")

    print(f"sweep done: {failures} failures")
    sys.exit(1 if failures else 0)


if __name__ == "__main__":
    main()
