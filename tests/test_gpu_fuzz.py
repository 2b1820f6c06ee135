"""Randomized parity fuzz (marked gpu): 20 random batch shapes — dtype mixes, null
densities, partition counts (pow2 and not), sizes including tiny/ragged — every one must
match the CPU oracle bit-exactly through the full partition path. Catches configuration
corners the fixed tests miss (both staged and v1 kernels get hit depending on shape)."""

import numpy as np
import pytest

from tests.test_gpu_parity import check_against_oracle

pytestmark = pytest.mark.gpu

FIXED = ["u8", "i16", "i32", "i64", "f32", "f64", "bool"]


def random_col(rng, n, dtype, null_p):
    valid = (rng.random(n) >= null_p).astype(np.uint8) if null_p > 0 else None
    if dtype == "utf8":
        lens = rng.integers(0, rng.integers(1, 40), n)
        off = np.zeros(n + 1, dtype=np.int32)
        off[1:] = np.cumsum(lens)
        data = rng.integers(0, 256, int(off[-1]), dtype=np.int64).astype(np.uint8)
        return {"dtype": "utf8", "data": data, "offsets": off, "valid": valid}
    if dtype == "dict32":
        nvals = int(rng.integers(1, 50))
        vals = [bytes(rng.integers(0, 256, rng.integers(0, 12), dtype=np.int64)
                      .astype(np.uint8)) for _ in range(nvals)]
        doff = np.zeros(nvals + 1, dtype=np.int32)
        doff[1:] = np.cumsum([len(v) for v in vals])
        return {"dtype": "dict32",
                "data": rng.integers(0, nvals, n).astype(np.int32),
                "dict_bytes": np.frombuffer(b"".join(vals), dtype=np.uint8).copy()
                if doff[-1] else np.zeros(0, dtype=np.uint8),
                "dict_offsets": doff, "valid": valid}
    npdt = {"u8": np.uint8, "bool": np.uint8, "i16": np.int16, "i32": np.int32,
            "i64": np.int64, "f32": np.float32, "f64": np.float64}[dtype]
    if dtype in ("f32", "f64"):
        data = rng.normal(size=n).astype(npdt)
        data[rng.random(n) < 0.01] = 0.0
        data[rng.random(n) < 0.01] = -0.0
        data[rng.random(n) < 0.005] = np.nan
    elif dtype == "bool":
        data = (rng.random(n) > 0.5).astype(np.uint8)
    else:
        info = np.iinfo(npdt)
        data = rng.integers(info.min, int(info.max) + 1, n, dtype=np.int64).astype(npdt)
    return {"dtype": dtype, "data": data, "valid": valid}


@pytest.mark.parametrize("case", range(20))
def test_fuzz_parity(case):
    rng = np.random.default_rng(1000 + case)
    n = int(rng.choice([1, 2, 63, 64, 65, 1000, 4096, 30000, 250000]))
    ncols = int(rng.integers(1, 9))
    dtypes = list(rng.choice(FIXED + ["utf8", "dict32"], ncols))
    cols = [random_col(rng, n, dt, float(rng.choice([0, 0, 0.1, 0.5]))) for dt in dtypes]
    nkeys = int(rng.integers(1, min(ncols, 4) + 1))
    key_idx = list(rng.choice(ncols, nkeys, replace=False))
    nparts = int(rng.choice([1, 2, 3, 7, 8, 16, 100, 128, 777, 2048]))
    check_against_oracle(cols, [int(k) for k in key_idx], nparts)


KNOBS = [  # (wpb, gmax, dd_v2_var)
    ("4", "2", "0"), ("4", "8", "1"), ("8", "2", "1"),
    ("8", "8", "0"), ("16", "2", "1"), ("16", "4", "0"),
]


@pytest.mark.parametrize("knob", KNOBS)
def test_fuzz_parity_tuning_knobs(knob, monkeypatch):
    """Every tuning configuration (waves-per-block, rows-per-round, var path) must stay
    bit-exact — the knobs change scheduling, never results."""
    wpb, gmax, var = knob
    monkeypatch.setenv("DD_V2_WPB", wpb)
    monkeypatch.setenv("DD_V2_GMAX", gmax)
    monkeypatch.setenv("DD_V2_VAR", var)
    rng = np.random.default_rng(7000 + int(wpb) * 10 + int(gmax))
    n = 120000
    cols = [
        random_col(rng, n, "i64", 0.1),
        random_col(rng, n, "utf8", 0.2),
        random_col(rng, n, "f64", 0),
        random_col(rng, n, "bool", 0.3),
    ]
    check_against_oracle(cols, [0, 3], 32)

def test_rhash_ab(monkeypatch):
    """The register-recompute spec path (rhash: no pid array; opt-in DD_RHASH=1 — a
    measured perf negative as a default) and the default pid-array path
    must both match the oracle bit-exactly on the same rhash-eligible batch. Covers
    multi-key, u8 keys, non-pow2 P."""
    rng = np.random.default_rng(555)
    n = 300000
    cols = [random_col(rng, n, "i64", 0), random_col(rng, n, "u8", 0),
            random_col(rng, n, "f64", 0), random_col(rng, n, "i32", 0)]
    for keys, nparts in [([0], 128), ([1, 0], 100), ([3, 1], 777)]:
        monkeypatch.setenv("DD_RHASH", "1")
        check_against_oracle(cols, keys, nparts)
        monkeypatch.delenv("DD_RHASH")
        check_against_oracle(cols, keys, nparts)

def test_hl_ab(monkeypatch):
    """Hidden-load scatter (default for its gated shape: inline-asm preload +
    hand-counted s_waitcnt; dd_kernels.hip HL header) must match the oracle
    bit-exactly — including ragged last rounds and non-pow2 P — and so must the plain
    spec path on the same batch (DD_K3_HL=0). Also covers a mixed 4/8-elem shape."""
    rng = np.random.default_rng(777)
    for n in [4096 * 3, 100_000, 250_001]:  # multiple of R, ragged, very ragged
        cols = [random_col(rng, n, "i64", 0), random_col(rng, n, "f64", 0),
                random_col(rng, n, "f64", 0), random_col(rng, n, "i32", 0)]
        for keys, nparts in [([0], 128), ([3, 0], 100)]:
            monkeypatch.setenv("DD_K3_PRE", "0")  # force the HL tier (pre outranks it)
            monkeypatch.delenv("DD_K3_HL", raising=False)  # default-within-tier: HL
            check_against_oracle(cols, keys, nparts)
            monkeypatch.setenv("DD_K3_HL", "0")  # plain spec path
            check_against_oracle(cols, keys, nparts)
    n = 123_457
    monkeypatch.setenv("DD_K3_PRE", "0")
    cols = [random_col(rng, n, "i32", 0), random_col(rng, n, "i64", 0),
            random_col(rng, n, "f32", 0), random_col(rng, n, "f64", 0)]
    monkeypatch.delenv("DD_K3_HL", raising=False)
    check_against_oracle(cols, [0, 1], 128)
    check_against_oracle(cols, [2, 3], 100)

def test_hlg_ab(monkeypatch):
    """Generalized hidden-load scatter (k_scatter_hlg, whitelisted 5-col multikey and
    7-col q1 shapes) vs the plain spec path (DD_K3_HL=0), both bit-exact vs the
    oracle, including ragged rounds and non-pow2 P."""
    rng = np.random.default_rng(999)
    for n in [100_000, 250_001]:
        mk = [random_col(rng, n, "i64", 0), random_col(rng, n, "f64", 0),
              random_col(rng, n, "f64", 0), random_col(rng, n, "i32", 0),
              random_col(rng, n, "i32", 0)]
        q1 = [random_col(rng, n, "u8", 0), random_col(rng, n, "bool", 0),
              random_col(rng, n, "f64", 0), random_col(rng, n, "f64", 0),
              random_col(rng, n, "f64", 0), random_col(rng, n, "f64", 0),
              random_col(rng, n, "i32", 0)]
        for cols, keys, nparts in [(mk, [0, 4], 128), (mk, [0], 100),
                                   (q1, [0, 1], 128), (q1, [0, 1], 7)]:
            monkeypatch.setenv("DD_K3_PRE", "0")  # force the hlg tier (pre outranks it)
            monkeypatch.delenv("DD_K3_HL", raising=False)  # default-within-tier: hlg
            check_against_oracle(cols, keys, nparts)
            monkeypatch.setenv("DD_K3_HL", "0")
            check_against_oracle(cols, keys, nparts)


def test_pre_ab(monkeypatch):
    '''K3-P precomputed-layout scatter (k_scatter_pre, the round-2 default for its
    instantiated shapes; dd_kernels.hip K3-P header) must match the oracle bit-exactly
    across every NBG tier (P<=128 / <=256 / <=512), ragged and tiny inputs, non-pow2 P,
    and the multikey/q1 shapes; the same batches through DD_K3_PRE=0 (HL/spec tier)
    must agree too.'''
    rng = np.random.default_rng(20_24)
    for n in [4096 * 3, 100_000, 250_001, 65, 1]:  # round-multiple, ragged, tiny
        cols = [random_col(rng, n, "i64", 0), random_col(rng, n, "f64", 0),
                random_col(rng, n, "f64", 0), random_col(rng, n, "i32", 0)]
        for keys, nparts in [([0], 128), ([0], 100), ([0], 1), ([3, 0], 256),
                             ([0], 200), ([0], 512), ([0], 300)]:
            monkeypatch.delenv("DD_K3_PRE", raising=False)  # default: pre
            check_against_oracle(cols, keys, nparts)
            monkeypatch.setenv("DD_K3_PRE", "0")
            check_against_oracle(cols, keys, nparts)
    # mixed 4/8 shapes (tier 1) + multikey + q1 shapes through the pre path
    n = 123_457
    monkeypatch.delenv("DD_K3_PRE", raising=False)
    mixed = [random_col(rng, n, "i32", 0), random_col(rng, n, "i64", 0),
             random_col(rng, n, "f32", 0), random_col(rng, n, "f64", 0)]
    check_against_oracle(mixed, [0, 1], 128)
    check_against_oracle(mixed, [2, 3], 97)
    mk = [random_col(rng, n, "i64", 0), random_col(rng, n, "f64", 0),
          random_col(rng, n, "f64", 0), random_col(rng, n, "i32", 0),
          random_col(rng, n, "i32", 0)]
    check_against_oracle(mk, [0, 4], 128)
    check_against_oracle(mk, [0, 4], 512)  # multikey P<=512 tier (G2, NBG8)
    check_against_oracle(mk, [0, 4], 300)
    q1 = [random_col(rng, n, "u8", 0), random_col(rng, n, "bool", 0),
          random_col(rng, n, "f64", 0), random_col(rng, n, "f64", 0),
          random_col(rng, n, "f64", 0), random_col(rng, n, "f64", 0),
          random_col(rng, n, "i32", 0)]
    check_against_oracle(q1, [0, 1], 128)
    check_against_oracle(q1, [0, 1], 7)
