/* dd_reduce.hip — GPU partial aggregation below the shuffle.
 *
 * Mirrors the reference's partial-reduce pass (/root/reference/src/distributed_planner/
 * partial_reduce_below_network_shuffles.rs, enabled by `distributed.partial_reduce`,
 * distributed_config.rs:50-54): an AggregateExec mode=Partial is placed BELOW the
 * repartition so the exchange moves per-group partials instead of raw rows. Semantics are
 * DataFusion's partial aggregate: each invocation may emit DUPLICATE groups (one per
 * block here, one per input partition there); the downstream final aggregate merges them.
 * Output row order is unspecified (the reference's is stream-order; both are merged by
 * the order-insensitive final aggregate — test_utils/property_based.rs:15-41).
 *
 * Design: per-block LDS open-addressing table keyed by the normative row hash
 * (dd_row_hash — the same hash the shuffle uses). Claim a slot with an LDS CAS on the
 * hash word, publish key values behind a ready flag, aggregate with LDS atomics
 * (f64/u64 add). Rows that do not fit after a bounded probe spill to the output as
 * singleton groups (sum = value, count = 1) — correct for any cardinality, fast for the
 * low-cardinality keys partial-reduce targets (q1: 6 groups).
 *
 * Coverage: fixed-width key columns (nullable), aggregate ops
 * sum(f64) / sum(i64) / count(*) / min / max.
 *
 * NULL semantics: each aggregate carries a NON-NULL INPUT COUNT in the partial state
 * (out_nn), so the downstream final merge can emit NULL for a group whose inputs were all
 * NULL — matching DataFusion, where SUM/MIN/MAX over an all-null group is NULL, not the
 * op identity. (COUNT counts every row it sees; its merge can never be NULL.) */

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "dd_hash_device.h"
#include "dd_internal.h"

#define RWAVE 64
#define R_THREADS 256
#define R_CAP 1024      /* table slots per block */
#define R_PROBE 64      /* bounded probe; then spill */

/* agg op codes come from include/dd_shuffle.h (via dd_internal.h) */

/* order-preserving f64 <-> u64 map (IEEE total order; NaN sorts above +inf, matching
 * Arrow's max semantics): x >= 0 -> bits | sign, x < 0 -> ~bits */
__device__ __forceinline__ uint64_t dd_f64_key(double v) {
    uint64_t b = __double_as_longlong(v);
    return (b & 0x8000000000000000ULL) ? ~b : (b | 0x8000000000000000ULL);
}
__device__ __forceinline__ double dd_f64_unkey(uint64_t k) {
    uint64_t b = (k & 0x8000000000000000ULL) ? (k & 0x7fffffffffffffffULL) : ~k;
    return __longlong_as_double((long long)b);
}
__device__ __forceinline__ uint64_t dd_i64_key(int64_t v) {
    return (uint64_t)v ^ 0x8000000000000000ULL; /* order-preserving signed -> unsigned */
}
__device__ __forceinline__ int64_t dd_i64_unkey(uint64_t k) {
    return (int64_t)(k ^ 0x8000000000000000ULL);
}

__device__ __forceinline__ unsigned long long dd_agg_identity(int op) {
    switch (op) {
    case DD_AGG_SUM_F64: return (unsigned long long)__double_as_longlong(0.0);
    case DD_AGG_MIN_F64:
    case DD_AGG_MIN_I64: return ~0ULL; /* min over mapped-u64 keys */
    case DD_AGG_MAX_F64:
    case DD_AGG_MAX_I64: return 0ULL;
    default: return 0ULL; /* COUNT / SUM_I64 */
    }
}

/* canonical 64-bit key bits (zero-extended; floats canonicalized like the hash) */
__device__ __forceinline__ uint64_t dd_key_bits(const dd_kcol &c, int64_t i) {
    switch (c.dtype) {
    case DD_KDT_U8:
    case DD_KDT_BOOL:
        return (uint64_t)((const uint8_t *)c.data)[i];
    case DD_KDT_I16:
        return (uint64_t)((const uint16_t *)c.data)[i];
    case DD_KDT_I32:
        return (uint64_t)((const uint32_t *)c.data)[i];
    case DD_KDT_I64:
        return ((const uint64_t *)c.data)[i];
    case DD_KDT_F32: {
        float v = ((const float *)c.data)[i];
        if (v == 0.0f) v = 0.0f;
        uint32_t b = __float_as_uint(v);
        if (v != v) b = 0x7fc00000u;
        return (uint64_t)b;
    }
    case DD_KDT_F64: {
        double v = ((const double *)c.data)[i];
        if (v == 0.0) v = 0.0;
        uint64_t b = __double_as_longlong(v);
        if (v != v) b = 0x7ff8000000000000ULL;
        return b;
    }
    case DD_KDT_DICT32:
        /* group by dictionary INDEX: within one batch the dict is shared, so equal
         * indices == equal values; duplicate values under different indices would only
         * split a group, which partial aggregation permits (merged downstream by value) */
        return (uint64_t)(uint32_t)((const int32_t *)c.data)[i];
    default:
        return 0;
    }
}

extern "C" __global__ __launch_bounds__(R_THREADS) void k_partial_reduce(
    dd_kargs a, int64_t chunk_rows, int n_aggs, const int32_t *agg_cols,
    const int32_t *agg_ops, uint64_t *out_keys /* [max_rows][n_keys] interleaved */,
    uint32_t *out_keynull /* [max_rows] bitmask */,
    double *out_aggs /* [max_rows][n_aggs] interleaved (i64 sums bit-cast) */,
    uint64_t *out_nn /* [max_rows][n_aggs] non-null input counts */,
    uint64_t *out_n /* global row counter */) {
    __shared__ uint64_t t_hash[R_CAP];
    __shared__ uint32_t t_ready[R_CAP];
    __shared__ uint64_t t_keys[DD_KMAX_KEYS > 4 ? 4 : DD_KMAX_KEYS][R_CAP];
    __shared__ uint32_t t_null[R_CAP];
    __shared__ unsigned long long t_agg[4][R_CAP]; /* f64 or i64 state, bit pattern */
    __shared__ uint32_t t_nn[4][R_CAP];            /* non-null input count per agg */

    const int nk = a.n_keys > 4 ? 4 : a.n_keys;
    for (int s = threadIdx.x; s < R_CAP; s += R_THREADS) {
        t_hash[s] = 0;
        t_ready[s] = 0;
        t_null[s] = 0;
        for (int k = 0; k < nk; k++) t_keys[k][s] = 0;
        for (int g = 0; g < n_aggs; g++) {
            t_agg[g][s] = dd_agg_identity(agg_ops[g]);
            t_nn[g][s] = 0;
        }
    }
    __syncthreads();

    const int64_t start = (int64_t)blockIdx.x * chunk_rows;
    const int64_t end = (start + chunk_rows < a.n_rows) ? (start + chunk_rows) : a.n_rows;

    for (int64_t row = start + threadIdx.x; row < end; row += R_THREADS) {
        uint64_t h = dd_row_hash(a, row);
        if (h == 0) h = 1; /* 0 is the empty sentinel */
        uint64_t kb[4];
        uint32_t knull = 0;
        for (int k = 0; k < nk; k++) {
            const dd_kcol &c = a.cols[a.key_idx[k]];
            const bool isnull = c.valid && !c.valid[row];
            if (isnull) knull |= 1u << k;
            kb[k] = isnull ? 0 : dd_key_bits(c, row);
        }
        int slot = -1;
        /* Two probe attempts. Within one divergent loop iteration a follower lane may not
         * observe its own wave's claimer publish (no independent thread scheduling, and
         * masked branch bodies need not interleave); after the first attempt's loop
         * reconverges, every publish from this wave IS visible, so attempt 2 matches
         * deterministically instead of claiming a duplicate slot per lane. */
        for (int attempt = 0; attempt < 2 && slot < 0; attempt++) {
        uint32_t s = (uint32_t)(h % R_CAP);
        for (int probe = 0; probe < R_PROBE; probe++, s = (s + 1) % R_CAP) {
            if (attempt == 0 && probe > 0) break; /* attempt 1: claim-or-match at home slot
                                                     only; defer the walk to attempt 2 */
            uint64_t cur = atomicCAS((unsigned long long *)&t_hash[s], 0ull,
                                     (unsigned long long)h);
            if (cur == 0) { /* claimed: publish keys, then ready */
                for (int k = 0; k < nk; k++) t_keys[k][s] = kb[k];
                t_null[s] = knull;
                __threadfence_block();
                atomicExch(&t_ready[s], 1u);
                slot = (int)s;
                break;
            }
            if (cur == h) {
                /* NEVER spin unboundedly on the publisher: the claimer may be a divergent
                 * lane of THIS wave (no independent thread scheduling on CDNA — an
                 * unbounded spin would deadlock the wave). A BOUNDED recheck drains the
                 * common race (claimer in another wave); if still unpublished, keep
                 * probing — worst case the row lands as a duplicate group, which partial
                 * aggregation permits. */
                uint32_t rdy = atomicAdd(&t_ready[s], 0u);
                if (rdy) {
                    __threadfence_block();
                    bool eq = t_null[s] == knull;
                    for (int k = 0; k < nk && eq; k++) eq = t_keys[k][s] == kb[k];
                    if (eq) {
                        slot = (int)s;
                        break;
                    }
                }
            }
            /* different group (or unpublished slot): keep probing */
        }
        }
        if (slot >= 0) {
            for (int g = 0; g < n_aggs; g++) {
                switch (agg_ops[g]) {
                case DD_AGG_SUM_F64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    if (c.valid && !c.valid[row]) break;
                    atomicAdd((double *)&t_agg[g][slot], ((const double *)c.data)[row]);
                    atomicAdd(&t_nn[g][slot], 1u);
                    break;
                }
                case DD_AGG_SUM_I64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    if (c.valid && !c.valid[row]) break;
                    atomicAdd(&t_agg[g][slot],
                              (unsigned long long)((const uint64_t *)c.data)[row]);
                    atomicAdd(&t_nn[g][slot], 1u);
                    break;
                }
                case DD_AGG_MIN_F64:
                case DD_AGG_MAX_F64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    if (c.valid && !c.valid[row]) break;
                    uint64_t k = dd_f64_key(((const double *)c.data)[row]);
                    if (agg_ops[g] == DD_AGG_MIN_F64)
                        atomicMin(&t_agg[g][slot], (unsigned long long)k);
                    else
                        atomicMax(&t_agg[g][slot], (unsigned long long)k);
                    atomicAdd(&t_nn[g][slot], 1u);
                    break;
                }
                case DD_AGG_MIN_I64:
                case DD_AGG_MAX_I64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    if (c.valid && !c.valid[row]) break;
                    uint64_t k = dd_i64_key(((const int64_t *)c.data)[row]);
                    if (agg_ops[g] == DD_AGG_MIN_I64)
                        atomicMin(&t_agg[g][slot], (unsigned long long)k);
                    else
                        atomicMax(&t_agg[g][slot], (unsigned long long)k);
                    atomicAdd(&t_nn[g][slot], 1u);
                    break;
                }
                case DD_AGG_COUNT:
                    atomicAdd(&t_agg[g][slot], 1ull);
                    atomicAdd(&t_nn[g][slot], 1u);
                    break;
                }
            }
        } else {
            /* spill: emit a singleton group */
            uint64_t o = atomicAdd((unsigned long long *)out_n, 1ull);
            for (int k = 0; k < nk; k++) out_keys[o * nk + k] = kb[k];
            out_keynull[o] = knull;
            for (int g = 0; g < n_aggs; g++) {
                double v = 0;
                uint64_t nn = 1;
                switch (agg_ops[g]) {
                case DD_AGG_SUM_F64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    const bool null = c.valid && !c.valid[row];
                    if (null) nn = 0;
                    v = null ? 0.0 : ((const double *)c.data)[row];
                    out_aggs[o * n_aggs + g] = v;
                    break;
                }
                case DD_AGG_MIN_F64:
                case DD_AGG_MAX_F64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    const bool null = c.valid && !c.valid[row];
                    if (null) nn = 0;
                    out_aggs[o * n_aggs + g] =
                        null ? dd_f64_unkey(dd_agg_identity(agg_ops[g]))
                             : ((const double *)c.data)[row];
                    break;
                }
                case DD_AGG_MIN_I64:
                case DD_AGG_MAX_I64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    const bool null = c.valid && !c.valid[row];
                    if (null) nn = 0;
                    out_aggs[o * n_aggs + g] = __longlong_as_double(
                        null ? dd_i64_unkey(dd_agg_identity(agg_ops[g]))
                             : ((const int64_t *)c.data)[row]);
                    break;
                }
                case DD_AGG_SUM_I64: {
                    const dd_kcol &c = a.cols[agg_cols[g]];
                    const bool null = c.valid && !c.valid[row];
                    if (null) nn = 0;
                    uint64_t iv = null ? 0 : ((const uint64_t *)c.data)[row];
                    out_aggs[o * n_aggs + g] = __longlong_as_double((long long)iv);
                    break;
                }
                case DD_AGG_COUNT:
                    out_aggs[o * n_aggs + g] = __longlong_as_double(1ll);
                    break;
                }
                out_nn[o * n_aggs + g] = nn;
            }
        }
    }
    __syncthreads();

    /* flush occupied slots */
    for (int s = threadIdx.x; s < R_CAP; s += R_THREADS) {
        if (t_hash[s] == 0) continue;
        uint64_t o = atomicAdd((unsigned long long *)out_n, 1ull);
        for (int k = 0; k < nk; k++) out_keys[o * nk + k] = t_keys[k][s];
        out_keynull[o] = t_null[s];
        for (int g = 0; g < n_aggs; g++) {
            unsigned long long st = t_agg[g][s];
            switch (agg_ops[g]) {
            case DD_AGG_MIN_F64:
            case DD_AGG_MAX_F64:
                out_aggs[o * n_aggs + g] = dd_f64_unkey(st);
                break;
            case DD_AGG_MIN_I64:
            case DD_AGG_MAX_I64:
                out_aggs[o * n_aggs + g] = __longlong_as_double(dd_i64_unkey(st));
                break;
            default:
                out_aggs[o * n_aggs + g] = __longlong_as_double((long long)st);
            }
            out_nn[o * n_aggs + g] = (uint64_t)t_nn[g][s];
        }
    }
}

extern "C" hipError_t dd_launch_partial_reduce(const dd_kargs *a, int64_t nblocks,
                                               int64_t chunk_rows, int n_aggs,
                                               const int32_t *agg_cols,
                                               const int32_t *agg_ops, uint64_t *out_keys,
                                               uint32_t *out_keynull, double *out_aggs,
                                               uint64_t *out_nn, uint64_t *out_n,
                                               hipStream_t s) {
    hipLaunchKernelGGL(k_partial_reduce, dim3((unsigned)nblocks), dim3(R_THREADS), 0, s, *a,
                       chunk_rows, n_aggs, agg_cols, agg_ops, out_keys, out_keynull,
                       out_aggs, out_nn, out_n);
    return hipGetLastError();
}
