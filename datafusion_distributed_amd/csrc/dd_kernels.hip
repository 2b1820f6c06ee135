/* dd_kernels.hip — CDNA4 (gfx950) kernels for the hash-repartition hot path.
 *
 * Replaces the per-batch hot loop of DataFusion 55's RepartitionExec hash partitioning
 * (constructed at /root/reference/src/execution_plans/network_shuffle.rs:121-127 and
 * /root/reference/src/distributed_planner/network_boundary.rs:100-103): per row, hash the
 * key columns, part = hash % P_total, stable gather of every column into per-partition
 * contiguous outputs. Semantics are the normative spec of DESIGN.md §3 (bit-exact vs
 * oracle/dd_oracle.c).
 *
 * Two kernel paths share the scan infrastructure (host picks in dd_host.cpp):
 *
 *   v2 "staged" (default; fixed-width columns, and var-width via synthetic VARLEN/ROWID
 *   columns + the K4 byte pass): one contiguous row TILE per block; rounds of
 *   R = gmax*wpb*64 rows ranked with ballot-multisplit, staged partition-major in LDS,
 *   flushed with run-granular coalesced stores. k_hash_count_tile + k_scan_* +
 *   k_scatter_staged (+ K4). Stability: tiles ordered, wave segments contiguous within a
 *   round, lane order == row order inside a 64-row group.
 *
 *   v1 "direct" (fallback beyond DD_STAGE_MAXC columns): chunk per WAVE, per-wave LDS
 *   bases, direct scatter. k_hash_count + k_scan_* + k_scatter. Same stability guarantees.
 *
 * All integer/byte work — HBM-bound; no MFMA (indexing, not GEMM-shaped work).
 * Optimization history and PMC evidence: DESIGN.md §9, profiles/.
 */

#include <hip/hip_runtime.h>
#include <stdint.h>

#include <type_traits> /* integral_constant for the hlg compile-time column walk */

#include "dd_internal.h"

#define WAVE 64
#define WAVES_PER_BLOCK 4
#define BLOCK_THREADS (WAVE * WAVES_PER_BLOCK)

#include "dd_hash_device.h"

__global__ void k_dict_hashes(const uint8_t *bytes, const int32_t *offsets, int64_t n,
                              uint64_t *out) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i < n) out[i] = dd_hash_bytes_dev(bytes + offsets[i], (int64_t)(offsets[i + 1] - offsets[i]));
}

/* ---------------- K1: hash + per-chunk histogram ---------------- */

__global__ __launch_bounds__(BLOCK_THREADS) void k_hash_count(
    dd_kargs a, int64_t chunk_rows, uint32_t nparts, int nbits, uint32_t *pid_out,
    uint32_t *counts /* [nchunks][P] */, uint32_t *bcounts /* [nvar][nchunks][P] or null */) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int chunk = blockIdx.x * WAVES_PER_BLOCK + wid;

    /* LDS carve: hist[WPB][P] u32, then bhist[nvar][WPB][P] u32 */
    uint32_t *hist = (uint32_t *)smem + (size_t)wid * nparts;
    uint32_t *bhist0 = (uint32_t *)smem + (size_t)WAVES_PER_BLOCK * nparts;

    for (uint32_t p = lane; p < nparts; p += WAVE) {
        hist[p] = 0;
        for (int v = 0; v < a.n_var; v++)
            bhist0[((size_t)v * WAVES_PER_BLOCK + wid) * nparts + p] = 0;
    }
    /* single wave: LDS ops are program-ordered; no barrier needed */

    const int64_t start = (int64_t)chunk * chunk_rows;
    const int64_t end = (start + chunk_rows < a.n_rows) ? (start + chunk_rows) : a.n_rows;

    for (int64_t base = start; base < end; base += WAVE) {
        const int64_t row = base + lane;
        const bool active = row < end;
        uint32_t pid = 0;
        if (active) {
            uint64_t h = dd_row_hash(a, row);
            /* pid = (h % pid_total) >> pid_shift (coarse-bucket mode when shift > 0);
             * mask == mod for power-of-two totals (the common case) */
            const uint32_t tot = a.pid_total;
            uint32_t fine = ((tot & (tot - 1)) == 0) ? (uint32_t)(h & (uint64_t)(tot - 1))
                                                     : (uint32_t)(h % (uint64_t)tot);
            pid = fine >> a.pid_shift;
            pid_out[row] = pid;
        }
        if (active) {
            atomicAdd(&hist[pid], 1u); /* LDS atomic counting (tools/ablate_k1.cpp) */
            /* var-col byte sums: LDS atomic add (order-free) */
            for (int v = 0; v < a.n_var; v++) {
                const dd_kcol &c = a.cols[a.var_idx[v]];
                uint32_t len = (uint32_t)(c.offsets[row + 1] - c.offsets[row]);
                atomicAdd(&bhist0[((size_t)v * WAVES_PER_BLOCK + wid) * nparts + pid], len);
            }
        }
    }

    for (uint32_t p = lane; p < nparts; p += WAVE) {
        counts[(size_t)chunk * nparts + p] = hist[p];
        for (int v = 0; v < a.n_var; v++)
            bcounts[((size_t)v * gridDim.x * WAVES_PER_BLOCK + chunk) * nparts + p] =
                bhist0[((size_t)v * WAVES_PER_BLOCK + wid) * nparts + p];
    }
}

/* ---------------- K2: column-major exclusive scan of counts ----------------
 * counts is [nchunks][P] row-major; we scan down each partition column. 3 coalesced passes:
 *   K2a: each (range r, lane-partition p) sums its chunk range        -> partials[R][P]
 *   K2b: one block: scan partials per partition; scan totals across P -> part_offsets[P+1]
 *   K2c: each (r, p) re-walks, rewriting counts to global-exclusive (within partition)
 * For byte matrices the same kernels run with their own buffers. DD_SCAN_RANGES ranges. */

__global__ void k_scan_partial(const uint32_t *counts, int64_t nchunks, uint32_t nparts,
                               int nranges, uint32_t *partials /* [nranges][P] */) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t total = (int64_t)nranges * nparts;
    if (tid >= total) return;
    const int r = (int)(tid / nparts);
    const uint32_t p = (uint32_t)(tid % nparts);
    const int64_t c0 = nchunks * r / nranges, c1 = nchunks * (r + 1) / nranges;
    uint32_t s = 0;
    for (int64_t c = c0; c < c1; c++) s += counts[c * nparts + p];
    partials[(size_t)r * nparts + p] = s;
}

/* one block; scans partials in place (exclusive within partition), then exclusive-scans the
 * per-partition totals across partitions into part_offsets[P+1] (u64). */
__global__ void k_scan_combine(uint32_t *partials, int nranges, uint32_t nparts,
                               uint64_t *part_offsets) {
    __shared__ uint64_t totals[DD_MAX_P];
    for (uint32_t p = threadIdx.x; p < nparts; p += blockDim.x) {
        uint32_t run = 0;
        for (int r = 0; r < nranges; r++) {
            uint32_t v = partials[(size_t)r * nparts + p];
            partials[(size_t)r * nparts + p] = run;
            run += v;
        }
        totals[p] = run;
    }
    __syncthreads();
    if (threadIdx.x == 0) { /* P <= 2048: serial scan is microseconds */
        uint64_t run = 0;
        for (uint32_t p = 0; p < nparts; p++) {
            uint64_t v = totals[p];
            part_offsets[p] = run;
            run += v;
        }
        part_offsets[nparts] = run;
    }
}

__global__ void k_scan_rewrite(uint32_t *counts, int64_t nchunks, uint32_t nparts,
                               int nranges, const uint32_t *partials,
                               const uint64_t *fold_offsets /* null, or part_offsets to
                                                               make bases GLOBAL (pre) */) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t total = (int64_t)nranges * nparts;
    if (tid >= total) return;
    const int r = (int)(tid / nparts);
    const uint32_t p = (uint32_t)(tid % nparts);
    const int64_t c0 = nchunks * r / nranges, c1 = nchunks * (r + 1) / nranges;
    uint32_t run = partials[(size_t)r * nparts + p];
    if (fold_offsets) run += (uint32_t)fold_offsets[p];
    for (int64_t c = c0; c < c1; c++) {
        uint32_t v = counts[c * nparts + p];
        counts[c * nparts + p] = run;
        run += v;
    }
}

/* ---------------- K3: stable scatter ---------------- */

__global__ __launch_bounds__(BLOCK_THREADS) void k_scatter(
    dd_kargs a, int64_t chunk_rows, uint32_t nparts, int nbits, const uint32_t *pid_in,
    const uint32_t *chunk_off /* [nchunks][P] exclusive within partition */,
    const uint64_t *part_offsets /* [P+1] */,
    const uint32_t *chunk_boff /* [nvar][nchunks][P] or null */,
    const uint64_t *part_boffsets /* [nvar][P+1] or null */) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int chunk = blockIdx.x * WAVES_PER_BLOCK + wid;
    const int64_t nchunksx = (int64_t)gridDim.x * WAVES_PER_BLOCK;

    /* LDS carve (per wave): row_base u64[P]; per var: byte_base u64[P], brun u32[P];
     * then lens u32[WPB][64] staging for intra-group byte prefix */
    char *ws = smem;
    uint64_t *row_base = (uint64_t *)ws + (size_t)wid * nparts;
    ws += sizeof(uint64_t) * WAVES_PER_BLOCK * nparts;
    uint64_t *byte_base0 = (uint64_t *)ws; /* [nvar][WPB][P] */
    ws += sizeof(uint64_t) * (size_t)a.n_var * WAVES_PER_BLOCK * nparts;
    uint32_t *lens_stage = (uint32_t *)ws + (size_t)wid * WAVE; /* [WPB][64] */

    const int64_t start = (int64_t)chunk * chunk_rows;
    const int64_t end = (start + chunk_rows < a.n_rows) ? (start + chunk_rows) : a.n_rows;

    /* stage bases: global row slot base per partition for this chunk */
    for (uint32_t p = lane; p < nparts; p += WAVE) {
        row_base[p] = part_offsets[p] + chunk_off[(size_t)chunk * nparts + p];
        for (int v = 0; v < a.n_var; v++)
            byte_base0[((size_t)v * WAVES_PER_BLOCK + wid) * nparts + p] =
                part_boffsets[(size_t)v * (nparts + 1) + p] +
                chunk_boff[((size_t)v * nchunksx + chunk) * nparts + p];
    }

    for (int64_t base = start; base < end; base += WAVE) {
        const int64_t row = base + lane;
        const bool active = row < end;
        const uint32_t pid = active ? pid_in[row] : 0u;
        const uint64_t act = __ballot(active);
        const uint64_t eq = active ? dd_eq_mask(pid, act, nbits) : 0;
        const uint64_t lt = ((uint64_t)1 << lane) - 1;
        const int rank = active ? (int)__popcll((unsigned long long)(eq & lt)) : 0;
        const int leader = active ? __ffsll((unsigned long long)eq) - 1 : 0;
        const int gsize = active ? (int)__popcll((unsigned long long)eq) : 0;

        uint64_t dst = 0;
        if (active) {
            /* leader reads + advances the base; others get it via shuffle */
            uint64_t b = 0;
            if (lane == leader) {
                b = row_base[pid];
                row_base[pid] = b + (uint64_t)gsize;
            }
            b = (uint64_t)__shfl((long long)b, leader);
            dst = b + (uint64_t)rank;

            /* fixed-width columns + validity */
            for (int c = 0; c < a.n_cols; c++) {
                const dd_kcol &col = a.cols[c];
                switch (col.elem) {
                case 1:
                    ((uint8_t *)col.out_data)[dst] = ((const uint8_t *)col.data)[row];
                    break;
                case 2:
                    ((uint16_t *)col.out_data)[dst] = ((const uint16_t *)col.data)[row];
                    break;
                case 4:
                    ((uint32_t *)col.out_data)[dst] = ((const uint32_t *)col.data)[row];
                    break;
                case 8:
                    ((uint64_t *)col.out_data)[dst] = ((const uint64_t *)col.data)[row];
                    break;
                default:
                    break; /* var col: handled below */
                }
                if (col.valid) col.out_valid[dst] = col.valid[row];
            }
        }

        /* var columns: intra-group byte prefix via LDS-staged lengths */
        for (int v = 0; v < a.n_var; v++) {
            const dd_kcol &col = a.cols[a.var_idx[v]];
            uint32_t len = 0;
            int32_t o0 = 0;
            if (active) {
                o0 = col.offsets[row];
                len = (uint32_t)(col.offsets[row + 1] - o0);
            }
            lens_stage[lane] = len; /* single wave: program-ordered LDS */
            uint32_t bpre = 0, bsum = 0;
            if (active) {
                uint64_t m = eq;
                while (m) {
                    int j = __ffsll((unsigned long long)m) - 1;
                    uint32_t lj = lens_stage[j];
                    if (j < lane) bpre += lj;
                    bsum += lj;
                    m &= m - 1;
                }
                uint64_t *bb = &byte_base0[((size_t)v * WAVES_PER_BLOCK + wid) * nparts + pid];
                uint64_t bbase = 0;
                if (lane == leader) {
                    bbase = *bb;
                    *bb = bbase + bsum;
                }
                bbase = (uint64_t)__shfl((long long)bbase, leader);
                uint64_t bdst = bbase + bpre;
                col.out_lengths[dst] = len;
                const uint8_t *src = (const uint8_t *)col.data + o0;
                uint8_t *d = (uint8_t *)col.out_data + bdst;
                /* 8-byte unaligned chunks (native on gfx950); destinations of different
                 * rows are disjoint so wide stores cannot race */
                uint32_t b = 0;
                for (; b + 8 <= len; b += 8) {
                    uint64_t t;
                    __builtin_memcpy(&t, src + b, 8);
                    __builtin_memcpy(d + b, &t, 8);
                }
                for (; b < len; b++) d[b] = src[b];
            }
        }
    }
}

/* ================= v2: block-tile, LDS-staged scatter (fixed-width batches) =========
 * Motivation (profiles/r01_k3v1_pmc_summary.json): v1's direct scatter writes 3.8x its
 * algorithmic bytes — scattered 8 B stores dirty 128 B lines that are evicted from the
 * XCD L2 before they fill. v2 stages each round of R rows in partition-major order in
 * LDS, then flushes with run-granular coalesced stores (run = R/P rows per partition),
 * so stores leave the CU in >=line-sized pieces at the bench shape (R=2048, P=128:
 * 16-row runs = 128 B per 8 B column).
 *
 * Work decomposition: one contiguous row tile per BLOCK (nchunks == gridDim.x); within a
 * tile, rounds of R rows; within a round, wave w owns rows [w*R/4, (w+1)*R/4) (so row
 * order == (wave, group, lane) order — stability preserved vs the oracle). Fixed-width
 * columns only; batches with var-width columns take the v1 path. */

__global__ __launch_bounds__(BLOCK_THREADS) void k_hash_count_tile(
    dd_kargs a, int64_t tile_rows, uint32_t nparts, int nbits, uint32_t *pid_out,
    uint32_t *counts /* [nblocks][P] */) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    uint32_t *hist = (uint32_t *)smem; /* [4][P] per-wave */
    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *myhist = hist + (size_t)wid * nparts;

    for (uint32_t p = tid; p < WAVES_PER_BLOCK * nparts; p += BLOCK_THREADS) hist[p] = 0;
    __syncthreads();

    const int64_t tstart = (int64_t)blockIdx.x * tile_rows;
    const int64_t tend = (tstart + tile_rows < a.n_rows) ? (tstart + tile_rows) : a.n_rows;

    /* 2 independent row-groups per iteration: doubles loads in flight per wave and
     * halves the serialized LDS-histogram chain per row */
    for (int64_t base = tstart; base < tend; base += 2 * BLOCK_THREADS) {
        uint32_t pidu[2];
        bool actu[2];
#pragma unroll
        for (int u = 0; u < 2; u++) {
            const int64_t row = base + u * BLOCK_THREADS + tid;
            actu[u] = row < tend;
            pidu[u] = 0;
            if (actu[u]) {
                uint64_t h = dd_row_hash(a, row);
                /* pid = (h % pid_total) >> pid_shift (coarse-bucket mode) */
                const uint32_t tot = a.pid_total;
                uint32_t fine = ((tot & (tot - 1)) == 0)
                                    ? (uint32_t)(h & (uint64_t)(tot - 1))
                                    : (uint32_t)(h % (uint64_t)tot);
                pidu[u] = fine >> a.pid_shift;
                if (pid_out) pid_out[row] = pidu[u]; /* null when K3 recomputes (rhash) */
            }
        }
#pragma unroll
        for (int u = 0; u < 2; u++) {
            /* LDS atomicAdd counting: measured faster than ballot-multisplit
             * (tools/ablate_k1.cpp) */
            if (actu[u]) atomicAdd(&myhist[pidu[u]], 1u);
        }
    }
    __syncthreads();
    for (uint32_t p = tid; p < nparts; p += BLOCK_THREADS) {
        uint32_t s = 0;
        for (int w = 0; w < WAVES_PER_BLOCK; w++) s += hist[(size_t)w * nparts + p];
        counts[(size_t)blockIdx.x * nparts + p] = s;
    }
}

/* exclusive scan of vals[0..P) into out[0..P); tmp is u32[BLOCK_THREADS]; 3 barriers:
 * per-thread span sums -> ONE wave shfl-scans the 256 span totals -> spans rewritten */
template <int BT>
__device__ __forceinline__ void dd_block_excl_scan(const uint32_t *vals, uint32_t *out,
                                                   uint32_t P, uint32_t *tmp) {
    const int tid = threadIdx.x;
    const uint32_t span = (P + BT - 1) / BT;
    const uint32_t lo = tid * span;
    const uint32_t hi = (lo + span < P) ? lo + span : P;
    uint32_t s = 0;
    for (uint32_t i = lo; i < hi; i++) s += vals[i];
    tmp[tid] = s;
    __syncthreads();
    if (tid < WAVE) {
        uint32_t carry = 0;
        for (int k = 0; k < BT / WAVE; k++) {
            uint32_t v = tmp[k * WAVE + tid];
#pragma unroll
            for (int d = 1; d < WAVE; d <<= 1) {
                uint32_t u = (uint32_t)__shfl_up((int)v, d);
                if (tid >= d) v += u;
            }
            v += carry;
            tmp[k * WAVE + tid] = v; /* inclusive */
            carry = (uint32_t)__shfl((int)v, WAVE - 1);
        }
    }
    __syncthreads();
    uint32_t run = (tid > 0) ? tmp[tid - 1] : 0;
    for (uint32_t i = lo; i < hi; i++) {
        out[i] = run;
        run += vals[i];
    }
    __syncthreads();
}

/* NC > 0: compile-time column count, no validity, no var (specialized fast path — the
 * generic runtime column loop with break-guards and validity checks measured ~8-15%
 * slower at the bench shape, tools/ablate_k3.cpp). NC == 0: generic.
 * RHASH: recompute the row hash from the preloaded register values instead of reading
 * pid_in — the key columns are already in colv, so the pid array's HBM round trip
 * (write in K1 + read here) disappears. Spec paths only (host gate: ka.rhash). */
template <int GMAX, int WPB, int MAXC, bool HASVAR, int NC = 0, bool RHASH = false>
__global__ __launch_bounds__(WPB * WAVE) void k_scatter_staged(
    dd_kargs a, int64_t tile_rows, uint32_t nparts, int nbits, const uint32_t *pid_in,
    const uint32_t *tile_off /* [nblocks][P] excl within partition */,
    const uint64_t *part_offsets /* [P+1] */) {
    constexpr int BT = WPB * WAVE;
    constexpr int R = GMAX * BT;
    constexpr int SEG = R / WPB;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* carve: dstbase u64[P] | seghist u32[WPB][P] | roundcnt u32[P] | round_off u32[P] |
     * scan_tmp u32[BT] | dstg u32[R] | per-col staging (walked incrementally) */
    char *ws = smem;
    uint64_t *dstbase = (uint64_t *)ws;
    ws += sizeof(uint64_t) * nparts;
    uint32_t *seghist = (uint32_t *)ws;
    ws += sizeof(uint32_t) * WPB * nparts;
    uint32_t *roundcnt = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *round_off = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *scan_tmp = (uint32_t *)ws;
    ws += sizeof(uint32_t) * BT;
    uint32_t *dstg = (uint32_t *)ws;
    ws += sizeof(uint32_t) * R;
    char *const stage0 = ws;

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *myseg = seghist + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    const int64_t tstart = (int64_t)blockIdx.x * tile_rows;
    const int64_t tend = (tstart + tile_rows < a.n_rows) ? (tstart + tile_rows) : a.n_rows;

    for (uint32_t p = tid; p < nparts; p += BT)
        dstbase[p] = part_offsets[p] + tile_off[(size_t)blockIdx.x * nparts + p];
    __syncthreads();

    /* per-round register state; (re)loaded by `preload` one round AHEAD so the global
     * loads overlap the previous round's flush */
    uint32_t pidr[GMAX], rankr[GMAX];
    bool actr[GMAX];
    uint64_t colv[GMAX][MAXC];
    uint8_t valv[GMAX][MAXC];

    auto preload = [&](int64_t rstart, int64_t rend) {
        const int64_t segstart = rstart + (int64_t)wid * SEG;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = segstart + g * WAVE + lane;
            const bool active = row < rend;
            actr[g] = active;
            if (!active) {
                pidr[g] = 0;
                continue;
            }
            if constexpr (!RHASH) {
                pidr[g] = pid_in[row];
            } else {
                /* KEY loads issued first and hashed immediately: pidr then depends only
                 * on these loads, not the whole preload — hashing after the full colv
                 * loop made rank() wait on every column's load (measured +30% K3). The
                 * colv loop below re-reads the key columns from the just-fetched line.
                 * Same normative pid as K1 (dd_row_hash over always-valid int keys). */
                uint64_t h = 0;
                for (int k = 0; k < a.n_keys; k++) {
                    const dd_kcol &col = a.cols[a.key_idx[k]];
                    uint64_t bits = 0;
                    switch (col.elem) {
                    case 1: bits = ((const uint8_t *)col.data)[row]; break;
                    case 2: bits = ((const uint16_t *)col.data)[row]; break;
                    case 4: bits = ((const uint32_t *)col.data)[row]; break;
                    case 8: bits = ((const uint64_t *)col.data)[row]; break;
                    }
                    uint64_t vh = dd_mix64(bits);
                    h = h ^ (vh + 0x9e3779b97f4a7c15ULL + (h << 6) + (h >> 2));
                }
                const uint32_t tot = a.pid_total;
                uint32_t fine = ((tot & (tot - 1)) == 0)
                                    ? (uint32_t)(h & (uint64_t)(tot - 1))
                                    : (uint32_t)(h % (uint64_t)tot);
                pidr[g] = fine >> a.pid_shift;
            }
#pragma unroll
            for (int c = 0; c < (NC > 0 ? NC : MAXC); c++) {
                if (NC == 0 && c >= a.n_cols) break;
                const dd_kcol &col = a.cols[c];
                bool synthetic = false;
                if constexpr (HASVAR) {
                    if (col.dtype == DD_KDT_VARLEN) {
                        /* synthetic: the var column's per-row byte length (K4 header) */
                        const int32_t *off = (const int32_t *)col.data;
                        colv[g][c] = (uint32_t)(off[row + 1] - off[row]);
                        synthetic = true;
                    } else if (col.dtype == DD_KDT_ROWID) {
                        colv[g][c] = (uint32_t)row; /* synthetic: permutation for K4d */
                        synthetic = true;
                    }
                }
                if (!synthetic) {
                    switch (col.elem) {
                    case 1: colv[g][c] = ((const uint8_t *)col.data)[row]; break;
                    case 2: colv[g][c] = ((const uint16_t *)col.data)[row]; break;
                    case 4: colv[g][c] = ((const uint32_t *)col.data)[row]; break;
                    case 8: colv[g][c] = ((const uint64_t *)col.data)[row]; break;
                    }
                }
                if (NC == 0 && col.valid) valv[g][c] = col.valid[row];
            }
        }
    };

    /* rank: LDS-only, consumes the preloaded pidr. Each wave zeroes and owns its own
     * seghist row. Runs for round r+1 BEFORE the flush of round r: the compiler's
     * vmcnt(0) drain at the first pidr use then hits only stores issued a full round ago
     * (the ablation showed flush-store time fully serialized with the next round
     * otherwise — profiles/r01_k3_ablation.json). */
    auto rank = [&]() {
        for (uint32_t p = lane; p < nparts; p += WAVE) myseg[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = dd_eq_mask(pid, act, nbits);
                int leader = __ffsll((unsigned long long)eq) - 1;
                uint32_t base = 0;
                if (lane == leader) {
                    base = myseg[pid];
                    myseg[pid] = base + (uint32_t)__popcll((unsigned long long)eq);
                }
                base = (uint32_t)__shfl((int)base, leader);
                rk = base + (uint32_t)__popcll((unsigned long long)(eq & lt));
            }
            rankr[g] = rk;
        }
    };

    {
        const int64_t rend0 = (tstart + R < tend) ? (tstart + R) : tend;
        if (tstart < tend) {
            preload(tstart, rend0);
            rank();
        }
    }
    __syncthreads();

    for (int64_t rstart = tstart; rstart < tend; rstart += R) {
        const int64_t rend = (rstart + R < tend) ? (rstart + R) : tend;
        const int round_rows = (int)(rend - rstart);

        /* cross-wave exclusive scan per partition + round totals, fused with the
         * partition-offset scan (contiguous spans; one wave shfl-scans the span sums) */
        {
            const uint32_t span = (nparts + BT - 1) / BT;
            const uint32_t plo = tid * span;
            const uint32_t phi = (plo + span < nparts) ? plo + span : nparts;
            uint32_t ssum = 0;
            for (uint32_t p = plo; p < phi; p++) {
                uint32_t run = 0;
#pragma unroll
                for (int w = 0; w < WPB; w++) {
                    uint32_t v = seghist[(size_t)w * nparts + p];
                    seghist[(size_t)w * nparts + p] = run;
                    run += v;
                }
                roundcnt[p] = run;
                ssum += run;
            }
            scan_tmp[tid] = ssum;
            __syncthreads();
            if (tid < WAVE) {
                uint32_t carry = 0;
                for (int k = 0; k < BT / WAVE; k++) {
                    uint32_t v = scan_tmp[k * WAVE + tid];
#pragma unroll
                    for (int d = 1; d < WAVE; d <<= 1) {
                        uint32_t u = (uint32_t)__shfl_up((int)v, d);
                        if (tid >= d) v += u;
                    }
                    v += carry;
                    scan_tmp[k * WAVE + tid] = v;
                    carry = (uint32_t)__shfl((int)v, WAVE - 1);
                }
            }
            __syncthreads();
            uint32_t run = (tid > 0) ? scan_tmp[tid - 1] : 0;
            for (uint32_t p = plo; p < phi; p++) {
                round_off[p] = run;
                run += roundcnt[p];
            }
            __syncthreads();
        }

        /* place rows into the partition-major LDS image (registers -> LDS) */
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (!actr[g]) continue;
            const uint32_t pid = pidr[g];
            const uint32_t rank_r = myseg[pid] + rankr[g];
            const uint32_t slot = round_off[pid] + rank_r;
            dstg[slot] = (uint32_t)(dstbase[pid] + rank_r);
            char *stage = stage0;
#pragma unroll
            for (int c = 0; c < (NC > 0 ? NC : MAXC); c++) {
                if (NC == 0 && c >= a.n_cols) break;
                const dd_kcol &col = a.cols[c];
                switch (col.elem) {
                case 1: ((uint8_t *)stage)[slot] = (uint8_t)colv[g][c]; break;
                case 2: ((uint16_t *)stage)[slot] = (uint16_t)colv[g][c]; break;
                case 4: ((uint32_t *)stage)[slot] = (uint32_t)colv[g][c]; break;
                case 8: ((uint64_t *)stage)[slot] = colv[g][c]; break;
                }
                stage += (size_t)R * col.elem;
                if (NC == 0 && col.valid) {
                    ((uint8_t *)stage)[slot] = valv[g][c];
                    stage += R;
                }
            }
        }
        __syncthreads();

        /* round r+1: issue its loads AND rank it before the flush — the loads overlap
         * the stores, and the pid-consumption drain lands on old stores */
        if (rstart + R < tend) {
            const int64_t nrend = (rstart + 2 * R < tend) ? (rstart + 2 * R) : tend;
            preload(rstart + R, nrend);
            rank();
        }

        /* flush: consecutive LDS slots -> consecutive global rows within each partition
         * run; stores coalesce into run-sized segments */
        for (int i = tid; i < round_rows; i += BT) {
            const uint64_t dst = dstg[i];
            char *stage = stage0;
#pragma unroll
            for (int c = 0; c < (NC > 0 ? NC : MAXC); c++) {
                if (NC == 0 && c >= a.n_cols) break;
                const dd_kcol &col = a.cols[c];
                switch (col.elem) {
                case 1: ((uint8_t *)col.out_data)[dst] = ((const uint8_t *)stage)[i]; break;
                case 2:
                    ((uint16_t *)col.out_data)[dst] = ((const uint16_t *)stage)[i];
                    break;
                case 4:
                    ((uint32_t *)col.out_data)[dst] = ((const uint32_t *)stage)[i];
                    break;
                case 8:
                    ((uint64_t *)col.out_data)[dst] = ((const uint64_t *)stage)[i];
                    break;
                }
                stage += (size_t)R * col.elem;
                if (NC == 0 && col.valid) {
                    col.out_valid[dst] = ((const uint8_t *)stage)[i];
                    stage += R;
                }
            }
        }
        /* advance per-partition bases (disjoint from the flush's LDS regions) */
        for (uint32_t p = tid; p < nparts; p += BT) dstbase[p] += roundcnt[p];
        __syncthreads();
    }
}

/* ================= HL: hidden-load scatter experiment (opt-in DD_K3_HL=1) ==========
 * The plain spec kernel's ISA carries 194 `s_waitcnt vmcnt(0)` FULL drains and zero
 * counted waits (llvm-objdump of k_scatter_staged<4,16,4,false,4>): every consumer wait
 * also drains the in-flight flush stores, serializing stores against the next round's
 * loads (SQ_WAIT 75 %, latency-bound). This clone hides the preload loads
 * from hipcc in inline asm and counts the VMEM queue BY HAND (guide §5 trap 4b: "hide
 * the register operand's loads in inline asm and count both queues by hand"):
 *   - per round: 4 pid loads issued first, then 16 column loads, then (next iteration)
 *     16 flush stores — vmcnt decrements in issue order, so
 *   - place waits vmcnt(16): column loads retired, flush stores of the previous round
 *     still in flight (never drained in-loop);
 *   - rank runs AFTER the flush and waits vmcnt(32): pid loads retired, this round's
 *     column loads + stores in flight.
 * Loads are unconditional with the row clamped to the round end so the instruction
 * count (and therefore the hand counts) is exact in ragged rounds; the flush is
 * likewise padded (clamped index rewrites the same value — idempotent). Gated shape:
 * GMAX=4, WPB=16, 4 fixed columns, no validity, element sizes in {4,8}^4 (16
 * instantiations). Waitcnt asm ties the loaded values as "+v" operands so no use can
 * be scheduled before the wait. DEFAULT for its shape since the A/B: K3 1.015 vs
 * 1.166 ms, headline +10.6 % on the same box (DD_K3_HL=0 reverts; profiles/).
 * Parity: tests/test_gpu_fuzz.py::test_hl_ab. */

#define HL_LD64(dst, p) \
    asm volatile("global_load_dwordx2 %0, %1, off" : "=v"(dst) : "v"(p))
#define HL_LD32(dst, p) asm volatile("global_load_dword %0, %1, off" : "=v"(dst) : "v"(p))

template <int E> struct hl_elem { using T = uint32_t; };
template <> struct hl_elem<8> { using T = uint64_t; };

template <int E, typename T>
__device__ __forceinline__ void hl_load(T &dst, const void *base, uint32_t row) {
    if constexpr (E == 8) {
        HL_LD64(dst, (const uint64_t *)base + row);
    } else {
        static_assert(E == 4, "HL instantiations cover elem 4/8 only");
        HL_LD32(dst, (const uint32_t *)base + row);
    }
}

template <int E0, int E1, int E2, int E3>
__global__ __launch_bounds__(16 * WAVE) void k_scatter_hl(
    dd_kargs a, int64_t tile_rows, uint32_t nparts, int nbits, const uint32_t *pid_in,
    const uint32_t *tile_off, const uint64_t *part_offsets) {
    constexpr int WPB = 16, GMAX = 4;
    constexpr int BT = WPB * WAVE;
    constexpr int R = GMAX * BT;
    constexpr int SEG = R / WPB;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    char *ws = smem;
    uint64_t *dstbase = (uint64_t *)ws;
    ws += sizeof(uint64_t) * nparts;
    uint32_t *seghist = (uint32_t *)ws;
    ws += sizeof(uint32_t) * WPB * nparts;
    uint32_t *roundcnt = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *round_off = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *scan_tmp = (uint32_t *)ws;
    ws += sizeof(uint32_t) * BT;
    uint32_t *dstg = (uint32_t *)ws;
    ws += sizeof(uint32_t) * R;
    char *const stage0 = ws;

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *myseg = seghist + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    const int64_t tstart = (int64_t)blockIdx.x * tile_rows;
    const int64_t tend = (tstart + tile_rows < a.n_rows) ? (tstart + tile_rows) : a.n_rows;

    for (uint32_t p = tid; p < nparts; p += BT)
        dstbase[p] = part_offsets[p] + tile_off[(size_t)blockIdx.x * nparts + p];
    __syncthreads();

    uint32_t pidr[GMAX], rankr[GMAX];
    bool actr[GMAX];
    typename hl_elem<E0>::T c0v[GMAX];
    typename hl_elem<E1>::T c1v[GMAX];
    typename hl_elem<E2>::T c2v[GMAX];
    typename hl_elem<E3>::T c3v[GMAX];

/* waits tie every value they guard as "+v" so uses cannot move above them */
#define HL_WAIT_PID(N)                                                                       \
    asm volatile("s_waitcnt vmcnt(" #N ")"                                                   \
                 : "+v"(pidr[0]), "+v"(pidr[1]), "+v"(pidr[2]), "+v"(pidr[3])::"memory")
#define HL_WAIT_COLV(N)                                                                      \
    asm volatile("s_waitcnt vmcnt(" #N ")"                                                   \
                 : "+v"(c0v[0]), "+v"(c0v[1]), "+v"(c0v[2]), "+v"(c0v[3]), "+v"(c1v[0]),     \
                   "+v"(c1v[1]), "+v"(c1v[2]), "+v"(c1v[3]), "+v"(c2v[0]), "+v"(c2v[1]),     \
                   "+v"(c2v[2]), "+v"(c2v[3]), "+v"(c3v[0]), "+v"(c3v[1]), "+v"(c3v[2]),     \
                   "+v"(c3v[3])::"memory")

    auto preload = [&](int64_t rstart, int64_t rend) {
        const int64_t segstart = rstart + (int64_t)wid * SEG;
        uint32_t rowc[GMAX];
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = segstart + g * WAVE + lane;
            actr[g] = row < rend;
            /* clamp: every load always issues so the hand counts stay exact */
            rowc[g] = (uint32_t)(actr[g] ? row : rend - 1);
        }
#pragma unroll
        for (int g = 0; g < GMAX; g++) HL_LD32(pidr[g], pid_in + rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_load<E0>(c0v[g], a.cols[0].data, rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_load<E1>(c1v[g], a.cols[1].data, rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_load<E2>(c2v[g], a.cols[2].data, rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_load<E3>(c3v[g], a.cols[3].data, rowc[g]);
    };

    auto rank = [&]() {
        for (uint32_t p = lane; p < nparts; p += WAVE) myseg[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = dd_eq_mask(pid, act, nbits);
                int leader = __ffsll((unsigned long long)eq) - 1;
                uint32_t base = 0;
                if (lane == leader) {
                    base = myseg[pid];
                    myseg[pid] = base + (uint32_t)__popcll((unsigned long long)eq);
                }
                base = (uint32_t)__shfl((int)base, leader);
                rk = base + (uint32_t)__popcll((unsigned long long)(eq & lt));
            }
            rankr[g] = rk;
        }
    };

    if (tstart < tend) {
        const int64_t rend0 = (tstart + R < tend) ? (tstart + R) : tend;
        preload(tstart, rend0);
        HL_WAIT_PID(16); /* 16 column loads stay in flight */
        rank();
        HL_WAIT_COLV(0); /* tile prologue only: drain once */
    }
    __syncthreads();

    for (int64_t rstart = tstart; rstart < tend; rstart += R) {
        const int64_t rend = (rstart + R < tend) ? (rstart + R) : tend;
        const int round_rows = (int)(rend - rstart);

        { /* fused cross-wave + partition-offset scan (same as k_scatter_staged) */
            const uint32_t span = (nparts + BT - 1) / BT;
            const uint32_t plo = tid * span;
            const uint32_t phi = (plo + span < nparts) ? plo + span : nparts;
            uint32_t ssum = 0;
            for (uint32_t p = plo; p < phi; p++) {
                uint32_t run = 0;
#pragma unroll
                for (int w = 0; w < WPB; w++) {
                    uint32_t v = seghist[(size_t)w * nparts + p];
                    seghist[(size_t)w * nparts + p] = run;
                    run += v;
                }
                roundcnt[p] = run;
                ssum += run;
            }
            scan_tmp[tid] = ssum;
            __syncthreads();
            if (tid < WAVE) {
                uint32_t carry = 0;
                for (int k = 0; k < BT / WAVE; k++) {
                    uint32_t v = scan_tmp[k * WAVE + tid];
#pragma unroll
                    for (int d = 1; d < WAVE; d <<= 1) {
                        uint32_t u = (uint32_t)__shfl_up((int)v, d);
                        if (tid >= d) v += u;
                    }
                    v += carry;
                    scan_tmp[k * WAVE + tid] = v;
                    carry = (uint32_t)__shfl((int)v, WAVE - 1);
                }
            }
            __syncthreads();
            uint32_t run = (tid > 0) ? scan_tmp[tid - 1] : 0;
            for (uint32_t p = plo; p < phi; p++) {
                round_off[p] = run;
                run += roundcnt[p];
            }
            __syncthreads();
        }

        /* place: column loads retired at vmcnt(16) — the previous round's 16 flush
         * stores are the only VMEM allowed to remain in flight */
        HL_WAIT_COLV(16);
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (!actr[g]) continue;
            const uint32_t pid = pidr[g];
            const uint32_t rank_r = myseg[pid] + rankr[g];
            const uint32_t slot = round_off[pid] + rank_r;
            dstg[slot] = (uint32_t)(dstbase[pid] + rank_r);
            char *stage = stage0;
            ((typename hl_elem<E0>::T *)stage)[slot] = c0v[g];
            stage += (size_t)R * E0;
            ((typename hl_elem<E1>::T *)stage)[slot] = c1v[g];
            stage += (size_t)R * E1;
            ((typename hl_elem<E2>::T *)stage)[slot] = c2v[g];
            stage += (size_t)R * E2;
            ((typename hl_elem<E3>::T *)stage)[slot] = c3v[g];
        }
        __syncthreads();

        const bool more = rstart + R < tend;
        if (more) {
            const int64_t nrend = (rstart + 2 * R < tend) ? (rstart + 2 * R) : tend;
            preload(rstart + R, nrend);
        }

        /* flush: padded to exactly GMAX iterations (clamped index rewrites the same
         * value) so the store count per lane is always GMAX*4 = 16 */
#pragma unroll
        for (int u = 0; u < GMAX; u++) {
            const int i = tid + u * BT;
            const int ic = (i < round_rows) ? i : (round_rows - 1);
            const uint64_t dst = dstg[ic];
            char *stage = stage0;
            ((typename hl_elem<E0>::T *)a.cols[0].out_data)[dst] =
                ((const typename hl_elem<E0>::T *)stage)[ic];
            stage += (size_t)R * E0;
            ((typename hl_elem<E1>::T *)a.cols[1].out_data)[dst] =
                ((const typename hl_elem<E1>::T *)stage)[ic];
            stage += (size_t)R * E1;
            ((typename hl_elem<E2>::T *)a.cols[2].out_data)[dst] =
                ((const typename hl_elem<E2>::T *)stage)[ic];
            stage += (size_t)R * E2;
            ((typename hl_elem<E3>::T *)a.cols[3].out_data)[dst] =
                ((const typename hl_elem<E3>::T *)stage)[ic];
        }

        if (more) {
            /* pid loads retired at vmcnt(32): this round's 16 column loads + 16 flush
             * stores stay in flight across the barrier */
            HL_WAIT_PID(32);
            rank();
        }
        for (uint32_t p = tid; p < nparts; p += BT) dstbase[p] += roundcnt[p];
        __syncthreads();
    }
#undef HL_WAIT_PID
#undef HL_WAIT_COLV
}

/* ---- generalized hidden-load scatter (k_scatter_hlg): same discipline as
 * k_scatter_hl for other column counts / widths / GMAX, with the wait counts computed
 * at compile time: per round L = GMAX*NC column loads and L padded flush stores, so
 *   place waits vmcnt(L)   (prev round's stores in flight),
 *   rank  waits vmcnt(2L)  (this round's col loads + stores in flight; needs 2L <= 63).
 * Ties are one tiny s_waitcnt per guarded value (the first does the real wait, the
 * rest are satisfied single-cycle SALU ops) — that sidesteps variadic asm operand
 * lists. Whitelisted instantiations only (launcher): the multikey bench shape
 * (8,8,8,4,4)@G4 and the q1 shape (1,1,8,8,8,8,4)@G2. */

template <int CNT, typename T>
__device__ __forceinline__ void hl_tie_wait(T &v) {
    asm volatile("s_waitcnt vmcnt(%1)" : "+v"(v) : "n"(CNT) : "memory");
}

template <int J, int NC, typename F>
__device__ __forceinline__ void hl_for(F &&f) {
    if constexpr (J < NC) {
        f(std::integral_constant<int, J>{});
        hl_for<J + 1, NC>(f);
    }
}

template <int G, int... Es>
__global__ __launch_bounds__(16 * WAVE) void k_scatter_hlg(
    dd_kargs a, int64_t tile_rows, uint32_t nparts, int nbits, const uint32_t *pid_in,
    const uint32_t *tile_off, const uint64_t *part_offsets) {
    constexpr int WPB = 16, GMAX = G;
    constexpr int BT = WPB * WAVE;
    constexpr int R = GMAX * BT;
    constexpr int SEG = R / WPB;
    constexpr int NC = sizeof...(Es);
    constexpr int EL[NC] = {Es...};
    constexpr int L = GMAX * NC;
    static_assert(2 * L <= 63, "vmcnt immediate is 6 bits");
    /* compile-time storage split: 8-byte columns -> u64 slots, narrower -> u32 slots */
    struct slots {
        int idx[NC];
        int n8, n4;
        constexpr slots() : idx{}, n8(0), n4(0) {
            for (int j = 0; j < NC; j++) idx[j] = (EL[j] == 8) ? n8++ : n4++;
        }
    };
    constexpr slots S{};

    extern __shared__ __attribute__((aligned(16))) char smem[];
    char *ws = smem;
    uint64_t *dstbase = (uint64_t *)ws;
    ws += sizeof(uint64_t) * nparts;
    uint32_t *seghist = (uint32_t *)ws;
    ws += sizeof(uint32_t) * WPB * nparts;
    uint32_t *roundcnt = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *round_off = (uint32_t *)ws;
    ws += sizeof(uint32_t) * nparts;
    uint32_t *scan_tmp = (uint32_t *)ws;
    ws += sizeof(uint32_t) * BT;
    uint32_t *dstg = (uint32_t *)ws;
    ws += sizeof(uint32_t) * R;
    char *const stage0 = ws;

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *myseg = seghist + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    const int64_t tstart = (int64_t)blockIdx.x * tile_rows;
    const int64_t tend = (tstart + tile_rows < a.n_rows) ? (tstart + tile_rows) : a.n_rows;

    for (uint32_t p = tid; p < nparts; p += BT)
        dstbase[p] = part_offsets[p] + tile_off[(size_t)blockIdx.x * nparts + p];
    __syncthreads();

    uint32_t pidr[GMAX], rankr[GMAX];
    bool actr[GMAX];
    uint64_t big[S.n8 > 0 ? S.n8 : 1][GMAX];
    uint32_t sml[S.n4 > 0 ? S.n4 : 1][GMAX];

    auto preload = [&](int64_t rstart, int64_t rend) {
        const int64_t segstart = rstart + (int64_t)wid * SEG;
        uint32_t rowc[GMAX];
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = segstart + g * WAVE + lane;
            actr[g] = row < rend;
            rowc[g] = (uint32_t)(actr[g] ? row : rend - 1); /* always issue (counts) */
        }
#pragma unroll
        for (int g = 0; g < GMAX; g++) HL_LD32(pidr[g], pid_in + rowc[g]);
        hl_for<0, NC>([&](auto jc) {
            constexpr int J = jc.value;
            const void *base = a.cols[J].data;
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if constexpr (EL[J] == 8) {
                    HL_LD64(big[S.idx[J]][g], (const uint64_t *)base + rowc[g]);
                } else if constexpr (EL[J] == 4) {
                    HL_LD32(sml[S.idx[J]][g], (const uint32_t *)base + rowc[g]);
                } else if constexpr (EL[J] == 2) {
                    asm volatile("global_load_ushort %0, %1, off"
                                 : "=v"(sml[S.idx[J]][g])
                                 : "v"((const uint16_t *)base + rowc[g]));
                } else {
                    asm volatile("global_load_ubyte %0, %1, off"
                                 : "=v"(sml[S.idx[J]][g])
                                 : "v"((const uint8_t *)base + rowc[g]));
                }
            }
        });
    };

    auto wait_pid = [&](auto cnt) {
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_tie_wait<cnt.value>(pidr[g]);
    };
    auto wait_colv = [&](auto cnt) {
        hl_for<0, NC>([&](auto jc) {
            constexpr int J = jc.value;
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if constexpr (EL[J] == 8) hl_tie_wait<cnt.value>(big[S.idx[J]][g]);
                else hl_tie_wait<cnt.value>(sml[S.idx[J]][g]);
            }
        });
    };

    auto rank = [&]() {
        for (uint32_t p = lane; p < nparts; p += WAVE) myseg[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = dd_eq_mask(pid, act, nbits);
                int leader = __ffsll((unsigned long long)eq) - 1;
                uint32_t base = 0;
                if (lane == leader) {
                    base = myseg[pid];
                    myseg[pid] = base + (uint32_t)__popcll((unsigned long long)eq);
                }
                base = (uint32_t)__shfl((int)base, leader);
                rk = base + (uint32_t)__popcll((unsigned long long)(eq & lt));
            }
            rankr[g] = rk;
        }
    };

    if (tstart < tend) {
        const int64_t rend0 = (tstart + R < tend) ? (tstart + R) : tend;
        preload(tstart, rend0);
        wait_pid(std::integral_constant<int, L>{});
        rank();
        wait_colv(std::integral_constant<int, 0>{}); /* prologue-only drain */
    }
    __syncthreads();

    for (int64_t rstart = tstart; rstart < tend; rstart += R) {
        const int64_t rend = (rstart + R < tend) ? (rstart + R) : tend;
        const int round_rows = (int)(rend - rstart);

        { /* fused cross-wave + partition-offset scan (same as k_scatter_staged) */
            const uint32_t span = (nparts + BT - 1) / BT;
            const uint32_t plo = tid * span;
            const uint32_t phi = (plo + span < nparts) ? plo + span : nparts;
            uint32_t ssum = 0;
            for (uint32_t p = plo; p < phi; p++) {
                uint32_t run = 0;
#pragma unroll
                for (int w = 0; w < WPB; w++) {
                    uint32_t v = seghist[(size_t)w * nparts + p];
                    seghist[(size_t)w * nparts + p] = run;
                    run += v;
                }
                roundcnt[p] = run;
                ssum += run;
            }
            scan_tmp[tid] = ssum;
            __syncthreads();
            if (tid < WAVE) {
                uint32_t carry = 0;
                for (int k = 0; k < BT / WAVE; k++) {
                    uint32_t v = scan_tmp[k * WAVE + tid];
#pragma unroll
                    for (int d = 1; d < WAVE; d <<= 1) {
                        uint32_t u = (uint32_t)__shfl_up((int)v, d);
                        if (tid >= d) v += u;
                    }
                    v += carry;
                    scan_tmp[k * WAVE + tid] = v;
                    carry = (uint32_t)__shfl((int)v, WAVE - 1);
                }
            }
            __syncthreads();
            uint32_t run = (tid > 0) ? scan_tmp[tid - 1] : 0;
            for (uint32_t p = plo; p < phi; p++) {
                round_off[p] = run;
                run += roundcnt[p];
            }
            __syncthreads();
        }

        wait_colv(std::integral_constant<int, L>{});
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (!actr[g]) continue;
            const uint32_t pid = pidr[g];
            const uint32_t rank_r = myseg[pid] + rankr[g];
            const uint32_t slot = round_off[pid] + rank_r;
            dstg[slot] = (uint32_t)(dstbase[pid] + rank_r);
            char *stage = stage0;
            hl_for<0, NC>([&](auto jc) {
                constexpr int J = jc.value;
                if constexpr (EL[J] == 8)
                    ((uint64_t *)stage)[slot] = big[S.idx[J]][g];
                else if constexpr (EL[J] == 4)
                    ((uint32_t *)stage)[slot] = sml[S.idx[J]][g];
                else if constexpr (EL[J] == 2)
                    ((uint16_t *)stage)[slot] = (uint16_t)sml[S.idx[J]][g];
                else
                    ((uint8_t *)stage)[slot] = (uint8_t)sml[S.idx[J]][g];
                stage += (size_t)R * EL[J];
            });
        }
        __syncthreads();

        const bool more = rstart + R < tend;
        if (more) {
            const int64_t nrend = (rstart + 2 * R < tend) ? (rstart + 2 * R) : tend;
            preload(rstart + R, nrend);
        }

        /* flush padded to GMAX iterations (clamped rewrite is idempotent): store count
         * per lane is always L */
#pragma unroll
        for (int u = 0; u < GMAX; u++) {
            const int i = tid + u * BT;
            const int ic = (i < round_rows) ? i : (round_rows - 1);
            const uint64_t dst = dstg[ic];
            char *stage = stage0;
            hl_for<0, NC>([&](auto jc) {
                constexpr int J = jc.value;
                void *out = a.cols[J].out_data;
                if constexpr (EL[J] == 8)
                    ((uint64_t *)out)[dst] = ((const uint64_t *)stage)[ic];
                else if constexpr (EL[J] == 4)
                    ((uint32_t *)out)[dst] = ((const uint32_t *)stage)[ic];
                else if constexpr (EL[J] == 2)
                    ((uint16_t *)out)[dst] = ((const uint16_t *)stage)[ic];
                else
                    ((uint8_t *)out)[dst] = ((const uint8_t *)stage)[ic];
                stage += (size_t)R * EL[J];
            });
        }

        if (more) {
            wait_pid(std::integral_constant<int, 2 * L>{});
            rank();
        }
        for (uint32_t p = tid; p < nparts; p += BT) dstbase[p] += roundcnt[p];
        __syncthreads();
    }
}

/* ================= K3-P: precomputed-layout hidden-load scatter (round 2) ==========
 * The HL kernel still spends ~0.4-0.5 ms/launch (of 1.1) on per-round machinery: the
 * 16-deep serial cross-wave seghist scan, the fused partition-offset scan (3 of the 5
 * barriers per round), and the serial dstbase accumulation chaining every round to the
 * previous one (profiles/r01_k3_ablation.json: "residual ~0.5 ms"). This kernel removes
 * ALL of it by precomputing the layout at WAVE-SEGMENT granularity in the K2 family:
 *
 *   - rounds are GLOBALLY aligned: round r = rows [r*R, (r+1)*R); segment s = r*WPB + w
 *     = rows [r*R + w*SEG, +SEG). K1 (k_hash_count_seg) writes per-SEGMENT histograms.
 *   - the K2 scan runs at segment granularity and REWRITES counts to global slot bases
 *     (within-partition exclusive + part_offsets fold): gbase[s][p] = first output slot
 *     of segment s's rows for partition p.
 *   - k_round_roff precomputes roff[r][p] (u16, ~4 KB/round): the partition-major
 *     LDS-image base of partition p within round r (exclusive scan of the round's
 *     partition counts) — reading only the ROUND-BOUNDARY rows of counts (a per-segment
 *     imgb table was measured first: +120 MB of K2 traffic for no K3 gain).
 *
 * K3 then needs NO cross-wave LDS traffic outside the two image barriers: each wave
 * hidden-loads its own gbase row + the round's roff row (wave-private LDS), ranks its
 * rows intra-segment with a per-wave LDS counter, and places at
 *   img  = roff[pid] + (gbase_w[pid] - gbase_w0[pid]) + rank    (w0 = wave 0's row,
 *                                     cross-wave LDS READ ordered by the round barrier)
 *   gdst = gbase_w[pid] + rank.
 * 2 barriers per round instead of 5, no serial scan, no cross-round dependency, uniform
 * full rounds everywhere except the single global tail; one round per BLOCK by default
 * (rpb=1 measured best — cross-block overlap beats in-block software pipelining here).
 * Hidden-load discipline identical to k_scatter_hl (issue order per round: G pid loads,
 * L=G*NC column loads, NB base loads, L flush stores); a single counted wait per round:
 *   wait vmcnt(L) after the flush (pid+cols+bases retired, stores stay in flight),
 * then the base rows are written to LDS and rank runs before the round barrier.
 * NBG/NBI are compile-time dword counts for the gbase/roff row loads (host gates P to
 * the instantiated tier); roff rows are padded to even u16 (sP2) so dword loads stay
 * aligned. Parity: bit-exact vs the oracle (tests/test_gpu_fuzz.py::test_pre_ab). */

template <int G, int W, int NBG, int NBI, int... Es>
__global__ __launch_bounds__(W * WAVE) void k_scatter_pre(
    dd_kargs a, int64_t nrounds, int rpb, uint32_t nparts, int nbits,
    const uint32_t *pid_in, const uint32_t *gbase /* [nseg][P] global slot bases */,
    const uint16_t *rofftab /* [nrounds][sP2] round image bases */, uint32_t sP2) {
    constexpr int WPB = W, GMAX = G;
    constexpr int BT = WPB * WAVE;
    constexpr int R = GMAX * BT;
    constexpr int SEG = R / WPB;
    constexpr int NC = sizeof...(Es);
    constexpr int EL[NC] = {Es...};
    constexpr int L = GMAX * NC;
    constexpr int NB = NBG + NBI;
    static_assert(2 * L + NB <= 63, "vmcnt immediate is 6 bits");
    struct slots {
        int idx[NC];
        int n8, n4;
        constexpr slots() : idx{}, n8(0), n4(0) {
            for (int j = 0; j < NC; j++) idx[j] = (EL[j] == 8) ? n8++ : n4++;
        }
        constexpr int rowb() const {
            int s = 0;
            for (int j = 0; j < NC; j++) s += EL[j];
            return s;
        }
    };
    constexpr slots S{};

    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* carve: stage | dstg u32[R] | per-wave gb u32[P] | per-wave rf u16[sP2] |
     * per-wave ms u16[P] */
    char *ws = smem;
    char *const stage0 = ws;
    ws += (size_t)R * S.rowb();
    uint32_t *dstg = (uint32_t *)ws;
    ws += sizeof(uint32_t) * R;
    uint32_t *gb_all = (uint32_t *)ws;
    ws += sizeof(uint32_t) * WPB * nparts;
    uint16_t *ib_all = (uint16_t *)ws;
    ws += sizeof(uint16_t) * WPB * sP2;
    uint16_t *ms_all = (uint16_t *)ws;

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *gb = gb_all + (size_t)wid * nparts;
    uint32_t *gb0 = gb_all; /* wave 0's row == the round's first-segment bases */
    uint32_t *ib32 = (uint32_t *)(ib_all + (size_t)wid * sP2); /* dword view */
    uint16_t *rf = ib_all + (size_t)wid * sP2;
    uint16_t *ms = ms_all + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    const int64_t r0 = (int64_t)blockIdx.x * rpb;
    const int64_t r1 = (r0 + rpb < nrounds) ? r0 + rpb : nrounds;
    if (r0 >= nrounds || a.n_rows == 0) return;

    uint32_t pidr[GMAX], rankr[GMAX];
    bool actr[GMAX];
    uint64_t big[S.n8 > 0 ? S.n8 : 1][GMAX];
    uint32_t sml[S.n4 > 0 ? S.n4 : 1][GMAX];
    uint32_t baser[NB]; /* NBG gbase dwords then NBI roff dwords */

    const uint32_t ndw = (nparts + 1) / 2; /* roff row dwords */

    auto preload = [&](int64_t r) {
        const int64_t rstart = r * R;
        const int64_t rend = (rstart + R < a.n_rows) ? rstart + R : a.n_rows;
        const int64_t segstart = rstart + (int64_t)wid * SEG;
        const int64_t seg = r * WPB + wid;
        uint32_t rowc[GMAX];
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = segstart + g * WAVE + lane;
            actr[g] = row < rend;
            rowc[g] = (uint32_t)(actr[g] ? row : rend - 1); /* always issue (counts) */
        }
        if (a.pid8) { /* uniform branch; exactly G pid loads issue either way */
#pragma unroll
            for (int g = 0; g < GMAX; g++)
                asm volatile("global_load_ubyte %0, %1, off"
                             : "=v"(pidr[g])
                             : "v"((const uint8_t *)pid_in + rowc[g]));
        } else {
#pragma unroll
            for (int g = 0; g < GMAX; g++) HL_LD32(pidr[g], pid_in + rowc[g]);
        }
        hl_for<0, NC>([&](auto jc) {
            constexpr int J = jc.value;
            const void *base = a.cols[J].data;
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if constexpr (EL[J] == 8) {
                    HL_LD64(big[S.idx[J]][g], (const uint64_t *)base + rowc[g]);
                } else if constexpr (EL[J] == 4) {
                    HL_LD32(sml[S.idx[J]][g], (const uint32_t *)base + rowc[g]);
                } else if constexpr (EL[J] == 2) {
                    asm volatile("global_load_ushort %0, %1, off"
                                 : "=v"(sml[S.idx[J]][g])
                                 : "v"((const uint16_t *)base + rowc[g]));
                } else {
                    asm volatile("global_load_ubyte %0, %1, off"
                                 : "=v"(sml[S.idx[J]][g])
                                 : "v"((const uint8_t *)base + rowc[g]));
                }
            }
        });
        /* base rows (wave-private): gbase u32[P] for MY segment, then the round's roff
         * u16[sP2] as dwords (same row for all 16 waves; the table is ~4 KB/round and
         * L2-hot, so the duplicate loads are free) */
        const uint32_t *grow = gbase + (size_t)seg * nparts;
#pragma unroll
        for (int k = 0; k < NBG; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx >= nparts) idx = nparts - 1; /* redundant re-load keeps counts exact */
            HL_LD32(baser[k], grow + idx);
        }
        const uint32_t *irow = (const uint32_t *)(rofftab + (size_t)r * sP2);
#pragma unroll
        for (int k = 0; k < NBI; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx >= ndw) idx = ndw - 1;
            HL_LD32(baser[NBG + k], irow + idx);
        }
    };

    auto wait_all = [&](auto cnt) { /* pid + columns + bases retired */
#pragma unroll
        for (int g = 0; g < GMAX; g++) hl_tie_wait<cnt.value>(pidr[g]);
        hl_for<0, NC>([&](auto jc) {
            constexpr int J = jc.value;
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if constexpr (EL[J] == 8) hl_tie_wait<cnt.value>(big[S.idx[J]][g]);
                else hl_tie_wait<cnt.value>(sml[S.idx[J]][g]);
            }
        });
#pragma unroll
        for (int k = 0; k < NB; k++) hl_tie_wait<cnt.value>(baser[k]);
    };

    auto write_base_rows = [&]() { /* wave-private LDS; ordered for OTHER waves (gb0
                                      readers) by the round barrier that follows */
#pragma unroll
        for (int k = 0; k < NBG; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx < nparts) gb[idx] = baser[k];
        }
#pragma unroll
        for (int k = 0; k < NBI; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx < ndw) ib32[idx] = baser[NBG + k];
        }
    };

    auto rank = [&]() {
        for (uint32_t p = lane; p < nparts; p += WAVE) ms[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = dd_eq_mask(pid, act, nbits);
                int leader = __ffsll((unsigned long long)eq) - 1;
                uint32_t base = 0;
                if (lane == leader) {
                    base = ms[pid];
                    ms[pid] = (uint16_t)(base + (uint32_t)__popcll((unsigned long long)eq));
                }
                base = (uint32_t)__shfl((int)base, leader);
                rk = base + (uint32_t)__popcll((unsigned long long)(eq & lt));
            }
            rankr[g] = rk;
        }
    };

    /* prologue: first round's loads, base rows, rank; barrier orders gb0 for place */
    preload(r0);
    wait_all(std::integral_constant<int, 0>{});
    write_base_rows();
    rank();
    __syncthreads();

    for (int64_t r = r0; r < r1; r++) {
        const int64_t rstart = r * R;
        const int round_rows =
            (int)(((rstart + R < a.n_rows) ? rstart + R : a.n_rows) - rstart);

        /* place: everything this round needs was waited on and LDS-staged before the
         * preceding barrier; the only VMEM in flight is the previous round's stores */
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (!actr[g]) continue;
            const uint32_t pid = pidr[g];
            const uint32_t rk = rankr[g];
            const uint32_t gd = gb[pid] + rk;
            const uint32_t slot = (uint32_t)rf[pid] + (gd - gb0[pid]);
            dstg[slot] = gd;
            char *stage = stage0;
            hl_for<0, NC>([&](auto jc) {
                constexpr int J = jc.value;
                if constexpr (EL[J] == 8)
                    ((uint64_t *)stage)[slot] = big[S.idx[J]][g];
                else if constexpr (EL[J] == 4)
                    ((uint32_t *)stage)[slot] = sml[S.idx[J]][g];
                else if constexpr (EL[J] == 2)
                    ((uint16_t *)stage)[slot] = (uint16_t)sml[S.idx[J]][g];
                else
                    ((uint8_t *)stage)[slot] = (uint8_t)sml[S.idx[J]][g];
                stage += (size_t)R * EL[J];
            });
        }
        __syncthreads();

        const bool more = r + 1 < r1;
        if (more) preload(r + 1);

        /* flush padded to GMAX iterations (clamped rewrite is idempotent). Stores are
         * PLAIN (cached): true non-temporal stores (global_store ... nt) measured 30%
         * SLOWER in the standalone replica (tools/ablate_pre.cpp: 1.51 vs 1.16 ms) —
         * partition runs average ~R/P bytes and their 128 B line edges only coalesce
         * through L2, which nt bypasses. (The earlier a.nt knob never actually emitted
         * nt bits: hipcc merged the branch arms and dropped the hint — the measured
         * "nt wins" were box noise. Knob removed.) */
#pragma unroll
        for (int u = 0; u < GMAX; u++) {
            const int i = tid + u * BT;
            const int ic = (i < round_rows) ? i : (round_rows - 1);
            const uint64_t dst = dstg[ic];
            char *stage = stage0;
            hl_for<0, NC>([&](auto jc) {
                constexpr int J = jc.value;
                void *out = a.cols[J].out_data;
                if constexpr (EL[J] == 8)
                    ((uint64_t *)out)[dst] = ((const uint64_t *)stage)[ic];
                else if constexpr (EL[J] == 4)
                    ((uint32_t *)out)[dst] = ((const uint32_t *)stage)[ic];
                else if constexpr (EL[J] == 2)
                    ((uint16_t *)out)[dst] = ((const uint16_t *)stage)[ic];
                else
                    ((uint8_t *)out)[dst] = ((const uint8_t *)stage)[ic];
                stage += (size_t)R * EL[J];
            });
        }

        if (more) {
            /* one counted wait per round: pid+cols+bases of round r+1 retired, the L
             * flush stores of round r stay in flight (never drained in-loop) */
            wait_all(std::integral_constant<int, L>{});
            write_base_rows();
            rank();
        }
        __syncthreads();
    }
}

/* K1 for the pre path: per-SEGMENT histograms (segment = one wave's SEG contiguous rows
 * of a globally-aligned round). 4 segments per 256-thread block; per-wave LDS hist.
 * Fuses the first scan pass: each wave atomicAdds its histogram into its range row of
 * `partials` (zeroed by the host first), so dd_launch_scan_deep skips k_scan_partial's
 * full re-read of counts. a.pid8=1 stores pids as u8. */
__global__ __launch_bounds__(BLOCK_THREADS) void k_hash_count_seg(
    dd_kargs a, int64_t nseg, int64_t seg_rows, uint32_t nparts, int nbits,
    uint32_t *pid_out, uint16_t *counts /* [nseg][P]; seg counts <= SEG <= 1024: u16
                                           halves the count-stream traffic */,
    uint32_t *partials /* [nranges][P], pre-zeroed */, int nranges) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    uint32_t *hist = (uint32_t *)smem + (size_t)wid * nparts;

    /* each wave owns a CONSECUTIVE run of segments (grid is capped: one block per
     * segment-quad was ~58k tiny dispatches, a measurable share of K1) */
    const int64_t wave_g = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
    const int64_t nwaves = (int64_t)gridDim.x * WAVES_PER_BLOCK;
    const int64_t per = (nseg + nwaves - 1) / nwaves;
    const int64_t s0 = wave_g * per;
    const int64_t s1 = (s0 + per < nseg) ? s0 + per : nseg;
    for (int64_t seg = s0; seg < s1; seg++) {
    for (uint32_t p = lane; p < nparts; p += WAVE) hist[p] = 0;
    /* single wave: LDS program order; no barrier */

    const int64_t start = seg * seg_rows;
    const int64_t end = (start + seg_rows < a.n_rows) ? start + seg_rows : a.n_rows;

    /* 4 independent row-groups per iteration (SEG = 256 at G4: the whole segment in one
     * iteration) — doubles the loads in flight vs the tile kernel's 2-group form */
    for (int64_t base = start; base < end; base += 4 * WAVE) {
        uint32_t pidu[4];
        bool actu[4];
#pragma unroll
        for (int u = 0; u < 4; u++) {
            const int64_t row = base + u * WAVE + lane;
            actu[u] = row < end;
            pidu[u] = 0;
            if (actu[u]) {
                uint64_t h = dd_row_hash(a, row);
                const uint32_t tot = a.pid_total;
                uint32_t fine = ((tot & (tot - 1)) == 0)
                                    ? (uint32_t)(h & (uint64_t)(tot - 1))
                                    : (uint32_t)(h % (uint64_t)tot);
                pidu[u] = fine >> a.pid_shift;
                if (a.pid8) ((uint8_t *)pid_out)[row] = (uint8_t)pidu[u];
                else pid_out[row] = pidu[u];
            }
        }
#pragma unroll
        for (int u = 0; u < 4; u++) {
            /* LDS atomicAdd counting: measured 30% FASTER than the ballot-multisplit
             * leader-add on this shape (tools/ablate_k1.cpp: 0.143 vs 0.202 ms) — the
             * log2(P) ballot chain costs more than uniform-key LDS atomic contention */
            if (actu[u]) atomicAdd(&hist[pidu[u]], 1u);
        }
    }
    /* range of this segment under k_scan_partial's fair-division mapping
     * c0(r) = floor(nseg*r/nranges): r = ceil((seg+1)*nranges/nseg) - 1 */
    const int64_t rw = ((seg + 1) * (int64_t)nranges + nseg - 1) / nseg - 1;
    uint32_t *prow = partials + (size_t)rw * nparts;
    for (uint32_t p = lane; p < nparts; p += WAVE) {
        const uint32_t h = hist[p];
        counts[(size_t)seg * nparts + p] = (uint16_t)h;
        if (h) atomicAdd(&prow[p], h);
    }
    } /* segment loop */
}

/* K2d for the pre path: per-round image bases. counts must already be rewritten to
 * GLOBAL slot bases (k_scan_rewrite with the part_offsets fold). One WAVE per round
 * (shfl scan, zero barriers — the one-block-per-round form with dd_block_excl_scan's 3
 * barriers measured ~0.1 ms of pure launch/latency for 14.6k tiny blocks), touching only
 * the ROUND-BOUNDARY rows of counts:
 *   roundcnt[p]  = gbase[seg0_{r+1}][p] - gbase[seg0_r][p]   (tail: part_offsets[p+1])
 *   roff[r][p]   = exclusive scan over p of roundcnt          (u16, <= R) */
__global__ __launch_bounds__(BLOCK_THREADS) void k_round_roff(
    const uint32_t *counts, const uint64_t *part_offsets, int64_t nrounds, int wpb,
    uint32_t nparts, uint32_t sP2, uint16_t *rofftab) {
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int64_t r = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
    if (r >= nrounds) return;
    const uint32_t *rb = counts + (size_t)r * wpb * nparts;
    const uint32_t *re = (r + 1 < nrounds) ? counts + (size_t)(r + 1) * wpb * nparts : nullptr;
    uint16_t *orow = rofftab + (size_t)r * sP2;
    uint32_t carry = 0;
    for (uint32_t p0 = 0; p0 < nparts; p0 += WAVE) {
        const uint32_t p = p0 + lane;
        uint32_t c = 0;
        if (p < nparts) c = (re ? re[p] : (uint32_t)part_offsets[p + 1]) - rb[p];
        uint32_t v = c;
#pragma unroll
        for (int d = 1; d < WAVE; d <<= 1) {
            uint32_t u = (uint32_t)__shfl_up((int)v, d);
            if (lane >= d) v += u;
        }
        if (p < nparts) orow[p] = (uint16_t)(carry + v - c); /* exclusive */
        carry += (uint32_t)__shfl((int)v, WAVE - 1);
    }
}

/* ================= K5: LDS-staged var-byte scatter (round 2) =================
 * Replaces K4's output-slot GATHER (k4_copy) for var batches whose strings are small:
 * the gather reads a random ~64 B string per slot and fetches whole 128 B lines —
 * measured 2.79 GB HBM reads for 1.51 GB algorithmic on ClickBench (read amp ~1.9x,
 * profiles/r02_clickbench_k4.json) and grid-depth-insensitive (not latency-bound).
 * K5 instead walks the INPUT in row order (coalesced reads), places each row's bytes
 * into a partition-major LDS byte image, and flushes per-partition byte runs with
 * shift-aligned 8 B stores (coalesced writes) — the K3-P precompute discipline applied
 * to bytes:
 *   - rounds of R5 = 1024 rows (16 waves x one 64-row group); bcounts[seg][p] = bytes
 *     of segment seg for partition p (k5_count), scanned to GLOBAL byte bases (the
 *     shared dd_launch_scan_deep + part_offsets fold), per-round image bases
 *     roffB[r][p..P] incl. total (k5_roff);
 *   - eligibility: ONE var column, max string length <= DD_K5_MAXLEN (the LDS image is
 *     R5 * maxlen-bounded; k5_maxlen computes it at create), var bytes < 2^31; longer
 *     strings keep the K4 gather (they are line-dense already).
 * Byte-granular LDS writes use head-bytes / aligned-dword / tail-bytes splits (LDS
 * requires natural alignment, unlike gfx950 global unaligned 8 B ops). Stability and
 * agreement with K4c's slot-side offsets hold by construction: both are prefix sums of
 * the same lengths in the same partition-major, row-stable order. */

__global__ void k5_maxlen(const int32_t *offsets, int64_t n, uint32_t *out_max) {
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    uint32_t len = 0;
    if (i < n) len = (uint32_t)(offsets[i + 1] - offsets[i]);
#pragma unroll
    for (int d = 32; d >= 1; d >>= 1) {
        uint32_t o = (uint32_t)__shfl_down((int)len, d);
        if (o > len) len = o;
    }
    if ((threadIdx.x % WAVE) == 0 && len) atomicMax(out_max, len);
}

/* per-64-row-group byte histograms: seg = one wave's group; counts[seg][p] (u32) */
__global__ __launch_bounds__(BLOCK_THREADS) void k5_count(
    int64_t n, uint32_t nparts, const uint32_t *pid, const int32_t *offsets,
    uint16_t *bcounts /* [nseg5][P]; 64-row-seg bytes <= 64*maxlen(128) fits u16 */,
    uint32_t *partials /* [nranges] fused L1 */, int nranges, int64_t nseg5) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int64_t seg = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
    if (seg >= nseg5) return;
    uint32_t *hist = (uint32_t *)smem + (size_t)wid * nparts;
    for (uint32_t p = lane; p < nparts; p += WAVE) hist[p] = 0;
    const int64_t row = seg * WAVE + lane;
    if (row < n) {
        const uint32_t len = (uint32_t)(offsets[row + 1] - offsets[row]);
        if (len) atomicAdd(&hist[pid[row]], len);
    }
    /* fair-division range of this segment (k_scan_partial mapping) */
    const int64_t rw = ((seg + 1) * (int64_t)nranges + nseg5 - 1) / nseg5 - 1;
    uint32_t *prow = partials + (size_t)rw * nparts;
    for (uint32_t p = lane; p < nparts; p += WAVE) {
        const uint32_t h = hist[p];
        bcounts[(size_t)seg * nparts + p] = (uint16_t)h;
        if (h) atomicAdd(&prow[p], h);
    }
}

/* per-round image byte bases incl. total: roffB[r][p] for p in [0,P], stride P+1 */
__global__ __launch_bounds__(BLOCK_THREADS) void k5_roff(
    const uint32_t *bcounts /* GLOBAL byte bases after the scan fold */,
    const uint64_t *part_boffsets, int64_t nrounds, int wpb, uint32_t nparts,
    uint32_t *roffB /* [nrounds][P+1] */) {
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int64_t r = (int64_t)blockIdx.x * WAVES_PER_BLOCK + wid;
    if (r >= nrounds) return;
    const uint32_t *rb = bcounts + (size_t)r * wpb * nparts;
    const uint32_t *re = (r + 1 < nrounds) ? bcounts + (size_t)(r + 1) * wpb * nparts : nullptr;
    uint32_t *orow = roffB + (size_t)r * (nparts + 1);
    uint32_t carry = 0;
    for (uint32_t p0 = 0; p0 < nparts; p0 += WAVE) {
        const uint32_t p = p0 + lane;
        uint32_t c = 0;
        if (p < nparts) c = (re ? re[p] : (uint32_t)part_boffsets[p + 1]) - rb[p];
        uint32_t v = c;
#pragma unroll
        for (int d = 1; d < WAVE; d <<= 1) {
            uint32_t u = (uint32_t)__shfl_up((int)v, d);
            if (lane >= d) v += u;
        }
        if (p < nparts) orow[p] = carry + v - c; /* exclusive */
        carry += (uint32_t)__shfl((int)v, WAVE - 1);
    }
    if (lane == 0) orow[nparts] = carry; /* round byte total */
}

/* byte copy helpers: LDS is natural-alignment-only; global unaligned 8B is native */
__device__ __forceinline__ void k5_bytes_to_lds(char *dst, const uint8_t *src,
                                                uint32_t len) {
    uint32_t b = 0;
    /* head: bytes until dst is 4-aligned */
    while (b < len && ((uintptr_t)(dst + b) & 3)) {
        dst[b] = (char)src[b];
        b++;
    }
    /* 8-B unaligned global reads (native on gfx950), two aligned dword LDS stores */
    for (; b + 8 <= len; b += 8) {
        uint64_t t;
        __builtin_memcpy(&t, src + b, 8);
        *(uint32_t *)(dst + b) = (uint32_t)t;
        *(uint32_t *)(dst + b + 4) = (uint32_t)(t >> 32);
    }
    for (; b + 4 <= len; b += 4) {
        uint32_t t;
        __builtin_memcpy(&t, src + b, 4);
        *(uint32_t *)(dst + b) = t;
    }
    for (; b < len; b++) dst[b] = (char)src[b];
}

template <int WPB5>
__global__ __launch_bounds__(WPB5 * WAVE) void k5_scatter(
    int64_t n, uint32_t nparts, int nbits, const uint32_t *pid, const int32_t *offsets,
    const uint8_t *in_bytes, const uint32_t *gbaseB /* [nseg5][P] global byte bases */,
    const uint32_t *roffB /* [nrounds][P+1] */, int64_t nrounds, uint8_t *out_bytes) {
    constexpr int R5 = WPB5 * WAVE;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    /* carve: per-wave gbase u32[P] | shared roff u32[P+1] | image */
    char *ws = smem;
    uint32_t *gb_all = (uint32_t *)ws;
    ws += sizeof(uint32_t) * WPB5 * nparts;
    uint32_t *roff = (uint32_t *)ws;
    ws += sizeof(uint32_t) * (nparts + 1);
    char *const img = ws;

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *gb = gb_all + (size_t)wid * nparts;
    uint32_t *gb0 = gb_all;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;
    __shared__ uint32_t lens_s[WPB5][WAVE];
    __shared__ uint32_t srcb_s[WPB5][WAVE];

    const int64_t r = blockIdx.x;
    if (r >= nrounds) return;
    const int64_t seg = r * WPB5 + wid;
    const int64_t row = seg * WAVE + lane;

    /* base rows -> LDS (wave-private write; cross-wave gb0 reads ordered by barrier) */
    {
        const uint32_t *grow = gbaseB + (size_t)seg * nparts;
        for (uint32_t p = lane; p < nparts; p += WAVE) gb[p] = grow[p];
        const uint32_t *rrow = roffB + (size_t)r * (nparts + 1);
        for (uint32_t p = tid; p <= nparts; p += WPB5 * WAVE) roff[p] = rrow[p];
    }
    __syncthreads();

    /* place: per-row byte range into the partition-major image */
    uint32_t len = 0, sb = 0, pidv = 0;
    if (row < n) {
        const int32_t o0 = offsets[row];
        len = (uint32_t)(offsets[row + 1] - o0);
        sb = (uint32_t)o0;
        pidv = pid[row];
    }
    lens_s[wid][lane] = len;
    srcb_s[wid][lane] = sb;
    /* group byte prefix among same-partition lanes (v1 k_scatter idiom) */
    const uint64_t act = __ballot(row < n);
    if (row < n) {
        uint64_t eq = dd_eq_mask(pidv, act, nbits);
        uint32_t bpre = 0, bsum = 0;
        uint64_t m = eq;
        while (m) {
            int j = __ffsll((unsigned long long)m) - 1;
            uint32_t lj = lens_s[wid][j];
            if (j < lane) bpre += lj;
            bsum += lj;
            m &= m - 1;
        }
        (void)bsum;
        if (len) {
            const uint32_t gdst = gb[pidv] + bpre;       /* global byte pos */
            const uint32_t slot = roff[pidv] + (gdst - gb0[pidv]); /* image byte pos */
            k5_bytes_to_lds(img + slot, in_bytes + sb, len);
        }
    }
    __syncthreads();

    /* flush: per-partition byte runs, lanes striding 8 B; LDS reads are shift-aligned
     * dword pairs, global stores unaligned 8 B (native). Partitions are wave-strided
     * for balance. */
    const uint32_t total = roff[nparts];
    (void)total;
    for (uint32_t p = wid; p < nparts; p += WPB5) {
        const uint32_t i0 = roff[p];
        const uint32_t cnt = roff[p + 1] - i0;
        if (!cnt) continue;
        /* global destination of this run: round base for p = gb0[p] */
        uint8_t *gdst = out_bytes + gb0[p];
        uint32_t b = lane * 8;
        for (; b + 8 <= cnt; b += WAVE * 8) {
            /* read 8 unaligned LDS bytes as two aligned dwords + funnel shift */
            const uint32_t src = i0 + b;
            const uint32_t a0 = (uint32_t)(src & ~3u);
            const uint32_t sh = (src & 3u) * 8;
            uint32_t w0 = *(const uint32_t *)(img + a0);
            uint32_t w1 = *(const uint32_t *)(img + a0 + 4);
            uint32_t w2 = *(const uint32_t *)(img + a0 + 8);
            uint64_t v;
            if (sh == 0) {
                v = (uint64_t)w0 | ((uint64_t)w1 << 32);
            } else {
                uint32_t lo = (w0 >> sh) | (w1 << (32 - sh));
                uint32_t hi = (w1 >> sh) | (w2 << (32 - sh));
                v = (uint64_t)lo | ((uint64_t)hi << 32);
            }
            __builtin_memcpy(gdst + b, &v, 8); /* unaligned global store: native */
        }
        /* tail < 8 B: after the strided loop exactly one lane has b < cnt */
        if (b < cnt) {
            for (uint32_t t = b; t < cnt; t++) gdst[t] = (uint8_t)img[i0 + t];
        }
    }
}
/* ================= K4: var-width bytes for the staged path =================
 * The staged scatter (v2) handles var columns' LENGTHS and a ROWID permutation as
 * synthetic fixed u32 columns (DD_KDT_VARLEN / DD_KDT_ROWID, set up by dd_host.cpp).
 * K4 then materializes the byte buffers in partition-major output order:
 *   K4a: per-wave partial sums of lengths[s] over contiguous slot ranges
 *   K4b: one block exclusive-scans the partials (u64)
 *   K4c: per-wave rewrite: out_off[s] = global exclusive byte offset (u64, n+1)
 *        == the Arrow offsets of the partition-major output (rebuilt, not just lengths)
 *   K4d: grid-stride gather-copy: slot s copies input row src_row[s]'s bytes to
 *        out_off[s] (consecutive slots -> consecutive output bytes: coalesced writes)
 *   K4e: part_byte_offsets[p] = out_off[part_row_offsets[p]]
 * No barriers anywhere except K4b. */

#define K4_RANGES 8192

__global__ void k4_len_partials(const uint32_t *lens, int64_t n, uint64_t *partials) {
    const int64_t w = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= K4_RANGES) return;
    const int64_t lo = n * w / K4_RANGES, hi = n * (w + 1) / K4_RANGES;
    uint64_t s = 0;
    for (int64_t i = lo + lane; i < hi; i += WAVE) s += lens[i];
#pragma unroll
    for (int d = 32; d >= 1; d >>= 1) s += (uint64_t)__shfl_down((long long)s, d);
    if (lane == 0) partials[w] = s;
}

__global__ void k4_scan_partials(uint64_t *partials, int nranges, int64_t total_bytes,
                                 uint64_t *out_off, int64_t n) {
    /* 256 threads: span sums -> one-wave shfl scan -> span rewrite */
    __shared__ uint64_t tmp[256];
    const int tid = threadIdx.x;
    const int span = (nranges + 255) / 256;
    const int lo = tid * span, hi = (lo + span < nranges) ? lo + span : nranges;
    uint64_t s = 0;
    for (int i = lo; i < hi; i++) s += partials[i];
    tmp[tid] = s;
    __syncthreads();
    if (tid < WAVE) {
        uint64_t carry = 0;
        for (int k = 0; k < 256 / WAVE; k++) {
            uint64_t v = tmp[k * WAVE + tid];
#pragma unroll
            for (int d = 1; d < WAVE; d <<= 1) {
                uint64_t u = (uint64_t)__shfl_up((long long)v, d);
                if (tid >= d) v += u;
            }
            v += carry;
            tmp[k * WAVE + tid] = v;
            carry = (uint64_t)__shfl((long long)v, WAVE - 1);
        }
    }
    __syncthreads();
    uint64_t run = (tid > 0) ? tmp[tid - 1] : 0;
    for (int i = lo; i < hi; i++) {
        uint64_t v = partials[i];
        partials[i] = run;
        run += v;
    }
    if (tid == 0) out_off[n] = (uint64_t)total_bytes;
}

__global__ void k4_off_rewrite(const uint32_t *lens, int64_t n, const uint64_t *partials,
                               uint64_t *out_off) {
    const int64_t w = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= K4_RANGES) return;
    const int64_t lo = n * w / K4_RANGES, hi = n * (w + 1) / K4_RANGES;
    uint64_t carry = partials[w];
    for (int64_t base = lo; base < hi; base += WAVE) {
        const int64_t s = base + lane;
        uint32_t len = (s < hi) ? lens[s] : 0;
        uint64_t v = len;
#pragma unroll
        for (int d = 1; d < WAVE; d <<= 1) {
            uint64_t u = (uint64_t)__shfl_up((long long)v, d);
            if (lane >= d) v += u;
        }
        if (s < hi) out_off[s] = carry + v - len; /* exclusive */
        carry += (uint64_t)__shfl((long long)v, WAVE - 1);
    }
}

/* Wave-cooperative gather-copy: a wave owns 64 consecutive slots and its 64 lanes copy
 * the group's CONTIGUOUS output byte range in 8-byte chunks (lane l writes bytes
 * [l*8, l*8+8) of the range, striding 512) — stores are fully coalesced; loads are
 * within-string contiguous (neighboring lanes usually read the same string). A chunk that
 * crosses string boundaries falls back to sub-chunk copies. One-string-per-thread copying
 * scattered every store instruction across 64 strings (9.8 GB written for 1.28 GB of
 * payload); a 32-B-per-lane variant quartered store coalescing (3.2 -> 5.0 ms); a 4-deep
 * pre-issued-load ILP variant measured neutral — this simple form stands. */
__device__ __forceinline__ void k4_copy_group(
    const uint32_t *src_row, const uint64_t *out_off, const int32_t *in_offsets,
    const uint8_t *in_bytes, int64_t n, uint8_t *out_bytes, uint32_t *loc, uint32_t *srcb,
    int lane, int64_t g) {
    const int64_t s0 = g * WAVE;
    const int64_t s = s0 + lane;
    uint32_t len = 0, sb = 0;
    if (s < n) {
        const uint32_t r = src_row[s];
        const int32_t o0 = in_offsets[r];
        len = (uint32_t)(in_offsets[r + 1] - o0);
        sb = (uint32_t)o0;
    }
    /* exclusive scan of lens across the wave -> group-local byte offsets */
    uint32_t v = len;
#pragma unroll
    for (int d = 1; d < WAVE; d <<= 1) {
        uint32_t u = (uint32_t)__shfl_up((int)v, d);
        if (lane >= d) v += u;
    }
    const uint32_t T = (uint32_t)__shfl((int)v, WAVE - 1);
    loc[lane] = v - len; /* exclusive */
    srcb[lane] = sb;
    if (lane == 0) loc[WAVE] = T;
    /* single wave: LDS program order; no barrier */
    const uint64_t obase = out_off[s0];
    for (uint32_t p0 = lane * 8; p0 < T; p0 += WAVE * 8) {
        uint32_t rem = (T - p0 < 8) ? (T - p0) : 8;
        uint32_t p = p0;
        /* binary search: largest j with loc[j] <= p */
        int lo = 0, hi = WAVE;
        while (lo < hi) {
            int mid = (lo + hi + 1) >> 1;
            if (loc[mid] <= p) lo = mid;
            else hi = mid - 1;
        }
        int j = lo;
        while (rem > 0) {
            while (loc[j + 1] <= p) j++; /* skip empty strings */
            const uint32_t within = p - loc[j];
            const uint32_t avail = loc[j + 1] - p;
            const uint32_t m = (rem < avail) ? rem : avail;
            const uint8_t *sp = in_bytes + srcb[j] + within;
            uint8_t *dp = out_bytes + obase + p;
            if (m == 8) {
                uint64_t t;
                __builtin_memcpy(&t, sp, 8);
                __builtin_memcpy(dp, &t, 8);
            } else {
                for (uint32_t b = 0; b < m; b++) dp[b] = sp[b];
            }
            p += m;
            rem -= m;
        }
    }
}

__global__ __launch_bounds__(BLOCK_THREADS) void k4_copy(
    const uint32_t *src_row, const uint64_t *out_off, const int32_t *in_offsets,
    const uint8_t *in_bytes, int64_t n, uint8_t *out_bytes) {
    __shared__ uint32_t loc_s[WAVES_PER_BLOCK][WAVE + 1];
    __shared__ uint32_t srcb_s[WAVES_PER_BLOCK][WAVE];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int64_t ngroups = (n + WAVE - 1) / WAVE;
    const int64_t wave_id = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
    const int64_t nwaves = ((int64_t)gridDim.x * blockDim.x) / WAVE;
    for (int64_t g = wave_id; g < ngroups; g += nwaves)
        k4_copy_group(src_row, out_off, in_offsets, in_bytes, n, out_bytes, loc_s[wid],
                      srcb_s[wid], lane, g);
}

__global__ void k_off64_to_off32(const uint64_t *off64, int64_t lo, int64_t n,
                                 int32_t *out32) {
    const int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (i <= n) out32[i] = (int32_t)(off64[lo + i] - off64[lo]);
}

__global__ void k4_part_boffsets(const uint64_t *out_off, const uint64_t *part_offsets,
                                 uint32_t nparts, int64_t n, int64_t total_bytes,
                                 uint64_t *part_boffsets) {
    const uint32_t p = blockIdx.x * blockDim.x + threadIdx.x;
    if (p > nparts) return;
    if (p == nparts) {
        part_boffsets[p] = (uint64_t)total_bytes;
        return;
    }
    const int64_t s = (int64_t)part_offsets[p];
    part_boffsets[p] = (s < n) ? out_off[s] : (uint64_t)total_bytes;
}

/* ---------------- launchers (called from dd_host.cpp) ---------------- */

/* one spec-path launch, fully typed; shared by the RHASH and pid variants (C++ linkage:
 * templates cannot live inside the extern "C" block below) */
template <int G, int C, int N, bool RH, bool HV = false>
static hipError_t dd_launch_spec1(const dd_kargs *a, dim3 grid, int64_t tile_rows,
                                  uint32_t nparts, int nbits, const uint32_t *pid_in,
                                  const uint32_t *tile_off, const uint64_t *part_offsets,
                                  size_t lds_bytes, hipStream_t s) {
    if (lds_bytes > 65536) {
        hipError_t e =
            hipFuncSetAttribute((const void *)k_scatter_staged<G, 16, C, HV, N, RH>,
                                hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL((k_scatter_staged<G, 16, C, HV, N, RH>), grid, dim3(16 * WAVE),
                       lds_bytes, s, *a, tile_rows, nparts, nbits, pid_in, tile_off,
                       part_offsets);
    return hipGetLastError();
}

extern "C" {

hipError_t dd_launch_off64_to_off32(const uint64_t *off64, int64_t lo, int64_t n,
                                    int32_t *out32, hipStream_t s) {
    int64_t total = n + 1;
    int blocks = (int)((total + 255) / 256);
    hipLaunchKernelGGL(k_off64_to_off32, dim3(blocks), dim3(256), 0, s, off64, lo, n, out32);
    return hipGetLastError();
}

hipError_t dd_launch_var_bytes(const uint32_t *lens, const uint32_t *src_row,
                               const int32_t *in_offsets, const uint8_t *in_bytes,
                               int64_t n, int64_t total_bytes, uint64_t *partials,
                               uint64_t *out_off, uint8_t *out_bytes,
                               const uint64_t *part_offsets, uint32_t nparts,
                               uint64_t *part_boffsets, uint32_t *k4w_meta,
                               uint32_t *k4w_order, int skip_copy, hipStream_t s) {
    const int wavegrid = K4_RANGES * WAVE / 256;
    hipLaunchKernelGGL(k4_len_partials, dim3(wavegrid), dim3(256), 0, s, lens, n, partials);
    hipLaunchKernelGGL(k4_scan_partials, dim3(1), dim3(256), 0, s, partials, K4_RANGES,
                       total_bytes, out_off, n);
    hipLaunchKernelGGL(k4_off_rewrite, dim3(wavegrid), dim3(256), 0, s, lens, n, partials,
                       out_off);
    /* Slot-order gather (k4_copy). MEASURED NEGATIVES (kept out of the tree, see
     * DESIGN.md §13): (a) 32 B-per-lane chunks — strided stores quarter write
     * coalescing (3.3 -> 5.0 ms); (b) 4-deep pre-issued-load ILP — neutral; (c) MALL
     * source-windowing, both as per-(partition,window) blocks (Zipf imbalance, 5.1 ms)
     * and as a balanced window-bucketed group order (randomized group-level write
     * order, 10 ms): the write side loses more than the read side gains.
     * DD_K4_BLOCKS overrides the grid cap (latency-hiding depth experiment). */
    if (!skip_copy) { /* skip_copy: K5 (k5_scatter) writes the bytes instead */
        int copy_blocks = (int)((n + 255) / 256);
        int cap = 8192;
        if (const char *e = getenv("DD_K4_BLOCKS")) {
            int v = atoi(e);
            if (v >= 256 && v <= 65536) cap = v;
        }
        if (copy_blocks > cap) copy_blocks = cap; /* latency-bound: deep
                                                     oversubscription hides the random
                                                     string reads */
        if (copy_blocks < 1) copy_blocks = 1;
        hipLaunchKernelGGL(k4_copy, dim3(copy_blocks), dim3(256), 0, s, src_row, out_off,
                           in_offsets, in_bytes, n, out_bytes);
    }
    (void)k4w_meta;
    (void)k4w_order;
    (void)total_bytes;
    hipLaunchKernelGGL(k4_part_boffsets, dim3((nparts + 256) / 256 + 1), dim3(256), 0, s,
                       out_off, part_offsets, nparts, n, total_bytes, part_boffsets);
    return hipGetLastError();
}

hipError_t dd_launch_k5_maxlen(const int32_t *offsets, int64_t n, uint32_t *out_max,
                               hipStream_t s) {
    int blocks = (int)((n + 255) / 256);
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(k5_maxlen, dim3(blocks), dim3(256), 0, s, offsets, n, out_max);
    return hipGetLastError();
}

hipError_t dd_launch_k5_count(int64_t n, uint32_t nparts, const uint32_t *pid,
                              const int32_t *offsets, uint16_t *bcounts,
                              uint32_t *partials, int nranges, int64_t nseg5,
                              size_t lds_bytes, hipStream_t s) {
    int64_t blocks = nseg5 / WAVES_PER_BLOCK;
    if (blocks < 1) blocks = 1;
    if (lds_bytes > 65536) {
        hipError_t e = hipFuncSetAttribute((const void *)k5_count,
                                           hipFuncAttributeMaxDynamicSharedMemorySize,
                                           (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(k5_count, dim3((unsigned)blocks), dim3(BLOCK_THREADS), lds_bytes,
                       s, n, nparts, pid, offsets, bcounts, partials, nranges, nseg5);
    return hipGetLastError();
}

hipError_t dd_launch_k5_roff(const uint32_t *bcounts, const uint64_t *part_boffsets,
                             int64_t nrounds, int wpb, uint32_t nparts, uint32_t *roffB,
                             hipStream_t s) {
    dim3 grid((unsigned)((nrounds + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK));
    hipLaunchKernelGGL(k5_roff, grid, dim3(BLOCK_THREADS), 0, s, bcounts, part_boffsets,
                       nrounds, wpb, nparts, roffB);
    return hipGetLastError();
}

hipError_t dd_launch_k5_scatter(int64_t n, uint32_t nparts, int nbits, const uint32_t *pid,
                                const int32_t *offsets, const uint8_t *in_bytes,
                                const uint32_t *gbaseB, const uint32_t *roffB,
                                int64_t nrounds, uint8_t *out_bytes, int wpb5,
                                size_t lds_bytes, hipStream_t s) {
#define DD_K5S(W)                                                                            \
    if (wpb5 == W) {                                                                         \
        if (lds_bytes > 65536) {                                                             \
            hipError_t e = hipFuncSetAttribute((const void *)k5_scatter<W>,                  \
                                               hipFuncAttributeMaxDynamicSharedMemorySize,   \
                                               (int)lds_bytes);                              \
            if (e != hipSuccess) return e;                                                   \
        }                                                                                    \
        hipLaunchKernelGGL((k5_scatter<W>), dim3((unsigned)nrounds), dim3(W * WAVE),         \
                           lds_bytes, s, n, nparts, nbits, pid, offsets, in_bytes, gbaseB,   \
                           roffB, nrounds, out_bytes);                                       \
        return hipGetLastError();                                                            \
    }
    DD_K5S(16)
    DD_K5S(8) /* half-size blocks: ~62 KB LDS -> 2 blocks/CU, so one block's flush
                 overlaps the co-resident block's place loads */
#undef DD_K5S
    return hipErrorInvalidValue;
}

hipError_t dd_launch_dict_hashes(const uint8_t *bytes, const int32_t *offsets, int64_t n,
                                 uint64_t *out, hipStream_t s) {
    int threads = 256;
    int blocks = (int)((n + threads - 1) / threads);
    if (blocks == 0) blocks = 1;
    hipLaunchKernelGGL(k_dict_hashes, dim3(blocks), dim3(threads), 0, s, bytes, offsets, n, out);
    return hipGetLastError();
}

hipError_t dd_launch_hash_count(const dd_kargs *a, int64_t nchunks, int64_t chunk_rows,
                                uint32_t nparts, int nbits, uint32_t *pid_out,
                                uint32_t *counts, uint32_t *bcounts, size_t lds_bytes,
                                hipStream_t s) {
    dim3 grid((unsigned)(nchunks / WAVES_PER_BLOCK));
    if (lds_bytes > 65536) {
        hipError_t e = hipFuncSetAttribute((const void *)k_hash_count,
                                           hipFuncAttributeMaxDynamicSharedMemorySize,
                                           (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(k_hash_count, grid, dim3(BLOCK_THREADS), lds_bytes, s, *a, chunk_rows,
                       nparts, nbits, pid_out, counts, bcounts);
    return hipGetLastError();
}

hipError_t dd_launch_scan(uint32_t *counts, int64_t nchunks, uint32_t nparts, int nranges,
                          uint32_t *partials, uint64_t *part_offsets, int fold_global,
                          hipStream_t s) {
    int threads = 256;
    int64_t total = (int64_t)nranges * nparts;
    int blocks = (int)((total + threads - 1) / threads);
    hipLaunchKernelGGL(k_scan_partial, dim3(blocks), dim3(threads), 0, s, counts, nchunks,
                       nparts, nranges, partials);
    hipLaunchKernelGGL(k_scan_combine, dim3(1), dim3(256), 0, s, partials, nranges, nparts,
                       part_offsets);
    hipLaunchKernelGGL(k_scan_rewrite, dim3(blocks), dim3(threads), 0, s, counts, nchunks,
                       nparts, nranges, partials,
                       fold_global ? (const uint64_t *)part_offsets : nullptr);
    return hipGetLastError();
}

__global__ void k_scan_rewrite16(const uint16_t *counts16, int64_t nchunks,
                                 uint32_t nparts, int nranges, const uint32_t *partials,
                                 const uint64_t *fold_offsets, uint32_t *gbase) {
    const int64_t tid = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t total = (int64_t)nranges * nparts;
    if (tid >= total) return;
    const int r = (int)(tid / nparts);
    const uint32_t p = (uint32_t)(tid % nparts);
    const int64_t c0 = nchunks * r / nranges, c1 = nchunks * (r + 1) / nranges;
    uint32_t run = partials[(size_t)r * nparts + p] + (uint32_t)fold_offsets[p];
    for (int64_t c = c0; c < c1; c++) {
        gbase[c * nparts + p] = run;
        run += counts16[c * nparts + p];
    }
}

/* two-level scan for segment-granular counts (pre path): k_scan_combine walking 2048
 * ranges serially in one block measured 0.6 ms (latency-bound); recurse instead:
 * counts -> partials[nr1] -> partials2[nr2] -> combine(nr2) -> rewrite back up. */
hipError_t dd_launch_scan_deep(uint16_t *counts16, int64_t nchunks, uint32_t nparts,
                               int nr1, int nr2, uint32_t *partials, uint32_t *partials2,
                               uint64_t *part_offsets, uint32_t *gbase, hipStream_t s) {
    int threads = 256;
    int b1 = (int)(((int64_t)nr1 * nparts + threads - 1) / threads);
    int b2 = (int)(((int64_t)nr2 * nparts + threads - 1) / threads);
    /* level-1 partials were fused into k_hash_count_seg / k5_count (atomicAdd) */
    hipLaunchKernelGGL(k_scan_partial, dim3(b2), dim3(threads), 0, s, partials, (int64_t)nr1,
                       nparts, nr2, partials2);
    hipLaunchKernelGGL(k_scan_combine, dim3(1), dim3(256), 0, s, partials2, nr2, nparts,
                       part_offsets);
    hipLaunchKernelGGL(k_scan_rewrite, dim3(b2), dim3(threads), 0, s, partials, (int64_t)nr1,
                       nparts, nr2, partials2, nullptr);
    hipLaunchKernelGGL(k_scan_rewrite16, dim3(b1), dim3(threads), 0, s, counts16, nchunks,
                       nparts, nr1, partials, (const uint64_t *)part_offsets, gbase);
    return hipGetLastError();
}

hipError_t dd_launch_hash_count_seg(const dd_kargs *a, int64_t nseg, int64_t seg_rows,
                                    uint32_t nparts, int nbits, uint32_t *pid_out,
                                    uint16_t *counts, uint32_t *partials, int nranges,
                                    size_t lds_bytes, hipStream_t s) {
    /* full grid (one segment per wave): a capped grid with consecutive-segment runs
     * per wave measured K1 0.26 -> 0.40 ms — the dense all-waves-in-one-row-window
     * layout wins (DRAM page locality), dispatch overhead is not the cost */
    int64_t nblk = nseg / WAVES_PER_BLOCK;
    if (nblk < 1) nblk = 1;
    dim3 grid((unsigned)nblk);
    if (lds_bytes > 65536) {
        hipError_t e = hipFuncSetAttribute((const void *)k_hash_count_seg,
                                           hipFuncAttributeMaxDynamicSharedMemorySize,
                                           (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(k_hash_count_seg, grid, dim3(BLOCK_THREADS), lds_bytes, s, *a, nseg,
                       seg_rows, nparts, nbits, pid_out, counts, partials, nranges);
    return hipGetLastError();
}

hipError_t dd_launch_round_layout(const uint32_t *counts, const uint64_t *part_offsets,
                                  int64_t nrounds, int wpb, uint32_t nparts, uint32_t sP2,
                                  uint16_t *rofftab, hipStream_t s) {
    dim3 grid((unsigned)((nrounds + WAVES_PER_BLOCK - 1) / WAVES_PER_BLOCK));
    hipLaunchKernelGGL(k_round_roff, grid, dim3(BLOCK_THREADS), 0, s, counts,
                       part_offsets, nrounds, wpb, nparts, sP2, rofftab);
    return hipGetLastError();
}

hipError_t dd_launch_scatter_pre(const dd_kargs *a, int64_t nblocks, int64_t nrounds,
                                 int rpb, uint32_t nparts, int nbits,
                                 const uint32_t *pid_in, const uint32_t *gbase,
                                 const uint16_t *rofftab, uint32_t sP2, int gmax, int wpb,
                                 size_t lds_bytes, hipStream_t s) {
    dim3 grid((unsigned)nblocks);
    const int n = a->n_cols;
    auto el = [&](int c) { return (int)a->cols[c].elem; };
#define DD_PREW(GM, W_, NBG_, NBI_, ...)                                                     \
    {                                                                                        \
        const int want[] = {__VA_ARGS__};                                                    \
        const int wn = (int)(sizeof(want) / sizeof(want[0]));                                \
        bool m = (gmax == GM && wpb == W_ && n == wn &&                                      \
                  nparts <= (uint32_t)(NBG_ * WAVE));                                        \
        for (int c = 0; c < wn && m; c++) m = el(c) == want[c];                              \
        if (m) {                                                                             \
            if (lds_bytes > 65536) {                                                         \
                hipError_t e = hipFuncSetAttribute(                                          \
                    (const void *)k_scatter_pre<GM, W_, NBG_, NBI_, __VA_ARGS__>,            \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds_bytes);             \
                if (e != hipSuccess) return e;                                               \
            }                                                                                \
            hipLaunchKernelGGL((k_scatter_pre<GM, W_, NBG_, NBI_, __VA_ARGS__>), grid,       \
                               dim3(W_ * WAVE), lds_bytes, s, *a, nrounds, rpb, nparts,      \
                               nbits, pid_in, gbase, rofftab, sP2);                          \
            return hipGetLastError();                                                        \
        }                                                                                    \
    }
#define DD_PRE(GM, NBG_, NBI_, ...) DD_PREW(GM, 16, NBG_, NBI_, __VA_ARGS__)
    /* WPB=8 tier (2 blocks/CU) — headline shape experiment */
    DD_PREW(4, 8, 2, 1, 8, 8, 8, 4)
    DD_PREW(4, 8, 2, 1, 8, 8, 8, 4, 4)
    /* P <= 128 tier (NBG=2, NBI=1): all {4,8}^4 combos + the multikey and q1 shapes */
    DD_PRE(4, 2, 1, 4, 4, 4, 4)
    DD_PRE(4, 2, 1, 4, 4, 4, 8)
    DD_PRE(4, 2, 1, 4, 4, 8, 4)
    DD_PRE(4, 2, 1, 4, 4, 8, 8)
    DD_PRE(4, 2, 1, 4, 8, 4, 4)
    DD_PRE(4, 2, 1, 4, 8, 4, 8)
    DD_PRE(4, 2, 1, 4, 8, 8, 4)
    DD_PRE(4, 2, 1, 4, 8, 8, 8)
    DD_PRE(4, 2, 1, 8, 4, 4, 4)
    DD_PRE(4, 2, 1, 8, 4, 4, 8)
    DD_PRE(4, 2, 1, 8, 4, 8, 4)
    DD_PRE(4, 2, 1, 8, 4, 8, 8)
    DD_PRE(4, 2, 1, 8, 8, 4, 4)
    DD_PRE(4, 2, 1, 8, 8, 4, 8)
    DD_PRE(4, 2, 1, 8, 8, 8, 4)
    DD_PRE(4, 2, 1, 8, 8, 8, 8)
    DD_PRE(4, 2, 1, 8, 8, 8, 4, 4)       /* multikey bench shape */
    DD_PRE(2, 2, 1, 8, 8, 8, 4)          /* G2 experiment (2 blocks/CU) */
    DD_PRE(2, 2, 1, 8, 8, 8, 8)
    DD_PRE(2, 2, 1, 4, 4, 4, 4)
    DD_PRE(2, 2, 1, 8, 8, 8, 4, 4)
    DD_PRE(2, 2, 1, 1, 1, 8, 8, 8, 8, 4) /* q1 shape */
    /* 129..256 tier (NBG=4, NBI=2): headline + multikey */
    DD_PRE(4, 4, 2, 8, 8, 8, 4)
    DD_PRE(4, 4, 2, 8, 8, 8, 4, 4)
    /* 257..512 tier (NBG=8, NBI=4, G=2 for LDS): headline + multikey shapes */
    DD_PRE(2, 8, 4, 8, 8, 8, 4)
    DD_PRE(2, 8, 4, 8, 8, 8, 4, 4)
#undef DD_PRE
#undef DD_PREW
    return hipErrorInvalidValue;
}

hipError_t dd_launch_hash_count_tile(const dd_kargs *a, int64_t nblocks, int64_t tile_rows,
                                     uint32_t nparts, int nbits, uint32_t *pid_out,
                                     uint32_t *counts, size_t lds_bytes, hipStream_t s) {
    if (lds_bytes > 65536) {
        hipError_t e = hipFuncSetAttribute((const void *)k_hash_count_tile,
                                           hipFuncAttributeMaxDynamicSharedMemorySize,
                                           (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(k_hash_count_tile, dim3((unsigned)nblocks), dim3(BLOCK_THREADS),
                       lds_bytes, s, *a, tile_rows, nparts, nbits, pid_out, counts);
    return hipGetLastError();
}

hipError_t dd_launch_scatter_staged(const dd_kargs *a, int64_t nblocks, int64_t tile_rows,
                                    uint32_t nparts, int nbits, const uint32_t *pid_in,
                                    const uint32_t *tile_off, const uint64_t *part_offsets,
                                    int gmax, int wpb, size_t lds_bytes, hipStream_t s) {
    dim3 grid((unsigned)nblocks);
    if (a->hl == 2) { /* generalized hidden-load (host whitelists the exact shapes) */
        const int n = a->n_cols;
        auto el = [&](int c) { return (int)a->cols[c].elem; };
#define DD_HLG(GM, ...)                                                                      \
    {                                                                                        \
        const int want[] = {__VA_ARGS__};                                                    \
        const int wn = (int)(sizeof(want) / sizeof(want[0]));                                \
        bool m = (gmax == GM && wpb == 16 && n == wn);                                       \
        for (int c = 0; c < wn && m; c++) m = el(c) == want[c];                              \
        if (m) {                                                                             \
            if (lds_bytes > 65536) {                                                         \
                hipError_t e =                                                               \
                    hipFuncSetAttribute((const void *)k_scatter_hlg<GM, __VA_ARGS__>,        \
                                        hipFuncAttributeMaxDynamicSharedMemorySize,          \
                                        (int)lds_bytes);                                     \
                if (e != hipSuccess) return e;                                               \
            }                                                                                \
            hipLaunchKernelGGL((k_scatter_hlg<GM, __VA_ARGS__>), grid, dim3(16 * WAVE),      \
                               lds_bytes, s, *a, tile_rows, nparts, nbits, pid_in,           \
                               tile_off, part_offsets);                                      \
            return hipGetLastError();                                                        \
        }                                                                                    \
    }
        DD_HLG(4, 8, 8, 8, 4, 4)       /* multikey bench shape */
        DD_HLG(2, 1, 1, 8, 8, 8, 8, 4) /* q1 shape */
#undef DD_HLG
        return hipErrorInvalidValue;
    }
    if (a->hl) { /* hidden-load, 4-column kernel (host gate: ka.hl) */
        if (wpb != 16 || gmax != 4 || a->n_cols != 4) return hipErrorInvalidValue;
        const int e0 = a->cols[0].elem, e1 = a->cols[1].elem, e2 = a->cols[2].elem,
                  e3 = a->cols[3].elem;
#define DD_HL(A, B, C, D)                                                                    \
    if (e0 == A && e1 == B && e2 == C && e3 == D) {                                          \
        if (lds_bytes > 65536) {                                                             \
            hipError_t e = hipFuncSetAttribute((const void *)k_scatter_hl<A, B, C, D>,       \
                                               hipFuncAttributeMaxDynamicSharedMemorySize,   \
                                               (int)lds_bytes);                              \
            if (e != hipSuccess) return e;                                                   \
        }                                                                                    \
        hipLaunchKernelGGL((k_scatter_hl<A, B, C, D>), grid, dim3(16 * WAVE), lds_bytes, s,  \
                           *a, tile_rows, nparts, nbits, pid_in, tile_off, part_offsets);    \
        return hipGetLastError();                                                            \
    }
        DD_HL(4, 4, 4, 4)
        DD_HL(4, 4, 4, 8)
        DD_HL(4, 4, 8, 4)
        DD_HL(4, 4, 8, 8)
        DD_HL(4, 8, 4, 4)
        DD_HL(4, 8, 4, 8)
        DD_HL(4, 8, 8, 4)
        DD_HL(4, 8, 8, 8)
        DD_HL(8, 4, 4, 4)
        DD_HL(8, 4, 4, 8)
        DD_HL(8, 4, 8, 4)
        DD_HL(8, 4, 8, 8)
        DD_HL(8, 8, 4, 4)
        DD_HL(8, 8, 4, 8)
        DD_HL(8, 8, 8, 4)
        DD_HL(8, 8, 8, 8)
#undef DD_HL
        return hipErrorInvalidValue;
    }
    const int maxc = (a->n_cols <= 4) ? 4 : 8;
    const bool hasvar = a->n_var > 0;
    /* specialized fast path: wpb 16, <=4 fixed columns, no validity, no var */
    bool can_spec = !hasvar && wpb == 16 && a->n_cols <= 4;
    for (int c = 0; c < a->n_cols && can_spec; c++)
        if (a->cols[c].valid || a->cols[c].elem == 0) can_spec = false;
#define DD_SPEC(G, N)                                                                        \
    if (can_spec && gmax == G && a->n_cols == N) {                                           \
        return a->rhash ? dd_launch_spec1<G, 4, N, true>(a, grid, tile_rows, nparts, nbits,  \
                                                         pid_in, tile_off, part_offsets,    \
                                                         lds_bytes, s)                       \
                        : dd_launch_spec1<G, 4, N, false>(a, grid, tile_rows, nparts, nbits, \
                                                          pid_in, tile_off, part_offsets,   \
                                                          lds_bytes, s);                     \
    }
    DD_SPEC(2, 1)
    DD_SPEC(2, 2)
    DD_SPEC(2, 3)
    DD_SPEC(2, 4)
    DD_SPEC(4, 1)
    DD_SPEC(4, 2)
    DD_SPEC(4, 3)
    DD_SPEC(4, 4)
#undef DD_SPEC
    /* 5-8 fixed columns: same fast path at MAXC=8 */
#define DD_SPEC8(G, N)                                                                       \
    if (can_spec8 && gmax == G && a->n_cols == N) {                                          \
        return a->rhash ? dd_launch_spec1<G, 8, N, true>(a, grid, tile_rows, nparts, nbits,  \
                                                         pid_in, tile_off, part_offsets,    \
                                                         lds_bytes, s)                       \
                        : dd_launch_spec1<G, 8, N, false>(a, grid, tile_rows, nparts, nbits, \
                                                          pid_in, tile_off, part_offsets,   \
                                                          lds_bytes, s);                     \
    }
    bool can_spec8 = !hasvar && wpb == 16 && a->n_cols >= 5 && a->n_cols <= 8;
    for (int c = 0; c < a->n_cols && can_spec8; c++)
        if (a->cols[c].valid || a->cols[c].elem == 0) can_spec8 = false;
    DD_SPEC8(2, 5)
    DD_SPEC8(2, 6)
    DD_SPEC8(2, 7)
    DD_SPEC8(2, 8)
    DD_SPEC8(4, 5)
    DD_SPEC8(4, 6)
    DD_SPEC8(4, 7)
    DD_SPEC8(4, 8)
#undef DD_SPEC8
    /* NC-specialized staged-var: compile-time column-count unroll for var batches with
     * no validity (the original var columns ride as elem-0 no-op entries; the synthetic
     * VARLEN/ROWID columns are handled by the HASVAR branch). The generic runtime-column
     * loop measured 8-15% slower on the spec path in round 1 — same reasoning here. */
    bool can_specv = hasvar && wpb == 16 && a->n_cols <= 8;
    for (int c = 0; c < a->n_cols && can_specv; c++)
        if (a->cols[c].valid) can_specv = false;
#define DD_SPECV(G, N)                                                                       \
    if (can_specv && gmax == G && a->n_cols == N) {                                          \
        return dd_launch_spec1<G, 8, N, false, true>(a, grid, tile_rows, nparts, nbits,      \
                                                     pid_in, tile_off, part_offsets,         \
                                                     lds_bytes, s);                          \
    }
    DD_SPECV(2, 2)
    DD_SPECV(2, 3)
    DD_SPECV(2, 4)
    DD_SPECV(2, 5)
    DD_SPECV(2, 6)
    DD_SPECV(2, 7)
    DD_SPECV(2, 8)
    DD_SPECV(4, 2)
    DD_SPECV(4, 3)
    DD_SPECV(4, 4)
    DD_SPECV(4, 5)
    DD_SPECV(4, 6)
    DD_SPECV(4, 7)
    DD_SPECV(4, 8)
#undef DD_SPECV
    /* rhash batches MUST take a spec path above (no pid array exists to read) — fail
     * loudly rather than fall through to a pid-consuming variant */
    if (a->rhash) return hipErrorInvalidValue;
#define DD_CASE(G, W, C, V)                                                                  \
    if (gmax == G && wpb == W && maxc == C && hasvar == V) {                                 \
        if (lds_bytes > 65536) {                                                             \
            hipError_t e = hipFuncSetAttribute((const void *)k_scatter_staged<G, W, C, V>,   \
                                               hipFuncAttributeMaxDynamicSharedMemorySize,   \
                                               (int)lds_bytes);                              \
            if (e != hipSuccess) return e;                                                   \
        }                                                                                    \
        hipLaunchKernelGGL((k_scatter_staged<G, W, C, V>), grid, dim3(W * WAVE), lds_bytes,  \
                           s, *a, tile_rows, nparts, nbits, pid_in, tile_off,                \
                           part_offsets);                                                    \
        return hipGetLastError();                                                            \
    }
    DD_CASE(2, 4, 4, false)
    DD_CASE(4, 4, 4, false)
    DD_CASE(8, 4, 4, false)
    DD_CASE(2, 8, 4, false)
    DD_CASE(4, 8, 4, false)
    DD_CASE(8, 8, 4, false)
    DD_CASE(2, 4, 8, false)
    DD_CASE(4, 4, 8, false)
    DD_CASE(8, 4, 8, false)
    DD_CASE(2, 8, 8, false)
    DD_CASE(4, 8, 8, false)
    DD_CASE(8, 8, 8, false)
    DD_CASE(2, 4, 4, true)
    DD_CASE(4, 4, 4, true)
    DD_CASE(8, 4, 4, true)
    DD_CASE(2, 8, 4, true)
    DD_CASE(4, 8, 4, true)
    DD_CASE(8, 8, 4, true)
    DD_CASE(2, 4, 8, true)
    DD_CASE(4, 4, 8, true)
    DD_CASE(8, 4, 8, true)
    DD_CASE(2, 8, 8, true)
    DD_CASE(4, 8, 8, true)
    DD_CASE(8, 8, 8, true)
    DD_CASE(2, 16, 4, false)
    DD_CASE(4, 16, 4, false)
    DD_CASE(2, 16, 8, false)
    DD_CASE(4, 16, 8, false)
    DD_CASE(2, 16, 4, true)
    DD_CASE(4, 16, 4, true)
    DD_CASE(2, 16, 8, true)
    DD_CASE(4, 16, 8, true)
#undef DD_CASE
    return hipErrorInvalidValue;
}

hipError_t dd_launch_scatter(const dd_kargs *a, int64_t nchunks, int64_t chunk_rows,
                             uint32_t nparts, int nbits, const uint32_t *pid_in,
                             const uint32_t *chunk_off, const uint64_t *part_offsets,
                             const uint32_t *chunk_boff, const uint64_t *part_boffsets,
                             size_t lds_bytes, hipStream_t s) {
    dim3 grid((unsigned)(nchunks / WAVES_PER_BLOCK));
    if (lds_bytes > 65536) {
        hipError_t e = hipFuncSetAttribute((const void *)k_scatter,
                                           hipFuncAttributeMaxDynamicSharedMemorySize,
                                           (int)lds_bytes);
        if (e != hipSuccess) return e;
    }
    hipLaunchKernelGGL(k_scatter, grid, dim3(BLOCK_THREADS), lds_bytes, s, *a, chunk_rows,
                       nparts, nbits, pid_in, chunk_off, part_offsets, chunk_boff,
                       part_boffsets);
    return hipGetLastError();
}

} /* extern "C" */
