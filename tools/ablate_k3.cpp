/* tools/ablate_k3.cpp — standalone ablation microbench for the staged scatter (K3).
 *
 * Perf-engineering infrastructure, not part of the product library: a trimmed copy of
 * k_scatter_staged (dd_kernels.hip) with compile-time ablations to attribute the kernel's
 * time between its phases at the bench shape (60M rows, P=128, 4 cols of 8/8/8/4 B).
 *
 *   ABLATE=0 full kernel
 *   ABLATE=1 no flush stores (pass4 skipped; LDS still read into a sink)
 *   ABLATE=2 no column loads (preload synthesizes values; pid still loaded)
 *   ABLATE=3 no rank machinery (fake rank = lane; WRONG results, timing only)
 *   ABLATE=4 no LDS staging (direct scatter from registers; pass3/pass4 merged)
 *
 * Build+run (GPU box):
 *   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ablate_k3.cpp -o /tmp/ablate && /tmp/ablate
 */

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>

#define WAVE 64
#define WPB 16
#define GMAX 4
#define BT (WPB * WAVE)
#define R (GMAX * BT)
#define NCOLS 4

static const int ELEM[NCOLS] = {8, 8, 8, 4};

#define HC(x)                                                                                \
    do {                                                                                     \
        hipError_t e_ = (x);                                                                 \
        if (e_ != hipSuccess) {                                                              \
            printf("HIP error %s at %d\n", hipGetErrorString(e_), __LINE__);                 \
            exit(1);                                                                         \
        }                                                                                    \
    } while (0)

__device__ __forceinline__ uint64_t eq_mask(uint32_t pid, uint64_t act, int nbits) {
    uint64_t eq = act;
    for (int b = 0; b < nbits; b++) {
        uint64_t bal = __ballot((pid >> b) & 1u);
        eq &= ((pid >> b) & 1u) ? bal : ~bal;
    }
    return eq;
}

template <int ABLATE>
__global__ __launch_bounds__(BT) void k_ablate(
    int64_t n_rows, int64_t tile_rows, uint32_t nparts, int nbits, const uint32_t *pid_in,
    const uint32_t *tile_off, const uint64_t *part_offsets, const uint64_t *in0,
    const uint64_t *in1, const uint64_t *in2, const uint32_t *in3, uint64_t *o0,
    uint64_t *o1, uint64_t *o2, uint32_t *o3, uint32_t *sink) {
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
    char *const stage0 = ws; /* [R*8][R*8][R*8][R*4] */

    const int tid = threadIdx.x;
    const int wid = tid / WAVE;
    const int lane = tid % WAVE;
    uint32_t *myseg = seghist + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;
    constexpr int SEG = R / WPB;

    const int64_t tstart = (int64_t)blockIdx.x * tile_rows;
    const int64_t tend = (tstart + tile_rows < n_rows) ? (tstart + tile_rows) : n_rows;
    for (uint32_t p = tid; p < nparts; p += BT)
        dstbase[p] = part_offsets[p] + tile_off[(size_t)blockIdx.x * nparts + p];
    __syncthreads();

    uint32_t pidr[GMAX], rankr[GMAX];
    bool actr[GMAX];
    uint64_t c0[GMAX], c1[GMAX], c2[GMAX];
    uint32_t c3[GMAX];

    auto preload = [&](int64_t rstart, int64_t rend) {
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = rstart + (int64_t)wid * SEG + g * WAVE + lane;
            const bool active = row < rend;
            actr[g] = active;
            pidr[g] = 0;
            if (!active) continue;
            pidr[g] = pid_in[row];
            if (ABLATE == 2) {
                c0[g] = row;
                c1[g] = row;
                c2[g] = row;
                c3[g] = (uint32_t)row;
            } else {
                c0[g] = in0[row];
                c1[g] = in1[row];
                c2[g] = in2[row];
                c3[g] = in3[row];
            }
        }
    };

    {
        const int64_t re = (tstart + R < tend) ? tstart + R : tend;
        if (tstart < tend) preload(tstart, re);
    }

    for (int64_t rstart = tstart; rstart < tend; rstart += R) {
        const int64_t rend = (rstart + R < tend) ? (rstart + R) : tend;
        const int round_rows = (int)(rend - rstart);

        for (uint32_t p = lane; p < nparts; p += WAVE) myseg[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (ABLATE == 3) {
                rankr[g] = lane;
                continue;
            }
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = eq_mask(pid, act, nbits);
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
        __syncthreads();
        for (uint32_t p = tid; p < nparts; p += BT) {
            uint32_t run = 0;
#pragma unroll
            for (int w = 0; w < WPB; w++) {
                uint32_t v = seghist[(size_t)w * nparts + p];
                seghist[(size_t)w * nparts + p] = run;
                run += v;
            }
            roundcnt[p] = run;
        }
        __syncthreads();
        /* block exclusive scan of roundcnt -> round_off */
        {
            const uint32_t span = (nparts + BT - 1) / BT;
            const uint32_t lo = tid * span;
            const uint32_t hi = (lo + span < nparts) ? lo + span : nparts;
            uint32_t ssum = 0;
            for (uint32_t p = lo; p < hi; p++) ssum += roundcnt[p];
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
            for (uint32_t p = lo; p < hi; p++) {
                round_off[p] = run;
                run += roundcnt[p];
            }
            __syncthreads();
        }

        if (ABLATE == 4) {
            /* direct scatter, no LDS staging */
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if (!actr[g]) continue;
                const uint32_t pid = pidr[g];
                uint64_t dst = dstbase[pid] + myseg[pid] + rankr[g];
                if (dst >= (uint64_t)n_rows) dst = 0; /* harness safety (ABLATE=3/4) */
                o0[dst] = c0[g];
                o1[dst] = c1[g];
                o2[dst] = c2[g];
                o3[dst] = c3[g];
            }
            __syncthreads();
            if (rstart + R < tend) {
                const int64_t nre = (rstart + 2 * R < tend) ? rstart + 2 * R : tend;
                preload(rstart + R, nre);
            }
        } else {
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                if (!actr[g]) continue;
                const uint32_t pid = pidr[g];
                const uint32_t rank_r = myseg[pid] + rankr[g];
                const uint32_t slot = round_off[pid] + rank_r;
                dstg[slot] = (uint32_t)(dstbase[pid] + rank_r);
                ((uint64_t *)stage0)[slot] = c0[g];
                ((uint64_t *)(stage0 + (size_t)R * 8))[slot] = c1[g];
                ((uint64_t *)(stage0 + (size_t)R * 16))[slot] = c2[g];
                ((uint32_t *)(stage0 + (size_t)R * 24))[slot] = c3[g];
            }
            __syncthreads();
            if (rstart + R < tend) {
                const int64_t nre = (rstart + 2 * R < tend) ? rstart + 2 * R : tend;
                preload(rstart + R, nre);
            }
            if (ABLATE == 1) {
                uint32_t acc = 0;
                for (int i = tid; i < round_rows; i += BT) {
                    acc += dstg[i] + (uint32_t)((uint64_t *)stage0)[i];
                }
                if (acc == 0xFFFFFFFFu) sink[0] = acc; /* keep the reads alive */
            } else {
                for (int i = tid; i < round_rows; i += BT) {
                    /* harness safety: ABLATE=3 leaves dstg partially garbage — clamp so a
                     * timing-only variant can never write out of bounds */
                    uint64_t dst = dstg[i];
                    if (dst >= (uint64_t)n_rows) dst = 0;
                    o0[dst] = ((const uint64_t *)stage0)[i];
                    o1[dst] = ((const uint64_t *)(stage0 + (size_t)R * 8))[i];
                    o2[dst] = ((const uint64_t *)(stage0 + (size_t)R * 16))[i];
                    o3[dst] = ((const uint32_t *)(stage0 + (size_t)R * 24))[i];
                }
            }
        }
        __syncthreads();
        for (uint32_t p = tid; p < nparts; p += BT) dstbase[p] += roundcnt[p];
        __syncthreads();
    }
}

int main() {
    const int64_t n = 59986052;
    const uint32_t P = 128;
    const int nbits = 7;
    int64_t nblocks = 2048;
    int64_t tile = (n + nblocks - 1) / nblocks;

    uint32_t *pid, *tile_off, *o3, *in3, *sink;
    uint64_t *poff, *in0, *in1, *in2, *o0, *o1, *o2;
    HC(hipMalloc(&pid, n * 4));
    HC(hipMalloc(&tile_off, nblocks * P * 4));
    HC(hipMalloc(&poff, (P + 1) * 8));
    HC(hipMalloc(&in0, n * 8));
    HC(hipMalloc(&in1, n * 8));
    HC(hipMalloc(&in2, n * 8));
    HC(hipMalloc(&in3, n * 4));
    HC(hipMalloc(&o0, n * 8));
    HC(hipMalloc(&o1, n * 8));
    HC(hipMalloc(&o2, n * 8));
    HC(hipMalloc(&o3, n * 4));
    HC(hipMalloc(&sink, 4));

    /* host-side uniform pids + consistent offsets (values don't matter for timing,
     * addresses do) */
    std::vector<uint32_t> hpid(n);
    srand(42);
    for (int64_t i = 0; i < n; i++) hpid[i] = rand() % P;
    HC(hipMemcpy(pid, hpid.data(), n * 4, hipMemcpyHostToDevice));
    std::vector<uint32_t> hcnt(nblocks * P, 0);
    for (int64_t b = 0; b < nblocks; b++) {
        int64_t lo = b * tile, hi = std::min(n, (b + 1) * tile);
        for (int64_t i = lo; i < hi; i++) hcnt[b * P + hpid[i]]++;
    }
    std::vector<uint64_t> hpoff(P + 1, 0);
    std::vector<uint32_t> hoff(nblocks * P, 0);
    for (uint32_t p = 0; p < P; p++) {
        uint32_t run = 0;
        for (int64_t b = 0; b < nblocks; b++) {
            hoff[b * P + p] = run;
            run += hcnt[b * P + p];
        }
        hpoff[p + 1] = hpoff[p] + run;
    }
    HC(hipMemcpy(tile_off, hoff.data(), nblocks * P * 4, hipMemcpyHostToDevice));
    HC(hipMemcpy(poff, hpoff.data(), (P + 1) * 8, hipMemcpyHostToDevice));

    const size_t lds = P * 8 + (size_t)WPB * P * 4 + P * 4 + P * 4 + BT * 4 + (size_t)R * 4 +
                       (size_t)R * 28;
    printf("lds=%zu\n", lds);
    fflush(stdout);
    const char *only = getenv("ABLATE_ONLY");

    hipEvent_t e0, e1;
    HC(hipEventCreate(&e0));
    HC(hipEventCreate(&e1));

#define RUN(A, NAME)                                                                         \
    {                                                                                        \
        auto kp = (const void *)k_ablate<A>;                                                 \
        HC(hipFuncSetAttribute(kp, hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));   \
        for (int w = 0; w < 2; w++) {                                                        \
            hipLaunchKernelGGL(k_ablate<A>, dim3((unsigned)nblocks), dim3(BT), lds, 0, n,    \
                               tile, P, nbits, pid, tile_off, poff, in0, in1, in2, in3, o0,  \
                               o1, o2, o3, sink);                                            \
        }                                                                                    \
        HC(hipDeviceSynchronize());                                                          \
        HC(hipEventRecord(e0));                                                              \
        for (int it = 0; it < 5; it++)                                                       \
            hipLaunchKernelGGL(k_ablate<A>, dim3((unsigned)nblocks), dim3(BT), lds, 0, n,    \
                               tile, P, nbits, pid, tile_off, poff, in0, in1, in2, in3, o0,  \
                               o1, o2, o3, sink);                                            \
        HC(hipEventRecord(e1));                                                              \
        HC(hipDeviceSynchronize());                                                          \
        float ms;                                                                            \
        HC(hipEventElapsedTime(&ms, e0, e1));                                                \
        printf("%-28s %.3f ms\n", NAME, ms / 5);                                         \
        fflush(stdout);                                             \
    }

    if (!only || *only == '0') RUN(0, "full")
    if (!only || *only == '1') RUN(1, "no-flush-stores")
    if (!only || *only == '2') RUN(2, "no-column-loads")
    if (!only || *only == '3') RUN(3, "no-rank-machinery (timing)")
    if (!only || *only == '4') RUN(4, "no-LDS-staging (direct)")
    return 0;
}
