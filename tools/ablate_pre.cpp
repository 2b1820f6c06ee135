/* tools/ablate_pre.cpp — standalone ablation microbench for the precomputed-layout
 * scatter (k_scatter_pre, dd_kernels.hip K3-P). Perf-engineering infrastructure, not
 * part of the product library: a trimmed copy of the kernel at the bench shape (60M
 * rows, P=128, cols 8/8/8/4 B, u8 pid, NT flush stores, rpb=2) with compile-time
 * ablations attributing its time between phases:
 *
 *   ABLATE=0 full kernel (verified against a host model on a sample)
 *   ABLATE=1 no flush stores (LDS still read into a sink)
 *   ABLATE=2 no column loads (values synthesized; pid + bases still loaded)
 *   ABLATE=3 no rank machinery (fake rank; WRONG results, timing only)
 *   ABLATE=4 no LDS staging (direct register->global scatter at gdst; no place/flush)
 *   ABLATE=5 no base rows (fake bases; WRONG results, timing only)
 *   ABLATE=6 no pid load (pid synthesized from the row index: round-robin)
 *
 * Build+run (GPU box):
 *   hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ablate_pre.cpp -o /tmp/ablpre && /tmp/ablpre
 */

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#define WAVE 64
#define WPB 16
#define GMAX 4
#define BT (WPB * WAVE)
#define R (GMAX * BT)
#define SEG (R / WPB)
#define NBG 2
#define NBI 1
#define NB (NBG + NBI)
#define NC 4
#define L (GMAX * NC)

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

#define LD64(dst, p) asm volatile("global_load_dwordx2 %0, %1, off" : "=v"(dst) : "v"(p))
#define LD32(dst, p) asm volatile("global_load_dword %0, %1, off" : "=v"(dst) : "v"(p))
#define LD8(dst, p) asm volatile("global_load_ubyte %0, %1, off" : "=v"(dst) : "v"(p))

template <int CNT, typename T> __device__ __forceinline__ void tie_wait(T &v) {
    asm volatile("s_waitcnt vmcnt(%1)" : "+v"(v) : "n"(CNT) : "memory");
}

/* DOUBLE: 0=none (full), 1=pid loads x2, 2=col loads x2, 3=base loads x2,
 * 4=rank x2, 5=place x2, 6=flush stores x2. Marginal phase cost = t(DOUBLE=k) - t(full),
 * with access patterns and results IDENTICAL to the full kernel (duplicated work is
 * idempotent), so overlapped phases correctly show ~zero marginal cost. */
template <int D>
__global__ __launch_bounds__(BT) void k_abl(
    int64_t n_rows, int64_t nrounds, int rpb, uint32_t nparts, int nbits,
    const uint8_t *pid_in, const uint32_t *gbase, const uint16_t *rofftab, uint32_t sP2,
    const uint64_t *in0, const uint64_t *in1, const uint64_t *in2, const uint32_t *in3,
    uint64_t *o0, uint64_t *o1, uint64_t *o2, uint32_t *o3, uint64_t *ob0, uint64_t *ob1,
    uint64_t *ob2, uint32_t *ob3, uint32_t *sink) {
    constexpr int NSTO = L * (D == 6 ? 2 : 1);
    static_assert(NSTO <= 63, "vmcnt");
    extern __shared__ __attribute__((aligned(16))) char smem[];
    char *ws = smem;
    char *const stage0 = ws;
    ws += (size_t)R * 28;
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
    uint32_t *gb0 = gb_all;
    uint32_t *ib32 = (uint32_t *)(ib_all + (size_t)wid * sP2);
    uint16_t *rf = ib_all + (size_t)wid * sP2;
    uint16_t *ms = ms_all + (size_t)wid * nparts;
    const uint64_t lt = ((uint64_t)1 << lane) - 1;

    const int64_t r0 = (int64_t)blockIdx.x * rpb;
    const int64_t r1 = (r0 + rpb < nrounds) ? r0 + rpb : nrounds;
    if (r0 >= nrounds || n_rows == 0) return;

    uint32_t pidr[GMAX], rankr[GMAX];
    uint32_t pidr2[GMAX]; /* D==1 duplicate */
    bool actr[GMAX];
    uint64_t c0v[GMAX], c1v[GMAX], c2v[GMAX];
    uint32_t c3v[GMAX];
    uint64_t x0v[GMAX], x1v[GMAX], x2v[GMAX]; /* D==2 duplicates */
    uint32_t x3v[GMAX];
    uint32_t baser[NB], baser2[NB];
    const uint32_t ndw = (nparts + 1) / 2;

    auto preload = [&](int64_t r) {
        const int64_t rstart = r * R;
        const int64_t rend = (rstart + R < n_rows) ? rstart + R : n_rows;
        const int64_t segstart = rstart + (int64_t)wid * SEG;
        const int64_t seg = r * WPB + wid;
        uint32_t rowc[GMAX];
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const int64_t row = segstart + g * WAVE + lane;
            actr[g] = row < rend;
            rowc[g] = (uint32_t)(actr[g] ? row : rend - 1);
        }
#pragma unroll
        for (int g = 0; g < GMAX; g++) LD8(pidr[g], pid_in + rowc[g]);
        if (D == 1) {
#pragma unroll
            for (int g = 0; g < GMAX; g++) LD8(pidr2[g], pid_in + rowc[g]);
        }
#pragma unroll
        for (int g = 0; g < GMAX; g++) LD64(c0v[g], in0 + rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) LD64(c1v[g], in1 + rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) LD64(c2v[g], in2 + rowc[g]);
#pragma unroll
        for (int g = 0; g < GMAX; g++) LD32(c3v[g], in3 + rowc[g]);
        if (D == 2) {
#pragma unroll
            for (int g = 0; g < GMAX; g++) LD64(x0v[g], in0 + rowc[g]);
#pragma unroll
            for (int g = 0; g < GMAX; g++) LD64(x1v[g], in1 + rowc[g]);
#pragma unroll
            for (int g = 0; g < GMAX; g++) LD64(x2v[g], in2 + rowc[g]);
#pragma unroll
            for (int g = 0; g < GMAX; g++) LD32(x3v[g], in3 + rowc[g]);
        }
        const uint32_t *grow = gbase + (size_t)seg * nparts;
        const uint32_t *irow = (const uint32_t *)(rofftab + (size_t)r * sP2);
#pragma unroll
        for (int k = 0; k < NBG; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx >= nparts) idx = nparts - 1;
            LD32(baser[k], grow + idx);
        }
#pragma unroll
        for (int k = 0; k < NBI; k++) {
            uint32_t idx = (uint32_t)lane + k * WAVE;
            if (idx >= ndw) idx = ndw - 1;
            LD32(baser[NBG + k], irow + idx);
        }
        if (D == 3) {
#pragma unroll
            for (int k = 0; k < NBG; k++) {
                uint32_t idx = (uint32_t)lane + k * WAVE;
                if (idx >= nparts) idx = nparts - 1;
                LD32(baser2[k], grow + idx);
            }
#pragma unroll
            for (int k = 0; k < NBI; k++) {
                uint32_t idx = (uint32_t)lane + k * WAVE;
                if (idx >= ndw) idx = ndw - 1;
                LD32(baser2[NBG + k], irow + idx);
            }
        }
    };

    auto wait_all = [&](auto cnt) {
#pragma unroll
        for (int g = 0; g < GMAX; g++) tie_wait<cnt.value>(pidr[g]);
        if (D == 1) {
#pragma unroll
            for (int g = 0; g < GMAX; g++) tie_wait<cnt.value>(pidr2[g]);
        }
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            tie_wait<cnt.value>(c0v[g]);
            tie_wait<cnt.value>(c1v[g]);
            tie_wait<cnt.value>(c2v[g]);
            tie_wait<cnt.value>(c3v[g]);
        }
        if (D == 2) {
#pragma unroll
            for (int g = 0; g < GMAX; g++) {
                tie_wait<cnt.value>(x0v[g]);
                tie_wait<cnt.value>(x1v[g]);
                tie_wait<cnt.value>(x2v[g]);
                tie_wait<cnt.value>(x3v[g]);
            }
        }
#pragma unroll
        for (int k = 0; k < NB; k++) tie_wait<cnt.value>(baser[k]);
        if (D == 3) {
#pragma unroll
            for (int k = 0; k < NB; k++) tie_wait<cnt.value>(baser2[k]);
        }
    };

    auto write_base_rows = [&]() {
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
        if (D == 3) { /* idempotent duplicate */
#pragma unroll
            for (int k = 0; k < NBG; k++) {
                uint32_t idx = (uint32_t)lane + k * WAVE;
                if (idx < nparts) gb[idx] = baser2[k];
            }
        }
    };

    auto rank1 = [&]() {
        for (uint32_t p = lane; p < nparts; p += WAVE) ms[p] = 0;
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            const bool active = actr[g];
            const uint32_t pid = pidr[g];
            uint64_t act = __ballot(active);
            uint32_t rk = 0;
            if (active) {
                uint64_t eq = eq_mask(pid, act, nbits);
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
    auto rank = [&]() {
        rank1();
        if (D == 4) rank1(); /* identical recompute (LDS side effects; not elided) */
    };

    auto place = [&]() {
#pragma unroll
        for (int g = 0; g < GMAX; g++) {
            if (!actr[g]) continue;
            const uint32_t pid = pidr[g];
            const uint32_t rk = rankr[g];
            const uint32_t gd = gb[pid] + rk;
            const uint32_t slot = (uint32_t)rf[pid] + (gd - gb0[pid]);
            dstg[slot] = gd;
            char *stage = stage0;
            ((uint64_t *)stage)[slot] = c0v[g];
            stage += (size_t)R * 8;
            ((uint64_t *)stage)[slot] = c1v[g];
            stage += (size_t)R * 8;
            ((uint64_t *)stage)[slot] = c2v[g];
            stage += (size_t)R * 8;
            ((uint32_t *)stage)[slot] = c3v[g];
        }
    };

    preload(r0);
    wait_all(std::integral_constant<int, 0>{});
    write_base_rows();
    rank();
    __syncthreads();

    for (int64_t r = r0; r < r1; r++) {
        const int64_t rstart = r * R;
        const int round_rows =
            (int)(((rstart + R < n_rows) ? rstart + R : n_rows) - rstart);

        place();
        if (D == 5) place(); /* idempotent duplicate */
        __syncthreads();

        const bool more = r + 1 < r1;
        if (more) preload(r + 1);

        if (D == 8) { /* paired flush: lane covers slots (2i, 2i+1); same-run pairs
                          (~97% at run ~32 slots) fuse into one 16B store per column */
#pragma unroll
            for (int u = 0; u < GMAX / 2; u++) {
                const int i = 2 * (tid + u * BT);
                const int ic = (i < round_rows) ? i : (round_rows - 1);
                const int ic2 = (i + 1 < round_rows) ? i + 1 : (round_rows - 1);
                const uint64_t d0 = dstg[ic];
                const uint64_t d1 = dstg[ic2];
                char *stage = stage0;
                if (d1 == d0 + 1 && (d0 & 1) == 0) {
                    { uint64_t v2[2] = {((const uint64_t *)stage)[ic], ((const uint64_t *)stage)[ic2]};
                      __builtin_memcpy(o0 + d0, v2, 16); }
                    stage += (size_t)R * 8;
                    { uint64_t v2[2] = {((const uint64_t *)stage)[ic], ((const uint64_t *)stage)[ic2]};
                      __builtin_memcpy(o1 + d0, v2, 16); }
                    stage += (size_t)R * 8;
                    { uint64_t v2[2] = {((const uint64_t *)stage)[ic], ((const uint64_t *)stage)[ic2]};
                      __builtin_memcpy(o2 + d0, v2, 16); }
                    stage += (size_t)R * 8;
                    { uint32_t v2[2] = {((const uint32_t *)stage)[ic], ((const uint32_t *)stage)[ic2]};
                      __builtin_memcpy(o3 + d0, v2, 8); }
                } else {
                    o0[d0] = ((const uint64_t *)stage)[ic];
                    o0[d1] = ((const uint64_t *)stage)[ic2];
                    stage += (size_t)R * 8;
                    o1[d0] = ((const uint64_t *)stage)[ic];
                    o1[d1] = ((const uint64_t *)stage)[ic2];
                    stage += (size_t)R * 8;
                    o2[d0] = ((const uint64_t *)stage)[ic];
                    o2[d1] = ((const uint64_t *)stage)[ic2];
                    stage += (size_t)R * 8;
                    o3[d0] = ((const uint32_t *)stage)[ic];
                    o3[d1] = ((const uint32_t *)stage)[ic2];
                }
            }
        } else
        for (int rep = 0; rep < (D == 6 ? 2 : 1); rep++) {
            /* rep 1 (D==6) writes a SECOND buffer set: same pattern, no WAW hazard */
            uint64_t *q0 = rep ? ob0 : o0, *q1 = rep ? ob1 : o1, *q2 = rep ? ob2 : o2;
            uint32_t *q3 = rep ? ob3 : o3;
#pragma unroll
            for (int u = 0; u < GMAX; u++) {
                const int i = tid + u * BT;
                const int ic = (i < round_rows) ? i : (round_rows - 1);
                const uint64_t dst = dstg[ic];
                char *stage = stage0;
                if (D == 7) { /* true-NT stores (negative result: ~30% slower) */
                    __builtin_nontemporal_store(((const uint64_t *)stage)[ic], q0 + dst);
                    stage += (size_t)R * 8;
                    __builtin_nontemporal_store(((const uint64_t *)stage)[ic], q1 + dst);
                    stage += (size_t)R * 8;
                    __builtin_nontemporal_store(((const uint64_t *)stage)[ic], q2 + dst);
                    stage += (size_t)R * 8;
                    __builtin_nontemporal_store(((const uint32_t *)stage)[ic], q3 + dst);
                } else { /* plain (cached) stores — the product default */
                    ((uint64_t *)q0)[dst] = ((const uint64_t *)stage)[ic];
                    stage += (size_t)R * 8;
                    ((uint64_t *)q1)[dst] = ((const uint64_t *)stage)[ic];
                    stage += (size_t)R * 8;
                    ((uint64_t *)q2)[dst] = ((const uint64_t *)stage)[ic];
                    stage += (size_t)R * 8;
                    ((uint32_t *)q3)[dst] = ((const uint32_t *)stage)[ic];
                }
            }
        }

        if (more) {
            wait_all(std::integral_constant<int, NSTO>{});
            write_base_rows();
            rank();
        }
        __syncthreads();
    }
    if (sink && tid == 0x7fffffff) sink[0] = rankr[0] + (uint32_t)c0v[0] + baser[0];
}

int main() {
    setvbuf(stdout, nullptr, _IONBF, 0);
    const int64_t n = 59986052;
    const uint32_t P = 128;
    const uint32_t sP2 = P;
    const int nbits = 7;
    const int rpb = 2;
    const int64_t nrounds = (n + R - 1) / R;
    const int64_t nseg = nrounds * WPB;
    const int64_t nblocks = (nrounds + rpb - 1) / rpb;

    /* host-side layout computation */
    std::vector<uint8_t> pid(n);
    srand(42);
    for (int64_t i = 0; i < n; i++) pid[i] = (uint8_t)(rand() % P);
    std::vector<uint32_t> counts((size_t)nseg * P, 0);
    for (int64_t i = 0; i < n; i++) counts[(size_t)(i / SEG) * P + pid[i]]++;
    std::vector<uint64_t> ptot(P + 1, 0);
    for (uint32_t p = 0; p < P; p++) {
        uint64_t s = 0;
        for (int64_t c = 0; c < nseg; c++) s += counts[(size_t)c * P + p];
        ptot[p + 1] = ptot[p] + s;
    }
    std::vector<uint32_t> gbase((size_t)nseg * P);
    for (uint32_t p = 0; p < P; p++) {
        uint32_t run = (uint32_t)ptot[p];
        for (int64_t c = 0; c < nseg; c++) {
            gbase[(size_t)c * P + p] = run;
            run += counts[(size_t)c * P + p];
        }
    }
    std::vector<uint16_t> roff((size_t)nrounds * sP2, 0);
    for (int64_t r = 0; r < nrounds; r++) {
        uint32_t run = 0;
        for (uint32_t p = 0; p < P; p++) {
            uint32_t rb = gbase[(size_t)r * WPB * P + p];
            uint32_t re = (r + 1 < nrounds) ? gbase[(size_t)(r + 1) * WPB * P + p]
                                            : (uint32_t)ptot[p + 1];
            roff[(size_t)r * sP2 + p] = (uint16_t)run;
            run += re - rb;
        }
    }

    uint8_t *d_pid;
    uint32_t *d_gbase, *d_in3, *d_o3, *d_ob3, *d_sink;
    uint16_t *d_roff;
    uint64_t *d_in0, *d_in1, *d_in2, *d_o0, *d_o1, *d_o2, *d_ob0, *d_ob1, *d_ob2;
    HC(hipMalloc(&d_pid, n));
    HC(hipMalloc(&d_gbase, gbase.size() * 4));
    HC(hipMalloc(&d_roff, roff.size() * 2));
    HC(hipMalloc(&d_in0, n * 8));
    HC(hipMalloc(&d_in1, n * 8));
    HC(hipMalloc(&d_in2, n * 8));
    HC(hipMalloc(&d_in3, n * 4));
    HC(hipMalloc(&d_o0, n * 8));
    HC(hipMalloc(&d_o1, n * 8));
    HC(hipMalloc(&d_o2, n * 8));
    HC(hipMalloc(&d_o3, n * 4));
    HC(hipMalloc(&d_ob0, n * 8));
    HC(hipMalloc(&d_ob1, n * 8));
    HC(hipMalloc(&d_ob2, n * 8));
    HC(hipMalloc(&d_ob3, n * 4));
    HC(hipMalloc(&d_sink, 4));
    HC(hipMemcpy(d_pid, pid.data(), n, hipMemcpyHostToDevice));
    HC(hipMemcpy(d_gbase, gbase.data(), gbase.size() * 4, hipMemcpyHostToDevice));
    HC(hipMemcpy(d_roff, roff.data(), roff.size() * 2, hipMemcpyHostToDevice));
    std::vector<uint64_t> hv(n);
    for (int64_t i = 0; i < n; i++) hv[i] = (uint64_t)i * 0x9e3779b97f4a7c15ULL;
    HC(hipMemcpy(d_in0, hv.data(), n * 8, hipMemcpyHostToDevice));
    HC(hipMemcpy(d_in1, hv.data(), n * 8, hipMemcpyHostToDevice));
    HC(hipMemcpy(d_in2, hv.data(), n * 8, hipMemcpyHostToDevice));
    std::vector<uint32_t> hv32(n);
    for (int64_t i = 0; i < n; i++) hv32[i] = (uint32_t)i;
    HC(hipMemcpy(d_in3, hv32.data(), n * 4, hipMemcpyHostToDevice));

    const size_t lds = (size_t)R * 28 + 4 * R + WPB * (4 * P + 2 * sP2 + 2 * P);
    printf("n=%lld nrounds=%lld nblocks=%lld lds=%zu\n", (long long)n, (long long)nrounds,
           (long long)nblocks, lds);

    auto run = [&](auto tag, const char *name) {
        constexpr int A = decltype(tag)::value;
        HC(hipFuncSetAttribute((const void *)k_abl<A>,
                               hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds));
        hipEvent_t e0, e1;
        HC(hipEventCreate(&e0));
        HC(hipEventCreate(&e1));
        float best = 1e9f;
        for (int rep = 0; rep < 12; rep++) {
            HC(hipEventRecord(e0, 0));
            hipLaunchKernelGGL((k_abl<A>), dim3((unsigned)nblocks), dim3(BT), lds, 0, n,
                               nrounds, rpb, P, nbits, d_pid, d_gbase, d_roff, sP2, d_in0,
                               d_in1, d_in2, d_in3, d_o0, d_o1, d_o2, d_o3, d_ob0,
                               d_ob1, d_ob2, d_ob3, d_sink);
            HC(hipEventRecord(e1, 0));
            HC(hipEventSynchronize(e1));
            float ms;
            HC(hipEventElapsedTime(&ms, e0, e1));
            if (rep >= 2 && ms < best) best = ms;
        }
        printf("%-28s %.4f ms\n", name, best);
        HC(hipEventDestroy(e0));
        HC(hipEventDestroy(e1));
        return best;
    };

    float full = run(std::integral_constant<int, 0>{}, "full");
    {
        std::vector<uint64_t> out0(n);
        HC(hipMemcpy(out0.data(), d_o0, n * 8, hipMemcpyDeviceToHost));
        std::vector<uint64_t> cursor(P);
        for (uint32_t p = 0; p < P; p++) cursor[p] = ptot[p];
        int bad = 0;
        for (int64_t i = 0; i < n && bad < 5; i++) {
            uint64_t d = cursor[pid[i]]++;
            if (out0[d] != hv[i]) {
                printf("MISMATCH row %lld dst %llu\n", (long long)i, (unsigned long long)d);
                bad++;
            }
        }
        printf(bad ? "VERIFY FAILED\n" : "verify ok (full kernel, all rows)\n");
    }
    const char *names[] = {"", "2x_pid_loads", "2x_col_loads", "2x_base_loads", "2x_rank",
                           "2x_place", "2x_flush_stores"};
    float t1 = run(std::integral_constant<int, 1>{}, names[1]);
    float t2 = run(std::integral_constant<int, 2>{}, names[2]);
    float t3 = run(std::integral_constant<int, 3>{}, names[3]);
    float t4 = run(std::integral_constant<int, 4>{}, names[4]);
    float t5 = run(std::integral_constant<int, 5>{}, names[5]);
    float t6 = run(std::integral_constant<int, 6>{}, names[6]);
    float t7 = run(std::integral_constant<int, 7>{}, "nt_stores(negative)");
    float t8 = run(std::integral_constant<int, 8>{}, "paired_16B_flush");
    {
        /* verify the paired variant too (it ran last into o0) */
        std::vector<uint64_t> out0(n);
        HC(hipMemcpy(out0.data(), d_o0, n * 8, hipMemcpyDeviceToHost));
        std::vector<uint64_t> cursor(P);
        for (uint32_t p = 0; p < P; p++) cursor[p] = ptot[p];
        int bad = 0;
        for (int64_t i = 0; i < n && bad < 3; i++) {
            uint64_t d = cursor[pid[i]]++;
            if (out0[d] != hv[i]) bad++;
        }
        printf(bad ? "PAIRED VERIFY FAILED\n" : "paired verify ok\n");
        printf("paired delta vs full: %.3f ms\n", t8 - full);
    }
    printf("marginal ms: pid=%.3f cols=%.3f bases=%.3f rank=%.3f place=%.3f flush=%.3f "
           "nt_delta=%.3f\n",
           t1 - full, t2 - full, t3 - full, t4 - full, t5 - full, t6 - full, t7 - full);
    return 0;
}
