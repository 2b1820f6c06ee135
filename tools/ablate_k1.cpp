/* tools/ablate_k1.cpp — doubling ablation for K1 (k_hash_count_seg) at the bench shape:
 * 60M rows, one i64 key, P=128, u8 pids, per-64-row-segment histograms + fused level-1
 * scan partials. Marginal phase cost = t(2x phase) - t(full); patterns preserved.
 * Variants: 1=2x key loads, 2=2x hash, 3=2x pid stores, 4=2x counts stores,
 * 5=2x ballot-histogram, 6=LDS-atomicAdd counting instead of ballot-multisplit,
 * 7=no fused partials atomics.
 * Build+run: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ablate_k1.cpp -o /tmp/ak1 && /tmp/ak1
 */
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define WAVE 64
#define WPB 4
#define BT (WPB * WAVE)
#define SEG 256

#define HC(x)                                                                                \
    do {                                                                                     \
        hipError_t e_ = (x);                                                                 \
        if (e_ != hipSuccess) {                                                              \
            printf("HIP error %s at %d\n", hipGetErrorString(e_), __LINE__);                 \
            exit(1);                                                                         \
        }                                                                                    \
    } while (0)

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
    x ^= x >> 30;
    x *= 0xbf58476d1ce4e5b9ULL;
    x ^= x >> 27;
    x *= 0x94d049bb133111ebULL;
    x ^= x >> 31;
    return x;
}

__device__ __forceinline__ uint64_t eq_mask(uint32_t pid, uint64_t act, int nbits) {
    uint64_t eq = act;
    for (int b = 0; b < nbits; b++) {
        uint64_t bal = __ballot((pid >> b) & 1u);
        eq &= ((pid >> b) & 1u) ? bal : ~bal;
    }
    return eq;
}

template <int D>
__global__ __launch_bounds__(BT) void k1_abl(
    const uint64_t *keys, int64_t n, int64_t nseg, uint32_t P, int nbits, uint8_t *pid_out,
    uint32_t *counts, uint32_t *partials, int nranges, uint32_t *sink) {
    extern __shared__ char smem[];
    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int64_t seg = (int64_t)blockIdx.x * WPB + wid;
    if (seg >= nseg) return;
    uint32_t *hist = (uint32_t *)smem + (size_t)wid * P;
    for (uint32_t p = lane; p < P; p += WAVE) hist[p] = 0;
    const int64_t start = seg * SEG;
    const int64_t end = (start + SEG < n) ? start + SEG : n;
    uint32_t acc = 0;
    for (int64_t base = start; base < end; base += 4 * WAVE) {
        uint32_t pidu[4];
        bool actu[4];
#pragma unroll
        for (int u = 0; u < 4; u++) {
            const int64_t row = base + u * WAVE + lane;
            actu[u] = row < end;
            pidu[u] = 0;
            if (actu[u]) {
                uint64_t kv = keys[row];
                if (D == 1) acc += (uint32_t)keys[row]; /* 2x key load */
                uint64_t h = mix64(kv);
                h = 0 ^ (h + 0x9e3779b97f4a7c15ULL + 0 + 0);
                if (D == 2) { /* 2x hash */
                    uint64_t h2 = mix64(kv ^ 1);
                    acc += (uint32_t)(h2 ^ (h2 + 0x9e3779b97f4a7c15ULL));
                }
                pidu[u] = (uint32_t)(h & (uint64_t)(P - 1));
                pid_out[row] = (uint8_t)pidu[u];
                if (D == 3) pid_out[row] = (uint8_t)pidu[u]; /* 2x pid store (WAW) */
            }
        }
#pragma unroll
        for (int u = 0; u < 4; u++) {
            if (D == 6) { /* LDS atomics instead of ballot-multisplit */
                if (actu[u]) atomicAdd(&hist[pidu[u]], 1u);
                continue;
            }
            uint64_t act = __ballot(actu[u]);
            if (actu[u]) {
                uint64_t eq = eq_mask(pidu[u], act, nbits);
                int leader = __ffsll((unsigned long long)eq) - 1;
                if (lane == leader) hist[pidu[u]] += (uint32_t)__popcll((unsigned long long)eq);
                if (D == 5) { /* 2x ballot machinery (result discarded via acc) */
                    uint64_t eq2 = eq_mask(pidu[u] ^ 1, act, nbits);
                    acc += (uint32_t)__popcll((unsigned long long)eq2);
                }
            }
        }
    }
    const int64_t rw = ((seg + 1) * (int64_t)nranges + nseg - 1) / nseg - 1;
    uint32_t *prow = partials + (size_t)rw * P;
    for (uint32_t p = lane; p < P; p += WAVE) {
        const uint32_t h = hist[p];
        counts[(size_t)seg * P + p] = h;
        if (D == 4) counts[(size_t)seg * P + p] = h; /* 2x counts store */
        if (D != 7 && h) atomicAdd(&prow[p], h);
        if (D == 4 && h) atomicAdd(&prow[p], 0u);
    }
    if (acc == 0xdeadbeefu) sink[0] = acc;
}

int main() {
    const int64_t n = 59986052;
    const uint32_t P = 128;
    const int nbits = 7;
    const int64_t nseg = (n + SEG - 1) / SEG;
    const int64_t nseg_pad = (nseg + 3) & ~3LL;
    std::vector<uint64_t> keys(n);
    srand(7);
    for (int64_t i = 0; i < n; i++)
        keys[i] = ((uint64_t)rand() << 32) ^ (uint64_t)rand();
    uint64_t *d_keys;
    uint8_t *d_pid;
    uint32_t *d_counts, *d_part, *d_sink;
    HC(hipMalloc(&d_keys, n * 8));
    HC(hipMalloc(&d_pid, n));
    HC(hipMalloc(&d_counts, (size_t)nseg_pad * P * 4));
    HC(hipMalloc(&d_part, (size_t)2048 * P * 4));
    HC(hipMalloc(&d_sink, 4));
    HC(hipMemcpy(d_keys, keys.data(), n * 8, hipMemcpyHostToDevice));
    const size_t lds = (size_t)WPB * P * 4;
    auto run = [&](auto tag, const char *name) {
        constexpr int A = decltype(tag)::value;
        hipEvent_t e0, e1;
        HC(hipEventCreate(&e0));
        HC(hipEventCreate(&e1));
        float best = 1e9f;
        for (int rep = 0; rep < 12; rep++) {
            HC(hipMemsetAsync(d_part, 0, (size_t)2048 * P * 4, 0));
            HC(hipEventRecord(e0, 0));
            hipLaunchKernelGGL((k1_abl<A>), dim3((unsigned)(nseg_pad / WPB)), dim3(BT),
                               lds, 0, d_keys, n, nseg_pad, P, nbits, d_pid, d_counts,
                               d_part, 2048, d_sink);
            HC(hipEventRecord(e1, 0));
            HC(hipEventSynchronize(e1));
            float ms;
            HC(hipEventElapsedTime(&ms, e0, e1));
            if (rep >= 2 && ms < best) best = ms;
        }
        printf("%-26s %.4f ms\n", name, best);
        HC(hipEventDestroy(e0));
        HC(hipEventDestroy(e1));
        return best;
    };
    setvbuf(stdout, nullptr, _IONBF, 0);
    float full = run(std::integral_constant<int, 0>{}, "full");
    float t1 = run(std::integral_constant<int, 1>{}, "2x_key_loads");
    float t2 = run(std::integral_constant<int, 2>{}, "2x_hash");
    float t3 = run(std::integral_constant<int, 3>{}, "2x_pid_stores");
    float t4 = run(std::integral_constant<int, 4>{}, "2x_counts_stores");
    float t5 = run(std::integral_constant<int, 5>{}, "2x_ballot_hist");
    float t6 = run(std::integral_constant<int, 6>{}, "lds_atomic_counting");
    float t7 = run(std::integral_constant<int, 7>{}, "no_fused_partials");
    printf("marginal: keys=%.3f hash=%.3f pid=%.3f counts=%.3f ballot=%.3f | "
           "atomic_vs_ballot=%.3f fused_partials=%.3f\n",
           t1 - full, t2 - full, t3 - full, t4 - full, t5 - full, t6 - full, full - t7);
    return 0;
}
