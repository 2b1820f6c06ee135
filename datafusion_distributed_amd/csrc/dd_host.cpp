/* dd_host.cpp — host side of libdd_shuffle.so: the C ABI (include/dd_shuffle.h), the task
 * cache mirroring the reference's worker execute path, and the RCCL/xGMI exchange replacing
 * the Arrow-Flight data plane. See header citations in include/dd_shuffle.h and DESIGN.md.
 *
 * Product path only: no CPU fallback anywhere — a missing/failed HIP device is a hard error
 * (DD_ERR_NO_DEVICE), by design.
 */

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <cmath>
#include <cstdlib>
#include <cstdio>
#include <cstring>
#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "../../include/dd_shuffle.h"
#include "dd_internal.h"

/* ---------------- error plumbing ---------------- */

static thread_local std::string g_last_error;

static dd_status set_err(dd_status s, const std::string &msg) {
    g_last_error = msg;
    return s;
}

extern "C" const char *dd_last_error(void) { return g_last_error.c_str(); }
/* shared with dd_proto.cpp (declared in dd_internal.h) */
extern "C" dd_status dd_set_error(dd_status s, const char *msg) {
    g_last_error = msg;
    return s;
}
extern "C" const char *dd_version(void) { return "dd_shuffle 0.1 (gfx950)"; }

#define HIP_TRY(expr)                                                                        \
    do {                                                                                     \
        hipError_t _e = (expr);                                                              \
        if (_e != hipSuccess)                                                                \
            return set_err(DD_ERR_HIP, std::string(#expr) + ": " + hipGetErrorString(_e));   \
    } while (0)

#define NCCL_TRY(expr)                                                                       \
    do {                                                                                     \
        ncclResult_t _r = (expr);                                                            \
        if (_r != ncclSuccess)                                                               \
            return set_err(DD_ERR_RCCL, std::string(#expr) + ": " + ncclGetErrorString(_r)); \
    } while (0)

/* Dedicated stream for pooled (stream-ordered) allocations: hipMallocAsync/hipFreeAsync
 * on the NULL stream never reuse the pool on ROCm 7.2 (measured: 50x 16 MB alloc/free
 * cycles leak ~190 MB on NULL stream, 0 on a real stream), so all pool traffic goes
 * through this stream and users of the buffers sync it once after allocating. */
static hipStream_t dd_pool_stream(void) {
    static hipStream_t s = nullptr;
    static std::once_flag once;
    std::call_once(once, [] { (void)hipStreamCreate(&s); });
    return s;
}

extern "C" int dd_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

/* ---------------- device helpers ---------------- */

extern "C" dd_status dd_dev_alloc(int64_t bytes, void **out) {
    if (dd_device_count() == 0) return set_err(DD_ERR_NO_DEVICE, "no HIP device");
    HIP_TRY(hipMalloc(out, bytes > 0 ? (size_t)bytes : 1));
    return DD_OK;
}
extern "C" dd_status dd_dev_free(void *p) {
    HIP_TRY(hipFree(p));
    return DD_OK;
}
extern "C" dd_status dd_memcpy_h2d(void *dst, const void *src, int64_t bytes) {
    HIP_TRY(hipMemcpy(dst, src, (size_t)bytes, hipMemcpyHostToDevice));
    return DD_OK;
}
extern "C" dd_status dd_memcpy_d2h(void *dst, const void *src, int64_t bytes) {
    HIP_TRY(hipMemcpy(dst, src, (size_t)bytes, hipMemcpyDeviceToHost));
    return DD_OK;
}
extern "C" dd_status dd_memcpy_d2d(void *dst, const void *src, int64_t bytes) {
    HIP_TRY(hipMemcpy(dst, src, (size_t)bytes, hipMemcpyDeviceToDevice));
    return DD_OK;
}
extern "C" dd_status dd_device_sync(void) {
    HIP_TRY(hipDeviceSynchronize());
    return DD_OK;
}

/* ---------------- partitioner ---------------- */

static int fixed_elem_size(int dtype) {
    switch (dtype) {
    case DD_DT_U8:
    case DD_DT_BOOL:
        return 1;
    case DD_DT_I16:
        return 2;
    case DD_DT_I32:
    case DD_DT_F32:
    case DD_DT_DICT32:
        return 4;
    case DD_DT_I64:
    case DD_DT_F64:
        return 8;
    default:
        return 0; /* utf8 */
    }
}

struct dd_partitioner {
    dd_batch_desc batch;
    dd_kargs ka;
    uint32_t nparts = 0;
    int nbits = 0;
    int64_t nchunks = 0, chunk_rows = 0;
    size_t lds_k1 = 0, lds_k3 = 0;
    bool staged = false; /* v2 path: block-tile + LDS-staged scatter (fixed-width only) */
    int gmax = 0;        /* v2 groups per wave per round */
    int wpb = 4;         /* v2 waves per block; rows per round R = gmax * wpb * 64 */
    bool pre = false;    /* K3-P: precomputed round layout (k_scatter_pre) */
    int64_t nrounds = 0, nseg_pad = 0; /* pre: globally-aligned rounds / padded segments */
    int rpb = 1;                       /* pre: rounds per block */
    int pre_nranges = 0;               /* pre: scan ranges (segment-granular counts) */
    uint32_t sP2 = 0;                  /* pre: imgb row stride (u16, even) */

    /* device buffers (owned) */
    uint32_t *pid = nullptr;
    uint32_t *counts = nullptr;          /* [nchunks][P] */
    uint32_t *partials = nullptr;        /* [RANGES][P] */
    uint64_t *part_offsets = nullptr;    /* [P+1] */
    uint32_t *bcounts = nullptr;         /* v1: [nvar][nchunks][P] */
    uint32_t *bpartials = nullptr;       /* v1: [nvar][RANGES][P] */
    uint64_t *part_boffsets = nullptr;   /* [nvar][P+1] */
    uint16_t *imgb = nullptr;            /* pre: [nrounds][sP2] round image bases (roff) */
    uint32_t *partials2 = nullptr;       /* pre: second-level scan partials [64][P] */
    uint16_t *counts16 = nullptr;        /* pre: [nseg][P] u16 segment counts */
    uint32_t *gbase = nullptr;           /* pre: [nseg][P] u32 global slot bases */
    uint32_t *src_row = nullptr;         /* staged-var: permutation out[slot] = input row */
    uint32_t *k4w_meta = nullptr;        /* staged-var: window hist/base (k4_copy_ord) */
    uint32_t *k4w_order = nullptr;       /* staged-var: window-bucketed group order */
    int k5 = 0;                          /* LDS-staged var-byte scatter (small strings) */
    int k5_wpb = 16;                     /* waves/block (8 = 2 blocks/CU overlap) */
    int k5_nranges = 2048;               /* fused-partials ranges (scales with nseg5) */
    int64_t k5_nrounds = 0, k5_nseg = 0;
    uint32_t k5_maxlen = 0;
    size_t lds_k5 = 0, lds_k5c = 0;
    uint16_t *k5_bcounts = nullptr;      /* [nseg5][P] u16 byte counts */
    uint32_t *k5_gbase = nullptr;        /* [nseg5][P] u32 global byte bases */
    uint32_t *k5_partials = nullptr;     /* [2048][P] */
    uint32_t *k5_partials2 = nullptr;    /* [64][P] */
    uint32_t *k5_roffB = nullptr;        /* [nrounds5][P+1] */
    uint64_t *out_off[DD_KMAX_VAR] = {}; /* staged-var: Arrow byte offsets [n+1] per var */
    uint64_t *k4_partials = nullptr;     /* staged-var scan scratch */
    uint64_t *dict_hashes[DD_KMAX_COLS] = {};
    void *out_data[DD_KMAX_COLS] = {};
    uint8_t *out_valid[DD_KMAX_COLS] = {};
    uint32_t *out_lengths[DD_KMAX_COLS] = {};

    hipEvent_t ev[5] = {}; /* 0 K1start 1 K1end 2 scans-end 3 K3end 4 K3start */
    std::atomic<bool> has_run{false}; /* release/acquire pairs with dd_execute_task */

    ~dd_partitioner() {
        (void)hipFree(pid);
        (void)hipFree(imgb);
        (void)hipFree(partials2);
        (void)hipFree(counts);
        (void)hipFree(partials);
        (void)hipFree(part_offsets);
        (void)hipFree(bcounts);
        (void)hipFree(bpartials);
        (void)hipFree(part_boffsets);
        (void)hipFree(src_row);
        (void)hipFree(k4w_meta);
        (void)hipFree(k4w_order);
        (void)hipFree(counts16);
        (void)hipFree(gbase);
        (void)hipFree(k5_bcounts);
        (void)hipFree(k5_gbase);
        (void)hipFree(k5_partials);
        (void)hipFree(k5_partials2);
        (void)hipFree(k5_roffB);
        for (auto &o : out_off) hipFree(o);
        (void)hipFree(k4_partials);
        for (int i = 0; i < DD_KMAX_COLS; i++) {
            (void)hipFree(dict_hashes[i]);
            (void)hipFree(out_data[i]);
            (void)hipFree(out_valid[i]);
            (void)hipFree(out_lengths[i]);
        }
        for (auto &e : ev)
            if (e) (void)hipEventDestroy(e);
    }
};

extern "C" dd_status dd_partitioner_create(const dd_batch_desc *batch, const int32_t *key_cols,
                                           int32_t n_keys, uint32_t n_partitions,
                                           dd_partitioner **out) {
    if (!batch || !key_cols || !out) return set_err(DD_ERR_INVALID, "null argument");
    if (dd_device_count() == 0)
        return set_err(DD_ERR_NO_DEVICE,
                       "no HIP device: the product path has no CPU fallback (DESIGN.md §2)");
    if (batch->n_cols < 1 || batch->n_cols > DD_MAX_COLS)
        return set_err(DD_ERR_INVALID, "n_cols out of range");
    if (n_keys < 1 || n_keys > DD_MAX_KEYS) return set_err(DD_ERR_INVALID, "n_keys out of range");
    if (n_partitions < 1 || n_partitions > DD_MAX_PARTITIONS)
        return set_err(DD_ERR_UNSUPPORTED, "n_partitions outside round-1 cap");
    if (batch->n_rows < 0 || batch->n_rows >= (int64_t)UINT32_MAX)
        return set_err(DD_ERR_UNSUPPORTED, "n_rows >= 2^32: chunk the batch");

    auto p = new dd_partitioner();
    p->batch = *batch;
    p->nparts = n_partitions;
    p->nbits = 0;
    while ((1u << p->nbits) < n_partitions) p->nbits++;

    const int64_t n = batch->n_rows;
    dd_kargs &ka = p->ka;
    memset(&ka, 0, sizeof(ka));
    ka.n_rows = n;
    ka.n_cols = batch->n_cols;
    ka.n_keys = n_keys;
    ka.pid_total = n_partitions;
    ka.pid_shift = 0;

    auto fail = [&](dd_status s, const char *m) {
        delete p;
        return set_err(s, m);
    };

    for (int k = 0; k < n_keys; k++) {
        if (key_cols[k] < 0 || key_cols[k] >= batch->n_cols)
            return fail(DD_ERR_INVALID, "key column index out of range");
        ka.key_idx[k] = key_cols[k];
    }

    for (int c = 0; c < batch->n_cols; c++) {
        const dd_col_desc &cd = batch->cols[c];
        dd_kcol &kc = ka.cols[c];
        kc.dtype = cd.dtype;
        kc.elem = fixed_elem_size(cd.dtype);
        kc.data = cd.data;
        kc.valid = cd.validity;
        kc.offsets = cd.offsets;
        if (cd.dtype == DD_DT_UTF8) {
            if (ka.n_var >= DD_KMAX_VAR) return fail(DD_ERR_UNSUPPORTED, "too many var columns");
            /* Arrow utf8 offsets are int32: a column past INT32_MAX bytes has already
             * overflowed its own offsets — reject rather than read garbage (chunked.py
             * splits larger inputs before they get here) */
            if (cd.data_len < 0 || cd.data_len > (int64_t)INT32_MAX)
                return fail(DD_ERR_UNSUPPORTED, "utf8 data > 2 GB (int32 Arrow offsets): chunk the batch");
            ka.var_idx[ka.n_var++] = c;
        }
    }

    const int64_t P = n_partitions;
    const int nvar = ka.n_var;

    /* v2 (staged) eligibility: <= DD_STAGE_MAXC columns including the synthetic staged-var
     * entries (per var col a VARLEN length column + one ROWID permutation column; the var
     * BYTES go through K4), staging fits LDS */
    /* staged-var (synthetic VARLEN/ROWID cols + wave-cooperative K4 gather-copy) measured
     * faster than the v1 direct path for string-heavy batches (457 vs 334 GB/s on the
     * ClickBench shape) — default on; DD_V2_VAR=0 forces the v1 path */
    const bool allow_staged_var =
        !(getenv("DD_V2_VAR") && atoi(getenv("DD_V2_VAR")) == 0);
    const int aug_cols = batch->n_cols + nvar + (nvar > 0 ? 1 : 0);
    if (aug_cols <= DD_STAGE_MAXC && (nvar == 0 || allow_staged_var)) {
        size_t row_stage = 4; /* dstg */
        int nvalid = 0;
        for (int c = 0; c < batch->n_cols; c++) {
            row_stage += fixed_elem_size(batch->cols[c].dtype);
            if (batch->cols[c].validity) nvalid++;
        }
        row_stage += nvalid;
        row_stage += 4 * nvar + (nvar > 0 ? 4 : 0); /* VARLEN cols + ROWID col */
        /* auto: 8 waves x 4 groups (R=2048, 2 blocks/CU at the bench shape) measured best;
         * DD_V2_GMAX / DD_V2_WPB override for experiments */
        /* wpb cascade: 16-wave blocks measured best at the bench shape (wpb=16 g=4:
         * K3 1.381 ms vs 1.408 at 8/4) but their partition arrays outgrow LDS at large P
         * — fall back to 8 then 4 waves. Aim for R >= 32P (16/4 measured best at P=128),
         * floor g=2. DD_V2_WPB / DD_V2_GMAX override for experiments. */
        int wpb_cands[3] = {16, 8, 4};
        if (const char *e = getenv("DD_V2_WPB")) {
            int v = atoi(e);
            if (v == 4 || v == 8 || v == 16) {
                wpb_cands[0] = v;
                wpb_cands[1] = wpb_cands[2] = 0;
            }
        }
        for (int wi = 0; wi < 3 && !p->staged; wi++) {
            const int wpb = wpb_cands[wi];
            if (!wpb) continue;
            /* k_scatter_staged is instantiated for (gmax, wpb) in {2,4,8}x{4,8} and
             * {2,4}x{16} */
            int gcap = (wpb == 16) ? 4 : 8;
            int gtop = 2;
            while (gtop < gcap && (size_t)gtop * wpb * 64 < 32 * (size_t)P) gtop *= 2;
            if (const char *e = getenv("DD_V2_GMAX")) {
                int v = atoi(e);
                if (v == 2 || v == 4 || v == 8) gtop = (v <= gcap) ? v : gcap;
            }
            for (int g = gtop; g >= 2; g /= 2) {
                const size_t part_lds =
                    (size_t)P * (8 + 4 * wpb + 4 + 4) + (size_t)wpb * 64 * 4;
                size_t lds = part_lds + (size_t)g * wpb * 64 * row_stage;
                if (lds <= 163840) {
                    p->staged = true;
                    p->gmax = g;
                    p->wpb = wpb;
                    p->lds_k3 = lds;
                    break;
                }
            }
        }
    }
    if (p->staged) {
        p->lds_k1 = (size_t)WAVES_PER_BLOCK_H * P * 4;
        int64_t nblocks = (n + 16383) / 16384;
        if (nblocks < 8) nblocks = 8;
        if (nblocks > 2048) nblocks = 2048;
        if (const char *e = getenv("DD_V2_BLOCKS")) { /* tile-count experiment knob */
            int64_t v = atoll(e);
            if (v >= 8 && v <= 8192) nblocks = v;
        }
        p->nchunks = nblocks;
        p->chunk_rows = (n + nblocks - 1) / nblocks;
        if (p->chunk_rows < 1) p->chunk_rows = 1;
    } else {
        p->lds_k1 = (size_t)WAVES_PER_BLOCK_H * P * 4 * (1 + nvar);
        p->lds_k3 =
            (size_t)WAVES_PER_BLOCK_H * P * 8 * (1 + nvar) + WAVES_PER_BLOCK_H * 64 * 4;
        if (p->lds_k1 > 163840 || p->lds_k3 > 163840)
            return fail(DD_ERR_UNSUPPORTED, "partition count x var columns exceeds LDS budget");
        int64_t nchunks = (n + 4095) / 4096;
        if (nchunks < 4) nchunks = 4;
        if (nchunks > 8192) nchunks = 8192;
        nchunks = (nchunks + 3) & ~3LL;
        p->nchunks = nchunks;
        p->chunk_rows = (n + nchunks - 1) / nchunks;
        if (p->chunk_rows < 1) p->chunk_rows = 1;
    }
    /* K3-P (precomputed round layout, round 2 — DESIGN.md §13): replaces the in-kernel
     * cross-wave scan / round_off scan / serial dstbase chain of the staged scatter with
     * segment-granular precomputed bases (k_hash_count_seg + global-base scan fold +
     * k_round_layout + k_scatter_pre). Default ON for its instantiated shapes;
     * DD_K3_PRE=0 reverts to the HL/spec path for A/B. Mutually exclusive with rhash
     * (pre reads the pid array). Shape whitelist mirrors dd_launch_scatter_pre. */
    if (p->staged && p->wpb == 16 && nvar == 0 &&
        !(getenv("DD_K3_PRE") && atoi(getenv("DD_K3_PRE")) == 0) &&
        !(getenv("DD_RHASH") && atoi(getenv("DD_RHASH")) == 1)) {
        bool fixed_ok = true;
        int rowb = 0;
        for (int c = 0; c < batch->n_cols && fixed_ok; c++) {
            const int e = fixed_elem_size(batch->cols[c].dtype);
            if (batch->cols[c].validity || e == 0) fixed_ok = false;
            rowb += e;
        }
        auto pre_is = [&](std::initializer_list<int> want) {
            if ((int)want.size() != batch->n_cols) return false;
            int c = 0;
            for (int w : want)
                if (fixed_elem_size(batch->cols[c++].dtype) != w) return false;
            return true;
        };
        int pgmax = 0;
        if (fixed_ok) {
            const uint32_t Pn = n_partitions;
            bool all48 = batch->n_cols == 4;
            for (int c = 0; c < batch->n_cols && all48; c++) {
                const int e = fixed_elem_size(batch->cols[c].dtype);
                if (e != 4 && e != 8) all48 = false;
            }
            if (Pn <= 128 && (all48 || pre_is({8, 8, 8, 4, 4})))
                pgmax = 4;
            else if (Pn <= 128 && pre_is({1, 1, 8, 8, 8, 8, 4}))
                pgmax = 2;
            else if (Pn <= 256 && pre_is({8, 8, 8, 4}))
                pgmax = 4;
            else if (Pn <= 512 && (pre_is({8, 8, 8, 4}) || pre_is({8, 8, 8, 4, 4})))
                pgmax = 2;
        }
        if (pgmax > 0) {
            if (const char *e = getenv("DD_PRE_GMAX")) { /* experiment knob */
                int v = atoi(e);
                if (v == 2 || v == 4) pgmax = v;
            }
            int wpb = 16;
            if (const char *e = getenv("DD_PRE_WPB")) { /* experiment knob (8: 2 blk/CU) */
                int v = atoi(e);
                if (v == 8 || v == 16) wpb = v;
            }
            const int64_t R = (int64_t)pgmax * wpb * 64;
            const uint32_t sP2 = (n_partitions + 1) & ~1u;
            const size_t lds = (size_t)R * rowb + 4 * (size_t)R +
                               (size_t)wpb * (4 * (size_t)n_partitions + 2 * (size_t)sP2 +
                                              2 * (size_t)n_partitions);
            if (lds <= 163840) {
                p->pre = true;
                p->gmax = pgmax;
                p->wpb = wpb;
                p->lds_k3 = lds;
                p->sP2 = sP2;
                p->nrounds = (n + R - 1) / R;
                if (p->nrounds < 1) p->nrounds = 1;
                const int64_t nseg = p->nrounds * wpb;
                p->nseg_pad = (nseg + 3) & ~3LL;
                p->rpb = 2; /* sweep (tools/sweep_pre.py): rpb2 K3 0.896 ms vs rpb1
                               0.953 / rpb4 0.924 / ceil(nrounds/2048)=8 1.079 on the
                               bench shape — small rpb beats in-block pipelining depth */
                /* u8 pid array when P fits: 3 B/row less HBM write (K1) + read (K3);
                   DD_PID8=0 reverts */
                if (n_partitions <= 256 &&
                    !(getenv("DD_PID8") && atoi(getenv("DD_PID8")) == 0))
                    ka.pid8 = 1;
                if (const char *e = getenv("DD_PRE_RPB")) { /* experiment knob */
                    int v = atoi(e);
                    if (v >= 1 && v <= 256) p->rpb = v;
                }
                /* scale the fused-partials range count with the segment count so the
                 * per-(range,partition) atomic contention stays ~constant (~114 adds):
                 * fixed 2048 measured K1 +28%/row at 600 M rows (SF100) */
                {
                    const int64_t nseg_tmp = ((n + R - 1) / R < 1 ? 1 : (n + R - 1) / R) * wpb;
                    int64_t nr = nseg_tmp / 114;
                    if (nr < 2048) nr = 2048;
                    if (nr > 16384) nr = 16384;
                    p->pre_nranges = (int)nr;
                }
                p->nchunks = p->nseg_pad;       /* counts rows (scan granularity) */
                p->chunk_rows = (int64_t)pgmax * 64; /* = SEG (K1seg rows per wave) */
                p->lds_k1 = (size_t)WAVES_PER_BLOCK_H * n_partitions * 4;
            }
        }
    }
    const int64_t nchunks = p->nchunks;

    /* rhash (opt-in, DD_RHASH=1): spec-path batches (wpb 16, all fixed, no validity)
     * with integer/bool keys recompute the row hash in K3 from its own key loads and
     * skip the pid array (4 B/row HBM write in K1 + read in K3). MEASURED NEGATIVE as a
     * default at the bench shape (K3 1.65-1.70 ms vs 1.30 ms, same box, both the
     * hash-after-preload and keys-first forms): the pid load's latency is fully hidden
     * in the staged pipeline, while pidr's dependency on hashed key loads stalls the
     * wave-synchronous rank() ballots — 0.24 GB less traffic does not buy back a 27%
     * stall. Kept parity-tested for re-evaluation on future silicon.
     * Must mirror the launcher's can_spec/can_spec8 conditions exactly
     * (dd_launch_scatter_staged fails loudly if not). */
    if (getenv("DD_RHASH") && atoi(getenv("DD_RHASH")) == 1 && p->staged && !p->pre &&
        p->wpb == 16 && nvar == 0 && batch->n_cols <= DD_STAGE_MAXC) {
        bool rhash = true;
        for (int c = 0; c < batch->n_cols && rhash; c++)
            if (batch->cols[c].validity || fixed_elem_size(batch->cols[c].dtype) == 0)
                rhash = false;
        for (int k = 0; k < n_keys && rhash; k++) {
            const int dt = batch->cols[key_cols[k]].dtype;
            if (!(dt == DD_DT_U8 || dt == DD_DT_BOOL || dt == DD_DT_I16 ||
                  dt == DD_DT_I32 || dt == DD_DT_I64))
                rhash = false; /* float needs canon, dict32 a table gather: pid path */
        }
        ka.rhash = rhash ? 1 : 0;
    }

    /* hidden-load scatter (dd_kernels.hip HL header): default ON for its gated shape —
     * measured K3 1.015 vs 1.166 ms (−13 %), headline +10.6 % on the same box
     * (profiles/). DD_K3_HL=0 reverts to the plain spec path for A/B.
     * Gate must match the instantiated shapes exactly (4 fixed cols, elems 4/8, no
     * validity, gmax 4, wpb 16); mutually exclusive with rhash (HL reads the pid
     * array). */
    if (!ka.rhash && !p->pre && !(getenv("DD_K3_HL") && atoi(getenv("DD_K3_HL")) == 0) &&
        p->staged && p->wpb == 16 && p->gmax == 4 && nvar == 0 && batch->n_cols == 4) {
        bool okhl = true;
        for (int c = 0; c < 4; c++) {
            const int e = fixed_elem_size(batch->cols[c].dtype);
            if (batch->cols[c].validity || (e != 4 && e != 8)) okhl = false;
        }
        if (okhl) ka.hl = 1;
    }
    /* generalized HL (k_scatter_hlg): exact whitelisted shapes only — must mirror the
     * launcher's DD_HLG table */
    if (!ka.rhash && !ka.hl && !p->pre &&
        !(getenv("DD_K3_HL") && atoi(getenv("DD_K3_HL")) == 0) &&
        p->staged && p->wpb == 16) {
        bool novalid = true;
        for (int c = 0; c < batch->n_cols && novalid; c++)
            if (batch->cols[c].validity) novalid = false;
        auto elems_are = [&](std::initializer_list<int> want) {
            if ((int)want.size() != batch->n_cols) return false;
            int c = 0;
            for (int w : want)
                if (fixed_elem_size(batch->cols[c++].dtype) != w) return false;
            return true;
        };
        if (novalid && nvar == 0 &&
            ((p->gmax == 4 && elems_are({8, 8, 8, 4, 4})) ||
             (p->gmax == 2 && elems_are({1, 1, 8, 8, 8, 8, 4}))))
            ka.hl = 2;
    }

    auto halloc = [&](void **ptr, size_t bytes) {
        return hipMalloc(ptr, bytes > 0 ? bytes : 1) == hipSuccess;
    };
    bool ok = (ka.rhash ? true
                        : halloc((void **)&p->pid, (size_t)n * (ka.pid8 ? 1 : 4))) &&
              halloc((void **)&p->counts, p->pre ? 4 : (size_t)nchunks * P * 4) &&
              halloc((void **)&p->partials,
                     (size_t)(p->pre ? p->pre_nranges : DD_SCAN_RANGES) * P * 4) &&
              halloc((void **)&p->part_offsets, (size_t)(P + 1) * 8);
    if (ok && p->pre)
        ok = halloc((void **)&p->imgb, (size_t)p->nrounds * p->sP2 * 2) &&
             halloc((void **)&p->partials2, (size_t)64 * P * 4) &&
             halloc((void **)&p->counts16, (size_t)p->nseg_pad * P * 2) &&
             halloc((void **)&p->gbase, (size_t)p->nseg_pad * P * 4);
    if (ok && nvar > 0) {
        ok = halloc((void **)&p->part_boffsets, (size_t)nvar * (P + 1) * 8);
        if (p->staged) {
            ok = ok && halloc((void **)&p->src_row, (size_t)n * 4) &&
                 halloc((void **)&p->k4_partials, (size_t)8192 * 8) &&
                 halloc((void **)&p->k4w_meta, (64 + 65) * 4) &&
                 halloc((void **)&p->k4w_order, ((size_t)(n + 63) / 64) * 4);
            for (int v = 0; v < nvar && ok; v++)
                ok = halloc((void **)&p->out_off[v], (size_t)(n + 1) * 8);
            /* K5 (LDS-staged var-byte scatter, dd_kernels.hip K5 header): one var
             * column with small strings. Needs the real max string length -> one tiny
             * create-time kernel + 4 B d2h (amortized across runs). DD_K5=0 reverts
             * to the K4 gather. */
            if (ok && nvar == 1 && n > 0 &&
                !(getenv("DD_K5") && atoi(getenv("DD_K5")) == 0)) {
                const dd_col_desc &vc = batch->cols[ka.var_idx[0]];
                uint32_t *mdev = nullptr;
                if (hipMalloc((void **)&mdev, 4) == hipSuccess) {
                    uint32_t maxlen = 0;
                    bool mok = hipMemset(mdev, 0, 4) == hipSuccess &&
                               dd_launch_k5_maxlen(vc.offsets, n, mdev, nullptr) ==
                                   hipSuccess &&
                               hipMemcpy(&maxlen, mdev, 4, hipMemcpyDeviceToHost) ==
                                   hipSuccess;
                    (void)hipFree(mdev);
                    int wpb5 = 16;
                    if (const char *e = getenv("DD_K5_WPB")) { /* experiment knob */
                        int v = atoi(e);
                        if (v == 8 || v == 16) wpb5 = v;
                    }
                    const int64_t R5 = (int64_t)wpb5 * 64;
                    const size_t img = (size_t)R5 * maxlen + 8;
                    const size_t lds5 =
                        (size_t)wpb5 * P * 4 + (size_t)(P + 1) * 4 + img;
                    if (mok && maxlen > 0 && maxlen <= 128 && lds5 <= 163840) {
                        p->k5_wpb = wpb5;
                        p->k5_nrounds = (n + R5 - 1) / R5;
                        p->k5_nseg = p->k5_nrounds * wpb5;
                        {
                            int64_t nr = p->k5_nseg / 114;
                            if (nr < 2048) nr = 2048;
                            if (nr > 16384) nr = 16384;
                            p->k5_nranges = (int)nr;
                        }
                        p->k5_maxlen = maxlen;
                        p->lds_k5 = lds5;
                        p->lds_k5c = (size_t)WAVES_PER_BLOCK_H * P * 4;
                        ok = halloc((void **)&p->k5_bcounts,
                                    (size_t)p->k5_nseg * P * 2) &&
                             halloc((void **)&p->k5_gbase,
                                    (size_t)p->k5_nseg * P * 4) &&
                             halloc((void **)&p->k5_partials,
                                    (size_t)p->k5_nranges * P * 4) &&
                             halloc((void **)&p->k5_partials2, (size_t)64 * P * 4) &&
                             halloc((void **)&p->k5_roffB,
                                    (size_t)p->k5_nrounds * (P + 1) * 4);
                        if (ok) p->k5 = 1;
                    }
                }
            }
        } else {
            ok = ok && halloc((void **)&p->bcounts, (size_t)nvar * nchunks * P * 4) &&
                 halloc((void **)&p->bpartials, (size_t)nvar * DD_SCAN_RANGES * P * 4);
        }
    }
    if (!ok) return fail(DD_ERR_HIP, "allocation failed (workspace)");

    for (int c = 0; c < batch->n_cols && ok; c++) {
        const dd_col_desc &cd = batch->cols[c];
        dd_kcol &kc = ka.cols[c];
        if (cd.dtype == DD_DT_UTF8) {
            ok = halloc(&p->out_data[c], (size_t)cd.data_len) &&
                 halloc((void **)&p->out_lengths[c], (size_t)n * 4);
            kc.out_lengths = p->out_lengths[c];
        } else {
            ok = halloc(&p->out_data[c], (size_t)n * kc.elem);
        }
        kc.out_data = p->out_data[c];
        if (ok && cd.validity) {
            ok = halloc((void **)&p->out_valid[c], (size_t)n);
            kc.out_valid = p->out_valid[c];
        }
        if (ok && cd.dtype == DD_DT_DICT32) {
            ok = halloc((void **)&p->dict_hashes[c], (size_t)cd.dict_n * 8);
            if (ok) {
                hipError_t e = dd_launch_dict_hashes((const uint8_t *)cd.dict_bytes,
                                                     cd.dict_offsets, cd.dict_n,
                                                     p->dict_hashes[c], nullptr);
                ok = (e == hipSuccess);
            }
            kc.dict_hashes = p->dict_hashes[c];
        }
    }
    if (!ok) return fail(DD_ERR_HIP, "allocation failed (outputs)");

    if (p->staged && nvar > 0) {
        /* synthetic staged columns: per var col its LENGTHS (u32, written partition-major
         * into out_lengths) and one ROWID permutation column (u32 -> src_row); the var
         * BYTES are materialized by K4 after the scatter (dd_kernels.hip K4 header) */
        for (int v = 0; v < nvar; v++) {
            const int ci = ka.var_idx[v];
            dd_kcol &kc = ka.cols[ka.n_cols];
            memset(&kc, 0, sizeof(kc));
            kc.dtype = DD_KDT_VARLEN;
            kc.elem = 4;
            kc.data = batch->cols[ci].offsets;
            kc.out_data = p->out_lengths[ci];
            ka.n_cols++;
        }
        if (!p->k5) { /* ROWID (src_row) only feeds the K4 gather; K5 replaces it */
            dd_kcol &kr = ka.cols[ka.n_cols];
            memset(&kr, 0, sizeof(kr));
            kr.dtype = DD_KDT_ROWID;
            kr.elem = 4;
            kr.out_data = p->src_row;
            ka.n_cols++;
        }
    }

    for (auto &e : p->ev)
        if (hipEventCreate(&e) != hipSuccess) return fail(DD_ERR_HIP, "event create failed");

    *out = p;
    return DD_OK;
}

/* coarse-bucket partitioner: pid = (h % pid_total) >> log2(coarse_div). The effective
 * partition count is pid_total/coarse_div CONTIGUOUS ranges of the final partition space
 * — pass A of the two-level var scatter. Power-of-two args only. */
extern "C" dd_status dd_partitioner_create_ranged(const dd_batch_desc *batch,
                                                  const int32_t *key_cols, int32_t n_keys,
                                                  uint32_t pid_total, uint32_t coarse_div,
                                                  dd_partitioner **out) {
    if (coarse_div == 0 || (coarse_div & (coarse_div - 1)) != 0 ||
        (pid_total & (pid_total - 1)) != 0 || pid_total % coarse_div != 0)
        return set_err(DD_ERR_INVALID, "pid_total and coarse_div must be powers of two "
                                       "with coarse_div | pid_total");
    dd_status st = dd_partitioner_create(batch, key_cols, n_keys, pid_total / coarse_div,
                                         out);
    if (st != DD_OK) return st;
    (*out)->ka.pid_total = pid_total;
    int shift = 0;
    while ((1u << shift) < coarse_div) shift++;
    (*out)->ka.pid_shift = shift;
    return DD_OK;
}

extern "C" dd_status dd_partitioner_run_phase1(dd_partitioner *p, void *stream) {
    if (!p) return set_err(DD_ERR_INVALID, "null partitioner");
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipEventRecord(p->ev[0], s));
    if (p->pre) {
        HIP_TRY(hipMemsetAsync(p->partials, 0, (size_t)p->pre_nranges * p->nparts * 4, s));
        HIP_TRY(dd_launch_hash_count_seg(&p->ka, p->nseg_pad, p->chunk_rows, p->nparts,
                                         p->nbits, p->pid, p->counts16, p->partials,
                                         p->pre_nranges, p->lds_k1, s));
    } else if (p->staged) {
        HIP_TRY(dd_launch_hash_count_tile(&p->ka, p->nchunks, p->chunk_rows, p->nparts,
                                          p->nbits, p->pid, p->counts, p->lds_k1, s));
    } else {
        HIP_TRY(dd_launch_hash_count(&p->ka, p->nchunks, p->chunk_rows, p->nparts, p->nbits,
                                     p->pid, p->counts, p->bcounts, p->lds_k1, s));
    }
    HIP_TRY(hipEventRecord(p->ev[1], s));
    if (p->pre) {
        HIP_TRY(dd_launch_scan_deep(p->counts16, p->nchunks, p->nparts, p->pre_nranges,
                                    64, p->partials, p->partials2, p->part_offsets,
                                    p->gbase, s));
    } else {
        HIP_TRY(dd_launch_scan(p->counts, p->nchunks, p->nparts, DD_SCAN_RANGES,
                               p->partials, p->part_offsets, 0, s));
    }
    if (p->pre) {
        HIP_TRY(dd_launch_round_layout(p->gbase, p->part_offsets, p->nrounds, p->wpb,
                                       p->nparts, p->sP2, p->imgb, s));
    }
    if (p->k5) { /* byte-base precompute for the K5 var scatter */
        const int ci = p->ka.var_idx[0];
        HIP_TRY(hipMemsetAsync(p->k5_partials, 0,
                               (size_t)p->k5_nranges * p->nparts * 4, s));
        HIP_TRY(dd_launch_k5_count(p->ka.n_rows, p->nparts, p->pid,
                                   p->batch.cols[ci].offsets, p->k5_bcounts,
                                   p->k5_partials, p->k5_nranges, p->k5_nseg,
                                   p->lds_k5c, s));
        HIP_TRY(dd_launch_scan_deep(p->k5_bcounts, p->k5_nseg, p->nparts, p->k5_nranges,
                                    64, p->k5_partials, p->k5_partials2,
                                    p->part_boffsets, p->k5_gbase, s));
        HIP_TRY(dd_launch_k5_roff(p->k5_gbase, p->part_boffsets, p->k5_nrounds,
                                  p->k5_wpb, p->nparts, p->k5_roffB, s));
    }
    if (!p->staged) {
        for (int v = 0; v < p->ka.n_var; v++) {
            HIP_TRY(dd_launch_scan(p->bcounts + (size_t)v * p->nchunks * p->nparts,
                                   p->nchunks, p->nparts, DD_SCAN_RANGES,
                                   p->bpartials + (size_t)v * DD_SCAN_RANGES * p->nparts,
                                   p->part_boffsets + (size_t)v * (p->nparts + 1), 0, s));
        }
    }
    HIP_TRY(hipEventRecord(p->ev[2], s));
    return DD_OK;
}

extern "C" dd_status dd_partitioner_run_phase2(dd_partitioner *p, void *stream) {
    if (!p) return set_err(DD_ERR_INVALID, "null partitioner");
    hipStream_t s = (hipStream_t)stream;
    HIP_TRY(hipEventRecord(p->ev[4], s)); /* K3 start on the phase2 stream */
    if (p->pre) {
        const int64_t nblocks = (p->nrounds + p->rpb - 1) / p->rpb;
        HIP_TRY(dd_launch_scatter_pre(&p->ka, nblocks, p->nrounds, p->rpb, p->nparts,
                                      p->nbits, p->pid, p->gbase, p->imgb, p->sP2,
                                      p->gmax, p->wpb, p->lds_k3, s));
    } else if (p->staged) {
        HIP_TRY(dd_launch_scatter_staged(&p->ka, p->nchunks, p->chunk_rows, p->nparts,
                                         p->nbits, p->pid, p->counts, p->part_offsets,
                                         p->gmax, p->wpb, p->lds_k3, s));
        for (int v = 0; v < p->ka.n_var; v++) {
            const int ci = p->ka.var_idx[v];
            const dd_col_desc &cd = p->batch.cols[ci];
            HIP_TRY(dd_launch_var_bytes(
                p->out_lengths[ci], p->src_row, cd.offsets, (const uint8_t *)cd.data,
                p->ka.n_rows, cd.data_len, p->k4_partials, p->out_off[v],
                (uint8_t *)p->out_data[ci], p->part_offsets, p->nparts,
                p->part_boffsets + (size_t)v * (p->nparts + 1), p->k4w_meta,
                p->k4w_order, p->k5 ? 1 : 0, s));
            if (p->k5) {
                HIP_TRY(dd_launch_k5_scatter(
                    p->ka.n_rows, p->nparts, p->nbits, p->pid, cd.offsets,
                    (const uint8_t *)cd.data, p->k5_gbase, p->k5_roffB, p->k5_nrounds,
                    (uint8_t *)p->out_data[ci], p->k5_wpb, p->lds_k5, s));
            }
        }
    } else {
        HIP_TRY(dd_launch_scatter(&p->ka, p->nchunks, p->chunk_rows, p->nparts, p->nbits,
                                  p->pid, p->counts, p->part_offsets, p->bcounts,
                                  p->part_boffsets, p->lds_k3, s));
    }
    HIP_TRY(hipEventRecord(p->ev[3], s));
    p->has_run.store(true, std::memory_order_release);
    return DD_OK;
}

extern "C" dd_status dd_partitioner_wait_phase1(dd_partitioner *p, void *stream) {
    HIP_TRY(hipStreamWaitEvent((hipStream_t)stream, p->ev[2], 0));
    return DD_OK;
}

extern "C" dd_status dd_partitioner_wait_phase2(dd_partitioner *p, void *stream) {
    HIP_TRY(hipStreamWaitEvent((hipStream_t)stream, p->ev[3], 0));
    return DD_OK;
}

extern "C" dd_status dd_partitioner_run(dd_partitioner *p, void *stream) {
    dd_status st = dd_partitioner_run_phase1(p, stream);
    if (st != DD_OK) return st;
    return dd_partitioner_run_phase2(p, stream);
}

extern "C" void dd_partitioner_destroy(dd_partitioner *p) { delete p; }

extern "C" int32_t dd_partitioner_pid_elem(const dd_partitioner *p) {
    return p->ka.pid8 ? 1 : 4;
}

extern "C" const uint32_t *dd_partitioner_pids(const dd_partitioner *p) { return p->pid; }
extern "C" const void *dd_partitioner_col_data(const dd_partitioner *p, int32_t c) {
    return (c >= 0 && c < p->batch.n_cols) ? p->out_data[c] : nullptr;
}
extern "C" const uint8_t *dd_partitioner_col_validity(const dd_partitioner *p, int32_t c) {
    return (c >= 0 && c < p->batch.n_cols) ? p->out_valid[c] : nullptr;
}
extern "C" const uint32_t *dd_partitioner_col_lengths(const dd_partitioner *p, int32_t c) {
    return (c >= 0 && c < p->batch.n_cols) ? p->out_lengths[c] : nullptr;
}

extern "C" dd_status dd_partitioner_row_offsets(const dd_partitioner *p, int64_t *host_out) {
    if (!p->has_run) return set_err(DD_ERR_INVALID, "partitioner has not run");
    HIP_TRY(hipMemcpy(host_out, p->part_offsets, (size_t)(p->nparts + 1) * 8,
                      hipMemcpyDeviceToHost));
    return DD_OK;
}

extern "C" const uint64_t *dd_partitioner_var_offsets64(const dd_partitioner *p,
                                                        int32_t col) {
    if (!p->staged) return nullptr;
    for (int v = 0; v < p->ka.n_var; v++)
        if (p->ka.var_idx[v] == col) return p->out_off[v];
    return nullptr;
}

extern "C" dd_status dd_make_offsets32(const uint64_t *off64, int64_t lo_row, int64_t n,
                                       int32_t *out32, void *stream) {
    if (!off64 || !out32) return set_err(DD_ERR_INVALID, "null argument");
    HIP_TRY(dd_launch_off64_to_off32(off64, lo_row, n, out32, (hipStream_t)stream));
    return DD_OK;
}

extern "C" dd_status dd_partitioner_byte_offsets(const dd_partitioner *p, int32_t col,
                                                 int64_t *host_out) {
    if (!p->has_run) return set_err(DD_ERR_INVALID, "partitioner has not run");
    for (int v = 0; v < p->ka.n_var; v++) {
        if (p->ka.var_idx[v] == col) {
            HIP_TRY(hipMemcpy(host_out, p->part_boffsets + (size_t)v * (p->nparts + 1),
                              (size_t)(p->nparts + 1) * 8, hipMemcpyDeviceToHost));
            return DD_OK;
        }
    }
    return set_err(DD_ERR_INVALID, "not a var column");
}

extern "C" dd_status dd_partitioner_kernel_ms(const dd_partitioner *p, float out_ms[3]) {
    if (!p->has_run) return set_err(DD_ERR_INVALID, "partitioner has not run");
    HIP_TRY(hipEventSynchronize(p->ev[3]));
    HIP_TRY(hipEventElapsedTime(&out_ms[0], p->ev[0], p->ev[1]));
    HIP_TRY(hipEventElapsedTime(&out_ms[1], p->ev[1], p->ev[2]));
    HIP_TRY(hipEventElapsedTime(&out_ms[2], p->ev[4], p->ev[3]));
    return DD_OK;
}

/* ---------------- task cache (mirrors src/worker/task_data.rs, worker_service.rs:12,31) --- */

struct TaskKeyCmp {
    bool operator()(const dd_task_key &a, const dd_task_key &b) const {
        return memcmp(&a, &b, sizeof(a)) < 0;
    }
};

/* Entries are shared_ptr so a concurrent dd_drop_task / dd_set_plan replacing the key
 * cannot free the partitioner out from under an in-flight dd_execute_task (the execute
 * call pins its own reference for the duration of the call — the reference worker keeps
 * task state alive the same way via Arc'd entries in its TTI cache, task_data.rs:16-29).
 * The raw pointer RETURNED to the caller is only guaranteed while the cache entry lives:
 * see the lifetime note on dd_execute_task in dd_shuffle.h. */
static std::mutex g_tasks_mu;
static std::map<dd_task_key, std::shared_ptr<dd_partitioner>, TaskKeyCmp> g_tasks;

extern "C" dd_status dd_set_plan(const dd_task_key *key, const dd_batch_desc *batch,
                                 const int32_t *key_cols, int32_t n_keys,
                                 uint32_t n_partitions) {
    if (!key) return set_err(DD_ERR_INVALID, "null key");
    dd_partitioner *p = nullptr;
    dd_status st = dd_partitioner_create(batch, key_cols, n_keys, n_partitions, &p);
    if (st != DD_OK) return st;
    std::shared_ptr<dd_partitioner> sp(p, dd_partitioner_destroy);
    std::lock_guard<std::mutex> g(g_tasks_mu);
    g_tasks[*key] = std::move(sp); /* replaces; pinned in-flight executes keep the old one */
    return DD_OK;
}

extern "C" dd_status dd_execute_task(const dd_task_key *key, uint32_t part_lo, uint32_t part_hi,
                                     void *stream, dd_partitioner **out) {
    if (!key || !out) return set_err(DD_ERR_INVALID, "null argument");
    std::shared_ptr<dd_partitioner> pin; /* keeps p alive for the whole call */
    {
        std::lock_guard<std::mutex> g(g_tasks_mu);
        auto it = g_tasks.find(*key);
        if (it == g_tasks.end())
            return set_err(DD_ERR_NOT_FOUND,
                           "unknown TaskKey (no plan was set: cf. impl_execute_task.rs:29-34)");
        pin = it->second;
    }
    dd_partitioner *p = pin.get();
    if (part_lo > part_hi || part_hi > p->nparts)
        return set_err(DD_ERR_INVALID, "partition range out of bounds");
    if (!p->has_run.load(std::memory_order_acquire)) {
        /* run under the cache lock: launches are async enqueues (cheap), and this keeps
         * two concurrent execute_task calls on the same key from double-launching */
        std::lock_guard<std::mutex> g(g_tasks_mu);
        if (!p->has_run.load(std::memory_order_relaxed)) {
            dd_status st = dd_partitioner_run(p, stream);
            if (st != DD_OK) return st;
        }
    }
    *out = p;
    return DD_OK;
}

extern "C" dd_status dd_drop_task(const dd_task_key *key) {
    std::shared_ptr<dd_partitioner> dead; /* destroy outside the lock */
    std::lock_guard<std::mutex> g(g_tasks_mu);
    auto it = g_tasks.find(*key);
    if (it == g_tasks.end()) return set_err(DD_ERR_NOT_FOUND, "unknown TaskKey");
    dead = std::move(it->second);
    g_tasks.erase(it);
    return DD_OK;
}

/* ---------------- RCCL exchange ---------------- */

struct dd_comm {
    ncclComm_t comm = nullptr;
    int rank = 0, nranks = 1;
    ~dd_comm() {
        if (comm) ncclCommDestroy(comm);
    }
};

extern "C" dd_status dd_comm_unique_id(void *bytes128) {
    if (!bytes128) return set_err(DD_ERR_INVALID, "null argument");
    ncclUniqueId id;
    NCCL_TRY(ncclGetUniqueId(&id));
    memcpy(bytes128, &id, sizeof(id));
    return DD_OK;
}

extern "C" dd_status dd_comm_init(const void *bytes128, int rank, int nranks, dd_comm **out) {
    if (dd_device_count() == 0) return set_err(DD_ERR_NO_DEVICE, "no HIP device");
    auto c = new dd_comm();
    c->rank = rank;
    c->nranks = nranks;
    ncclUniqueId id;
    memcpy(&id, bytes128, sizeof(id));
    ncclResult_t r = ncclCommInitRank(&c->comm, nranks, id, rank);
    if (r != ncclSuccess) {
        delete c;
        return set_err(DD_ERR_RCCL, std::string("ncclCommInitRank: ") + ncclGetErrorString(r));
    }
    *out = c;
    return DD_OK;
}

extern "C" void dd_comm_destroy(dd_comm *c) { delete c; }

/* allgather of the per-rank size matrix (exchange + coalesce); device staging buffers are
 * freed on every path, including RCCL/HIP failure */
static dd_status dd_meta_allgather(dd_comm *c, const std::vector<int64_t> &mine,
                                   std::vector<int64_t> &all, hipStream_t s) {
    const size_t meta_n = mine.size();
    int64_t *d_in = nullptr, *d_all = nullptr;
    HIP_TRY(hipMalloc(&d_in, meta_n * 8));
    if (hipMalloc(&d_all, all.size() * 8) != hipSuccess) {
        (void)hipFree(d_in);
        return set_err(DD_ERR_HIP, "meta allgather alloc");
    }
    hipError_t he = hipMemcpyAsync(d_in, mine.data(), meta_n * 8, hipMemcpyHostToDevice, s);
    ncclResult_t nr = ncclSuccess;
    if (he == hipSuccess) nr = ncclAllGather(d_in, d_all, meta_n, ncclInt64, c->comm, s);
    if (he == hipSuccess && nr == ncclSuccess) he = hipStreamSynchronize(s);
    if (he == hipSuccess && nr == ncclSuccess)
        he = hipMemcpy(all.data(), d_all, all.size() * 8, hipMemcpyDeviceToHost);
    (void)hipFree(d_in);
    (void)hipFree(d_all);
    if (nr != ncclSuccess)
        return set_err(DD_ERR_RCCL, std::string("size allgather: ") + ncclGetErrorString(nr));
    if (he != hipSuccess)
        return set_err(DD_ERR_HIP, std::string("size allgather: ") + hipGetErrorString(he));
    return DD_OK;
}

/* ---------------- broadcast (dd_bcast) ---------------- */

struct dd_bcast {
    int64_t n_rows = 0;
    int32_t n_cols = 0;
    int32_t dtypes[DD_MAX_COLS] = {};
    int64_t data_lens[DD_MAX_COLS] = {};
    void *data[DD_MAX_COLS] = {};
    uint8_t *valid[DD_MAX_COLS] = {};
    int32_t *offsets[DD_MAX_COLS] = {};
    void *dict_bytes[DD_MAX_COLS] = {};
    int32_t *dict_offsets[DD_MAX_COLS] = {};
    bool owns = false; /* non-root ranks own their buffers */
    ~dd_bcast() {
        if (!owns) return;
        for (int i = 0; i < DD_MAX_COLS; i++) {
            (void)hipFree(data[i]);
            (void)hipFree(valid[i]);
            (void)hipFree(offsets[i]);
            (void)hipFree(dict_bytes[i]);
            (void)hipFree(dict_offsets[i]);
        }
    }
};

/* header: [n_rows, n_cols] + per col [dtype, data_bytes, has_valid, n_offsets, dict_n,
 * dict_bytes_len] — broadcast first so non-root ranks can allocate */
#define DD_BHDR (2 + DD_MAX_COLS * 6)

extern "C" dd_status dd_broadcast_run(dd_comm *c, const dd_batch_desc *batch, int root,
                                      void *stream, dd_bcast **out) {
    if (!c || !out) return set_err(DD_ERR_INVALID, "null argument");
    hipStream_t s = (hipStream_t)stream;
    const bool is_root = c->rank == root;
    if (is_root && !batch) return set_err(DD_ERR_INVALID, "root needs a batch");

    int64_t hdr[DD_BHDR] = {};
    if (is_root) {
        hdr[0] = batch->n_rows;
        hdr[1] = batch->n_cols;
        for (int i = 0; i < batch->n_cols; i++) {
            const dd_col_desc &cd = batch->cols[i];
            int64_t *h = hdr + 2 + (size_t)i * 6;
            h[0] = cd.dtype;
            h[1] = (cd.dtype == DD_DT_UTF8) ? cd.data_len
                                            : batch->n_rows * fixed_elem_size(cd.dtype);
            h[2] = cd.validity ? 1 : 0;
            h[3] = cd.offsets ? batch->n_rows + 1 : 0;
            h[4] = cd.dict_n;
            h[5] = (cd.dtype == DD_DT_DICT32 && cd.dict_offsets)
                       ? 0 /* filled below via D2H of last offset */
                       : 0;
        }
        /* dict byte lengths require the last dict offset */
        for (int i = 0; i < batch->n_cols; i++) {
            const dd_col_desc &cd = batch->cols[i];
            if (cd.dtype == DD_DT_DICT32 && cd.dict_n > 0) {
                int32_t last = 0;
                HIP_TRY(hipMemcpy(&last, cd.dict_offsets + cd.dict_n, 4,
                                  hipMemcpyDeviceToHost));
                hdr[2 + (size_t)i * 6 + 5] = last;
            }
        }
    }
    int64_t *d_hdr = nullptr;
    HIP_TRY(hipMalloc(&d_hdr, sizeof(hdr)));
    {
        /* merged checks so d_hdr is freed on every path */
        hipError_t he = hipSuccess;
        ncclResult_t nr = ncclSuccess;
        if (is_root) he = hipMemcpyAsync(d_hdr, hdr, sizeof(hdr), hipMemcpyHostToDevice, s);
        if (he == hipSuccess)
            nr = ncclBroadcast(d_hdr, d_hdr, DD_BHDR, ncclInt64, root, c->comm, s);
        if (he == hipSuccess && nr == ncclSuccess) he = hipStreamSynchronize(s);
        if (he == hipSuccess && nr == ncclSuccess)
            he = hipMemcpy(hdr, d_hdr, sizeof(hdr), hipMemcpyDeviceToHost);
        (void)hipFree(d_hdr);
        if (nr != ncclSuccess)
            return set_err(DD_ERR_RCCL,
                           std::string("header broadcast: ") + ncclGetErrorString(nr));
        if (he != hipSuccess)
            return set_err(DD_ERR_HIP,
                           std::string("header broadcast: ") + hipGetErrorString(he));
    }

    std::unique_ptr<dd_bcast> b_own(new dd_bcast()); /* freed on every error return */
    dd_bcast *b = b_own.get();
    b->n_rows = hdr[0];
    b->n_cols = (int32_t)hdr[1];
    b->owns = !is_root;
    auto fail = [&](const char *m) { return set_err(DD_ERR_HIP, m); };
    for (int i = 0; i < b->n_cols; i++) {
        const int64_t *h = hdr + 2 + (size_t)i * 6;
        b->dtypes[i] = (int32_t)h[0];
        b->data_lens[i] = h[1];
        if (is_root) {
            const dd_col_desc &cd = batch->cols[i];
            b->data[i] = const_cast<void *>(cd.data);
            b->valid[i] = const_cast<uint8_t *>(cd.validity);
            b->offsets[i] = const_cast<int32_t *>(cd.offsets);
            b->dict_bytes[i] = const_cast<void *>(cd.dict_bytes);
            b->dict_offsets[i] = const_cast<int32_t *>(cd.dict_offsets);
        } else {
            if (hipMalloc(&b->data[i], (size_t)h[1] + 1) != hipSuccess)
                return fail("bcast alloc data");
            if (h[2] && hipMalloc((void **)&b->valid[i], (size_t)b->n_rows + 1) != hipSuccess)
                return fail("bcast alloc valid");
            if (h[3] && hipMalloc((void **)&b->offsets[i], (size_t)h[3] * 4) != hipSuccess)
                return fail("bcast alloc offsets");
            if (h[4]) {
                if (hipMalloc(&b->dict_bytes[i], (size_t)h[5] + 1) != hipSuccess ||
                    hipMalloc((void **)&b->dict_offsets[i], ((size_t)h[4] + 1) * 4) !=
                        hipSuccess)
                    return fail("bcast alloc dict");
            }
        }
    }
    NCCL_TRY(ncclGroupStart());
    for (int i = 0; i < b->n_cols; i++) {
        const int64_t *h = hdr + 2 + (size_t)i * 6;
        if (h[1] > 0)
            NCCL_TRY(ncclBroadcast(b->data[i], b->data[i], h[1], ncclUint8, root, c->comm, s));
        if (h[2] && b->n_rows > 0)
            NCCL_TRY(ncclBroadcast(b->valid[i], b->valid[i], b->n_rows, ncclUint8, root,
                                   c->comm, s));
        if (h[3])
            NCCL_TRY(ncclBroadcast(b->offsets[i], b->offsets[i], h[3] * 4, ncclUint8, root,
                                   c->comm, s));
        if (h[4]) {
            if (h[5] > 0)
                NCCL_TRY(ncclBroadcast(b->dict_bytes[i], b->dict_bytes[i], h[5], ncclUint8,
                                       root, c->comm, s));
            NCCL_TRY(ncclBroadcast(b->dict_offsets[i], b->dict_offsets[i], (h[4] + 1) * 4,
                                   ncclUint8, root, c->comm, s));
        }
    }
    NCCL_TRY(ncclGroupEnd());
    HIP_TRY(hipStreamSynchronize(s));
    *out = b_own.release();
    return DD_OK;
}

extern "C" void dd_bcast_destroy(dd_bcast *b) { delete b; }
extern "C" int64_t dd_bcast_n_rows(const dd_bcast *b) { return b->n_rows; }
extern "C" int32_t dd_bcast_n_cols(const dd_bcast *b) { return b->n_cols; }
extern "C" const void *dd_bcast_col_data(const dd_bcast *b, int32_t col) {
    return (col >= 0 && col < b->n_cols) ? b->data[col] : nullptr;
}
extern "C" const uint8_t *dd_bcast_col_validity(const dd_bcast *b, int32_t col) {
    return (col >= 0 && col < b->n_cols) ? b->valid[col] : nullptr;
}
extern "C" const int32_t *dd_bcast_col_offsets(const dd_bcast *b, int32_t col) {
    return (col >= 0 && col < b->n_cols) ? b->offsets[col] : nullptr;
}
extern "C" dd_status dd_bcast_col_meta(const dd_bcast *b, int32_t col, int32_t *dtype,
                                       int64_t *data_len) {
    if (col < 0 || col >= b->n_cols) return set_err(DD_ERR_INVALID, "col out of range");
    *dtype = b->dtypes[col];
    *data_len = b->data_lens[col];
    return DD_OK;
}


struct dd_exchanged {
    int32_t n_cols = 0;
    int nranks = 1;
    uint32_t P = 0;
    int64_t total_rows = 0;
    hipStream_t alloc_stream = nullptr; /* buffers are stream-ordered (hipMallocAsync):
                                           repeated exchanges reuse the pool instead of
                                           paying a device malloc per step */
    std::vector<int64_t> row_counts;                 /* [nranks][P] */
    std::vector<std::vector<int64_t>> byte_counts;   /* per col: [nranks][P] (var only) */
    void *data[DD_MAX_COLS] = {};
    uint8_t *valid[DD_MAX_COLS] = {};
    uint32_t *lengths[DD_MAX_COLS] = {};
    float ms = 0;
    int64_t egress = 0;
    hipEvent_t e0 = nullptr, e1 = nullptr;
    ~dd_exchanged() {
        for (int i = 0; i < DD_MAX_COLS; i++) {
            if (data[i]) (void)hipFreeAsync(data[i], alloc_stream);
            if (valid[i]) (void)hipFreeAsync(valid[i], alloc_stream);
            if (lengths[i]) (void)hipFreeAsync(lengths[i], alloc_stream);
        }
        if (e0) (void)hipEventDestroy(e0);
        if (e1) (void)hipEventDestroy(e1);
    }
};

extern "C" dd_status dd_exchange_run(dd_comm *c, const dd_partitioner *p, void *stream,
                                     dd_exchanged **out) {
    if (!c || !p || !out) return set_err(DD_ERR_INVALID, "null argument");
    if (!p->has_run) return set_err(DD_ERR_INVALID, "partitioner has not run");
    if (p->nparts % c->nranks != 0)
        return set_err(DD_ERR_INVALID, "P_total must be nranks * partitions-per-consumer "
                                       "(scale_partitioning, common.rs:18-30)");
    hipStream_t s = (hipStream_t)stream;
    const uint32_t P = p->nparts / c->nranks;
    const int R = c->nranks;
    const int nvar = p->ka.n_var;

    /* 1) size matrix: my per-partition row counts (+ per var col byte counts), allgathered.
     * meta layout per rank: [P_total rows][nvar * P_total bytes] (u64) */
    const size_t meta_n = (size_t)p->nparts * (1 + nvar);
    std::vector<int64_t> my_meta(meta_n);
    std::vector<int64_t> off_h(p->nparts + 1);
    dd_status st = dd_partitioner_row_offsets(p, off_h.data());
    if (st != DD_OK) return st;
    for (uint32_t q = 0; q < p->nparts; q++) my_meta[q] = off_h[q + 1] - off_h[q];
    std::vector<std::vector<int64_t>> boff_h(nvar, std::vector<int64_t>(p->nparts + 1));
    for (int v = 0; v < nvar; v++) {
        st = dd_partitioner_byte_offsets(p, p->ka.var_idx[v], boff_h[v].data());
        if (st != DD_OK) return st;
        for (uint32_t q = 0; q < p->nparts; q++)
            my_meta[(size_t)(1 + v) * p->nparts + q] = boff_h[v][q + 1] - boff_h[v][q];
    }
    std::vector<int64_t> all_meta((size_t)R * meta_n);
    {
        dd_status ms = dd_meta_allgather(c, my_meta, all_meta, s);
        if (ms != DD_OK) return ms;
    }

    std::unique_ptr<dd_exchanged> e_own(new dd_exchanged()); /* freed on error returns */
    dd_exchanged *e = e_own.get();
    e->alloc_stream = dd_pool_stream();
    e->n_cols = p->batch.n_cols;
    e->nranks = R;
    e->P = P;
    e->row_counts.assign((size_t)R * P, 0);
    e->byte_counts.assign(p->batch.n_cols, {});
    /* my window = partitions [P*rank, P*(rank+1)) of every producer */
    for (int r = 0; r < R; r++)
        for (uint32_t q = 0; q < P; q++) {
            e->row_counts[(size_t)r * P + q] =
                all_meta[(size_t)r * meta_n + (size_t)P * c->rank + q];
            e->total_rows += e->row_counts[(size_t)r * P + q];
        }
    std::vector<int64_t> recv_rows_per_producer(R, 0);
    for (int r = 0; r < R; r++)
        for (uint32_t q = 0; q < P; q++)
            recv_rows_per_producer[r] += e->row_counts[(size_t)r * P + q];

    auto fail = [&](dd_status sc, const char *m) { return set_err(sc, m); };

    /* allocate receive buffers */
    std::vector<std::vector<int64_t>> recv_bytes_per_producer(p->batch.n_cols,
                                                              std::vector<int64_t>(R, 0));
    for (int ci = 0; ci < p->batch.n_cols; ci++) {
        const dd_kcol &kc = p->ka.cols[ci];
        if (kc.elem > 0) {
            if (hipMallocAsync(&e->data[ci], (size_t)e->total_rows * kc.elem + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "recv alloc");
        } else {
            /* var col: find its v index, total bytes of my window */
            int v = -1;
            for (int vv = 0; vv < nvar; vv++)
                if (p->ka.var_idx[vv] == ci) v = vv;
            e->byte_counts[ci].assign((size_t)R * P, 0);
            int64_t total_b = 0;
            for (int r = 0; r < R; r++)
                for (uint32_t q = 0; q < P; q++) {
                    int64_t b = all_meta[(size_t)r * meta_n + (size_t)(1 + v) * p->nparts +
                                         (size_t)P * c->rank + q];
                    e->byte_counts[ci][(size_t)r * P + q] = b;
                    recv_bytes_per_producer[ci][r] += b;
                    total_b += b;
                }
            if (hipMallocAsync(&e->data[ci], (size_t)total_b + 1, e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "recv alloc (var)");
            if (hipMallocAsync((void **)&e->lengths[ci], (size_t)e->total_rows * 4 + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "recv alloc (lengths)");
        }
        if (kc.valid) {
            if (hipMallocAsync((void **)&e->valid[ci], (size_t)e->total_rows + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "recv alloc (validity)");
        }
    }

    if (hipStreamSynchronize(e->alloc_stream) != hipSuccess) /* allocations ready */
        return fail(DD_ERR_HIP, "pool stream sync");
    if (hipEventCreate(&e->e0) != hipSuccess || hipEventCreate(&e->e1) != hipSuccess)
        return fail(DD_ERR_HIP, "event create");

    /* 2) grouped send/recv: one contiguous slice per (column-stream, peer).
     * send slice for peer j = partitions [P*j, P*(j+1)) of my partition-major output;
     * recv placed producer-major. */
    HIP_TRY(hipEventRecord(e->e0, s));
    NCCL_TRY(ncclGroupStart());
    for (int j = 0; j < R; j++) {
        const int64_t srow0 = off_h[(size_t)P * j], srow1 = off_h[(size_t)P * (j + 1)];
        int64_t rrow0 = 0;
        for (int r = 0; r < j; r++) rrow0 += recv_rows_per_producer[r];
        const int64_t rrows = recv_rows_per_producer[j];
        for (int ci = 0; ci < p->batch.n_cols; ci++) {
            const dd_kcol &kc = p->ka.cols[ci];
            if (kc.elem > 0) {
                int64_t sb = (srow1 - srow0) * kc.elem;
                if (sb > 0)
                    NCCL_TRY(ncclSend((const uint8_t *)p->out_data[ci] + srow0 * kc.elem, sb,
                                      ncclUint8, j, c->comm, s));
                int64_t rb = rrows * kc.elem;
                if (rb > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->data[ci] + rrow0 * kc.elem, rb, ncclUint8,
                                      j, c->comm, s));
                if (j != c->rank) e->egress += sb;
            } else {
                int v = -1;
                for (int vv = 0; vv < nvar; vv++)
                    if (p->ka.var_idx[vv] == ci) v = vv;
                int64_t sb0 = boff_h[v][(size_t)P * j], sb1 = boff_h[v][(size_t)P * (j + 1)];
                if (sb1 > sb0)
                    NCCL_TRY(ncclSend((const uint8_t *)p->out_data[ci] + sb0, sb1 - sb0,
                                      ncclUint8, j, c->comm, s));
                int64_t rb0 = 0;
                for (int r = 0; r < j; r++) rb0 += recv_bytes_per_producer[ci][r];
                int64_t rb = recv_bytes_per_producer[ci][j];
                if (rb > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->data[ci] + rb0, rb, ncclUint8, j, c->comm, s));
                if ((srow1 - srow0) > 0)
                    NCCL_TRY(ncclSend(p->out_lengths[ci] + srow0, (srow1 - srow0) * 4,
                                      ncclUint8, j, c->comm, s));
                if (rrows > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->lengths[ci] + rrow0 * 4, rrows * 4,
                                      ncclUint8, j, c->comm, s));
                if (j != c->rank) e->egress += (sb1 - sb0) + (srow1 - srow0) * 4;
            }
            if (kc.valid) {
                if (srow1 > srow0)
                    NCCL_TRY(ncclSend(p->out_valid[ci] + srow0, srow1 - srow0, ncclUint8, j,
                                      c->comm, s));
                if (rrows > 0)
                    NCCL_TRY(ncclRecv(e->valid[ci] + rrow0, rrows, ncclUint8, j, c->comm, s));
                if (j != c->rank) e->egress += srow1 - srow0;
            }
        }
    }
    NCCL_TRY(ncclGroupEnd());
    HIP_TRY(hipEventRecord(e->e1, s));
    HIP_TRY(hipStreamSynchronize(s));
    HIP_TRY(hipEventElapsedTime(&e->ms, e->e0, e->e1));
    *out = e_own.release();
    return DD_OK;
}

extern "C" void dd_exchanged_destroy(dd_exchanged *e) { delete e; }
extern "C" int64_t dd_exchanged_total_rows(const dd_exchanged *e) { return e->total_rows; }
extern "C" const void *dd_exchanged_col_data(const dd_exchanged *e, int32_t c) {
    return (c >= 0 && c < e->n_cols) ? e->data[c] : nullptr;
}
extern "C" const uint8_t *dd_exchanged_col_validity(const dd_exchanged *e, int32_t c) {
    return (c >= 0 && c < e->n_cols) ? e->valid[c] : nullptr;
}
extern "C" const uint32_t *dd_exchanged_col_lengths(const dd_exchanged *e, int32_t c) {
    return (c >= 0 && c < e->n_cols) ? e->lengths[c] : nullptr;
}
extern "C" dd_status dd_exchanged_row_counts(const dd_exchanged *e, int64_t *host_out) {
    memcpy(host_out, e->row_counts.data(), e->row_counts.size() * 8);
    return DD_OK;
}
extern "C" dd_status dd_exchanged_byte_counts(const dd_exchanged *e, int32_t col,
                                              int64_t *host_out) {
    if (col < 0 || col >= e->n_cols || e->byte_counts[col].empty())
        return set_err(DD_ERR_INVALID, "not a var column");
    memcpy(host_out, e->byte_counts[col].data(), e->byte_counts[col].size() * 8);
    return DD_OK;
}
extern "C" dd_status dd_exchanged_stats(const dd_exchanged *e, float *ms,
                                        int64_t *egress_bytes) {
    *ms = e->ms;
    *egress_bytes = e->egress;
    return DD_OK;
}

/* ---------------- partial aggregation (dd_reducer) ---------------- */

struct dd_reducer {
    int32_t n_keys = 0, n_aggs = 0;
    uint64_t *dict_hashes[DD_KMAX_KEYS] = {}; /* per dict32 KEY col (hash input) */
    int64_t n_rows = 0; /* output rows */
    uint64_t *keys = nullptr;
    uint32_t *keynull = nullptr;
    double *aggs = nullptr;
    uint64_t *nn = nullptr; /* [max_rows][n_aggs] non-null input counts */
    uint64_t *n_dev = nullptr;
    int32_t *meta = nullptr; /* device: agg_cols + agg_ops */
    float kernel_ms = 0;     /* hipEvent time of the reduce kernel alone */
    ~dd_reducer() {
        for (auto &d : dict_hashes) (void)hipFree(d);
        (void)hipFree(keys);
        (void)hipFree(keynull);
        (void)hipFree(aggs);
        (void)hipFree(nn);
        (void)hipFree(n_dev);
        (void)hipFree(meta);
    }
};

extern "C" dd_status dd_partial_reduce_run(const dd_batch_desc *batch,
                                           const int32_t *key_cols, int32_t n_keys,
                                           const int32_t *agg_cols, const int32_t *agg_ops,
                                           int32_t n_aggs, void *stream, dd_reducer **out) {
    if (!batch || !key_cols || !agg_ops || !out)
        return set_err(DD_ERR_INVALID, "null argument");
    if (dd_device_count() == 0) return set_err(DD_ERR_NO_DEVICE, "no HIP device");
    if (n_keys < 1 || n_keys > 4) return set_err(DD_ERR_UNSUPPORTED, "1..4 key columns");
    if (n_aggs < 1 || n_aggs > 4) return set_err(DD_ERR_UNSUPPORTED, "1..4 aggregates");
    for (int k = 0; k < n_keys; k++) {
        if (key_cols[k] < 0 || key_cols[k] >= batch->n_cols)
            return set_err(DD_ERR_INVALID, "key column out of range");
        if (batch->cols[key_cols[k]].dtype == DD_DT_UTF8)
            return set_err(DD_ERR_UNSUPPORTED,
                           "partial reduce: fixed-width or dictionary keys only");
    }
    for (int g = 0; g < n_aggs; g++) {
        if (agg_ops[g] == DD_AGG_COUNT) continue;
        if (!agg_cols || agg_cols[g] < 0 || agg_cols[g] >= batch->n_cols)
            return set_err(DD_ERR_INVALID, "agg column out of range");
        int dt = batch->cols[agg_cols[g]].dtype;
        if ((agg_ops[g] == DD_AGG_SUM_F64 || agg_ops[g] == DD_AGG_MIN_F64 ||
             agg_ops[g] == DD_AGG_MAX_F64) &&
            dt != DD_DT_F64)
            return set_err(DD_ERR_UNSUPPORTED, "f64 aggregate needs an f64 column");
        if ((agg_ops[g] == DD_AGG_SUM_I64 || agg_ops[g] == DD_AGG_MIN_I64 ||
             agg_ops[g] == DD_AGG_MAX_I64) &&
            dt != DD_DT_I64)
            return set_err(DD_ERR_UNSUPPORTED, "i64 aggregate needs an i64 column");
        if (agg_ops[g] < 0 || agg_ops[g] > DD_AGG_MAX_I64)
            return set_err(DD_ERR_INVALID, "unknown aggregate op");
    }

    dd_kargs ka;
    memset(&ka, 0, sizeof(ka));
    const int64_t n = batch->n_rows;
    ka.n_rows = n;
    ka.n_cols = batch->n_cols;
    ka.n_keys = n_keys;
    for (int k = 0; k < n_keys; k++) ka.key_idx[k] = key_cols[k];
    for (int c = 0; c < batch->n_cols; c++) {
        const dd_col_desc &cd = batch->cols[c];
        dd_kcol &kc = ka.cols[c];
        kc.dtype = cd.dtype;
        kc.elem = fixed_elem_size(cd.dtype);
        kc.data = cd.data;
        kc.valid = cd.validity;
        kc.offsets = cd.offsets;
    }

    int64_t nblocks = (n + 16383) / 16384;
    if (nblocks < 8) nblocks = 8;
    if (nblocks > 1024) nblocks = 1024;
    const int64_t chunk = (n + nblocks - 1) / nblocks;
    const int64_t max_rows = n + nblocks * 1024 + 1; /* spill worst case + table flush */

    auto r = new dd_reducer();
    r->n_keys = n_keys;
    r->n_aggs = n_aggs;
    hipEvent_t e0 = nullptr, e1 = nullptr;
    auto fail = [&](const char *m) {
        if (e0) (void)hipEventDestroy(e0);
        if (e1) (void)hipEventDestroy(e1);
        delete r;
        return set_err(DD_ERR_HIP, m);
    };

    /* dict32 key columns: the row hash reads precomputed per-value hashes */
    for (int k = 0; k < n_keys; k++) {
        const dd_col_desc &cd = batch->cols[key_cols[k]];
        if (cd.dtype != DD_DT_DICT32) continue;
        if (hipMalloc((void **)&r->dict_hashes[k], (size_t)cd.dict_n * 8 + 8) != hipSuccess)
            return fail("dict hash alloc");
        if (dd_launch_dict_hashes((const uint8_t *)cd.dict_bytes, cd.dict_offsets,
                                  cd.dict_n, r->dict_hashes[k],
                                  (hipStream_t)stream) != hipSuccess)
            return fail("dict hash launch");
        ka.cols[key_cols[k]].dict_hashes = r->dict_hashes[k];
    }
    if (hipMalloc((void **)&r->keys, (size_t)max_rows * n_keys * 8) != hipSuccess ||
        hipMalloc((void **)&r->keynull, (size_t)max_rows * 4) != hipSuccess ||
        hipMalloc((void **)&r->aggs, (size_t)max_rows * n_aggs * 8) != hipSuccess ||
        hipMalloc((void **)&r->nn, (size_t)max_rows * n_aggs * 8) != hipSuccess ||
        hipMalloc((void **)&r->n_dev, 8) != hipSuccess ||
        hipMalloc((void **)&r->meta, 2 * 4 * 4) != hipSuccess)
        return fail("partial reduce alloc");
    hipStream_t s = (hipStream_t)stream;
    if (hipMemsetAsync(r->n_dev, 0, 8, s) != hipSuccess) return fail("memset");
    int32_t meta_h[8] = {};
    for (int g = 0; g < n_aggs; g++) {
        meta_h[g] = agg_cols ? agg_cols[g] : 0;
        meta_h[4 + g] = agg_ops[g];
    }
    if (hipMemcpyAsync(r->meta, meta_h, sizeof(meta_h), hipMemcpyHostToDevice, s) !=
        hipSuccess)
        return fail("meta upload");
    (void)hipEventCreate(&e0);
    (void)hipEventCreate(&e1);
    (void)hipEventRecord(e0, s);
    hipError_t e = dd_launch_partial_reduce(&ka, nblocks, chunk, n_aggs, r->meta,
                                            r->meta + 4, r->keys, r->keynull, r->aggs,
                                            r->nn, r->n_dev, s);
    (void)hipEventRecord(e1, s);
    if (e != hipSuccess) return fail(hipGetErrorString(e));
    if (hipStreamSynchronize(s) != hipSuccess) return fail("sync");
    (void)hipEventElapsedTime(&r->kernel_ms, e0, e1);
    (void)hipEventDestroy(e0);
    (void)hipEventDestroy(e1);
    e0 = e1 = nullptr; /* fail() must not double-destroy */
    uint64_t n_out = 0;
    if (hipMemcpy(&n_out, r->n_dev, 8, hipMemcpyDeviceToHost) != hipSuccess)
        return fail("n_out copy");
    r->n_rows = (int64_t)n_out;
    *out = r;
    return DD_OK;
}

extern "C" int64_t dd_reducer_n_rows(const dd_reducer *r) { return r->n_rows; }
extern "C" float dd_reducer_kernel_ms(const dd_reducer *r) { return r->kernel_ms; }

extern "C" dd_status dd_reducer_fetch(const dd_reducer *r, uint64_t *host_keys,
                                      uint32_t *host_keynull, double *host_aggs) {
    HIP_TRY(hipMemcpy(host_keys, r->keys, (size_t)r->n_rows * r->n_keys * 8,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(host_keynull, r->keynull, (size_t)r->n_rows * 4,
                      hipMemcpyDeviceToHost));
    HIP_TRY(hipMemcpy(host_aggs, r->aggs, (size_t)r->n_rows * r->n_aggs * 8,
                      hipMemcpyDeviceToHost));
    return DD_OK;
}

/* non-null input counts per output row per aggregate ([n_rows][n_aggs]): the final
 * merge sums these; total 0 => the merged aggregate is NULL (DataFusion semantics for
 * SUM/MIN/MAX over an all-null group) */
extern "C" dd_status dd_reducer_fetch_nn(const dd_reducer *r, uint64_t *host_nn) {
    HIP_TRY(hipMemcpy(host_nn, r->nn, (size_t)r->n_rows * r->n_aggs * 8,
                      hipMemcpyDeviceToHost));
    return DD_OK;
}

extern "C" void dd_reducer_destroy(dd_reducer *r) { delete r; }

/* ---------------- coalesce (dd_coalesce_run) ---------------- */

/* task_group (network_coalesce.rs:376-400): contiguous ceil-split of producers */
static void dd_task_group(int input_tasks, int task_index, int task_count, int *start,
                          int *len) {
    const int base = input_tasks / task_count;
    const int extra = input_tasks % task_count;
    *len = base + (task_index < extra ? 1 : 0);
    *start = task_index * base + (task_index < extra ? task_index : extra);
}

extern "C" dd_status dd_coalesce_run(dd_comm *c, const dd_partitioner *p,
                                     int32_t consumer_tasks, void *stream,
                                     dd_exchanged **out) {
    if (!c || !p || !out) return set_err(DD_ERR_INVALID, "null argument");
    if (!p->has_run) return set_err(DD_ERR_INVALID, "partitioner has not run");
    if (consumer_tasks < 1 || consumer_tasks > c->nranks)
        return set_err(DD_ERR_INVALID, "consumer_tasks must be in [1, nranks]");
    hipStream_t s = (hipStream_t)stream;
    const int R = c->nranks;
    const uint32_t P = p->nparts;
    const int nvar = p->ka.n_var;

    /* 1) size matrix allgather (same meta layout as dd_exchange_run) */
    const size_t meta_n = (size_t)P * (1 + nvar);
    std::vector<int64_t> my_meta(meta_n);
    std::vector<int64_t> off_h(P + 1);
    dd_status st = dd_partitioner_row_offsets(p, off_h.data());
    if (st != DD_OK) return st;
    for (uint32_t q = 0; q < P; q++) my_meta[q] = off_h[q + 1] - off_h[q];
    std::vector<std::vector<int64_t>> boff_h(nvar, std::vector<int64_t>(P + 1));
    for (int v = 0; v < nvar; v++) {
        st = dd_partitioner_byte_offsets(p, p->ka.var_idx[v], boff_h[v].data());
        if (st != DD_OK) return st;
        for (uint32_t q = 0; q < P; q++)
            my_meta[(size_t)(1 + v) * P + q] = boff_h[v][q + 1] - boff_h[v][q];
    }
    std::vector<int64_t> all_meta((size_t)R * meta_n);
    {
        dd_status ms = dd_meta_allgather(c, my_meta, all_meta, s);
        if (ms != DD_OK) return ms;
    }

    /* 2) my group as a consumer (ranks >= consumer_tasks consume nothing) */
    int gstart = 0, glen = 0;
    if (c->rank < consumer_tasks) dd_task_group(R, c->rank, consumer_tasks, &gstart, &glen);
    /* my consumer as a producer: the consumer whose group contains me */
    int my_consumer = -1;
    for (int t = 0; t < consumer_tasks; t++) {
        int s0, l0;
        dd_task_group(R, t, consumer_tasks, &s0, &l0);
        if (c->rank >= s0 && c->rank < s0 + l0) my_consumer = t;
    }

    std::unique_ptr<dd_exchanged> e_own(new dd_exchanged()); /* freed on error returns */
    dd_exchanged *e = e_own.get();
    e->alloc_stream = dd_pool_stream();
    e->n_cols = p->batch.n_cols;
    e->nranks = glen; /* producers I consume */
    e->P = P;
    e->row_counts.assign((size_t)(glen > 0 ? glen : 1) * P, 0);
    e->byte_counts.assign(p->batch.n_cols, {});
    std::vector<int64_t> recv_rows(glen, 0);
    for (int gi = 0; gi < glen; gi++) {
        const int r = gstart + gi;
        for (uint32_t q = 0; q < P; q++) {
            int64_t n = all_meta[(size_t)r * meta_n + q];
            e->row_counts[(size_t)gi * P + q] = n;
            recv_rows[gi] += n;
            e->total_rows += n;
        }
    }
    auto fail = [&](dd_status sc, const char *m) { return set_err(sc, m); };

    std::vector<std::vector<int64_t>> recv_bytes(p->batch.n_cols,
                                                 std::vector<int64_t>(glen, 0));
    for (int ci = 0; ci < p->batch.n_cols; ci++) {
        const dd_kcol &kc = p->ka.cols[ci];
        if (kc.elem > 0) {
            if (hipMallocAsync(&e->data[ci], (size_t)e->total_rows * kc.elem + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "coalesce recv alloc");
        } else {
            int v = -1;
            for (int vv = 0; vv < nvar; vv++)
                if (p->ka.var_idx[vv] == ci) v = vv;
            e->byte_counts[ci].assign((size_t)(glen > 0 ? glen : 1) * P, 0);
            int64_t total_b = 0;
            for (int gi = 0; gi < glen; gi++) {
                const int r = gstart + gi;
                for (uint32_t q = 0; q < P; q++) {
                    int64_t b = all_meta[(size_t)r * meta_n + (size_t)(1 + v) * P + q];
                    e->byte_counts[ci][(size_t)gi * P + q] = b;
                    recv_bytes[ci][gi] += b;
                    total_b += b;
                }
            }
            if (hipMallocAsync(&e->data[ci], (size_t)total_b + 1, e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "coalesce recv alloc (var)");
            if (hipMallocAsync((void **)&e->lengths[ci], (size_t)e->total_rows * 4 + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "coalesce recv alloc (lengths)");
        }
        if (kc.valid) {
            if (hipMallocAsync((void **)&e->valid[ci], (size_t)e->total_rows + 1,
                               e->alloc_stream) != hipSuccess)
                return fail(DD_ERR_HIP, "coalesce recv alloc (validity)");
        }
    }
    if (hipStreamSynchronize(e->alloc_stream) != hipSuccess) /* allocations ready */
        return fail(DD_ERR_HIP, "pool stream sync");
    if (hipEventCreate(&e->e0) != hipSuccess || hipEventCreate(&e->e1) != hipSuccess)
        return fail(DD_ERR_HIP, "event create");

    /* 3) grouped p2p: I SEND my whole output to my_consumer; as a consumer I RECV from
     * each producer in my group, producer-major. Post order per peer: per col (data,
     * [lengths], [valid]) — symmetric on both sides. */
    const int64_t my_rows = off_h[P];
    HIP_TRY(hipEventRecord(e->e0, s));
    NCCL_TRY(ncclGroupStart());
    for (int ci = 0; ci < p->batch.n_cols; ci++) {
        const dd_kcol &kc = p->ka.cols[ci];
        /* sends (every rank has exactly one consumer) */
        if (my_consumer >= 0) {
            int64_t sent = 0;
            if (kc.elem > 0) {
                if (my_rows > 0)
                    NCCL_TRY(ncclSend(p->out_data[ci], my_rows * kc.elem, ncclUint8,
                                      my_consumer, c->comm, s));
                sent = my_rows * kc.elem;
            } else {
                int v = -1;
                for (int vv = 0; vv < nvar; vv++)
                    if (p->ka.var_idx[vv] == ci) v = vv;
                const int64_t my_b = boff_h[v][P];
                if (my_b > 0)
                    NCCL_TRY(ncclSend(p->out_data[ci], my_b, ncclUint8, my_consumer,
                                      c->comm, s));
                if (my_rows > 0)
                    NCCL_TRY(ncclSend(p->out_lengths[ci], my_rows * 4, ncclUint8,
                                      my_consumer, c->comm, s));
                sent = my_b + my_rows * 4;
            }
            if (kc.valid && my_rows > 0) {
                NCCL_TRY(ncclSend(p->out_valid[ci], my_rows, ncclUint8, my_consumer,
                                  c->comm, s));
                sent += my_rows;
            }
            if (my_consumer != c->rank) e->egress += sent;
        }
        /* recvs */
        int64_t rrow0 = 0, rb0 = 0;
        for (int gi = 0; gi < glen; gi++) {
            const int r = gstart + gi;
            if (kc.elem > 0) {
                if (recv_rows[gi] > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->data[ci] + rrow0 * kc.elem,
                                      recv_rows[gi] * kc.elem, ncclUint8, r, c->comm, s));
            } else {
                if (recv_bytes[ci][gi] > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->data[ci] + rb0, recv_bytes[ci][gi],
                                      ncclUint8, r, c->comm, s));
                if (recv_rows[gi] > 0)
                    NCCL_TRY(ncclRecv((uint8_t *)e->lengths[ci] + rrow0 * 4,
                                      recv_rows[gi] * 4, ncclUint8, r, c->comm, s));
                rb0 += recv_bytes[ci][gi];
            }
            if (kc.valid && recv_rows[gi] > 0)
                NCCL_TRY(ncclRecv(e->valid[ci] + rrow0, recv_rows[gi], ncclUint8, r,
                                  c->comm, s));
            rrow0 += recv_rows[gi];
        }
    }
    NCCL_TRY(ncclGroupEnd());
    HIP_TRY(hipEventRecord(e->e1, s));
    HIP_TRY(hipStreamSynchronize(s));
    HIP_TRY(hipEventElapsedTime(&e->ms, e->e0, e->e1));
    *out = e_own.release();
    return DD_OK;
}

