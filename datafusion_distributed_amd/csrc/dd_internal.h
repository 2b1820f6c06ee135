/* dd_internal.h — shared between dd_kernels.hip and dd_host.cpp (not part of the ABI). */

#ifndef DD_INTERNAL_H
#define DD_INTERNAL_H

#include <hip/hip_runtime.h>
#include <stdint.h>

#define DD_MAX_P 2048
#define WAVES_PER_BLOCK_H 4 /* must match WAVES_PER_BLOCK in dd_kernels.hip */
#define DD_KMAX_COLS 24
#define DD_KMAX_KEYS 8
#define DD_KMAX_VAR 4
#define DD_SCAN_RANGES 64
#define DD_STAGE_MAXC 8 /* staged-path column cap (register prefetch arrays) */

/* dtype codes mirror dd_dtype in include/dd_shuffle.h */
enum {
    DD_KDT_U8 = 1,
    DD_KDT_I16 = 2,
    DD_KDT_I32 = 3,
    DD_KDT_I64 = 4,
    DD_KDT_F32 = 5,
    DD_KDT_F64 = 6,
    DD_KDT_BOOL = 7,
    DD_KDT_UTF8 = 8,
    DD_KDT_DICT32 = 9,
    DD_KDT_VARLEN = 100, /* synthetic (staged-var): data = var col's offsets; value = len */
    DD_KDT_ROWID = 101,  /* synthetic (staged-var): value = input row index */
};

struct dd_kcol {
    int32_t dtype;
    int32_t elem;               /* fixed elem size (1/2/4/8); 0 for var (utf8) data */
    const void *data;           /* input values / utf8 bytes */
    const uint8_t *valid;       /* unpacked u8, or null */
    const int32_t *offsets;     /* utf8 input offsets[n+1] */
    const uint64_t *dict_hashes;/* dict32: per-value hash */
    void *out_data;             /* partition-major output values / bytes */
    uint8_t *out_valid;         /* partition-major output validity (if valid != null) */
    uint32_t *out_lengths;      /* utf8: partition-major per-row byte lengths */
};

struct dd_kargs {
    int64_t n_rows;
    int32_t n_cols;
    int32_t n_keys;
    int32_t n_var;
    uint32_t pid_total; /* pid = (h % pid_total) >> pid_shift; nparts = pid_total >> shift */
    int32_t pid_shift;  /* 0 for plain partitioning; >0 = contiguous coarse buckets */
    int32_t rhash;      /* staged spec path recomputes the row hash from preloaded
                           registers in K3 (no pid array: K1 skips its store, K3 its
                           load). Host gates: all-fixed no-validity batch, integer/bool
                           keys, wpb==16 (DD_RHASH=0 disables). */
    int32_t nt;         /* pre path: non-temporal flush stores (default on) */
    int32_t pid8;       /* pre path, P <= 256: pid array is u8 (saves 3 B/row of HBM
                           write in K1 + read in K3); dd_partitioner_pid_elem reports */
    int32_t hl;         /* hidden-load scatter (default on for its shape; DD_K3_HL=0
                           reverts): preload loads in inline asm + hand-counted
                           s_waitcnt so flush stores never drain mid-loop. gmax 4,
                           wpb 16, 4 fixed cols, no validity, elems in {4,8};
                           mutually exclusive with rhash. */
    int32_t key_idx[DD_KMAX_KEYS];
    int32_t var_idx[DD_KMAX_VAR];
    dd_kcol cols[DD_KMAX_COLS];
};

#include "dd_shuffle.h" /* dd_status for dd_set_error */

extern "C" {
dd_status dd_set_error(dd_status s, const char *msg); /* thread-local dd_last_error */
hipError_t dd_launch_dict_hashes(const uint8_t *bytes, const int32_t *offsets, int64_t n,
                                 uint64_t *out, hipStream_t s);
hipError_t dd_launch_k5_maxlen(const int32_t *offsets, int64_t n, uint32_t *out_max,
                               hipStream_t s);
hipError_t dd_launch_k5_count(int64_t n, uint32_t nparts, const uint32_t *pid,
                              const int32_t *offsets, uint16_t *bcounts,
                              uint32_t *partials, int nranges, int64_t nseg5,
                              size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_k5_roff(const uint32_t *bcounts, const uint64_t *part_boffsets,
                             int64_t nrounds, int wpb, uint32_t nparts, uint32_t *roffB,
                             hipStream_t s);
hipError_t dd_launch_k5_scatter(int64_t n, uint32_t nparts, int nbits, const uint32_t *pid,
                                const int32_t *offsets, const uint8_t *in_bytes,
                                const uint32_t *gbaseB, const uint32_t *roffB,
                                int64_t nrounds, uint8_t *out_bytes, int wpb5,
                                size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_hash_count(const dd_kargs *a, int64_t nchunks, int64_t chunk_rows,
                                uint32_t nparts, int nbits, uint32_t *pid_out,
                                uint32_t *counts, uint32_t *bcounts, size_t lds_bytes,
                                hipStream_t s);
hipError_t dd_launch_scan(uint32_t *counts, int64_t nchunks, uint32_t nparts, int nranges,
                          uint32_t *partials, uint64_t *part_offsets, int fold_global,
                          hipStream_t s);
hipError_t dd_launch_scan_deep(uint16_t *counts16, int64_t nchunks, uint32_t nparts,
                               int nr1, int nr2, uint32_t *partials, uint32_t *partials2,
                               uint64_t *part_offsets, uint32_t *gbase, hipStream_t s);
hipError_t dd_launch_hash_count_seg(const dd_kargs *a, int64_t nseg, int64_t seg_rows,
                                    uint32_t nparts, int nbits, uint32_t *pid_out,
                                    uint16_t *counts, uint32_t *partials, int nranges,
                                    size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_round_layout(const uint32_t *counts, const uint64_t *part_offsets,
                                  int64_t nrounds, int wpb, uint32_t nparts, uint32_t sP2,
                                  uint16_t *imgb, hipStream_t s);
hipError_t dd_launch_scatter_pre(const dd_kargs *a, int64_t nblocks, int64_t nrounds,
                                 int rpb, uint32_t nparts, int nbits,
                                 const uint32_t *pid_in, const uint32_t *gbase,
                                 const uint16_t *imgb, uint32_t sP2, int gmax, int wpb,
                                 size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_hash_count_tile(const dd_kargs *a, int64_t nblocks, int64_t tile_rows,
                                     uint32_t nparts, int nbits, uint32_t *pid_out,
                                     uint32_t *counts, size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_scatter_staged(const dd_kargs *a, int64_t nblocks, int64_t tile_rows,
                                    uint32_t nparts, int nbits, const uint32_t *pid_in,
                                    const uint32_t *tile_off, const uint64_t *part_offsets,
                                    int gmax, int wpb, size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_scatter(const dd_kargs *a, int64_t nchunks, int64_t chunk_rows,
                             uint32_t nparts, int nbits, const uint32_t *pid_in,
                             const uint32_t *chunk_off, const uint64_t *part_offsets,
                             const uint32_t *chunk_boff, const uint64_t *part_boffsets,
                             size_t lds_bytes, hipStream_t s);
hipError_t dd_launch_off64_to_off32(const uint64_t *off64, int64_t lo, int64_t n,
                                    int32_t *out32, hipStream_t s);
hipError_t dd_launch_var_bytes(const uint32_t *lens, const uint32_t *src_row,
                               const int32_t *in_offsets, const uint8_t *in_bytes,
                               int64_t n, int64_t total_bytes, uint64_t *partials,
                               uint64_t *out_off, uint8_t *out_bytes,
                               const uint64_t *part_offsets, uint32_t nparts,
                               uint64_t *part_boffsets, uint32_t *k4w_meta,
                               uint32_t *k4w_order, int skip_copy, hipStream_t s);
hipError_t dd_launch_partial_reduce(const dd_kargs *a, int64_t nblocks, int64_t chunk_rows,
                                    int n_aggs, const int32_t *agg_cols,
                                    const int32_t *agg_ops, uint64_t *out_keys,
                                    uint32_t *out_keynull, double *out_aggs,
                                    uint64_t *out_nn, uint64_t *out_n, hipStream_t s);
}

#endif
