/* dd_hash_device.h — device-side normative hash, shared by the shuffle
 * kernels (dd_kernels.hip) and the partial-reduce kernel (dd_reduce.hip). Must match
 * oracle/dd_oracle.c bit-exactly. */

#ifndef DD_HASH_DEVICE_H
#define DD_HASH_DEVICE_H

#include <hip/hip_runtime.h>
#include <stdint.h>

#include "dd_internal.h"

/* ---------------- normative hash (must match oracle/dd_oracle.c bit-exactly) ------------ */

__device__ __forceinline__ uint64_t dd_mix64(uint64_t x) {
    x ^= x >> 30;
    x *= 0xbf58476d1ce4e5b9ULL;
    x ^= x >> 27;
    x *= 0x94d049bb133111ebULL;
    x ^= x >> 31;
    return x;
}

__device__ __forceinline__ uint64_t dd_hash_bytes_dev(const uint8_t *p, int64_t len) {
    uint64_t h = 0x9e3779b97f4a7c15ULL ^ ((uint64_t)len * 0xff51afd7ed558ccdULL);
    int64_t i = 0;
    for (; i + 8 <= len; i += 8) {
        uint64_t c;
        /* gfx950 supports unaligned wide loads: this lowers to global_load_dwordx2 */
        __builtin_memcpy(&c, p + i, 8);
        h = dd_mix64(h ^ c);
    }
    if (i < len) {
        uint64_t c = 0;
        for (int64_t b = 0; b < len - i; b++) c |= (uint64_t)p[i + b] << (8 * b);
        h = dd_mix64(h ^ c);
    }
    return h;
}

__device__ __forceinline__ uint64_t dd_canon_f64_dev(double v) {
    if (v == 0.0) v = 0.0;
    uint64_t b = __double_as_longlong(v);
    if (v != v) b = 0x7ff8000000000000ULL;
    return b;
}

__device__ __forceinline__ uint64_t dd_canon_f32_dev(float v) {
    if (v == 0.0f) v = 0.0f;
    uint32_t b = __float_as_uint(v);
    if (v != v) b = 0x7fc00000u;
    return (uint64_t)b;
}

/* value hash of a valid row (column described by a dd_kcol) */
__device__ __forceinline__ uint64_t dd_value_hash_dev(const dd_kcol &c, int64_t i) {
    switch (c.dtype) {
    case DD_KDT_U8:
    case DD_KDT_BOOL: /* unpacked u8 0/1 hashes like u8 (oracle dd_value_hash) */
        return dd_mix64((uint64_t)((const uint8_t *)c.data)[i]);
    case DD_KDT_I16:
        return dd_mix64((uint64_t)((const uint16_t *)c.data)[i]);
    case DD_KDT_I32:
        return dd_mix64((uint64_t)((const uint32_t *)c.data)[i]);
    case DD_KDT_I64:
        return dd_mix64(((const uint64_t *)c.data)[i]);
    case DD_KDT_F32:
        return dd_mix64(dd_canon_f32_dev(((const float *)c.data)[i]));
    case DD_KDT_F64:
        return dd_mix64(dd_canon_f64_dev(((const double *)c.data)[i]));
    case DD_KDT_UTF8: {
        int32_t o0 = c.offsets[i], o1 = c.offsets[i + 1];
        return dd_hash_bytes_dev((const uint8_t *)c.data + o0, (int64_t)(o1 - o0));
    }
    case DD_KDT_DICT32: {
        int32_t k = ((const int32_t *)c.data)[i];
        return c.dict_hashes[k]; /* precomputed by k_dict_hashes */
    }
    default:
        return 0;
    }
}


/* row hash over the key columns; create_hashes restatement */
__device__ __forceinline__ uint64_t dd_row_hash(const dd_kargs &a, int64_t i) {
    uint64_t h = 0;
    for (int k = 0; k < a.n_keys; k++) {
        const dd_kcol &c = a.cols[a.key_idx[k]];
        if (c.valid && !c.valid[i]) continue;
        uint64_t vh = dd_value_hash_dev(c, i);
        h = h ^ (vh + 0x9e3779b97f4a7c15ULL + (h << 6) + (h >> 2));
    }
    return h;
}

/* ballot-multisplit: lanes with equal pid (among `act`); returns the equal-mask */
__device__ __forceinline__ uint64_t dd_eq_mask(uint32_t pid, uint64_t act, int nbits) {
    uint64_t eq = act;
    for (int b = 0; b < nbits; b++) {
        uint64_t bal = __ballot((pid >> b) & 1u);
        eq &= ((pid >> b) & 1u) ? bal : ~bal;
    }
    return eq;
}


#endif /* DD_HASH_DEVICE_H */
