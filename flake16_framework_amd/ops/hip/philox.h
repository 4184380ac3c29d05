// Philox4x32-10 device implementation — bit-identical to the numpy
// reference in flake16_framework_amd/utils/philox.py.  All model/balancing
// randomness flows through this, keyed on deterministic identities
// (tag, node sample-range, draw index), never on scheduling order.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define PHILOX_M0 0xD2511F53u
#define PHILOX_M1 0xCD9E8D57u
#define PHILOX_W0 0x9E3779B9u
#define PHILOX_W1 0xBB67AE85u

// Domain-separation tags (must match utils/philox.py).
#define TAG_BOOTSTRAP 1u
#define TAG_FEATSEL 2u
#define TAG_THRESH 3u
#define TAG_SMOTE_PICK 4u
#define TAG_SMOTE_GAP 5u

struct Philox4 {
    uint32_t x0, x1, x2, x3;
};

__device__ __host__ __forceinline__ Philox4
philox4x32(uint32_t c0, uint32_t c1, uint32_t c2, uint32_t c3,
           uint32_t k0, uint32_t k1) {
    #pragma unroll
    for (int r = 0; r < 10; ++r) {
        uint64_t p0 = (uint64_t)PHILOX_M0 * c0;
        uint64_t p1 = (uint64_t)PHILOX_M1 * c2;
        uint32_t hi0 = (uint32_t)(p0 >> 32), lo0 = (uint32_t)p0;
        uint32_t hi1 = (uint32_t)(p1 >> 32), lo1 = (uint32_t)p1;
        uint32_t n0 = hi1 ^ c1 ^ k0;
        uint32_t n1 = lo1;
        uint32_t n2 = hi0 ^ c3 ^ k1;
        uint32_t n3 = lo0;
        c0 = n0; c1 = n1; c2 = n2; c3 = n3;
        k0 += PHILOX_W0;
        k1 += PHILOX_W1;
    }
    return {c0, c1, c2, c3};
}

// First output word only — mirrors utils/philox.py draws_u32.
__device__ __host__ __forceinline__ uint32_t
philox_draw(uint32_t tag, uint32_t c1, uint32_t c2, uint32_t c3,
            uint32_t k0, uint32_t k1) {
    return philox4x32(tag, c1, c2, c3, k0, k1).x0;
}

// uint32 -> [0, n) by multiply-shift (matches bounded_int).
__device__ __host__ __forceinline__ uint32_t
philox_bounded(uint32_t u, uint32_t n) {
    return (uint32_t)(((uint64_t)u * n) >> 32);
}

// uint32 -> float32 in [0, 1) (matches u32_to_unit: fp64 scale, then cast).
__device__ __host__ __forceinline__ float philox_unit(uint32_t u) {
    return (float)((double)u * (1.0 / 4294967296.0));
}
