// Device modular arithmetic for RNS-CKKS on gfx950 (CDNA4).
//
// Replaces the SEAL 2.3 C++ poly arithmetic behind Pyfhel that the
// reference invokes (FLPyfhelin.py:217,295,381,385). CDNA4 has no native
// 64-bit modmul; variable*variable uses Barrett reduction of the 128-bit
// product (two __umul64hi chains), constant*variable uses Shoup
// precomputed floor(w<<64 / q) — one mul_hi + one mul_lo per butterfly.
// All limb primes are < 2**60 (hefl/he/primes.py) so a+b never wraps and
// a lazy int64 SUM over <= 8 clients (the RCCL all-reduce) is overflow-free.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

__device__ __forceinline__ uint64_t addmod_u64(uint64_t a, uint64_t b, uint64_t q) {
    uint64_t r = a + b;
    return r >= q ? r - q : r;
}

__device__ __forceinline__ uint64_t submod_u64(uint64_t a, uint64_t b, uint64_t q) {
    return a >= b ? a - b : a + q - b;
}

// Shoup multiplication: w < q fixed with wsh = floor(w * 2^64 / q); x < q.
// Result < q. (Harvey butterfly form.)
__device__ __forceinline__ uint64_t mulmod_shoup(uint64_t x, uint64_t w,
                                                 uint64_t wsh, uint64_t q) {
    uint64_t hi = __umul64hi(x, wsh);
    uint64_t r = x * w - hi * q;  // mod 2^64; r < 2q
    return r >= q ? r - q : r;
}

// Barrett reduction of a 128-bit value z1:z0 mod q, with precomputed
// ratio r1:r0 = floor(2^128 / q) (SEAL barrett_reduce_128 structure).
__device__ __forceinline__ uint64_t barrett_red128(uint64_t z0, uint64_t z1,
                                                   uint64_t q, uint64_t r0,
                                                   uint64_t r1) {
    uint64_t carry = __umul64hi(z0, r0);
    uint64_t lo1 = z0 * r1;
    uint64_t hi1 = __umul64hi(z0, r1);
    uint64_t t1 = lo1 + carry;
    uint64_t t3 = hi1 + (t1 < lo1);
    uint64_t lo2 = z1 * r0;
    uint64_t hi2 = __umul64hi(z1, r0);
    uint64_t t1b = t1 + lo2;
    uint64_t carry2 = hi2 + (t1b < lo2);
    uint64_t qhat = z1 * r1 + t3 + carry2;
    uint64_t res = z0 - qhat * q;  // < 2q
    return res >= q ? res - q : res;
}

__device__ __forceinline__ uint64_t mulmod_barrett(uint64_t a, uint64_t b,
                                                   uint64_t q, uint64_t r0,
                                                   uint64_t r1) {
    uint64_t z0 = a * b;
    uint64_t z1 = __umul64hi(a, b);
    return barrett_red128(z0, z1, q, r0, r1);
}

// Host-side: floor(2^128 / q) for an odd prime q (q never divides 2^128,
// so floor((2^128 - 1)/q) is the same value and fits 128-bit arithmetic).
inline void barrett_ratio(uint64_t q, uint64_t* r0, uint64_t* r1) {
    unsigned __int128 num = ~(unsigned __int128)0;
    unsigned __int128 r = num / q;
    *r0 = (uint64_t)r;
    *r1 = (uint64_t)(r >> 64);
}
