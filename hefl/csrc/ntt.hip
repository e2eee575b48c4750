// Batched negacyclic NTT / INTT over RNS limbs + pointwise modular ops —
// the keystone HIP kernels of the CKKS layer (NTT row).
//
// Replaces SEAL's NTT inside every encrypt/mult the reference reaches via
// Pyfhel (FLPyfhelin.py:217,295,381,385). Layout matches the CPU oracle
// hefl/he/ntt_cpu.py exactly: Cooley-Tukey forward / Gentleman-Sande inverse
// with merged psi powers in bit-reversed table order (tables built in
// Python, uploaded once per context).
//
// Kernel index (in file order):
//   ntt_lds_kernel / intt_lds_kernel         single-q LDS-resident stages
//   ntt_global_stage_kernel / intt_...       strided stages for n=2^15
//   *_ml variants                            fused multi-limb (limb = row%L),
//                                            radix-4 in-LDS (2 butterfly
//                                            levels per barrier)
//   ntt/intt_global_radix4_ml_kernel         fused radix-4 global stage pair
//   modmul/modadd/modsub/modmul_scalar       Barrett/Shoup pointwise ops
//   modadd_limbs/modadd3_limbs/modsub_limbs  fused per-limb pointwise glue
//                                            (limb = (i/n) % L; modsub takes
//                                            a strided source for rescale)
//   bcast_center_mod_kernel                  [.., n] -> [.., L, n] center +
//                                            per-limb Barrett reduce
//   ct_mul_kernel                            ct x ct tensor product
//                                            (d0, d1, d2 in one launch)
//   ks_inner_kernel                          whole key-switch digit inner
//                                            product, both accumulators
//   modreduce_ kernel                        lazy-sum -> [0,q) (post
//                                            all-reduce), branchless
//   cbd21_kernel                             centered-binomial noise from
//                                            one 64-bit draw (popcounts)
//
// Structure per row of length n (n = 2^6 .. 2^15):
//  * stages whose butterfly span exceeds NBLK run in a strided global-memory
//    kernel (only n=2^15 needs any on MI355X with NBLK=8192);
//  * the remaining stages run inside one workgroup with the whole block
//    resident in LDS (NBLK * 8 B <= 64 KiB of the CU's 160 KiB), one
//    __syncthreads between stages. Twiddles come from global memory — they
//    are shared by every row of the batch, so they sit in the XCD L2s.
//
// Batch dimension (ciphertexts x limbs) is gridDim.y — HE workloads here
// launch hundreds of rows, comfortably above the 256-CU fill point.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>

#include "modmath.h"

#define CHECK_CUDA_OK(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")

namespace {

constexpr int kThreads = 256;
constexpr int kNblkMax = 8192;  // 64 KiB int64 in LDS

// ---------------------------------------------------------------------------
// Forward NTT
// ---------------------------------------------------------------------------

// One global-memory stage: group count m, butterfly stride t = n/(2m) > NBLK/2.
__global__ void ntt_global_stage_kernel(int64_t* __restrict__ x,
                                        const int64_t* __restrict__ w,
                                        const int64_t* __restrict__ wsh,
                                        uint64_t q, int n, int m) {
    const int64_t nhalf = n >> 1;
    const int64_t row = blockIdx.y;
    int64_t base_off = row * (int64_t)n;
    const uint32_t t = (uint32_t)(n / (2 * m));
    const int tlog = 31 - __clz(t);  // t is a power of two
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < nhalf;
         k += (int64_t)gridDim.x * blockDim.x) {
        uint32_t j = (uint32_t)(k >> tlog);
        uint32_t pos = (uint32_t)k & (t - 1);
        int64_t i0 = base_off + (int64_t)j * 2 * t + pos;
        uint64_t W = (uint64_t)w[m + j];
        uint64_t Wsh = (uint64_t)wsh[m + j];
        uint64_t U = (uint64_t)x[i0];
        uint64_t V = mulmod_shoup((uint64_t)x[i0 + t], W, Wsh, q);
        x[i0] = (int64_t)addmod_u64(U, V, q);
        x[i0 + t] = (int64_t)submod_u64(U, V, q);
    }
}

// LDS phase: all stages with butterfly span <= nblk. blockIdx.x = block
// within row, blockIdx.y = row. m_start = n / nblk (global group count at
// the first LDS stage).
__global__ void __launch_bounds__(kThreads)
ntt_lds_kernel(int64_t* __restrict__ x, const int64_t* __restrict__ w,
               const int64_t* __restrict__ wsh, uint64_t q, int n, int nblk) {
    extern __shared__ __attribute__((aligned(16))) int64_t smem[];
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    int64_t* xr = x + (int64_t)blockIdx.y * n + (int64_t)blk * nblk;
    for (int i = tid; i < nblk; i += kThreads) smem[i] = xr[i];
    __syncthreads();
    const int nb2 = nblk >> 1;
    for (int m = n / nblk; m < n; m <<= 1) {
        const uint32_t t = (uint32_t)(n / (2 * m));  // <= nblk/2 here
        const int tlog = 31 - __clz(t);              // t is a power of two
        for (int lb = tid; lb < nb2; lb += kThreads) {
            uint32_t jloc = (uint32_t)lb >> tlog;
            uint32_t pos = (uint32_t)lb & (t - 1);
            uint32_t base = jloc * 2 * t + pos;
            uint32_t jglob = (uint32_t)blk * ((uint32_t)nb2 >> tlog) + jloc;
            uint64_t W = (uint64_t)w[m + jglob];
            uint64_t Wsh = (uint64_t)wsh[m + jglob];
            uint64_t U = (uint64_t)smem[base];
            uint64_t V = mulmod_shoup((uint64_t)smem[base + t], W, Wsh, q);
            smem[base] = (int64_t)addmod_u64(U, V, q);
            smem[base + t] = (int64_t)submod_u64(U, V, q);
        }
        __syncthreads();
    }
    for (int i = tid; i < nblk; i += kThreads) xr[i] = smem[i];
}

// ---------------------------------------------------------------------------
// Inverse NTT (Gentleman-Sande). LDS phase first (small strides), then
// global stages, final scaling by n^-1 folded into the last stage.
// ---------------------------------------------------------------------------

__global__ void __launch_bounds__(kThreads)
intt_lds_kernel(int64_t* __restrict__ x, const int64_t* __restrict__ winv,
                const int64_t* __restrict__ winvsh, uint64_t q, int n,
                int nblk, uint64_t ninv, uint64_t ninvsh, int scale_here) {
    extern __shared__ __attribute__((aligned(16))) int64_t smem[];
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    int64_t* xr = x + (int64_t)blockIdx.y * n + (int64_t)blk * nblk;
    for (int i = tid; i < nblk; i += kThreads) smem[i] = xr[i];
    __syncthreads();
    const int nb2 = nblk >> 1;
    // iterate m = n, n/2, ..., down while 2t <= nblk i.e. m >= 2n/nblk
    for (int m = n; m >= 2 * (n / nblk); m >>= 1) {
        const int h = m >> 1;
        const uint32_t t = (uint32_t)(n / m);
        const int tlog = 31 - __clz(t);              // t is a power of two
        for (int lb = tid; lb < nb2; lb += kThreads) {
            uint32_t jloc = (uint32_t)lb >> tlog;
            uint32_t pos = (uint32_t)lb & (t - 1);
            uint32_t base = jloc * 2 * t + pos;
            uint32_t jglob = (uint32_t)blk * ((uint32_t)nb2 >> tlog) + jloc;
            uint64_t S = (uint64_t)winv[h + jglob];
            uint64_t Ssh = (uint64_t)winvsh[h + jglob];
            uint64_t U = (uint64_t)smem[base];
            uint64_t V = (uint64_t)smem[base + t];
            smem[base] = (int64_t)addmod_u64(U, V, q);
            smem[base + t] = (int64_t)mulmod_shoup(submod_u64(U, V, q), S, Ssh, q);
        }
        __syncthreads();
    }
    if (scale_here) {
        for (int i = tid; i < nblk; i += kThreads)
            smem[i] = (int64_t)mulmod_shoup((uint64_t)smem[i], ninv, ninvsh, q);
        __syncthreads();
    }
    for (int i = tid; i < nblk; i += kThreads) xr[i] = smem[i];
}

__global__ void intt_global_stage_kernel(int64_t* __restrict__ x,
                                         const int64_t* __restrict__ winv,
                                         const int64_t* __restrict__ winvsh,
                                         uint64_t q, int n, int m,
                                         uint64_t ninv, uint64_t ninvsh,
                                         int scale_here) {
    const int64_t nhalf = n >> 1;
    const int64_t row = blockIdx.y;
    int64_t base_off = row * (int64_t)n;
    const int h = m >> 1;
    const uint32_t t = (uint32_t)(n / m);
    const int tlog = 31 - __clz(t);  // t is a power of two
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < nhalf;
         k += (int64_t)gridDim.x * blockDim.x) {
        uint32_t j = (uint32_t)(k >> tlog);
        uint32_t pos = (uint32_t)k & (t - 1);
        int64_t i0 = base_off + (int64_t)j * 2 * t + pos;
        uint64_t S = (uint64_t)winv[h + j];
        uint64_t Ssh = (uint64_t)winvsh[h + j];
        uint64_t U = (uint64_t)x[i0];
        uint64_t V = (uint64_t)x[i0 + t];
        uint64_t a = addmod_u64(U, V, q);
        uint64_t b = mulmod_shoup(submod_u64(U, V, q), S, Ssh, q);
        if (scale_here) {
            a = mulmod_shoup(a, ninv, ninvsh, q);
            b = mulmod_shoup(b, ninv, ninvsh, q);
        }
        x[i0] = (int64_t)a;
        x[i0 + t] = (int64_t)b;
    }
}

// ---------------------------------------------------------------------------
// Multi-limb fused variants: one launch covers every RNS limb of a
// [R, L, n] tensor (flat row r -> limb r % L, tables indexed per limb).
// Cuts the per-limb Python/launch loop out of encrypt/decrypt/keyswitch.
// ---------------------------------------------------------------------------

__global__ void ntt_global_stage_ml_kernel(int64_t* __restrict__ x,
                                           const int64_t* __restrict__ w,
                                           const int64_t* __restrict__ wsh,
                                           const int64_t* __restrict__ qs,
                                           int L, int n, int m) {
    const int64_t nhalf = n >> 1;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = w + (int64_t)limb * n;
    const int64_t* wshl = wsh + (int64_t)limb * n;
    int64_t base_off = row * (int64_t)n;
    const uint32_t t = (uint32_t)(n / (2 * m));
    const int tlog = 31 - __clz(t);  // t is a power of two
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < nhalf;
         k += (int64_t)gridDim.x * blockDim.x) {
        uint32_t j = (uint32_t)(k >> tlog);
        uint32_t pos = (uint32_t)k & (t - 1);
        int64_t i0 = base_off + (int64_t)j * 2 * t + pos;
        uint64_t U = (uint64_t)x[i0];
        uint64_t V = mulmod_shoup((uint64_t)x[i0 + t], (uint64_t)wl[m + j],
                                  (uint64_t)wshl[m + j], q);
        x[i0] = (int64_t)addmod_u64(U, V, q);
        x[i0 + t] = (int64_t)submod_u64(U, V, q);
    }
}

__global__ void __launch_bounds__(kThreads)
ntt_lds_ml_kernel(int64_t* __restrict__ x, const int64_t* __restrict__ w,
                  const int64_t* __restrict__ wsh,
                  const int64_t* __restrict__ qs, int L, int n, int nblk) {
    extern __shared__ __attribute__((aligned(16))) int64_t smem[];
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = w + (int64_t)limb * n;
    const int64_t* wshl = wsh + (int64_t)limb * n;
    int64_t* xr = x + row * n + (int64_t)blk * nblk;
    for (int i = tid; i < nblk; i += kThreads) smem[i] = xr[i];
    __syncthreads();
    const int nb2 = nblk >> 1;
    const int nb4 = nblk >> 2;
    int m = n / nblk;
    // radix-4 passes: two CT stages (m, 2m) per barrier — halves the LDS
    // round trips and __syncthreads count of the radix-2 ladder
    for (; 2 * m < n; m <<= 2) {
        const uint32_t t = (uint32_t)(n / (2 * m));
        const uint32_t t2 = t >> 1;
        const int t2log = 31 - __clz(t2);
        for (int lb = tid; lb < nb4; lb += kThreads) {
            uint32_t g = (uint32_t)lb >> t2log;
            uint32_t p = (uint32_t)lb & (t2 - 1);
            uint32_t base = g * 2 * t + p;
            uint32_t jm = (uint32_t)blk * ((uint32_t)nb2 >> (t2log + 1)) + g;
            uint64_t x0 = (uint64_t)smem[base];
            uint64_t x1 = (uint64_t)smem[base + t2];
            uint64_t x2 = (uint64_t)smem[base + t];
            uint64_t x3 = (uint64_t)smem[base + t + t2];
            // stage m (twiddle w[m + jm] for both pairs)
            {
                uint64_t W = (uint64_t)wl[m + jm];
                uint64_t Wsh = (uint64_t)wshl[m + jm];
                uint64_t v = mulmod_shoup(x2, W, Wsh, q);
                x2 = submod_u64(x0, v, q);
                x0 = addmod_u64(x0, v, q);
                v = mulmod_shoup(x3, W, Wsh, q);
                x3 = submod_u64(x1, v, q);
                x1 = addmod_u64(x1, v, q);
            }
            // stage 2m: groups 2jm (x0,x1) and 2jm+1 (x2,x3)
            {
                uint64_t W = (uint64_t)wl[2 * m + 2 * jm];
                uint64_t Wsh = (uint64_t)wshl[2 * m + 2 * jm];
                uint64_t v = mulmod_shoup(x1, W, Wsh, q);
                x1 = submod_u64(x0, v, q);
                x0 = addmod_u64(x0, v, q);
                W = (uint64_t)wl[2 * m + 2 * jm + 1];
                Wsh = (uint64_t)wshl[2 * m + 2 * jm + 1];
                v = mulmod_shoup(x3, W, Wsh, q);
                x3 = submod_u64(x2, v, q);
                x2 = addmod_u64(x2, v, q);
            }
            smem[base] = (int64_t)x0;
            smem[base + t2] = (int64_t)x1;
            smem[base + t] = (int64_t)x2;
            smem[base + t + t2] = (int64_t)x3;
        }
        __syncthreads();
    }
    for (; m < n; m <<= 1) {  // at most one radix-2 tail stage
        const uint32_t t = (uint32_t)(n / (2 * m));
        const int tlog = 31 - __clz(t);
        for (int lb = tid; lb < nb2; lb += kThreads) {
            uint32_t jloc = (uint32_t)lb >> tlog;
            uint32_t pos = (uint32_t)lb & (t - 1);
            uint32_t base = jloc * 2 * t + pos;
            uint32_t jglob = (uint32_t)blk * ((uint32_t)nb2 >> tlog) + jloc;
            uint64_t U = (uint64_t)smem[base];
            uint64_t V = mulmod_shoup((uint64_t)smem[base + t],
                                      (uint64_t)wl[m + jglob],
                                      (uint64_t)wshl[m + jglob], q);
            smem[base] = (int64_t)addmod_u64(U, V, q);
            smem[base + t] = (int64_t)submod_u64(U, V, q);
        }
        __syncthreads();
    }
    for (int i = tid; i < nblk; i += kThreads) xr[i] = smem[i];
}

__global__ void __launch_bounds__(kThreads)
intt_lds_ml_kernel(int64_t* __restrict__ x, const int64_t* __restrict__ winv,
                   const int64_t* __restrict__ winvsh,
                   const int64_t* __restrict__ qs,
                   const int64_t* __restrict__ ninv,
                   const int64_t* __restrict__ ninvsh, int L, int n, int nblk,
                   int scale_here) {
    extern __shared__ __attribute__((aligned(16))) int64_t smem[];
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = winv + (int64_t)limb * n;
    const int64_t* wshl = winvsh + (int64_t)limb * n;
    int64_t* xr = x + row * n + (int64_t)blk * nblk;
    for (int i = tid; i < nblk; i += kThreads) smem[i] = xr[i];
    __syncthreads();
    const int nb2 = nblk >> 1;
    const int nb4 = nblk >> 2;
    int m = n;
    // radix-4 passes: two GS stages (m, m/2) per barrier — halves LDS
    // round trips and __syncthreads vs the radix-2 ladder
    for (; m >= 4 * (n / nblk); m >>= 2) {
        const int h = m >> 1;
        const int h2 = m >> 2;
        const uint32_t t = (uint32_t)(n / m);
        const int tlog = 31 - __clz(t);              // t is a power of two
        for (int lb = tid; lb < nb4; lb += kThreads) {
            uint32_t j2 = (uint32_t)lb >> tlog;
            uint32_t p = (uint32_t)lb & (t - 1);
            uint32_t base = j2 * 4 * t + p;
            uint32_t j2g = (uint32_t)blk * ((uint32_t)nb4 >> tlog) + j2;
            uint64_t x0 = (uint64_t)smem[base];
            uint64_t x1 = (uint64_t)smem[base + t];
            uint64_t x2 = (uint64_t)smem[base + 2 * t];
            uint64_t x3 = (uint64_t)smem[base + 3 * t];
            // stage m: (x0,x1) group 2*j2g, (x2,x3) group 2*j2g+1
            {
                uint64_t W = (uint64_t)wl[h + 2 * j2g];
                uint64_t Wsh = (uint64_t)wshl[h + 2 * j2g];
                uint64_t u = addmod_u64(x0, x1, q);
                x1 = mulmod_shoup(submod_u64(x0, x1, q), W, Wsh, q);
                x0 = u;
                W = (uint64_t)wl[h + 2 * j2g + 1];
                Wsh = (uint64_t)wshl[h + 2 * j2g + 1];
                u = addmod_u64(x2, x3, q);
                x3 = mulmod_shoup(submod_u64(x2, x3, q), W, Wsh, q);
                x2 = u;
            }
            // stage m/2: (x0,x2) and (x1,x3), both group j2g
            {
                uint64_t W = (uint64_t)wl[h2 + j2g];
                uint64_t Wsh = (uint64_t)wshl[h2 + j2g];
                uint64_t u = addmod_u64(x0, x2, q);
                x2 = mulmod_shoup(submod_u64(x0, x2, q), W, Wsh, q);
                x0 = u;
                u = addmod_u64(x1, x3, q);
                x3 = mulmod_shoup(submod_u64(x1, x3, q), W, Wsh, q);
                x1 = u;
            }
            smem[base] = (int64_t)x0;
            smem[base + t] = (int64_t)x1;
            smem[base + 2 * t] = (int64_t)x2;
            smem[base + 3 * t] = (int64_t)x3;
        }
        __syncthreads();
    }
    for (; m >= 2 * (n / nblk); m >>= 1) {  // at most one radix-2 tail stage
        const int h = m >> 1;
        const uint32_t t = (uint32_t)(n / m);
        const int tlog = 31 - __clz(t);
        for (int lb = tid; lb < nb2; lb += kThreads) {
            uint32_t jloc = (uint32_t)lb >> tlog;
            uint32_t pos = (uint32_t)lb & (t - 1);
            uint32_t base = jloc * 2 * t + pos;
            uint32_t jglob = (uint32_t)blk * ((uint32_t)nb2 >> tlog) + jloc;
            uint64_t U = (uint64_t)smem[base];
            uint64_t V = (uint64_t)smem[base + t];
            smem[base] = (int64_t)addmod_u64(U, V, q);
            smem[base + t] = (int64_t)mulmod_shoup(
                submod_u64(U, V, q), (uint64_t)wl[h + jglob],
                (uint64_t)wshl[h + jglob], q);
        }
        __syncthreads();
    }
    if (scale_here) {
        for (int i = tid; i < nblk; i += kThreads)
            smem[i] = (int64_t)mulmod_shoup((uint64_t)smem[i],
                                            (uint64_t)ninv[limb],
                                            (uint64_t)ninvsh[limb], q);
        __syncthreads();
    }
    for (int i = tid; i < nblk; i += kThreads) xr[i] = smem[i];
}

__global__ void intt_global_stage_ml_kernel(int64_t* __restrict__ x,
                                            const int64_t* __restrict__ winv,
                                            const int64_t* __restrict__ winvsh,
                                            const int64_t* __restrict__ qs,
                                            const int64_t* __restrict__ ninv,
                                            const int64_t* __restrict__ ninvsh,
                                            int L, int n, int m,
                                            int scale_here) {
    const int64_t nhalf = n >> 1;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = winv + (int64_t)limb * n;
    const int64_t* wshl = winvsh + (int64_t)limb * n;
    int64_t base_off = row * (int64_t)n;
    const int h = m >> 1;
    const uint32_t t = (uint32_t)(n / m);
    const int tlog = 31 - __clz(t);  // t is a power of two
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < nhalf;
         k += (int64_t)gridDim.x * blockDim.x) {
        uint32_t j = (uint32_t)(k >> tlog);
        uint32_t pos = (uint32_t)k & (t - 1);
        int64_t i0 = base_off + (int64_t)j * 2 * t + pos;
        uint64_t U = (uint64_t)x[i0];
        uint64_t V = (uint64_t)x[i0 + t];
        uint64_t a = addmod_u64(U, V, q);
        uint64_t b = mulmod_shoup(submod_u64(U, V, q), (uint64_t)wl[h + j],
                                  (uint64_t)wshl[h + j], q);
        if (scale_here) {
            a = mulmod_shoup(a, (uint64_t)ninv[limb], (uint64_t)ninvsh[limb], q);
            b = mulmod_shoup(b, (uint64_t)ninv[limb], (uint64_t)ninvsh[limb], q);
        }
        x[i0] = (int64_t)a;
        x[i0 + t] = (int64_t)b;
    }
}

// Radix-4 fused cross-block stages for n = 4*nblk (n = 2^15): ONE global
// pass instead of two. Forward: CT stages m=1 (w[1]) then m=2 (w[2], w[3]).
__global__ void ntt_global_radix4_ml_kernel(int64_t* __restrict__ x,
                                            const int64_t* __restrict__ w,
                                            const int64_t* __restrict__ wsh,
                                            const int64_t* __restrict__ qs,
                                            int L, int n) {
    const int64_t quarter = n >> 2;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = w + (int64_t)limb * n;
    const int64_t* wshl = wsh + (int64_t)limb * n;
    int64_t* xr = x + row * (int64_t)n;
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         k < quarter; k += (int64_t)gridDim.x * blockDim.x) {
        uint64_t x0 = (uint64_t)xr[k];
        uint64_t x1 = (uint64_t)xr[k + quarter];
        uint64_t x2 = (uint64_t)xr[k + 2 * quarter];
        uint64_t x3 = (uint64_t)xr[k + 3 * quarter];
        // stage m=1 (t = n/2, twiddle w[1]): pairs (x0,x2), (x1,x3)
        {
            uint64_t W = (uint64_t)wl[1], Wsh = (uint64_t)wshl[1];
            uint64_t v = mulmod_shoup(x2, W, Wsh, q);
            uint64_t u = x0;
            x0 = addmod_u64(u, v, q);
            x2 = submod_u64(u, v, q);
            v = mulmod_shoup(x3, W, Wsh, q);
            u = x1;
            x1 = addmod_u64(u, v, q);
            x3 = submod_u64(u, v, q);
        }
        // stage m=2 (t = n/4): pair (x0,x1) w[2]; pair (x2,x3) w[3]
        {
            uint64_t W = (uint64_t)wl[2], Wsh = (uint64_t)wshl[2];
            uint64_t v = mulmod_shoup(x1, W, Wsh, q);
            uint64_t u = x0;
            x0 = addmod_u64(u, v, q);
            x1 = submod_u64(u, v, q);
            W = (uint64_t)wl[3];
            Wsh = (uint64_t)wshl[3];
            v = mulmod_shoup(x3, W, Wsh, q);
            u = x2;
            x2 = addmod_u64(u, v, q);
            x3 = submod_u64(u, v, q);
        }
        xr[k] = (int64_t)x0;
        xr[k + quarter] = (int64_t)x1;
        xr[k + 2 * quarter] = (int64_t)x2;
        xr[k + 3 * quarter] = (int64_t)x3;
    }
}

// Inverse: GS stages m=4 (winv[2], winv[3]) then m=2 (winv[1]) + 1/n scale.
__global__ void intt_global_radix4_ml_kernel(int64_t* __restrict__ x,
                                             const int64_t* __restrict__ winv,
                                             const int64_t* __restrict__ winvsh,
                                             const int64_t* __restrict__ qs,
                                             const int64_t* __restrict__ ninv,
                                             const int64_t* __restrict__ ninvsh,
                                             int L, int n) {
    const int64_t quarter = n >> 2;
    const int64_t row = blockIdx.y;
    const int limb = (int)(row % L);
    const uint64_t q = (uint64_t)qs[limb];
    const int64_t* wl = winv + (int64_t)limb * n;
    const int64_t* wshl = winvsh + (int64_t)limb * n;
    const uint64_t nv = (uint64_t)ninv[limb], nvs = (uint64_t)ninvsh[limb];
    int64_t* xr = x + row * (int64_t)n;
    for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         k < quarter; k += (int64_t)gridDim.x * blockDim.x) {
        uint64_t x0 = (uint64_t)xr[k];
        uint64_t x1 = (uint64_t)xr[k + quarter];
        uint64_t x2 = (uint64_t)xr[k + 2 * quarter];
        uint64_t x3 = (uint64_t)xr[k + 3 * quarter];
        // stage m=4 (t = n/4): pairs (x0,x1) winv[2]; (x2,x3) winv[3]
        {
            uint64_t W = (uint64_t)wl[2], Wsh = (uint64_t)wshl[2];
            uint64_t u = x0, v = x1;
            x0 = addmod_u64(u, v, q);
            x1 = mulmod_shoup(submod_u64(u, v, q), W, Wsh, q);
            W = (uint64_t)wl[3];
            Wsh = (uint64_t)wshl[3];
            u = x2;
            v = x3;
            x2 = addmod_u64(u, v, q);
            x3 = mulmod_shoup(submod_u64(u, v, q), W, Wsh, q);
        }
        // stage m=2 (t = n/2): pairs (x0,x2) winv[1]; (x1,x3) winv[1]
        {
            uint64_t W = (uint64_t)wl[1], Wsh = (uint64_t)wshl[1];
            uint64_t u = x0, v = x2;
            x0 = addmod_u64(u, v, q);
            x2 = mulmod_shoup(submod_u64(u, v, q), W, Wsh, q);
            u = x1;
            v = x3;
            x1 = addmod_u64(u, v, q);
            x3 = mulmod_shoup(submod_u64(u, v, q), W, Wsh, q);
        }
        xr[k] = (int64_t)mulmod_shoup(x0, nv, nvs, q);
        xr[k + quarter] = (int64_t)mulmod_shoup(x1, nv, nvs, q);
        xr[k + 2 * quarter] = (int64_t)mulmod_shoup(x2, nv, nvs, q);
        xr[k + 3 * quarter] = (int64_t)mulmod_shoup(x3, nv, nvs, q);
    }
}

// a [.., L, n] x b (numel divides a's, same limb layout) with per-limb
// primes; ratios [L][2] = floor(2^128/q) words.
__global__ void modmul_limbs_kernel(const int64_t* __restrict__ a,
                                    const int64_t* __restrict__ b,
                                    int64_t* __restrict__ out, int64_t total,
                                    int64_t b_numel,
                                    const int64_t* __restrict__ qs,
                                    const int64_t* __restrict__ ratios, int L,
                                    int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int limb = (int)((i / n) % L);
        out[i] = (int64_t)mulmod_barrett(
            (uint64_t)a[i], (uint64_t)b[i % b_numel], (uint64_t)qs[limb],
            (uint64_t)ratios[limb * 2], (uint64_t)ratios[limb * 2 + 1]);
    }
}

__global__ void modmul_scalar_limbs_kernel(const int64_t* __restrict__ a,
                                           int64_t* __restrict__ out,
                                           int64_t total,
                                           const int64_t* __restrict__ scalars,
                                           const int64_t* __restrict__ shoups,
                                           const int64_t* __restrict__ qs,
                                           int L, int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int limb = (int)((i / n) % L);
        out[i] = (int64_t)mulmod_shoup((uint64_t)a[i],
                                       (uint64_t)scalars[limb],
                                       (uint64_t)shoups[limb],
                                       (uint64_t)qs[limb]);
    }
}

// ---------------------------------------------------------------------------
// Pointwise modular ops
// ---------------------------------------------------------------------------

__global__ void modmul_kernel(const int64_t* __restrict__ a,
                              const int64_t* __restrict__ b,
                              int64_t* __restrict__ out, int64_t total,
                              int64_t b_numel, uint64_t q, uint64_t r0,
                              uint64_t r1) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        out[i] = (int64_t)mulmod_barrett((uint64_t)a[i],
                                         (uint64_t)b[i % b_numel], q, r0, r1);
    }
}

__global__ void modmul_scalar_kernel(const int64_t* __restrict__ a,
                                     int64_t* __restrict__ out, int64_t total,
                                     uint64_t s, uint64_t ssh, uint64_t q) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        out[i] = (int64_t)mulmod_shoup((uint64_t)a[i], s, ssh, q);
    }
}

__global__ void modadd_kernel(const int64_t* __restrict__ a,
                              const int64_t* __restrict__ b,
                              int64_t* __restrict__ out, int64_t total,
                              uint64_t q) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        out[i] = (int64_t)addmod_u64((uint64_t)a[i], (uint64_t)b[i], q);
    }
}

__global__ void modsub_kernel(const int64_t* __restrict__ a,
                              const int64_t* __restrict__ b,
                              int64_t* __restrict__ out, int64_t total,
                              uint64_t q) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        out[i] = (int64_t)submod_u64((uint64_t)a[i], (uint64_t)b[i], q);
    }
}

// Reduce lazily-summed int64 values (< 8q, q < 2^60 so no wrap) back to
// [0, q) — the step right after the RCCL all-reduce on raw coefficients.
// x is [..., L, n]; limb selects q via qs[]. In place.
__global__ void modreduce_kernel(int64_t* __restrict__ x,
                                 const int64_t* __restrict__ qs, int64_t total,
                                 int64_t limb_stride, int L) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const uint64_t q = (uint64_t)qs[(i / limb_stride) % L];
        uint64_t v = (uint64_t)x[i];
        // v < 8q < 2^63 (lazy sum of <= 8 residues): three conditional
        // subtracts replace the software u64 division
        const uint64_t q4 = q << 2, q2 = q << 1;
        if (v >= q4) v -= q4;
        if (v >= q2) v -= q2;
        if (v >= q) v -= q;
        x[i] = (int64_t)v;
    }
}

// ---------------------------------------------------------------------------
// Fused per-limb pointwise ops over [..., L, n] tensors (limb = (i/n) % L).
// These replace the torch.remainder / int64 add glue that at::native was
// otherwise running between the NTT launches (round-1 profiles showed ~4%
// of the config #5 round in generic elementwise remainder kernels).
// ---------------------------------------------------------------------------

__global__ void modadd_limbs_kernel(const int64_t* __restrict__ a,
                                    const int64_t* __restrict__ b,
                                    int64_t* __restrict__ out, int64_t total,
                                    int64_t b_numel,
                                    const int64_t* __restrict__ qs, int L,
                                    int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const uint64_t q = (uint64_t)qs[(i / n) % L];
        out[i] = (int64_t)addmod_u64((uint64_t)a[i], (uint64_t)b[i % b_numel], q);
    }
}

// c0 of encrypt: (b*u + e + pt) with all three residues already in [0, q).
__global__ void modadd3_limbs_kernel(const int64_t* __restrict__ a,
                                     const int64_t* __restrict__ b,
                                     const int64_t* __restrict__ c,
                                     int64_t* __restrict__ out, int64_t total,
                                     const int64_t* __restrict__ qs, int L,
                                     int64_t n) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const uint64_t q = (uint64_t)qs[(i / n) % L];
        uint64_t v = (uint64_t)a[i] + (uint64_t)b[i];  // < 2q < 2^61
        if (v >= q) v -= q;
        out[i] = (int64_t)addmod_u64(v, (uint64_t)c[i], q);
    }
}

// a may carry MORE limbs than the output (aL >= L): rescale subtracts over
// the first L of a's aL limbs without materializing the slice first.
__global__ void modsub_limbs_kernel(const int64_t* __restrict__ a,
                                    const int64_t* __restrict__ b,
                                    int64_t* __restrict__ out, int64_t total,
                                    int64_t b_numel,
                                    const int64_t* __restrict__ qs, int L,
                                    int aL, int64_t n) {
    const int64_t Ln = (int64_t)L * n;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const uint64_t q = (uint64_t)qs[(i / n) % L];
        const int64_t ai = (aL == L) ? i
                                     : (i / Ln) * ((int64_t)aL * n) + (i % Ln);
        out[i] = (int64_t)submod_u64((uint64_t)a[ai], (uint64_t)b[i % b_numel], q);
    }
}

// Broadcast one coefficient row [.., n] into L limb residues [.., L, n]:
//   v = x (centered signed if qc == 0, else x in [0, qc) centered mod qc),
//   out[.., l, j] = v mod q_l  (Barrett; |v| can exceed q_l).
// One launch replaces torch.where + the [.., 1, n] -> [.., L, n] broadcast
// remainder in _to_rns_ntt / rescale / key-switch mod-down.
__global__ void bcast_center_mod_kernel(const int64_t* __restrict__ x,
                                        int64_t* __restrict__ out,
                                        int64_t total_out, int64_t n, int L,
                                        uint64_t qc,
                                        const int64_t* __restrict__ qs,
                                        const int64_t* __restrict__ ratios) {
    const int64_t Ln = (int64_t)L * n;
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         o < total_out; o += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = o % n;
        const int l = (int)((o / n) % L);
        const int64_t row = o / Ln;
        int64_t v = x[row * n + j];
        if (qc) {
            if ((uint64_t)v > (qc >> 1)) v -= (int64_t)qc;
        }
        const uint64_t q = (uint64_t)qs[l];
        const uint64_t r0 = (uint64_t)ratios[2 * l];
        const uint64_t r1 = (uint64_t)ratios[2 * l + 1];
        const uint64_t m = v < 0 ? (uint64_t)(-v) : (uint64_t)v;
        const uint64_t red = barrett_red128(m, 0, q, r0, r1);
        out[o] = (int64_t)((v < 0 && red) ? q - red : red);
    }
}

// ---------------------------------------------------------------------------
// Fused ct x ct + hybrid key-switch kernels (the Python
// per-digit/per-limb loop was O(L*(L+1)) small launches).
// ---------------------------------------------------------------------------

// Tensor product of two ciphertexts: d0 = a0*b0, d1 = a0*b1 + a1*b0,
// d2 = a1*b1, per limb. a is [R, 2, L, n]; b is [Rb, 2, L, n] with Rb == R
// or Rb == 1 (a batched ciphertext times one shared ct, e.g. the encrypted
// 1/n denominator). One launch replaces 3L modmuls + L remainders.
__global__ void ct_mul_kernel(const int64_t* __restrict__ a,
                              const int64_t* __restrict__ b,
                              int64_t* __restrict__ d0,
                              int64_t* __restrict__ d1,
                              int64_t* __restrict__ d2, int64_t total,
                              int64_t b_total, int L, int64_t n,
                              const int64_t* __restrict__ qs,
                              const int64_t* __restrict__ ratios) {
    const int64_t Ln = (int64_t)L * n;
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         o < total; o += (int64_t)gridDim.x * blockDim.x) {
        const int l = (int)((o / n) % L);
        const int64_t row = o / Ln;
        const int64_t inner = o % Ln;
        const uint64_t q = (uint64_t)qs[l];
        const uint64_t r0 = (uint64_t)ratios[2 * l];
        const uint64_t r1 = (uint64_t)ratios[2 * l + 1];
        const int64_t ia = row * 2 * Ln + inner;
        const int64_t ib = (row * 2 * Ln + inner) % b_total;
        const uint64_t a0 = (uint64_t)a[ia], a1 = (uint64_t)a[ia + Ln];
        const uint64_t b0 = (uint64_t)b[ib], b1 = (uint64_t)b[ib + Ln];
        d0[o] = (int64_t)mulmod_barrett(a0, b0, q, r0, r1);
        d1[o] = (int64_t)addmod_u64(mulmod_barrett(a0, b1, q, r0, r1),
                                    mulmod_barrett(a1, b0, q, r0, r1), q);
        d2[o] = (int64_t)mulmod_barrett(a1, b1, q, r0, r1);
    }
}

// Key-switch inner product: acc_c[row, i, j] = sum_d dig[row, d, i, j] *
// rlk[d, c, i, j] mod q_i, for c in {0, 1}. dig is the NTT-form digit
// decomposition [R, D, Lp, n]; rlk is [D, 2, Lp, n]. Reads dig once for
// both accumulators; one launch replaces 2*D*Lp modmul+add rounds.
__global__ void ks_inner_kernel(const int64_t* __restrict__ dig,
                                const int64_t* __restrict__ rlk,
                                int64_t* __restrict__ acc0,
                                int64_t* __restrict__ acc1, int64_t total,
                                int D, int Lp, int64_t n,
                                const int64_t* __restrict__ qs,
                                const int64_t* __restrict__ ratios) {
    const int64_t Lpn = (int64_t)Lp * n;
    for (int64_t o = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         o < total; o += (int64_t)gridDim.x * blockDim.x) {
        const int64_t j = o % n;
        const int i = (int)((o / n) % Lp);
        const int64_t row = o / Lpn;
        const uint64_t q = (uint64_t)qs[i];
        const uint64_t r0 = (uint64_t)ratios[2 * i];
        const uint64_t r1 = (uint64_t)ratios[2 * i + 1];
        uint64_t a0 = 0, a1 = 0;
        const int64_t dig_base = row * D * Lpn + (int64_t)i * n + j;
        const int64_t rlk_base = (int64_t)i * n + j;
        for (int d = 0; d < D; ++d) {
            const uint64_t x = (uint64_t)dig[dig_base + (int64_t)d * Lpn];
            const int64_t rb = rlk_base + (int64_t)d * 2 * Lpn;
            a0 = addmod_u64(a0, mulmod_barrett(x, (uint64_t)rlk[rb], q, r0, r1), q);
            a1 = addmod_u64(a1, mulmod_barrett(x, (uint64_t)rlk[rb + Lpn], q, r0, r1), q);
        }
        acc0[o] = (int64_t)a0;
        acc1[o] = (int64_t)a1;
    }
}

// Centered binomial eta=21 from ONE uniform 64-bit draw per coefficient:
// e = popcount(bits[0:21]) - popcount(bits[21:42]) — replaces 42 separate
// int8 Bernoulli draws (the sampling half of encrypt's cost at ResNet scale).
__global__ void cbd21_kernel(const int64_t* __restrict__ bits,
                             int64_t* __restrict__ out, int64_t total) {
    constexpr uint64_t M21 = (1ull << 21) - 1;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        uint64_t x = (uint64_t)bits[i];
        // __popcll returns unsigned: cast BEFORE subtracting or negative
        // differences wrap to ~2^32
        out[i] = (int64_t)((int)__popcll(x & M21) - (int)__popcll((x >> 21) & M21));
    }
}

int64_t shoup_of(uint64_t w, uint64_t q) {
    unsigned __int128 t = ((unsigned __int128)w) << 64;
    return (int64_t)(uint64_t)(t / q);
}

inline dim3 rows_grid(int blocks_per_row, int64_t rows) {
    TORCH_CHECK(rows <= 65535, "batch too large for gridDim.y");
    return dim3((unsigned)blocks_per_row, (unsigned)rows, 1);
}

}  // namespace

// ---------------------------------------------------------------------------
// Host entry points (registered in bindings.cpp)
// ---------------------------------------------------------------------------

// In-place forward NTT over x [rows, n] for a single limb prime q.
void ntt_batch(torch::Tensor x, torch::Tensor w, torch::Tensor wsh, int64_t q) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous() && x.dtype() == torch::kInt64);
    const int n = (int)x.size(-1);
    const int64_t rows = x.numel() / n;
    const int nblk = n < kNblkMax ? n : kNblkMax;
    auto stream = at::cuda::getCurrentCUDAStream();
    // global stages: m = 1 .. n/nblk/2... while span > nblk (t >= nblk/... )
    for (int m = 1; m < n / nblk; m <<= 1) {
        int blocks = (int)std::min<int64_t>((n / 2 + kThreads - 1) / kThreads, 1024);
        hipLaunchKernelGGL(ntt_global_stage_kernel, rows_grid(blocks, rows),
                           dim3(kThreads), 0, stream,
                           x.data_ptr<int64_t>(), w.data_ptr<int64_t>(),
                           wsh.data_ptr<int64_t>(), (uint64_t)q, n, m);
    }
    hipLaunchKernelGGL(ntt_lds_kernel, rows_grid(n / nblk, rows), dim3(kThreads),
                       nblk * sizeof(int64_t), stream, x.data_ptr<int64_t>(),
                       w.data_ptr<int64_t>(), wsh.data_ptr<int64_t>(),
                       (uint64_t)q, n, nblk);
}

// In-place inverse NTT over x [rows, n].
void intt_batch(torch::Tensor x, torch::Tensor winv, torch::Tensor winvsh,
                int64_t q, int64_t ninv, int64_t ninvsh) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous() && x.dtype() == torch::kInt64);
    const int n = (int)x.size(-1);
    const int64_t rows = x.numel() / n;
    const int nblk = n < kNblkMax ? n : kNblkMax;
    auto stream = at::cuda::getCurrentCUDAStream();
    const bool has_global = nblk < n;
    hipLaunchKernelGGL(intt_lds_kernel, rows_grid(n / nblk, rows), dim3(kThreads),
                       nblk * sizeof(int64_t), stream, x.data_ptr<int64_t>(),
                       winv.data_ptr<int64_t>(), winvsh.data_ptr<int64_t>(),
                       (uint64_t)q, n, nblk, (uint64_t)ninv, (uint64_t)ninvsh,
                       has_global ? 0 : 1);
    // remaining stages: m = n/nblk (after LDS ran m = n .. 2n/nblk) down to 2
    for (int m = n / nblk; m >= 2; m >>= 1) {
        int blocks = (int)std::min<int64_t>((n / 2 + kThreads - 1) / kThreads, 1024);
        hipLaunchKernelGGL(intt_global_stage_kernel, rows_grid(blocks, rows),
                           dim3(kThreads), 0, stream, x.data_ptr<int64_t>(),
                           winv.data_ptr<int64_t>(), winvsh.data_ptr<int64_t>(),
                           (uint64_t)q, n, m, (uint64_t)ninv, (uint64_t)ninvsh,
                           m == 2 ? 1 : 0);
    }
}

torch::Tensor modmul(torch::Tensor a, torch::Tensor b, int64_t q) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
    TORCH_CHECK(a.numel() % b.numel() == 0, "b must tile a");
    auto out = torch::empty_like(a);
    uint64_t r0, r1;
    barrett_ratio((uint64_t)q, &r0, &r1);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modmul_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       b.numel(), (uint64_t)q, r0, r1);
    return out;
}

torch::Tensor modmul_scalar(torch::Tensor a, int64_t s, int64_t q) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous());
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modmul_scalar_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       out.data_ptr<int64_t>(), total, (uint64_t)s,
                       (uint64_t)shoup_of((uint64_t)s, (uint64_t)q), (uint64_t)q);
    return out;
}

torch::Tensor modadd(torch::Tensor a, torch::Tensor b, int64_t q) {
    CHECK_CUDA_OK(a);
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modadd_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       (uint64_t)q);
    return out;
}

torch::Tensor modsub(torch::Tensor a, torch::Tensor b, int64_t q) {
    CHECK_CUDA_OK(a);
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modsub_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       (uint64_t)q);
    return out;
}

// In-place: reduce x [..., L, n] mod per-limb primes qs [L].
void modreduce_(torch::Tensor x, torch::Tensor qs) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous() && qs.is_contiguous());
    const int L = (int)qs.numel();
    const int64_t n = x.size(-1);
    TORCH_CHECK(x.size(-2) == L, "limb dim mismatch");
    int64_t total = x.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modreduce_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), x.data_ptr<int64_t>(),
                       qs.data_ptr<int64_t>(), total, n, L);
}

// In-place fused forward NTT over x [R, L, n] (limb = row % L).
void ntt_limbs(torch::Tensor x, torch::Tensor w, torch::Tensor wsh,
               torch::Tensor qs, int64_t L) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous());
    const int n = (int)x.size(-1);
    const int64_t rows = x.numel() / n;
    const int nblk = n < kNblkMax ? n : kNblkMax;
    auto stream = at::cuda::getCurrentCUDAStream();
    if (n / nblk == 4) {
        int blocks = (int)std::min<int64_t>((n / 4 + kThreads - 1) / kThreads, 1024);
        hipLaunchKernelGGL(ntt_global_radix4_ml_kernel, rows_grid(blocks, rows),
                           dim3(kThreads), 0, stream, x.data_ptr<int64_t>(),
                           w.data_ptr<int64_t>(), wsh.data_ptr<int64_t>(),
                           qs.data_ptr<int64_t>(), (int)L, n);
    } else {
        for (int m = 1; m < n / nblk; m <<= 1) {
            int blocks = (int)std::min<int64_t>((n / 2 + kThreads - 1) / kThreads, 1024);
            hipLaunchKernelGGL(ntt_global_stage_ml_kernel, rows_grid(blocks, rows),
                               dim3(kThreads), 0, stream, x.data_ptr<int64_t>(),
                               w.data_ptr<int64_t>(), wsh.data_ptr<int64_t>(),
                               qs.data_ptr<int64_t>(), (int)L, n, m);
        }
    }
    hipLaunchKernelGGL(ntt_lds_ml_kernel, rows_grid(n / nblk, rows),
                       dim3(kThreads), nblk * sizeof(int64_t), stream,
                       x.data_ptr<int64_t>(), w.data_ptr<int64_t>(),
                       wsh.data_ptr<int64_t>(), qs.data_ptr<int64_t>(), (int)L,
                       n, nblk);
}

void intt_limbs(torch::Tensor x, torch::Tensor winv, torch::Tensor winvsh,
                torch::Tensor qs, torch::Tensor ninv, torch::Tensor ninvsh,
                int64_t L) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous());
    const int n = (int)x.size(-1);
    const int64_t rows = x.numel() / n;
    const int nblk = n < kNblkMax ? n : kNblkMax;
    auto stream = at::cuda::getCurrentCUDAStream();
    const bool has_global = nblk < n;
    hipLaunchKernelGGL(intt_lds_ml_kernel, rows_grid(n / nblk, rows),
                       dim3(kThreads), nblk * sizeof(int64_t), stream,
                       x.data_ptr<int64_t>(), winv.data_ptr<int64_t>(),
                       winvsh.data_ptr<int64_t>(), qs.data_ptr<int64_t>(),
                       ninv.data_ptr<int64_t>(), ninvsh.data_ptr<int64_t>(),
                       (int)L, n, nblk, has_global ? 0 : 1);
    if (n / nblk == 4) {
        int blocks = (int)std::min<int64_t>((n / 4 + kThreads - 1) / kThreads, 1024);
        hipLaunchKernelGGL(intt_global_radix4_ml_kernel, rows_grid(blocks, rows),
                           dim3(kThreads), 0, stream, x.data_ptr<int64_t>(),
                           winv.data_ptr<int64_t>(), winvsh.data_ptr<int64_t>(),
                           qs.data_ptr<int64_t>(), ninv.data_ptr<int64_t>(),
                           ninvsh.data_ptr<int64_t>(), (int)L, n);
    } else {
        for (int m = n / nblk; m >= 2; m >>= 1) {
            int blocks = (int)std::min<int64_t>((n / 2 + kThreads - 1) / kThreads, 1024);
            hipLaunchKernelGGL(intt_global_stage_ml_kernel, rows_grid(blocks, rows),
                               dim3(kThreads), 0, stream, x.data_ptr<int64_t>(),
                               winv.data_ptr<int64_t>(), winvsh.data_ptr<int64_t>(),
                               qs.data_ptr<int64_t>(), ninv.data_ptr<int64_t>(),
                               ninvsh.data_ptr<int64_t>(), (int)L, n, m,
                               m == 2 ? 1 : 0);
        }
    }
}

torch::Tensor modmul_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           torch::Tensor ratios, int64_t L, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
    TORCH_CHECK(a.numel() % b.numel() == 0);
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modmul_limbs_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       b.numel(), qs.data_ptr<int64_t>(),
                       ratios.data_ptr<int64_t>(), (int)L, n);
    return out;
}

torch::Tensor modmul_scalar_limbs(torch::Tensor a, torch::Tensor scalars,
                                  torch::Tensor shoups, torch::Tensor qs,
                                  int64_t L, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous());
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modmul_scalar_limbs_kernel, dim3(blocks), dim3(kThreads),
                       0, at::cuda::getCurrentCUDAStream(),
                       a.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       scalars.data_ptr<int64_t>(), shoups.data_ptr<int64_t>(),
                       qs.data_ptr<int64_t>(), (int)L, n);
    return out;
}

torch::Tensor cbd21(torch::Tensor bits) {
    CHECK_CUDA_OK(bits);
    auto out = torch::empty_like(bits);
    int64_t total = bits.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(cbd21_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(),
                       bits.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total);
    return out;
}

torch::Tensor modadd_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           int64_t L, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
    TORCH_CHECK(a.numel() % b.numel() == 0);
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modadd_limbs_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       b.numel(), qs.data_ptr<int64_t>(), (int)L, n);
    return out;
}

torch::Tensor modadd3_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor c,
                            torch::Tensor qs, int64_t L, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous() && c.is_contiguous());
    TORCH_CHECK(a.numel() == b.numel() && a.numel() == c.numel());
    auto out = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modadd3_limbs_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), c.data_ptr<int64_t>(),
                       out.data_ptr<int64_t>(), total, qs.data_ptr<int64_t>(),
                       (int)L, n);
    return out;
}

torch::Tensor modsub_limbs(torch::Tensor a, torch::Tensor b, torch::Tensor qs,
                           int64_t L, int64_t aL, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
    TORCH_CHECK(a.size(-2) == aL && a.size(-1) == n);
    auto sizes = a.sizes().vec();
    sizes[sizes.size() - 2] = L;  // output carries the first L of a's limbs
    auto out = torch::empty(sizes, a.options());
    int64_t total = out.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 2048);
    hipLaunchKernelGGL(modsub_limbs_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), out.data_ptr<int64_t>(), total,
                       b.numel(), qs.data_ptr<int64_t>(), (int)L, (int)aL, n);
    return out;
}

// a [R, 2, L, n] x b [Rb, 2, L, n] (Rb == R or 1) -> (d0, d1, d2) each [R, L, n].
std::vector<torch::Tensor> ct_mul(torch::Tensor a, torch::Tensor b,
                                  torch::Tensor qs, torch::Tensor ratios,
                                  int64_t L, int64_t n) {
    CHECK_CUDA_OK(a);
    TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
    auto sizes = a.sizes().vec();       // [.., 2, L, n]
    sizes.erase(sizes.end() - 3);       // drop the (c0, c1) dim
    auto d0 = torch::empty(sizes, a.options());
    auto d1 = torch::empty(sizes, a.options());
    auto d2 = torch::empty(sizes, a.options());
    int64_t total = d0.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 4096);
    hipLaunchKernelGGL(ct_mul_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), a.data_ptr<int64_t>(),
                       b.data_ptr<int64_t>(), d0.data_ptr<int64_t>(),
                       d1.data_ptr<int64_t>(), d2.data_ptr<int64_t>(), total,
                       b.numel(), (int)L, n, qs.data_ptr<int64_t>(),
                       ratios.data_ptr<int64_t>());
    return {d0, d1, d2};
}

// dig [R, D, Lp, n] x rlk [D, 2, Lp, n] -> (acc0, acc1) each [R, Lp, n].
std::vector<torch::Tensor> ks_inner(torch::Tensor dig, torch::Tensor rlk,
                                    torch::Tensor qs, torch::Tensor ratios,
                                    int64_t D, int64_t Lp, int64_t n) {
    CHECK_CUDA_OK(dig);
    TORCH_CHECK(dig.is_contiguous() && rlk.is_contiguous());
    auto sizes = dig.sizes().vec();     // [.., D, Lp, n]
    sizes.erase(sizes.end() - 3);       // drop the digit dim
    auto acc0 = torch::empty(sizes, dig.options());
    auto acc1 = torch::empty(sizes, dig.options());
    int64_t total = acc0.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 4096);
    hipLaunchKernelGGL(ks_inner_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(),
                       dig.data_ptr<int64_t>(), rlk.data_ptr<int64_t>(),
                       acc0.data_ptr<int64_t>(), acc1.data_ptr<int64_t>(),
                       total, (int)D, (int)Lp, n, qs.data_ptr<int64_t>(),
                       ratios.data_ptr<int64_t>());
    return {acc0, acc1};
}

// x [..., n] -> out [..., L, n]: per-limb Barrett reduction of the (optionally
// qc-centered) coefficient row. qc = 0 means x is already centered signed.
torch::Tensor bcast_center_mod(torch::Tensor x, int64_t qc, torch::Tensor qs,
                               torch::Tensor ratios, int64_t L) {
    CHECK_CUDA_OK(x);
    TORCH_CHECK(x.is_contiguous());
    const int64_t n = x.size(-1);
    auto sizes = x.sizes().vec();
    sizes.insert(sizes.end() - 1, L);
    auto out = torch::empty(sizes, x.options());
    int64_t total = out.numel();
    int blocks = (int)std::min<int64_t>((total + kThreads - 1) / kThreads, 4096);
    hipLaunchKernelGGL(bcast_center_mod_kernel, dim3(blocks), dim3(kThreads), 0,
                       at::cuda::getCurrentCUDAStream(), x.data_ptr<int64_t>(),
                       out.data_ptr<int64_t>(), total, n, (int)L, (uint64_t)qc,
                       qs.data_ptr<int64_t>(), ratios.data_ptr<int64_t>());
    return out;
}
