// CNN training kernels for gfx950 (CDNA4): MFMA implicit-GEMM conv
// fwd/dgrad/wgrad, MFMA linear fwd/dgrad/wgrad, maxpool 2x2, fused
// softmax-CE, fused Adam, bias grad, ReLU bwd.
//
// Replaces the TF/Keras native surface the reference leans on for
// model.fit/predict (reference model at
// FLPyfhelin.py:118-141, Adam at :140, loss at :141).
//
// Layout: NHWC activations (bf16), [K,R,S,C] weights (bf16 on device,
// fp32 master copy in Python), fp32 accumulation via
// v_mfma_f32_16x16x32_bf16 (one 16x16 output fragment per wave,
// 4 waves per workgroup covering 64 output rows).
//
// Kernel index (in file order):
//   FastDiv / fdiv                 magic-multiply division (all index decode)
//   conv_fwd_kernel / tile_mfma    gather-form implicit-GEMM conv fwd
//   glds16                         global_load_lds 16-B DMA helper
//   conv_fwd_glds_kernel           fwd with LDS-DMA 3-buffer counted-vmcnt
//                                  pipeline (raw s_barrier, 1 stage in flight)
//   conv_dgrad_kernel/_glds_       dgrad, K reordered to (r,s,ko) for
//                                  contiguous dy gathers
//   conv_wgrad_kernel/_glds_       wgrad, pixel-major tiles, split-K slabs
//   conv_wgrad_small_kernel        Kout<=64 & RSC<=16 first-layer wgrad:
//                                  waves split over pixels, LDS reduce
//   sum_slabs_f32_kernel           lane-parallel split-K slab reduction
//   xcd_remap                      consecutive conv M-tiles -> one XCD's L2
//   conv_fwd_glds64_kernel         BK=64 two-buffer fwd (measured slower at
//                                  these short-K shapes; HEFL_GLDS64 A/B)
//   conv_fwd_tile3_kernel          direct tiled 3x3 s1 conv: input tile
//                                  staged once, all nine taps from LDS
//                                  (~144 MFMAs per barrier); also drives
//                                  dgrad via rot180_transpose_w_kernel
//   conv_wgrad_glds_k32_kernel     Kout<=32 wgrad tile, 128-pixel K-steps
//   pad_channels_kernel            C%8!=0 stem -> glds MFMA path
//   maxpool_fwd_kernel (2x2) / maxpool_fwd_oct_kernel
//                                  pool fwd + argmax byte (octet: 16-B
//                                  loads + packed 8-B argmax store)
//   maxpool2x2_bwd_gather_kernel   gather-form pool bwd (no scatter/fill)
//   pool_relu_bias_bwd2_kernel     FUSED conv->relu->pool backward: one
//                                  thread per 2x2 block x 8 channels,
//                                  ReLU gate on the POOLED output
//   pool_relu_bias_bwd_scalar_kernel  K%8!=0 fallback (LeNet), lane-reduce
//   dense_head2_bwd_kernel         single-launch 2-layer head backward
//                                  (measured slower end-to-end; HEFL_HEAD2)
//   relu_bias_bwd_kernel           fused ReLU-mask + bias grad (fc layers)
//   linear_splitk_kernel           skinny-M wave GEMM, direct-from-global,
//                                  unrolled wave-uniform fast path
//   linear_epilogue_kernel         slab sum + bias + relu + bf16 store
//   gemm_kernel<AT,BT,...>         general MFMA GEMM (fc d/wgrad)
//   softmax_xent_fwd/bwd_kernel    fused softmax-CE with in-kernel
//                                  loss/accuracy accumulators
//   adam_prep(_epoch)_kernel       device-side Keras-decay schedule
//   fused_adam(_mt)_kernel         fused Adam (+bf16 shadow refresh);
//                                  _mt = one launch over all params
//   pack_mt/unpack_mt_kernel       FedAvg weight vector pack/unpack
//   bn_partial/bn_finalize/bn_apply/bn_bwd_partial/bn_bwd_finalize/bn_dx
//                                  two-stage BatchNorm, bf16x8 octet paths,
//                                  ReLU gate fused into backward;
//                                  bn_partial_fused_kernel = last-block-
//                                  finalize probe (HEFL_BN_FUSE, -15%)
//   synth_batch_kernel<AUG>        one-pass synthetic data generation,
//                                  AUG = in-kernel zoom/shear/flip via
//                                  inverse-affine template sampling
//   maxpool_gen_fwd/bwd_kernel     generic k/s/p pooling (ResNet stem)
//   avgpool_global_fwd/bwd_kernel  global average pool
//   add_relu_kernel                residual add + ReLU (bf16x8)
//   relu_bwd_kernel, bias_grad_kernel, cast_f32_bf16_kernel   elementwise

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>
#include <unordered_map>

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")

namespace {

using bf16_t = __bf16;
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned short u16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
    union { float f; unsigned int i; } v;
    v.f = f;
    unsigned int x = v.i;
    unsigned int lsb = (x >> 16) & 1;           // round-to-nearest-even
    x += 0x7fffu + lsb;
    if ((v.i & 0x7fffffffu) > 0x7f800000u) return 0x7fc0u;  // NaN
    return (unsigned short)(x >> 16);
}

// ---------------------------------------------------------------------------
// Shared MFMA tile machinery: BM=64 (4 waves stacked on M), BN=16, BK=32.
// LDS images As[64][32] and Bs[16][32] (B stored col-major-as-rows so each
// lane's 8 k-consecutive bf16 are one 16-B read).
// Fragment maps for v_mfma_f32_16x16x32_bf16:
//   A: row = lane&15, k = (lane>>4)*8 + j   (j = 0..7)
//   B: col = lane&15, k = (lane>>4)*8 + j
//   D: col = lane&15, row = (lane>>4)*4 + r (r = 0..3)
// ---------------------------------------------------------------------------

constexpr int BM = 64;
constexpr int BN = 16;
constexpr int BK = 32;
constexpr int TPB = 256;  // 4 waves

struct TileSmem {
    unsigned short As[BM][BK];
    unsigned short Bs[BN][BK];
};

__device__ __forceinline__ f32x4 tile_mfma_step(const TileSmem& sm, int wave,
                                                int lane, f32x4 acc) {
    const int half = lane >> 4;          // 0..3 (k-slice of 8)
    const int sub = lane & 15;
    bf16x8 a = *reinterpret_cast<const bf16x8*>(&sm.As[wave * 16 + sub][half * 8]);
    bf16x8 b = *reinterpret_cast<const bf16x8*>(&sm.Bs[sub][half * 8]);
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// Generic GEMM: C[M,N] = A[M,K] x B + bias, optional relu.
// AT: A stored [K,M] (read transposed); BT: B stored [N,K].
// OUT_BF16: write bf16, else f32.
// ---------------------------------------------------------------------------

template <bool AT, bool BT, bool OUT_BF16, bool ACCUM>
__global__ void __launch_bounds__(TPB)
gemm_kernel(const unsigned short* __restrict__ A,
            const unsigned short* __restrict__ B, void* __restrict__ C,
            const float* __restrict__ bias, int M, int N, int K, int relu) {
    __shared__ TileSmem sm;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;

    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int k0 = 0; k0 < K; k0 += BK) {
        // stage A: 64x32 = 2048 elems, 8 per thread
        for (int i = tid; i < BM * BK / 8; i += TPB) {
            int row = i / (BK / 8);
            int kc = (i % (BK / 8)) * 8;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                int m = m0 + row, k = k0 + kc + j;
                unsigned short v = 0;
                if (m < M && k < K)
                    v = AT ? A[(int64_t)k * M + m] : A[(int64_t)m * K + k];
                sm.As[row][kc + j] = v;
            }
        }
        // stage B: 16x32 = 512 elems
        for (int i = tid; i < BN * BK / 8; i += TPB) {
            int col = i / (BK / 8);
            int kc = (i % (BK / 8)) * 8;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                int nn = n0 + col, k = k0 + kc + j;
                unsigned short v = 0;
                if (nn < N && k < K)
                    v = BT ? B[(int64_t)nn * K + k] : B[(int64_t)k * N + nn];
                sm.Bs[col][kc + j] = v;
            }
        }
        __syncthreads();
        acc = tile_mfma_step(sm, wave, lane, acc);
        __syncthreads();
    }
    // epilogue
    const int col = n0 + (lane & 15);
    if (col >= N) return;
    float bv = bias ? bias[col] : 0.f;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int row = m0 + wave * 16 + (lane >> 4) * 4 + r;
        if (row >= M) continue;
        float v = acc[r] + bv;
        if (relu) v = v > 0.f ? v : 0.f;
        if (OUT_BF16) {
            reinterpret_cast<unsigned short*>(C)[(int64_t)row * N + col] = f2bf(v);
        } else if (ACCUM) {
            atomicAdd(reinterpret_cast<float*>(C) + (int64_t)row * N + col, v);
        } else {
            reinterpret_cast<float*>(C)[(int64_t)row * N + col] = v;
        }
    }
}

// ---------------------------------------------------------------------------
// Conv2d implicit-GEMM kernels, v2.
//
// Shared structure (fwd / dgrad / wgrad): 64xBN output tile per 256-thread
// workgroup (BN=64: 2x2 waves each computing 2x2 MFMA fragments; BN=16:
// 4x1 waves, 1 fragment), BK=32 K-steps staged through LDS. Staging uses
// 16-B vector loads whenever the innermost gathered dim is a multiple of 8
// (channels for fwd A / wgrad B, output channels for dgrad A / wgrad A —
// dgrad reorders its K-dim to (r,s,ko) to make the dy gather contiguous);
// per-row im2col coordinate decode is hoisted out of the K loop.
// ---------------------------------------------------------------------------

// Branch-free unsigned division by a runtime constant (magic multiply):
// exact for all n < 2^31, d < 2^16 (Hacker's Delight round-up magic with a
// 64-bit multiply). The im2col index decode in the conv stage loops was a
// chain of ~6 u32 divisions per k-step — more VALU cycles than the MFMAs.
struct FastDiv {
    unsigned long long m;
    int p;
};

__device__ __forceinline__ unsigned fdiv(unsigned n, FastDiv f) {
    return (unsigned)(((unsigned long long)n * f.m) >> f.p);
}

struct ConvShape {
    int N, H, W, C, Kout, R, S, OH, OW, stride, pad;
    FastDiv fC, fS, fOW, fOHOW, fKout;
};

template <int BM, int BN, int WM, int WN, int FM, int FN>
struct ConvTile {
    static constexpr int BK = 32;
    unsigned short As[BM][BK];
    unsigned short Bs[BN][BK];
};

template <int BM, int BN, int WM, int WN, int FM, int FN>
__device__ __forceinline__ void tile_mfma(
    const unsigned short (&As)[BM][32], const unsigned short (&Bs)[BN][32],
    f32x4 (&acc)[FM][FN], int wave, int lane) {
    const int wm = wave / WN, wn = wave % WN;
    const int half = lane >> 4, sub = lane & 15;
    bf16x8 a[FM], b[FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
        a[i] = *reinterpret_cast<const bf16x8*>(
            &As[wm * FM * 16 + i * 16 + sub][half * 8]);
#pragma unroll
    for (int j = 0; j < FN; ++j)
        b[j] = *reinterpret_cast<const bf16x8*>(
            &Bs[wn * FN * 16 + j * 16 + sub][half * 8]);
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[i], b[j],
                                                                acc[i][j], 0, 0, 0);
}

__device__ __forceinline__ void copy16(unsigned short* dst,
                                       const unsigned short* src) {
    *reinterpret_cast<uint4*>(dst) = *reinterpret_cast<const uint4*>(src);
}

__device__ __forceinline__ void zero16(unsigned short* dst) {
    *reinterpret_cast<uint4*>(dst) = uint4{0, 0, 0, 0};
}

// ---- forward: y[m=(n,oh,ow), ko] = sum_k A(m,k) * w[ko, k],
//      k = (r, s, c) with c fastest ----
template <int BM, int BN, int WM, int WN, int FM, int FN>
__global__ void __launch_bounds__(TPB)
conv_fwd_kernel(const unsigned short* __restrict__ x,
                const unsigned short* __restrict__ w,
                const float* __restrict__ bias, unsigned short* __restrict__ y,
                ConvShape s, int relu) {
    __shared__ ConvTile<BM, BN, WM, WN, FM, FN> sm;
    constexpr int RPT = BM * 4 / TPB;  // A-chunks per thread
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int M = s.N * s.OH * s.OW;
    const int KK = s.R * s.S * s.C;
    const bool fast = (s.C % 8 == 0);

    // hoisted per-thread A-row coordinates (RPT rows of 4 chunks each)
    int arows[RPT], am_[RPT], a_n[RPT], a_oh[RPT], a_ow[RPT];
    const int akc = (tid & 3) * 8;
#pragma unroll
    for (int t = 0; t < RPT; ++t) {
        arows[t] = (tid + t * TPB) >> 2;
        const int am = m0 + arows[t];
        am_[t] = am;
        a_n[t] = a_oh[t] = a_ow[t] = 0;
        if (am < M) {
            a_n[t] = am / (s.OH * s.OW);
            int rem = am % (s.OH * s.OW);
            a_oh[t] = rem / s.OW;
            a_ow[t] = rem % s.OW;
        }
    }

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int k0 = 0; k0 < KK; k0 += 32) {
#pragma unroll
        for (int t = 0; t < RPT; ++t) {  // ---- stage A ----
            const int k = k0 + akc;
            unsigned short* dst = &sm.As[arows[t]][akc];
            if (am_[t] < M && k < KK) {
                if (fast) {
                    int rs = (int)fdiv((unsigned)k, s.fC), c = k - rs * s.C;
                    int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                    int ih = a_oh[t] * s.stride + r - s.pad;
                    int iw = a_ow[t] * s.stride + ss - s.pad;
                    if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                        copy16(dst, x + ((((int64_t)a_n[t] * s.H + ih) * s.W + iw)
                                             * s.C + c));
                    else
                        zero16(dst);
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        int kk = k + j;
                        unsigned short v = 0;
                        if (kk < KK) {
                            int rs = (int)fdiv((unsigned)kk, s.fC);
                            int c = kk - rs * s.C;
                            int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                            int ih = a_oh[t] * s.stride + r - s.pad;
                            int iw = a_ow[t] * s.stride + ss - s.pad;
                            if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                                v = x[((((int64_t)a_n[t] * s.H + ih) * s.W + iw)
                                           * s.C + c)];
                        }
                        dst[j] = v;
                    }
                }
            } else {
                zero16(dst);
            }
        }
        // ---- stage B: w[ko][k] contiguous in k ----
        for (int i = tid; i < BN * 4; i += TPB) {
            const int col = i >> 2;
            const int kc = (i & 3) * 8;
            const int ko = n0 + col, k = k0 + kc;
            unsigned short* dst = &sm.Bs[col][kc];
            if (ko < s.Kout && k < KK) {
                if ((KK & 7) == 0) {
                    copy16(dst, w + (int64_t)ko * KK + k);
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        dst[j] = (k + j < KK) ? w[(int64_t)ko * KK + k + j] : 0;
                }
            } else {
                zero16(dst);
            }
        }
        __syncthreads();
        tile_mfma<BM, BN, WM, WN, FM, FN>(sm.As, sm.Bs, acc, wave, lane);
        __syncthreads();
    }
    // ---- epilogue ----
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= s.Kout) continue;
        const float bv = bias ? bias[col] : 0.f;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
                float v = acc[i][j][r] + bv;
                if (relu) v = v > 0.f ? v : 0.f;
                y[(int64_t)row * s.Kout + col] = f2bf(v);
            }
    }
}

// ---- glds-pipelined forward: 2 LDS buffers, global_load_lds staging
// (dest is lane-linear: thread t's chunk lands at byte t*16 of its wave's
// 1 KiB slice), stage of tile k+1 issued BEFORE the MFMAs of tile k so the
// DMA flight hides under compute; __syncthreads drains it at the barrier.
// Out-of-range lanes (M tail, padding border, K tail) read a 16-B zero
// scratch buffer instead of branching around the DMA. Requires C % 8 == 0
// and KK % 8 == 0 (wrapper falls back to the gather kernel otherwise). ----

typedef uint32_t __attribute__((address_space(3))) lds_u32_t;
typedef const uint32_t __attribute__((address_space(1))) glb_u32_t;

// XCD-aware tile remap: hardware dispatch round-robins workgroups over the
// 8 XCDs (each with its own L2); adjacent conv M-tiles share R-1 input
// halo rows, so map CONSECUTIVE tile ranges onto one XCD to keep the halo
// reuse in a single L2. Bijective for any nwg (guide formula).
__device__ __forceinline__ int xcd_remap(int wg, int nwg) {
    if (nwg < 16) return wg;
    const int xcd = wg & 7;
    const int q = nwg >> 3, r = nwg & 7;
    const int base = xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
    return base + (wg >> 3);
}

__device__ __forceinline__ void glds16(const unsigned short* src, void* lds_base) {
    __builtin_amdgcn_global_load_lds((glb_u32_t*)src, (lds_u32_t*)lds_base, 16,
                                     0, 0);
}

template <int BM, int BN, int WM, int WN, int FM, int FN, int BK = 32>
__global__ void __launch_bounds__(TPB)
conv_fwd_glds_kernel(const unsigned short* __restrict__ x,
                     const unsigned short* __restrict__ w,
                     const float* __restrict__ bias,
                     unsigned short* __restrict__ y,
                     float* __restrict__ y32,  // split-K accumulator (or null)
                     const unsigned short* __restrict__ zbuf, ConvShape s,
                     int relu, int k_chunks) {
    // ONE shared object (two makes hipcc emit a vmcnt(0) drain before each
    // k-step's first ds_read, defeating the glds pipeline — guide trap 4a)
    __shared__ unsigned short smem[3 * (BM + BN) * BK];
    auto As = [&](int buf) -> unsigned short (*)[BK] {
        return reinterpret_cast<unsigned short(*)[BK]>(smem + buf * (BM + BN) * BK);
    };
    auto Bs = [&](int buf) -> unsigned short (*)[BK] {
        return reinterpret_cast<unsigned short(*)[BK]>(smem + buf * (BM + BN) * BK
                                                       + BM * BK);
    };
    constexpr int RPT = BM * (BK / 8) / TPB;
    constexpr int BPT = BN * (BK / 8) / TPB;  // B chunks per thread
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = xcd_remap(blockIdx.x, gridDim.x) * BM;
    const int n0 = blockIdx.y * BN;
    const int M = s.N * s.OH * s.OW;
    const int KK = s.R * s.S * s.C;

    constexpr int CPR = BK / 8;  // chunks per row
    int a_n[RPT], a_oh[RPT], a_ow[RPT];
    int akcs[RPT], arow3[RPT];
    bool a_ok[RPT];
#pragma unroll
    for (int t = 0; t < RPT; ++t) {
        const int arow = (tid + t * TPB) / CPR;
        // XOR-swizzled chunk assignment (rule 21: permute the SOURCE, read
        // with the same XOR): kills the 4-way ds_read_b128 bank conflict of
        // rows-at-same-column fragment reads. Only defined for CPR == 4.
        // key = (row>>2)&3: rows 4 apart share a 256-B LDS bank row quarter,
        // so the slot choice must differ among them (row&3 would alias)
        arow3[t] = (arow >> 2) & 3;
        akcs[t] = ((tid % CPR) ^ (CPR == 4 ? arow3[t] : 0)) * 8;
        const int am = m0 + arow;
        a_ok[t] = am < M;
        a_n[t] = a_oh[t] = a_ow[t] = 0;
        if (a_ok[t]) {
            a_n[t] = am / (s.OH * s.OW);
            int rem = am % (s.OH * s.OW);
            a_oh[t] = rem / s.OW;
            a_ow[t] = rem % s.OW;
        }
    }

    auto stage = [&](int buf, int k0) {
        // A: RPT chunks, each one 16-B DMA; lane-linear within the wave
#pragma unroll
        for (int t = 0; t < RPT; ++t) {
            const int k = k0 + akcs[t];
            const unsigned short* src = zbuf;
            if (a_ok[t] && k < KK) {
                int rs = (int)fdiv((unsigned)k, s.fC), c = k - rs * s.C;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                int ih = a_oh[t] * s.stride + r - s.pad;
                int iw = a_ow[t] * s.stride + ss - s.pad;
                if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                    src = x + ((((int64_t)a_n[t] * s.H + ih) * s.W + iw) * s.C
                               + c);
            }
            char* base = (char*)&As(buf)[0][0] + (wave + t * 4) * 1024;
            glds16(src, base);
        }
#pragma unroll
        for (int t = 0; t < BPT; ++t) {
            const int i = tid + t * TPB;
            const int korow = i / CPR;
            const int ko = n0 + korow;
            const int kc = ((i % CPR) ^ (CPR == 4 ? ((korow >> 2) & 3) : 0)) * 8;
            const int k = k0 + kc;
            const unsigned short* src = zbuf;
            if (ko < s.Kout && k < KK) src = w + (int64_t)ko * KK + k;
            char* base = (char*)&Bs(buf)[0][0] + (wave + t * 4) * 1024;
            glds16(src, base);
        }
    };

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int ksteps = (KK + BK - 1) / BK;
    const int kc_len = (ksteps + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * kc_len * BK;
    const int kend = min(kbeg + kc_len * BK, KK);
    // 3-buffer pipeline, one stage always in flight: counted s_waitcnt
    // vmcnt(2) (= 2 glds instructions per stage per wave) + raw s_barrier.
    // __syncthreads here would emit vmcnt(0) and drain the in-flight stage
    // (guide: 3-buf span +83% vs sync, 2-buf +40%).
    stage(0, kbeg);
    stage(1, kbeg + BK);
    int buf = 0;
    for (int k0 = kbeg; k0 < kend; k0 += BK) {
        if (k0 + BK < kend)
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_barrier" ::: "memory");
        if (k0 + 2 * BK < kend) stage(buf == 0 ? 2 : buf - 1, k0 + 2 * BK);
        {   // MFMA over the BK-deep tile, 32 k per instruction
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
            const int swz = (BK == 32) ? ((sub >> 2) & 3) : 0;  // stage XOR key
#pragma unroll
            for (int kk = 0; kk < BK; kk += 32) {
                bf16x8 a[FM], b[FN];
#pragma unroll
                for (int i = 0; i < FM; ++i)
                    a[i] = *reinterpret_cast<const bf16x8*>(
                        &As(buf)[wm * FM * 16 + i * 16 + sub]
                                [kk + ((half ^ swz) * 8)]);
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    b[j] = *reinterpret_cast<const bf16x8*>(
                        &Bs(buf)[wn * FN * 16 + j * 16 + sub]
                                [kk + ((half ^ swz) * 8)]);
#pragma unroll
                for (int i = 0; i < FM; ++i)
#pragma unroll
                    for (int j = 0; j < FN; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[i], b[j], acc[i][j], 0, 0, 0);
            }
        }
        buf = buf == 2 ? 0 : buf + 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // degenerate-range drain
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= s.Kout) continue;
        const float bv = bias ? bias[col] : 0.f;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
                if (k_chunks > 1) {
                    atomicAdd(y32 + (int64_t)row * s.Kout + col, acc[i][j][r]);
                } else {
                    float v = acc[i][j][r] + bv;
                    if (relu) v = v > 0.f ? v : 0.f;
                    y[(int64_t)row * s.Kout + col] = f2bf(v);
                }
            }
    }
}

// ---- BK=64 two-buffer glds forward (guide GEMM-ladder G15 structure:
// stage -> vmcnt(0) + __syncthreads -> issue next stage -> MFMA, so the
// next tile's DMA flies under the current tile's 16 MFMAs/wave; st_16x32
// LDS swizzle kills the 8-way ds_read_b128 conflict of linear 128-B rows).
// Tiles are sized to the conv shapes: BN=32 for 32-filter layers (the
// 64-wide tile wasted half its MFMA work there), BN=64 with FM=4 depth
// otherwise. Requires C % 8 == 0 and KK % 8 == 0 like the BK=32 kernel. ----

__device__ __forceinline__ int swz128(int byte_off) {
    // st_16x32 swizzle for 128-B LDS rows: XOR byte bit 5 with bit 9
    // (involutive: the XOR never changes bit 9)
    return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

template <int BM, int BN, int WM, int WN, int FM, int FN, bool PAD0>
__global__ void __launch_bounds__(TPB)
conv_fwd_glds64_kernel(const unsigned short* __restrict__ x,
                       const unsigned short* __restrict__ w,
                       const float* __restrict__ bias,
                       unsigned short* __restrict__ y,
                       float* __restrict__ y32,  // split-K accumulator or null
                       const unsigned short* __restrict__ zbuf, ConvShape s,
                       int relu, int k_chunks) {
    constexpr int BK = 64;
    constexpr int NW = TPB / 64;
    static_assert(WM * WN == NW, "wave grid must cover the block");
    static_assert(BM == WM * FM * 16 && BN == WN * FN * 16, "tile mismatch");
    constexpr int ABYTES = BM * BK * 2;
    constexpr int RPT = (BM * BK / 8) / TPB;  // A 16-B chunks per thread
    constexpr int BPT = (BN * BK / 8) / TPB;
    static_assert(RPT * TPB * 8 == BM * BK && BPT * TPB * 8 == BN * BK);
    __shared__ unsigned short smem[2 * (BM + BN) * BK];
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int M = s.N * s.OH * s.OW;
    const int KK = s.R * s.S * s.C;

    // Per-thread A-chunk decode: the DMA dest is lane-linear (physical
    // byte p), so the SOURCE chunk is the one whose swizzled logical
    // offset is p (rule 21: permute the source, read with the same XOR).
    int a_k8[RPT], a_oh[RPT], a_ow[RPT], a_n[RPT];
    int64_t a_base[RPT];
    bool a_ok[RPT];
#pragma unroll
    for (int t = 0; t < RPT; ++t) {
        const int p = (wave + t * NW) * 1024 + lane * 16;
        const int l = swz128(p);
        const int row = l >> 7;
        a_k8[t] = (l & 127) >> 1;  // k offset within the tile (multiple of 8)
        const int am = m0 + row;
        a_ok[t] = am < M;
        a_base[t] = 0;
        a_n[t] = a_oh[t] = a_ow[t] = 0;
        if (a_ok[t]) {
            a_n[t] = am / (s.OH * s.OW);
            const int rem = am - a_n[t] * (s.OH * s.OW);
            a_oh[t] = rem / s.OW;
            a_ow[t] = rem - a_oh[t] * s.OW;
            // pad == 0 (valid conv): ih = oh*stride + r <= H-1 always, so
            // the whole gather is branchless off one precomputed base
            a_base[t] = (((int64_t)a_n[t] * s.H + a_oh[t] * s.stride) * s.W
                         + a_ow[t] * s.stride) * s.C;
        }
    }

    auto stage = [&](int bufsel, int k0) {
        char* abase = (char*)smem + bufsel * (ABYTES + BN * BK * 2);
#pragma unroll
        for (int t = 0; t < RPT; ++t) {
            const int k = k0 + a_k8[t];
            const unsigned short* src = zbuf;
            if (a_ok[t] && k < KK) {
                const int rs = (int)fdiv((unsigned)k, s.fC);
                const int c = k - rs * s.C;
                const int r = (int)fdiv((unsigned)rs, s.fS);
                const int ss = rs - r * s.S;
                if (PAD0) {
                    src = x + a_base[t] + ((int64_t)r * s.W + ss) * s.C + c;
                } else {
                    const int ih = a_oh[t] * s.stride + r - s.pad;
                    const int iw = a_ow[t] * s.stride + ss - s.pad;
                    if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                        src = x + ((((int64_t)a_n[t] * s.H + ih) * s.W + iw)
                                   * s.C + c);
                }
            }
            glds16(src, abase + (wave + t * NW) * 1024 + lane * 16);
        }
        char* bbase = abase + ABYTES;
#pragma unroll
        for (int t = 0; t < BPT; ++t) {
            const int p = (wave + t * NW) * 1024 + lane * 16;
            const int l = swz128(p);
            const int korow = l >> 7;
            const int kc = (l & 127) >> 1;
            const int ko = n0 + korow;
            const int k = k0 + kc;
            const unsigned short* src = zbuf;
            if (ko < s.Kout && k < KK) src = w + (int64_t)ko * KK + k;
            glds16(src, bbase + p);
        }
    };

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int ksteps = (KK + BK - 1) / BK;
    const int kc_len = (ksteps + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * kc_len * BK;
    const int kend = min(kbeg + kc_len * BK, KK);
    const int wm = wave / WN, wn = wave % WN;
    const int half = lane >> 4, sub = lane & 15;
    stage(0, kbeg);
    int buf = 0;
    for (int k0 = kbeg; k0 < kend; k0 += BK) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();  // tile k0 visible everywhere; last tile consumed
        if (k0 + BK < kend) stage(buf ^ 1, k0 + BK);  // next DMA in flight
        const char* Ab = (const char*)smem + buf * (ABYTES + BN * BK * 2);
        const char* Bb = Ab + ABYTES;
#pragma unroll
        for (int kk = 0; kk < BK; kk += 32) {
            bf16x8 a[FM], b[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i) {
                const int row = wm * FM * 16 + i * 16 + sub;
                a[i] = *reinterpret_cast<const bf16x8*>(
                    Ab + swz128(row * 128 + (kk + half * 8) * 2));
            }
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int row = wn * FN * 16 + j * 16 + sub;
                b[j] = *reinterpret_cast<const bf16x8*>(
                    Bb + swz128(row * 128 + (kk + half * 8) * 2));
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        buf ^= 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + sub;
        if (col >= s.Kout) continue;
        const float bv = bias ? bias[col] : 0.f;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wm * FM * 16 + i * 16 + half * 4 + r;
                if (row >= M) continue;
                if (k_chunks > 1) {
                    atomicAdd(y32 + (int64_t)row * s.Kout + col, acc[i][j][r]);
                } else {
                    float v = acc[i][j][r] + bv;
                    if (relu) v = v > 0.f ? v : 0.f;
                    y[(int64_t)row * s.Kout + col] = f2bf(v);
                }
            }
    }
}

// ---- Direct tiled 3x3 stride-1 conv forward: one block owns a TH x TW
// output patch of one image and a BN-wide Kout tile; the (TH+2) x (TW+2) x
// CS input tile is staged in LDS ONCE per channel slab and all NINE kernel
// taps read from it. The implicit-GEMM kernel re-gathers the same input
// 9x through the im2col view and runs only 4 MFMAs between barriers; here
// one staging barrier covers 9 * FM * FN MFMAs per wave (~144), so the
// kernel is MFMA-issue rather than stage/barrier bound. Requires stride 1,
// R = S = 3, C % 32 == 0 (channel slabs of 32). pad 0 or 1.
template <int TH, int TW, int BN, int WM, int WN, int FM, int FN>
__global__ void __launch_bounds__(TPB)
conv_fwd_tile3_kernel(const unsigned short* __restrict__ x,
                      const unsigned short* __restrict__ w,
                      const float* __restrict__ bias,
                      unsigned short* __restrict__ y, ConvShape s, int relu,
                      int tiles_h, int tiles_w) {
    constexpr int CS = 32;               // channel slab (one MFMA K-step)
    constexpr int XH = TH + 2, XW = TW + 2;
    // +8 channel pad on the x-tile pixel stride: A-fragment lanes read 16
    // consecutive pixels at 64-B stride; padding to 80 B breaks the 4-way
    // bank alias (rule 21)
    constexpr int XP = CS + 8;
    __shared__ unsigned short xs[XH * XW * XP];
    constexpr int WP = CS + 8;  // same bank-alias pad as the x-tile
    __shared__ unsigned short ws[9 * BN * WP];
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    // tile decode: blockIdx.x = (n, th, tw)
    const int tw_ = blockIdx.x % tiles_w;
    const int th_ = (blockIdx.x / tiles_w) % tiles_h;
    const int n = blockIdx.x / (tiles_w * tiles_h);
    const int oh0 = th_ * TH, ow0 = tw_ * TW;
    const int n0 = blockIdx.y * BN;
    const int wm = wave / WN, wn = wave % WN;
    const int half = lane >> 4, sub = lane & 15;

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int KK = s.R * s.S * s.C;  // w rows are [ko][r][s][c], c fastest
    for (int c0 = 0; c0 < s.C; c0 += CS) {
        __syncthreads();  // previous slab's MFMAs done before overwrite
        // x-tile: XH*XW pixels x CS channels, 16-B chunks; border/halo
        // out-of-range rows load zeros
        for (int i = tid; i < XH * XW * (CS / 8); i += TPB) {
            const int pix = i / (CS / 8);
            const int cc = (i % (CS / 8)) * 8;
            const int gy = pix / XW, gx = pix - gy * XW;
            const int ih = oh0 + gy - s.pad;
            const int iw = ow0 + gx - s.pad;
            u16x8 v = {};
            if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                v = *reinterpret_cast<const u16x8*>(
                    &x[((((int64_t)n * s.H + ih) * s.W + iw) * s.C) + c0 + cc]);
            *reinterpret_cast<u16x8*>(&xs[pix * XP + cc]) = v;
        }
        // w-slab: [rs][ko][CS], ko-major rows so B-fragment reads are
        // 16-B at stride CS*2
        for (int i = tid; i < 9 * BN * (CS / 8); i += TPB) {
            const int rs = i / (BN * (CS / 8));
            const int rem = i - rs * BN * (CS / 8);
            const int ko = rem / (CS / 8);
            const int cc = (rem % (CS / 8)) * 8;
            u16x8 v = {};
            if (n0 + ko < s.Kout)
                v = *reinterpret_cast<const u16x8*>(
                    &w[(int64_t)(n0 + ko) * KK + rs * s.C + c0 + cc]);
            *reinterpret_cast<u16x8*>(&ws[(rs * BN + ko) * WP + cc]) = v;
        }
        __syncthreads();
#pragma unroll
        for (int rs = 0; rs < 9; ++rs) {
            const int r = rs / 3, ss = rs - r * 3;
            bf16x8 a[FM], b[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i) {
                const int p = wm * FM * 16 + i * 16 + sub;  // local pixel
                const int py = p / TW, px = p - py * TW;
                a[i] = *reinterpret_cast<const bf16x8*>(
                    &xs[((py + r) * XW + (px + ss)) * XP + half * 8]);
            }
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int ko = wn * FN * 16 + j * 16 + sub;
                b[j] = *reinterpret_cast<const bf16x8*>(
                    &ws[(rs * BN + ko) * WP + half * 8]);
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
    }
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + sub;
        if (col >= s.Kout) continue;
        const float bv = bias ? bias[col] : 0.f;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int p = wm * FM * 16 + i * 16 + half * 4 + r;
                const int py = p / TW, px = p - py * TW;
                const int oh = oh0 + py, ow = ow0 + px;
                if (oh >= s.OH || ow >= s.OW) continue;
                float v = acc[i][j][r] + bv;
                if (relu) v = v > 0.f ? v : 0.f;
                y[((((int64_t)n * s.OH + oh) * s.OW + ow) * s.Kout) + col]
                    = f2bf(v);
            }
    }
}

// wT[c][r][s][ko] = w[ko][2-r][2-s][c]: the 180-rotated transpose that
// turns dgrad into a plain forward conv of dy (tiny tensor; one launch).
__global__ void rot180_transpose_w_kernel(const unsigned short* __restrict__ w,
                                          unsigned short* __restrict__ wT,
                                          int Kout, int C, int64_t total) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total; i += (int64_t)gridDim.x * blockDim.x) {
        const int ko = (int)(i % Kout);
        const int64_t t = i / Kout;
        const int rs = (int)(t % 9);
        const int c = (int)(t / 9);
        const int r = rs / 3, ss = rs - r * 3;
        wT[i] = w[(((int64_t)ko * 3 + (2 - r)) * 3 + (2 - ss)) * C + c];
    }
}

// ---- dgrad: dx[m=(n,ih,iw), c] = sum_k A(m,k) * B(k,c),
//      k = (r, s, ko) with ko FASTEST so the dy gather is contiguous;
//      B(k, c) = w[ko, r, s, c] (strided, small tile) ----
template <int BM, int BN, int WM, int WN, int FM, int FN, bool S1 = false>
__global__ void __launch_bounds__(TPB)
conv_dgrad_kernel(const unsigned short* __restrict__ dy,
                  const unsigned short* __restrict__ w,
                  unsigned short* __restrict__ dx, ConvShape s) {
    __shared__ unsigned short As[BM][32];
    __shared__ unsigned short Bst[32][BN + 8];  // k-major: B(k, c) tile
    constexpr int RPT = BM * 4 / TPB;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int M = s.N * s.H * s.W;
    const int KK = s.Kout * s.R * s.S;
    const bool fast = (s.Kout % 8 == 0);

    int arows[RPT], am_[RPT], a_n[RPT], a_ih[RPT], a_iw[RPT];
    const int akc = (tid & 3) * 8;
#pragma unroll
    for (int t = 0; t < RPT; ++t) {
        arows[t] = (tid + t * TPB) >> 2;
        const int am = m0 + arows[t];
        am_[t] = am;
        a_n[t] = a_ih[t] = a_iw[t] = 0;
        if (am < M) {
            a_n[t] = am / (s.H * s.W);
            int rem = am % (s.H * s.W);
            a_ih[t] = rem / s.W;
            a_iw[t] = rem % s.W;
        }
    }

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int k0 = 0; k0 < KK; k0 += 32) {
#pragma unroll
        for (int t = 0; t < RPT; ++t) {  // ---- stage A from dy ----
            const int k = k0 + akc;
            unsigned short* dst = &As[arows[t]][akc];
            bool done = false;
            if (am_[t] < M && k < KK && fast) {
                // chunk shares (r,s): ko = k % Kout, rs = k / Kout
                int rs = (int)fdiv((unsigned)k, s.fKout), ko = k - rs * s.Kout;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                int oh_num = a_ih[t] + s.pad - r, ow_num = a_iw[t] + s.pad - ss;
                done = true;
                if (S1) {  // stride 1: no divisibility checks or divisions
                    if (oh_num >= 0 && ow_num >= 0 && oh_num < s.OH &&
                        ow_num < s.OW) {
                        copy16(dst, dy + ((((int64_t)a_n[t] * s.OH + oh_num)
                                              * s.OW + ow_num) * s.Kout + ko));
                    } else {
                        zero16(dst);
                    }
                } else if (oh_num >= 0 && ow_num >= 0 &&
                           oh_num % s.stride == 0 && ow_num % s.stride == 0 &&
                           oh_num / s.stride < s.OH &&
                           ow_num / s.stride < s.OW) {
                    copy16(dst, dy + ((((int64_t)a_n[t] * s.OH + oh_num / s.stride)
                                          * s.OW + ow_num / s.stride)
                                         * s.Kout + ko));
                } else {
                    zero16(dst);
                }
            }
            if (!done) {
                if (am_[t] < M && k < KK) {
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        int kk = k + j;
                        unsigned short v = 0;
                        if (kk < KK) {
                            int rs = (int)fdiv((unsigned)kk, s.fKout), ko = kk - rs * s.Kout;
                            int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                            int oh_num = a_ih[t] + s.pad - r;
                            int ow_num = a_iw[t] + s.pad - ss;
                            if (oh_num >= 0 && ow_num >= 0 &&
                                oh_num % s.stride == 0 &&
                                ow_num % s.stride == 0 &&
                                oh_num / s.stride < s.OH &&
                                ow_num / s.stride < s.OW)
                                v = dy[((((int64_t)a_n[t] * s.OH +
                                          oh_num / s.stride) * s.OW +
                                         ow_num / s.stride) * s.Kout + ko)];
                        }
                        dst[j] = v;
                    }
                } else {
                    zero16(dst);
                }
            }
        }
        // ---- stage B k-major: Bst[k][c..c+8] <- w[ko*RS*C + rs*C + c],
        //      contiguous in c (16-B loads when C % 8 == 0) ----
        for (int i = tid; i < 32 * (BN / 8); i += TPB) {
            const int kk = i / (BN / 8);
            const int cc = (i % (BN / 8)) * 8;
            const int k = k0 + kk;
            const int c = n0 + cc;
            unsigned short* dst = &Bst[kk][cc];
            if (k < KK && c < s.C) {
                int rs = (int)fdiv((unsigned)k, s.fKout), ko = k - rs * s.Kout;
                const unsigned short* src =
                    w + ((int64_t)ko * s.R * s.S + rs) * s.C + c;
                if (s.C % 8 == 0) {
                    copy16(dst, src);
                } else {
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        dst[j] = (c + j < s.C) ? src[j] : 0;
                }
            } else {
                zero16(dst);
            }
        }
        __syncthreads();
        {   // ---- MFMA: A from As (k-minor), B from Bst (k-major) ----
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
            bf16x8 a[FM], b[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i)
                a[i] = *reinterpret_cast<const bf16x8*>(
                    &As[wm * FM * 16 + i * 16 + sub][half * 8]);
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int col = wn * FN * 16 + j * 16 + sub;
#pragma unroll
                for (int t = 0; t < 8; ++t)
                    b[j][t] = *reinterpret_cast<const bf16_t*>(
                        &Bst[half * 8 + t][col]);
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= s.C) continue;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
                dx[(int64_t)row * s.C + col] = f2bf(acc[i][j][r]);
            }
    }
}

// ---- glds-pipelined dgrad (same 2-buffer DMA structure as fwd; B tile is
// k-major so its staging is also lane-linear). Requires stride==1 paths NOT
// needed: S1 handled in the source-address computation. ----
template <int BM, int BN, int WM, int WN, int FM, int FN, bool S1>
__global__ void __launch_bounds__(TPB)
conv_dgrad_glds_kernel(const unsigned short* __restrict__ dy,
                       const unsigned short* __restrict__ w,
                       unsigned short* __restrict__ dx,
                       float* __restrict__ dx32,
                       const unsigned short* __restrict__ zbuf, ConvShape s,
                       int k_chunks) {
    __shared__ unsigned short smem[3 * (BM + BN) * 32];
    auto As = [&](int buf) -> unsigned short (*)[32] {
        return reinterpret_cast<unsigned short(*)[32]>(smem + buf * (BM + BN) * 32);
    };
    auto Bst = [&](int buf) -> unsigned short (*)[BN] {
        return reinterpret_cast<unsigned short(*)[BN]>(smem + buf * (BM + BN) * 32
                                                        + BM * 32);
    };
    constexpr int RPT = BM * 4 / TPB;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = xcd_remap(blockIdx.x, gridDim.x) * BM;
    const int n0 = blockIdx.y * BN;
    const int M = s.N * s.H * s.W;
    const int KK = s.Kout * s.R * s.S;

    int a_n[RPT], a_ih[RPT], a_iw[RPT];
    bool a_ok[RPT];
    const int akc = (tid & 3) * 8;
#pragma unroll
    for (int t = 0; t < RPT; ++t) {
        const int am = m0 + ((tid + t * TPB) >> 2);
        a_ok[t] = am < M;
        a_n[t] = a_ih[t] = a_iw[t] = 0;
        if (a_ok[t]) {
            a_n[t] = am / (s.H * s.W);
            int rem = am % (s.H * s.W);
            a_ih[t] = rem / s.W;
            a_iw[t] = rem % s.W;
        }
    }

    auto stage = [&](int buf, int k0) {
#pragma unroll
        for (int t = 0; t < RPT; ++t) {
            const int k = k0 + akc;
            const unsigned short* src = zbuf;
            if (a_ok[t] && k < KK) {
                int rs = (int)fdiv((unsigned)k, s.fKout), ko = k - rs * s.Kout;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                int oh_num = a_ih[t] + s.pad - r, ow_num = a_iw[t] + s.pad - ss;
                if (S1) {
                    if (oh_num >= 0 && ow_num >= 0 && oh_num < s.OH &&
                        ow_num < s.OW)
                        src = dy + ((((int64_t)a_n[t] * s.OH + oh_num) * s.OW
                                     + ow_num) * s.Kout + ko);
                } else if (oh_num >= 0 && ow_num >= 0 &&
                           oh_num % s.stride == 0 && ow_num % s.stride == 0 &&
                           oh_num / s.stride < s.OH &&
                           ow_num / s.stride < s.OW) {
                    src = dy + ((((int64_t)a_n[t] * s.OH + oh_num / s.stride)
                                 * s.OW + ow_num / s.stride) * s.Kout + ko);
                }
            }
            char* base = (char*)&As(buf)[0][0] + (wave + t * 4) * 1024;
            glds16(src, base);
        }
        {   // B: Bst[k][c..c+8], chunk i = tid -> byte tid*16 (BN == 64)
            const int kk = tid >> 3;
            const int cc = (tid & 7) * 8;
            const int k = k0 + kk;
            const int c = n0 + cc;
            const unsigned short* src = zbuf;
            if (k < KK && c < s.C) {
                int rs = (int)fdiv((unsigned)k, s.fKout), ko = k - rs * s.Kout;
                src = w + ((int64_t)ko * s.R * s.S + rs) * s.C + c;
            }
            char* base = (char*)&Bst(buf)[0][0] + wave * 1024;
            glds16(src, base);
        }
    };

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    const int ksteps = (KK + 31) / 32;
    const int kc_len = (ksteps + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * kc_len * 32;
    const int kend = min(kbeg + kc_len * 32, KK);
    // 3-buffer counted-vmcnt pipeline (see conv_fwd_glds_kernel)
    stage(0, kbeg);
    stage(1, kbeg + 32);
    int buf = 0;
    for (int k0 = kbeg; k0 < kend; k0 += 32) {
        if (k0 + 32 < kend)
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_barrier" ::: "memory");
        if (k0 + 64 < kend) stage(buf == 0 ? 2 : buf - 1, k0 + 64);
        {
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
            bf16x8 a[FM], b[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i)
                a[i] = *reinterpret_cast<const bf16x8*>(
                    &As(buf)[wm * FM * 16 + i * 16 + sub][half * 8]);
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int col = wn * FN * 16 + j * 16 + sub;
#pragma unroll
                for (int t = 0; t < 8; ++t)
                    b[j][t] = *reinterpret_cast<const bf16_t*>(
                        &Bst(buf)[half * 8 + t][col]);
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        buf = buf == 2 ? 0 : buf + 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= s.C) continue;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (row >= M) continue;
                if (k_chunks > 1)
                    atomicAdd(dx32 + (int64_t)row * s.C + col, acc[i][j][r]);
                else
                    dx[(int64_t)row * s.C + col] = f2bf(acc[i][j][r]);
            }
    }
}

// f32 accumulator -> bf16 (no bias/relu): split-K epilogue for dgrad.
// clear=1: consume-and-clear — zero each element after reading so a pooled
// accumulator (acc_pool below) is zero again for its next user without a
// separate fill launch (the FillFunctor fills were 11.5% of config #2
// kernel time, profiles/r02_config2_final_kernel_stats.csv).
__global__ void cast_f32_bf16_kernel(float* __restrict__ src,
                                     unsigned short* __restrict__ dst,
                                     int64_t total, int clear) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        dst[i] = f2bf(src[i]);
        if (clear) src[i] = 0.f;
    }
}

// Zero-pad the channel (last) dim C -> C8: routes C % 8 != 0 stem convs
// (RefCNN6 conv1 C=3, ResNet stem, cnn4 conv1 C=1) onto the glds MFMA
// pipeline, which needs 16-B-aligned channel vectors. The padded FLOPs are
// multiply-by-zero; the pipeline is ~5-10x faster per FLOP than the
// generic-gather fallback at these shapes (profiles/r02_refcnn6_*).
__global__ void pad_channels_kernel(const unsigned short* __restrict__ x,
                                    unsigned short* __restrict__ out,
                                    int64_t total_out, int C, int C8) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total_out; i += (int64_t)gridDim.x * blockDim.x) {
        const int c = (int)(i % C8);
        out[i] = c < C ? x[(i / C8) * C + c] : 0;
    }
}

// ---- wgrad: dw[ko, rsc] = sum_pix dy[pix, ko] * xcol[pix, rsc].
// LDS tiles are PIXEL-major (Dys[32][64], Xs[32][BN]) so BOTH global
// gathers are 16-B vector loads (ko contiguous in dy; c contiguous in x);
// fragments read LDS with k-strided scalar loads instead. Split-K over
// grid.z with fp32 atomics. ----
template <int BN, int WM, int WN, int FM, int FN>
__global__ void __launch_bounds__(TPB)
conv_wgrad_kernel(const unsigned short* __restrict__ dy,
                  const unsigned short* __restrict__ x,
                  float* __restrict__ dw, ConvShape s, int k_chunks) {
    __shared__ unsigned short Dys[32][64 + 8];  // [pix][ko] (+pad)
    __shared__ unsigned short Xs[32][BN + 8];   // [pix][rsc]
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * 64;   // over Kout
    const int n0 = blockIdx.y * BN;   // over RSC
    const int M = s.Kout;
    const int NN = s.R * s.S * s.C;
    const int KK = s.N * s.OH * s.OW;
    const int chunk = (KK + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * chunk;
    const int kend = min(kbeg + chunk, KK);
    const bool fast_dy = (s.Kout % 8 == 0);
    const bool fast_x = (s.C % 8 == 0);

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int k0 = kbeg; k0 < kend; k0 += 32) {
        // ---- stage dy tile [32 pix][64 ko]: 256 chunks of 8 ko ----
        {
            const int pix = tid >> 3;
            const int koc = (tid & 7) * 8;
            const int kpix = k0 + pix;
            const int ko = m0 + koc;
            unsigned short* dst = &Dys[pix][koc];
            if (kpix < kend && ko + 8 <= M && fast_dy) {
                copy16(dst, dy + (int64_t)kpix * s.Kout + ko);
            } else if (kpix < kend && ko < M) {
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    dst[j] = (ko + j < M) ? dy[(int64_t)kpix * s.Kout + ko + j]
                                          : 0;
            } else {
                zero16(dst);
            }
        }
        // ---- stage x tile [32 pix][BN rsc]: BN*4 chunks of 8 c ----
        for (int i = tid; i < 32 * (BN / 8); i += TPB) {
            const int pix = i / (BN / 8);
            const int cc = (i % (BN / 8)) * 8;
            const int kpix = k0 + pix;
            const int nn = n0 + cc;
            unsigned short* dst = &Xs[pix][cc];
            bool done = false;
            if (kpix < kend && nn < NN) {
                int n_ = (int)fdiv((unsigned)kpix, s.fOHOW);
                int rem = kpix - n_ * (s.OH * s.OW);
                int oh = (int)fdiv((unsigned)rem, s.fOW), ow = rem - oh * s.OW;
                int rs = (int)fdiv((unsigned)nn, s.fC);
                int c = nn - rs * s.C;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                int ih = oh * s.stride + r - s.pad;
                int iw = ow * s.stride + ss - s.pad;
                if (fast_x && c + 8 <= s.C) {
                    done = true;
                    if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                        copy16(dst, x + ((((int64_t)n_ * s.H + ih) * s.W + iw)
                                             * s.C + c));
                    else
                        zero16(dst);
                }
                if (!done) {
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        int n2 = nn + j;
                        unsigned short v = 0;
                        if (n2 < NN) {
                            int rs2 = (int)fdiv((unsigned)n2, s.fC);
                            int c2 = n2 - rs2 * s.C;
                            int r2 = (int)fdiv((unsigned)rs2, s.fS), ss2 = rs2 - r2 * s.S;
                            int ih2 = oh * s.stride + r2 - s.pad;
                            int iw2 = ow * s.stride + ss2 - s.pad;
                            if (ih2 >= 0 && ih2 < s.H && iw2 >= 0 && iw2 < s.W)
                                v = x[((((int64_t)n_ * s.H + ih2) * s.W + iw2)
                                           * s.C + c2)];
                        }
                        dst[j] = v;
                    }
                    done = true;
                }
            }
            if (!done) zero16(dst);
        }
        __syncthreads();
        // ---- MFMA: A(m=ko, k=pix) from Dys[k][m]; B(k=pix, n) from Xs ----
        {
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
            bf16x8 a[FM], b[FN];
#pragma unroll
            for (int i = 0; i < FM; ++i) {
                const int mrow = wm * FM * 16 + i * 16 + sub;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    a[i][j] = *reinterpret_cast<const bf16_t*>(
                        &Dys[half * 8 + j][mrow]);
            }
#pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int col = wn * FN * 16 + j * 16 + sub;
#pragma unroll
                for (int t = 0; t < 8; ++t)
                    b[j][t] = *reinterpret_cast<const bf16_t*>(
                        &Xs[half * 8 + t][col]);
            }
#pragma unroll
            for (int i = 0; i < FM; ++i)
#pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();
    }
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= NN) continue;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ko = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (ko >= M) continue;
                if (k_chunks > 1)
                    atomicAdd(dw + (int64_t)ko * NN + col, acc[i][j][r]);
                else
                    dw[(int64_t)ko * NN + col] = acc[i][j][r];
            }
    }
}

// ---- glds-pipelined wgrad: both tiles (Dys pixel-major [32][64] and
// Xs pixel-major [32][BN]) stage as lane-linear 16-B DMAs; 2-buffer
// pipeline like fwd/dgrad. Requires Kout % 8 == 0 and C % 8 == 0. ----
template <int BN, int WM, int WN, int FM, int FN, int BKP = 32>
__global__ void __launch_bounds__(TPB)
conv_wgrad_glds_kernel(const unsigned short* __restrict__ dy,
                       const unsigned short* __restrict__ x,
                       float* __restrict__ dw,
                       const unsigned short* __restrict__ zbuf, ConvShape s,
                       int k_chunks) {
    __shared__ unsigned short smem[3 * (64 + BN) * BKP];
    auto Dys = [&](int buf) -> unsigned short (*)[64] {
        return reinterpret_cast<unsigned short(*)[64]>(smem + buf * (64 + BN) * BKP);
    };
    auto Xs = [&](int buf) -> unsigned short (*)[BN] {
        return reinterpret_cast<unsigned short(*)[BN]>(smem + buf * (64 + BN) * BKP
                                                        + 64 * BKP);
    };
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * 64;   // over Kout
    const int n0 = blockIdx.y * BN;   // over RSC
    const int M = s.Kout;
    const int NN = s.R * s.S * s.C;
    const int KK = s.N * s.OH * s.OW;
    const int chunk = (KK + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * chunk;
    const int kend = min(kbeg + chunk, KK);

    auto stage = [&](int buf, int k0) {
#pragma unroll
        for (int t = 0; t < BKP * 8 / TPB; ++t) {  // Dys[pix][ko] chunks
            const int i = tid + t * TPB;
            const int pix = i >> 3;
            const int koc = (i & 7) * 8;
            const int kpix = k0 + pix;
            const int ko = m0 + koc;
            const unsigned short* src = zbuf;
            if (kpix < kend && ko + 8 <= M)
                src = dy + (int64_t)kpix * s.Kout + ko;
            glds16(src, (char*)&Dys(buf)[0][0] + (wave + t * 4) * 1024);
        }
#pragma unroll
        for (int t = 0; t < BKP * (BN / 8) / TPB; ++t) {  // Xs chunks
            const int i = tid + t * TPB;
            const int pix = i / (BN / 8);
            const int cc = (i % (BN / 8)) * 8;
            const int kpix = k0 + pix;
            const int nn = n0 + cc;
            const unsigned short* src = zbuf;
            if (kpix < kend && nn < NN) {
                int n_ = (int)fdiv((unsigned)kpix, s.fOHOW);
                int rem = kpix - n_ * (s.OH * s.OW);
                int oh = (int)fdiv((unsigned)rem, s.fOW), ow = rem - oh * s.OW;
                int rs = (int)fdiv((unsigned)nn, s.fC);
                int c = nn - rs * s.C;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                if (c + 8 <= s.C) {
                    int ih = oh * s.stride + r - s.pad;
                    int iw = ow * s.stride + ss - s.pad;
                    if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                        src = x + ((((int64_t)n_ * s.H + ih) * s.W + iw) * s.C
                                   + c);
                }
            }
            glds16(src, (char*)&Xs(buf)[0][0] + (wave + t * 4) * 1024);
        }
    };

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    // 3-buffer counted-vmcnt pipeline (see conv_fwd_glds_kernel)
    stage(0, kbeg);
    stage(1, kbeg + BKP);
    int buf = 0;
    for (int k0 = kbeg; k0 < kend; k0 += BKP) {
        if (k0 + BKP < kend) {
            // one stage in flight = BKP*(8 + BN/8)/TPB glds per thread
            if constexpr (BKP == 32)
                asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        } else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_barrier" ::: "memory");
        if (k0 + 2 * BKP < kend) stage(buf == 0 ? 2 : buf - 1, k0 + 2 * BKP);
        {
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
#pragma unroll
            for (int kk = 0; kk < BKP; kk += 32) {
                bf16x8 a[FM], b[FN];
#pragma unroll
                for (int i = 0; i < FM; ++i) {
                    const int mrow = wm * FM * 16 + i * 16 + sub;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        a[i][j] = *reinterpret_cast<const bf16_t*>(
                            &Dys(buf)[kk + half * 8 + j][mrow]);
                }
#pragma unroll
                for (int j = 0; j < FN; ++j) {
                    const int col = wn * FN * 16 + j * 16 + sub;
#pragma unroll
                    for (int t = 0; t < 8; ++t)
                        b[j][t] = *reinterpret_cast<const bf16_t*>(
                            &Xs(buf)[kk + half * 8 + t][col]);
                }
#pragma unroll
                for (int i = 0; i < FM; ++i)
#pragma unroll
                    for (int j = 0; j < FN; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[i], b[j], acc[i][j], 0, 0, 0);
            }
        }
        buf = buf == 2 ? 0 : buf + 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= NN) continue;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ko = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (ko >= M) continue;
                if (k_chunks > 1)
                    atomicAdd(dw + (int64_t)ko * NN + col, acc[i][j][r]);
                else
                    dw[(int64_t)ko * NN + col] = acc[i][j][r];
            }
    }
}

// ---- Kout <= 32 wgrad: the 64-ko tile wasted HALF its MFMA work on
// 32-filter layers (RefCNN6 conv1-3 — the round's #1 kernel). This variant
// tiles [32 ko] x [BN cc] with a 64-PIXEL K-step: same 4 MFMAs per wave
// per barrier, no ko waste, half the barriers per pixel. Pipeline and
// gather structure mirror conv_wgrad_glds_kernel (3-buffer counted
// vmcnt(3): three 16-B glds per thread per stage). ----
template <int BN, int WM, int WN, int FM, int FN, int BKP = 64>
__global__ void __launch_bounds__(TPB)
conv_wgrad_glds_k32_kernel(const unsigned short* __restrict__ dy,
                           const unsigned short* __restrict__ x,
                           float* __restrict__ dw,
                           const unsigned short* __restrict__ zbuf,
                           ConvShape s, int k_chunks) {
    // BKP = pixels per K-step (64 or 128; 128 halves the barrier count
    // again at 3 LDS buffers x 24 KB)
    __shared__ unsigned short smem[3 * (32 + BN) * BKP];
    auto Dys = [&](int buf) -> unsigned short (*)[32] {
        return reinterpret_cast<unsigned short(*)[32]>(
            smem + buf * (32 + BN) * BKP);
    };
    auto Xs = [&](int buf) -> unsigned short (*)[BN] {
        return reinterpret_cast<unsigned short(*)[BN]>(
            smem + buf * (32 + BN) * BKP + 32 * BKP);
    };
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int m0 = blockIdx.x * 32;   // over Kout
    const int n0 = blockIdx.y * BN;   // over RSC
    const int M = s.Kout;
    const int NN = s.R * s.S * s.C;
    const int KK = s.N * s.OH * s.OW;
    const int chunk = (KK + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * chunk;
    const int kend = min(kbeg + chunk, KK);

    auto stage = [&](int buf, int k0) {
#pragma unroll
        for (int t = 0; t < BKP * 4 / TPB; ++t) {  // Dys[pix][ko] chunks
            const int i = tid + t * TPB;
            const int pix = i >> 2;
            const int koc = (i & 3) * 8;
            const int kpix = k0 + pix;
            const int ko = m0 + koc;
            const unsigned short* src = zbuf;
            if (kpix < kend && ko + 8 <= M)
                src = dy + (int64_t)kpix * s.Kout + ko;
            glds16(src, (char*)&Dys(buf)[0][0] + (wave + t * 4) * 1024);
        }
#pragma unroll
        for (int t = 0; t < BKP * (BN / 8) / TPB; ++t) {  // Xs[pix][cc] chunks
            const int i = tid + t * TPB;
            const int pix = i / (BN / 8);
            const int cc = (i % (BN / 8)) * 8;
            const int kpix = k0 + pix;
            const int nn = n0 + cc;
            const unsigned short* src = zbuf;
            if (kpix < kend && nn < NN) {
                int n_ = (int)fdiv((unsigned)kpix, s.fOHOW);
                int rem = kpix - n_ * (s.OH * s.OW);
                int oh = (int)fdiv((unsigned)rem, s.fOW), ow = rem - oh * s.OW;
                int rs = (int)fdiv((unsigned)nn, s.fC);
                int c = nn - rs * s.C;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                if (c + 8 <= s.C) {
                    int ih = oh * s.stride + r - s.pad;
                    int iw = ow * s.stride + ss - s.pad;
                    if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                        src = x + ((((int64_t)n_ * s.H + ih) * s.W + iw) * s.C
                                   + c);
                }
            }
            glds16(src, (char*)&Xs(buf)[0][0] + (wave + t * 4) * 1024);
        }
    };

    f32x4 acc[FM][FN];
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

    stage(0, kbeg);
    stage(1, kbeg + BKP);
    int buf = 0;
    for (int k0 = kbeg; k0 < kend; k0 += BKP) {
        if (k0 + BKP < kend) {
            // one stage in flight = (BKP*4 + BKP*8)/TPB glds per thread
            if constexpr (BKP == 64)
                asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        } else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        asm volatile("s_barrier" ::: "memory");
        if (k0 + 2 * BKP < kend) stage(buf == 0 ? 2 : buf - 1, k0 + 2 * BKP);
        {
            const int wm = wave / WN, wn = wave % WN;
            const int half = lane >> 4, sub = lane & 15;
#pragma unroll
            for (int kk = 0; kk < BKP; kk += 32) {
                bf16x8 a[FM], b[FN];
#pragma unroll
                for (int i = 0; i < FM; ++i) {
                    const int mrow = wm * FM * 16 + i * 16 + sub;
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        a[i][j] = *reinterpret_cast<const bf16_t*>(
                            &Dys(buf)[kk + half * 8 + j][mrow]);
                }
#pragma unroll
                for (int j = 0; j < FN; ++j) {
                    const int col = wn * FN * 16 + j * 16 + sub;
#pragma unroll
                    for (int t = 0; t < 8; ++t)
                        b[j][t] = *reinterpret_cast<const bf16_t*>(
                            &Xs(buf)[kk + half * 8 + t][col]);
                }
#pragma unroll
                for (int i = 0; i < FM; ++i)
#pragma unroll
                    for (int j = 0; j < FN; ++j)
                        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[i], b[j], acc[i][j], 0, 0, 0);
            }
        }
        buf = buf == 2 ? 0 : buf + 1;
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    const int wm = wave / WN, wn = wave % WN;
#pragma unroll
    for (int j = 0; j < FN; ++j) {
        const int col = n0 + wn * FN * 16 + j * 16 + (lane & 15);
        if (col >= NN) continue;
#pragma unroll
        for (int i = 0; i < FM; ++i)
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int ko = m0 + wm * FM * 16 + i * 16 + (lane >> 4) * 4 + r;
                if (ko >= M) continue;
                if (k_chunks > 1)
                    atomicAdd(dw + (int64_t)ko * NN + col, acc[i][j][r]);
                else
                    dw[(int64_t)ko * NN + col] = acc[i][j][r];
            }
    }
}

// ---- small-shape wgrad (first convs: Kout <= 64, R*S*C <= 16, e.g. the
// 28x28x1 model's conv1): the generic 64xBN tile wastes 3/4 of its waves on
// out-of-range rows. Here the 4 waves SPLIT over pixels (each owns 32 of a
// 128-pixel iteration) and reduce their partial accumulators through LDS at
// the end; grid.z chunks pixels further. ----
__global__ void __launch_bounds__(TPB)
conv_wgrad_small_kernel(const unsigned short* __restrict__ dy,
                        const unsigned short* __restrict__ x,
                        float* __restrict__ dw, ConvShape s, int k_chunks) {
    __shared__ unsigned short Dys[128][64];
    __shared__ unsigned short Xp[128][16];
    __shared__ float red[4][16 * 16];
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int M = s.Kout;            // <= 64
    const int NN = s.R * s.S * s.C;  // <= 16
    const int KK = s.N * s.OH * s.OW;
    const int FM = (M + 15) / 16;
    const int chunk = (KK + k_chunks - 1) / k_chunks;
    const int kbeg = blockIdx.z * chunk;
    const int kend = min(kbeg + chunk, KK);
    const bool fast_dy = (s.Kout % 8 == 0);

    f32x4 acc[4];  // up to FM=4 fragments
#pragma unroll
    for (int i = 0; i < 4; ++i) acc[i] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int p0 = kbeg; p0 < kend; p0 += 128) {
        // stage dy [128][Kout]
        for (int i = tid; i < 128 * (M / 8 > 0 ? M / 8 : 1); i += TPB) {
            if (fast_dy) {
                const int pix = i / (M / 8);
                const int koc = (i % (M / 8)) * 8;
                unsigned short* dst = &Dys[pix][koc];
                if (p0 + pix < kend)
                    copy16(dst, dy + (int64_t)(p0 + pix) * s.Kout + koc);
                else
                    zero16(dst);
            } else {
                const int pix = i;  // M < 8: scalar per pixel
                if (pix < 128) {
                    for (int ko = 0; ko < M; ++ko)
                        Dys[pix][ko] = (p0 + pix < kend)
                            ? dy[(int64_t)(p0 + pix) * s.Kout + ko] : 0;
                }
            }
        }
        // stage x patches [128][NN]
        for (int i = tid; i < 128 * NN; i += TPB) {
            const int pix = i / NN;
            const int nn = i % NN;
            unsigned short v = 0;
            const int kpix = p0 + pix;
            if (kpix < kend) {
                int n_ = (int)fdiv((unsigned)kpix, s.fOHOW);
                int rem = kpix - n_ * (s.OH * s.OW);
                int oh = (int)fdiv((unsigned)rem, s.fOW), ow = rem - oh * s.OW;
                int rs = (int)fdiv((unsigned)nn, s.fC);
                int c = nn - rs * s.C;
                int r = (int)fdiv((unsigned)rs, s.fS), ss = rs - r * s.S;
                int ih = oh * s.stride + r - s.pad;
                int iw = ow * s.stride + ss - s.pad;
                if (ih >= 0 && ih < s.H && iw >= 0 && iw < s.W)
                    v = x[((((int64_t)n_ * s.H + ih) * s.W + iw) * s.C + c)];
            }
            Xp[pix][nn] = v;
        }
        __syncthreads();
        // each wave: its own 32 pixels as the MFMA K dim
        {
            const int half = lane >> 4, sub = lane & 15;
            const int pbase = wave * 32 + half * 8;
            bf16x8 b{};
#pragma unroll
            for (int j = 0; j < 8; ++j)
                if (sub < 16)
                    b[j] = *reinterpret_cast<const bf16_t*>(&Xp[pbase + j][sub]);
#pragma unroll
            for (int fm = 0; fm < 4; ++fm) {
                if (fm >= FM) break;
                bf16x8 a;
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    a[j] = *reinterpret_cast<const bf16_t*>(
                        &Dys[pbase + j][fm * 16 + sub]);
                acc[fm] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[fm],
                                                                  0, 0, 0);
            }
        }
        __syncthreads();
    }
    // cross-wave reduction per fragment, then atomics into dw
    for (int fm = 0; fm < FM; ++fm) {
        const int sub = lane & 15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int row = (lane >> 4) * 4 + r;  // 0..15 within fragment
            red[wave][row * 16 + sub] = acc[fm][r];
        }
        __syncthreads();
        // wave 0 sums and writes
        if (wave == 0) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = (lane >> 4) * 4 + r;
                float v = red[0][row * 16 + sub] + red[1][row * 16 + sub] +
                          red[2][row * 16 + sub] + red[3][row * 16 + sub];
                int ko = fm * 16 + row;
                if (ko < M && sub < NN) {
                    // split-K: per-chunk slab slice (every [M*NN] cell is
                    // stored exactly once -> no atomics, no zero-fill; a
                    // tiny epilogue sums the slices)
                    float* dst = dw + (k_chunks > 1
                                           ? (int64_t)blockIdx.z * M * NN : 0);
                    dst[(int64_t)ko * NN + sub] = v;
                }
            }
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// MaxPool 2x2 stride 2 (NHWC bf16), argmax corner saved for backward.
// ---------------------------------------------------------------------------

// Synthetic data generation fused to one kernel: class-template gather +
// counter-hash Box-Muller Gaussian noise + sigmoid, bf16 store.
// Replaces the staging chain randn/index/mul/add/sigmoid/cast (~6 fp32
// passes over the epoch tensor) with a single bandwidth-bound pass
// (hefl/data/synthetic.py batch()).
__device__ __forceinline__ unsigned long long splitmix64(unsigned long long z) {
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

// AUG: per-sample random zoom/shear/h-flip applied by sampling the class
// template at inverse-affine coordinates (bilinear, border-clamped) — the
// MI355X-native analog of the reference's ImageDataGenerator augmentation
// (shear_range 0.2, zoom_range 0.2, horizontal_flip; FLPyfhelin.py:80-86),
// fused into the same one-pass generator so the epoch graph keeps staging
// data on-device. zr/sr = 0 and flip = 0 reproduces the plain path.
template <bool AUG>
__global__ void synth_batch_kernel(const float* __restrict__ T,
                                   const int64_t* __restrict__ lab,
                                   unsigned short* __restrict__ out,
                                   int64_t per_img, int64_t total,
                                   unsigned long long seed, FastDiv fPer,
                                   const long long* __restrict__ seed_buf,
                                   int H, int W, int C, FastDiv fC, FastDiv fW,
                                   float zr, float sr, int flip) {
    if (seed_buf) seed += (unsigned long long)seed_buf[0];  // graph-replay seed
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        const int64_t n = (int64_t)fdiv((unsigned)i, fPer);
        const int64_t off = i - n * per_img;
        float t;
        if (!AUG) {
            t = T[lab[n] * per_img + off];
        } else {
            // decode (h, w, c); per-SAMPLE transform params from their own
            // hash chain (seed ^ sample index)
            const unsigned hw = fdiv((unsigned)off, fC);
            const int c = (int)((unsigned)off - hw * C);
            const unsigned hh = fdiv(hw, fW);
            const int w = (int)(hw - hh * W);
            const int h = (int)hh;
            const unsigned long long pz =
                splitmix64(seed ^ (0xA5A5A5A5ull + (unsigned long long)n));
            const float uz = (unsigned int)pz * 2.3283064e-10f;
            const float us = (unsigned int)(pz >> 32) * 2.3283064e-10f;
            const float uf = (unsigned int)splitmix64(pz) * 2.3283064e-10f;
            const float zoom = 1.f + zr * (2.f * uz - 1.f);
            const float shear = sr * (2.f * us - 1.f);
            const float sf = (flip && uf < 0.5f) ? -1.f : 1.f;
            const float cy = 0.5f * (H - 1), cx = 0.5f * (W - 1);
            const float sy = cy + (h - cy) / zoom;
            const float sx = cx + (w - cx) * sf / zoom + shear * (h - cy);
            // border-clamped bilinear sample of the class template
            const float syc = fminf(fmaxf(sy, 0.f), (float)(H - 1));
            const float sxc = fminf(fmaxf(sx, 0.f), (float)(W - 1));
            const int y0 = (int)syc, x0 = (int)sxc;
            const int y1 = min(y0 + 1, H - 1), x1 = min(x0 + 1, W - 1);
            const float fy = syc - y0, fx = sxc - x0;
            const float* Tn = T + lab[n] * per_img;
            const float t00 = Tn[((int64_t)y0 * W + x0) * C + c];
            const float t01 = Tn[((int64_t)y0 * W + x1) * C + c];
            const float t10 = Tn[((int64_t)y1 * W + x0) * C + c];
            const float t11 = Tn[((int64_t)y1 * W + x1) * C + c];
            t = (t00 * (1 - fx) + t01 * fx) * (1 - fy)
                + (t10 * (1 - fx) + t11 * fx) * fy;
        }
        // splitmix64 counter hash -> two 32-bit uniforms -> Box-Muller
        unsigned long long z =
            splitmix64(seed + (unsigned long long)i * 0x9E3779B97F4A7C15ull);
        const float u1 = ((unsigned int)z + 1.0f) * 2.3283064e-10f;  // (0,1]
        const float u2 = (unsigned int)(z >> 32) * 2.3283064e-10f;
        const float nrm = sqrtf(-2.f * __logf(u1)) * __cosf(6.2831853f * u2);
        const float v = 0.6f * t + 0.4f * nrm;
        out[i] = f2bf(1.f / (1.f + __expf(-v)));  // sigmoid -> [0,1]
    }
}

__global__ void maxpool_fwd_kernel(const unsigned short* __restrict__ x,
                                   unsigned short* __restrict__ y,
                                   uint8_t* __restrict__ idx, int N, int H,
                                   int W, int C, int OH, int OW, FastDiv fC,
                                   FastDiv fOW, FastDiv fOH) {
    int64_t total = (int64_t)N * OH * OW * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t * C);
        unsigned t2 = fdiv(t, fOW);
        int ow = (int)(t - t2 * OW);
        unsigned t3 = fdiv(t2, fOH);
        int oh = (int)(t2 - t3 * OH);
        int n = (int)t3;
        int ih = oh * 2, iw = ow * 2;
        float best = -3.4e38f;
        int bi = 0;
#pragma unroll
        for (int d = 0; d < 4; ++d) {
            int dh = d >> 1, dw_ = d & 1;
            float v = bf2f(x[(((int64_t)n * H + ih + dh) * W + iw + dw_) * C + c]);
            if (v > best) { best = v; bi = d; }
        }
        y[i] = f2bf(best);
        idx[i] = (uint8_t)bi;
    }
}

// Gather form: one thread per INPUT cell, no zero-fill pass needed
// (2x2 stride-2 windows never overlap; odd tail rows/cols get 0).
__global__ void maxpool2x2_bwd_gather_kernel(const unsigned short* __restrict__ dy,
                                             const uint8_t* __restrict__ idx,
                                             unsigned short* __restrict__ dx,
                                             int N, int H, int W, int C,
                                             int OH, int OW, FastDiv fC,
                                             FastDiv fW, FastDiv fH) {
    int64_t total = (int64_t)N * H * W * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t * C);
        unsigned t2 = fdiv(t, fW);
        int iw = (int)(t - t2 * W);
        unsigned t3 = fdiv(t2, fH);
        int ih = (int)(t2 - t3 * H);
        int n = (int)t3;
        int oh = ih >> 1, ow = iw >> 1;
        unsigned short v = 0;
        if (oh < OH && ow < OW) {
            int64_t oidx = (((int64_t)n * OH + oh) * OW + ow) * C + c;
            int d = idx[oidx];
            if ((d >> 1) == (ih & 1) && (d & 1) == (iw & 1)) v = dy[oidx];
        }
        dx[i] = v;
    }
}

// Fused ReLU-mask + bias grad: dy' = dy * (y > 0), db[k] = sum dy'[., k].
// Row-major traversal (thread = (channel, row-lane) like the BN partials:
// consecutive threads touch consecutive channels => coalesced reads AND
// writes), per-block LDS lane reduction, one atomic per channel per block.
__global__ void relu_bias_bwd_kernel(const unsigned short* __restrict__ dy,
                                     const unsigned short* __restrict__ y,
                                     unsigned short* __restrict__ dym,
                                     float* __restrict__ db, int64_t M, int K,
                                     int rows_per_block) {
    __shared__ float red[256];
    const int64_t r0 = (int64_t)blockIdx.x * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    if (K >= (int)blockDim.x) {
        for (int c = threadIdx.x; c < K; c += blockDim.x) {
            float acc = 0.f;
            for (int64_t r = r0; r < r1; ++r) {
                int64_t i = r * K + c;
                unsigned short yv = y[i];
                unsigned short g =
                    ((yv & 0x7fffu) != 0 && !(yv & 0x8000u)) ? dy[i] : 0;
                dym[i] = g;
                acc += bf2f(g);
            }
            if (gridDim.x == 1) db[c] = acc;  // single block: direct store
            else atomicAdd(db + c, acc);
        }
        return;
    }
    int lanes = (int)blockDim.x / K;
    lanes = 1 << (31 - __clz(lanes));
    const int c = threadIdx.x % K;
    const int rl = threadIdx.x / K;
    float acc = 0.f;
    if (rl < lanes) {
        for (int64_t r = r0 + rl; r < r1; r += lanes) {
            int64_t i = r * K + c;
            unsigned short yv = y[i];
            unsigned short g =
                ((yv & 0x7fffu) != 0 && !(yv & 0x8000u)) ? dy[i] : 0;
            dym[i] = g;
            acc += bf2f(g);
        }
    }
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int off = lanes >> 1; off >= 1; off >>= 1) {
        if (rl < off) red[threadIdx.x] += red[threadIdx.x + off * K];
        __syncthreads();
    }
    if (rl == 0) {
        if (gridDim.x == 1) db[c] = red[threadIdx.x];
        else atomicAdd(db + c, red[threadIdx.x]);
    }
}

// Octet maxpool2x2 forward: one thread per (pixel, 8-channel group), four
// 16-B x loads + one 16-B y store + one 8-B packed idx store. The scalar
// form was load-instruction bound ~4x off the HBM roofline.
__global__ void maxpool_fwd_oct_kernel(const unsigned short* __restrict__ x,
                                       unsigned short* __restrict__ y,
                                       uint8_t* __restrict__ idx,
                                       int64_t total8, int H, int W, int C,
                                       int OH, int OW, FastDiv fOct,
                                       FastDiv fOW, FastDiv fOH) {
    const int noct = C >> 3;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total8; i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t = fdiv((unsigned)i, fOct);
        const int oct = (int)((unsigned)i - t * noct);
        unsigned t2 = fdiv(t, fOW);
        const int ow = (int)(t - t2 * OW);
        unsigned t3 = fdiv(t2, fOH);
        const int oh = (int)(t2 - t3 * OH);
        const int n = (int)t3;
        const int64_t r0 = ((((int64_t)n * H + 2 * oh) * W) + 2 * ow) * C
                           + oct * 8;
        const u16x8 a = *reinterpret_cast<const u16x8*>(&x[r0]);
        const u16x8 b = *reinterpret_cast<const u16x8*>(&x[r0 + C]);
        const u16x8 c8 = *reinterpret_cast<const u16x8*>(&x[r0 + (int64_t)W * C]);
        const u16x8 d = *reinterpret_cast<const u16x8*>(&x[r0 + (int64_t)W * C + C]);
        u16x8 best;
        uint64_t packed = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float bv = bf2f(a[j]);
            int bi = 0;
            const float v1 = bf2f(b[j]), v2 = bf2f(c8[j]), v3 = bf2f(d[j]);
            if (v1 > bv) { bv = v1; bi = 1; }
            if (v2 > bv) { bv = v2; bi = 2; }
            if (v3 > bv) { bv = v3; bi = 3; }
            best[j] = f2bf(bv);
            packed |= (uint64_t)bi << (8 * j);
        }
        const int64_t oo = (int64_t)t * C + oct * 8;
        *reinterpret_cast<u16x8*>(&y[oo]) = best;
        *reinterpret_cast<uint64_t*>(idx + oo) = packed;
    }
}

// 2x2-block trunk pool backward: one thread per (input 2x2 block, 8-channel
// group). The ReLU gate uses the POOLED output p (= y[argmax]; relu ran
// before pool, so p > 0 iff the argmax cell was active) — the full pre-pool
// activation y is neither read nor saved for backward anymore. dy/idx/p are
// read ONCE per output cell (the row-per-thread form re-read them 4x).
__global__ void pool_relu_bias_bwd2_kernel(
    const unsigned short* __restrict__ dy, const uint8_t* __restrict__ idx,
    const unsigned short* __restrict__ p, unsigned short* __restrict__ dym,
    float* __restrict__ db, int64_t total8, int K, int H, int W, int OH,
    int OW, int HB, int WB, FastDiv fOct, FastDiv fWB, FastDiv fHB) {
    extern __shared__ float dbs[];  // [K]
    const int noct = K >> 3;
    for (int c = threadIdx.x; c < K; c += blockDim.x) dbs[c] = 0.f;
    __syncthreads();
    float acc[8] = {0.f};
    int my_oct = -1;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total8; i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t = fdiv((unsigned)i, fOct);
        const int oct = (int)((unsigned)i - t * noct);
        my_oct = oct;  // constant per thread: stride % noct == 0 (noct pow2)
        unsigned t2 = fdiv(t, fWB);
        const int wb = (int)(t - t2 * WB);
        unsigned t3 = fdiv(t2, fHB);
        const int hb = (int)(t2 - t3 * HB);
        const int n = (int)t3;
        const int ih = 2 * hb, iw = 2 * wb;
        // four separate vectors with predicated selects: a g[pos][j] array
        // with a RUNTIME pos index would force the whole array to scratch
        // memory (dynamic indexing defeats register allocation)
        u16x8 g0 = {}, g1 = {}, g2 = {}, g3 = {};
        if (hb < OH && wb < OW) {
            const int64_t o = (((int64_t)n * OH + hb) * OW + wb) * K + oct * 8;
            const u16x8 d8 = *reinterpret_cast<const u16x8*>(&dy[o]);
            const u16x8 p8 = *reinterpret_cast<const u16x8*>(&p[o]);
            const uint64_t i8 = *reinterpret_cast<const uint64_t*>(idx + o);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const bool act = (p8[j] & 0x7fffu) != 0 && !(p8[j] & 0x8000u);
                const int pos = (int)((i8 >> (8 * j)) & 3);
                const unsigned short dv = act ? d8[j] : (unsigned short)0;
                if (act) acc[j] += bf2f(d8[j]);
                g0[j] = pos == 0 ? dv : (unsigned short)0;
                g1[j] = pos == 1 ? dv : (unsigned short)0;
                g2[j] = pos == 2 ? dv : (unsigned short)0;
                g3[j] = pos == 3 ? dv : (unsigned short)0;
            }
        }
        const int64_t r0 = (((int64_t)n * H + ih) * W + iw) * K + oct * 8;
        *reinterpret_cast<u16x8*>(&dym[r0]) = g0;
        if (iw + 1 < W)
            *reinterpret_cast<u16x8*>(&dym[r0 + K]) = g1;
        if (ih + 1 < H) {
            *reinterpret_cast<u16x8*>(&dym[r0 + (int64_t)W * K]) = g2;
            if (iw + 1 < W)
                *reinterpret_cast<u16x8*>(&dym[r0 + (int64_t)W * K + K]) = g3;
        }
    }
    if (my_oct >= 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
            if (acc[j] != 0.f) atomicAdd(dbs + my_oct * 8 + j, acc[j]);
    }
    __syncthreads();
    if (gridDim.x == 1) {  // small layers: direct store, db needs no zero-fill
        for (int c = threadIdx.x; c < K; c += blockDim.x) db[c] = dbs[c];
    } else {
        for (int c = threadIdx.x; c < K; c += blockDim.x)
            if (dbs[c] != 0.f) atomicAdd(db + c, dbs[c]);
    }
}

// Scalar fallback for K % 8 != 0 trunk layers// Scalar fallback for K % 8 != 0 trunk layers (LeNet-5's 6/16 filters):
// thread = (channel, row-lane) so each thread's bias partial stays in a
// register (naive per-element global atomics serialized on K=6 addresses
// and cost config #3 ~40%); ReLU gate on the POOLED output p (= the
// argmax cell's activation; relu ran before pool).
__global__ void pool_relu_bias_bwd_scalar_kernel(
    const unsigned short* __restrict__ dy, const uint8_t* __restrict__ idx,
    const unsigned short* __restrict__ p, unsigned short* __restrict__ dym,
    float* __restrict__ db, int64_t M, int K, int rows_per_block, int H,
    int W, int OH, int OW, FastDiv fW, FastDiv fH) {
    __shared__ float red[256];
    const int64_t r0 = (int64_t)blockIdx.x * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    int lanes = (int)blockDim.x / K;
    lanes = 1 << (31 - __clz(lanes));
    const int c = threadIdx.x % K;
    const int rl = threadIdx.x / K;
    float acc = 0.f;
    if (rl < lanes) {
        for (int64_t r = r0 + rl; r < r1; r += lanes) {
            const unsigned t1 = fdiv((unsigned)r, fW);
            const int iw = (int)((unsigned)r - t1 * W);
            const unsigned t2 = fdiv(t1, fH);
            const int ih = (int)(t1 - t2 * H);
            const int oh = ih >> 1, ow = iw >> 1;
            unsigned short g = 0;
            if (oh < OH && ow < OW) {
                const int64_t o = (((int64_t)t2 * OH + oh) * OW + ow) * K + c;
                const int d = idx[o];
                const unsigned short pv = p[o];
                if ((d >> 1) == (ih & 1) && (d & 1) == (iw & 1) &&
                    (pv & 0x7fffu) != 0 && !(pv & 0x8000u))
                    g = dy[o];
            }
            dym[r * K + c] = g;
            acc += bf2f(g);
        }
    }
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int off = lanes >> 1; off >= 1; off >>= 1) {
        if (rl < off) red[threadIdx.x] += red[threadIdx.x + off * K];
        __syncthreads();
    }
    if (rl == 0) {
        if (gridDim.x == 1) db[c] = red[threadIdx.x];
        else atomicAdd(db + c, red[threadIdx.x]);
    }
}


// ---------------------------------------------------------------------------
// Fused backward of the 2-layer dense head (Dense-relu -> Dense-logits) at
// M <= 32: dW2, db2, dh1 (relu-gated), dW1, db1 and dx in ONE single-
// workgroup launch. The six separate launches it replaces (2 gemm dgrads,
// 2 gemm wgrads, relu+bias bwd, bias grad) ran at the ~5-7 us replay floor
// each = ~34 us of the ~146 us config #2 step; every GEMM here has
// K = 32 (the batch), so each 16x16 MFMA tile is a single instruction and
// results stream straight to HBM with no accumulator carry.
// ---------------------------------------------------------------------------

// Fused Dense(relu) -> Dense(logits) FORWARD, row-parallel: one block per
// sample row; each row's logits depend only on its own h1, so there is no
// cross-block dependency (the single-workgroup fused BACKWARD measured
// -11%; the forward is ~1/3 of that work and parallelizes over M). x row
// staged in LDS, h1 kept in LDS for the second layer, h1 also stored
// (bf16) for the composed backward. Replaces two ~7 us linear_splitk
// launches per step on the 2-dense heads (cnn2).
__global__ void dense_head2_fwd_kernel(const unsigned short* __restrict__ x,
                                       const unsigned short* __restrict__ w1,
                                       const float* __restrict__ b1,
                                       const unsigned short* __restrict__ w2,
                                       const float* __restrict__ b2,
                                       unsigned short* __restrict__ h1,
                                       unsigned short* __restrict__ logits,
                                       int M, int K, int N1, int N2) {
    extern __shared__ unsigned short lds[];          // [K] x row (bf16)
    float* hbuf = reinterpret_cast<float*>(lds + ((K + 7) & ~7));  // [N1]
    const int r = blockIdx.x;
    if (r >= M) return;
    for (int k = threadIdx.x * 8; k < K; k += blockDim.x * 8)
        *reinterpret_cast<u16x8*>(&lds[k]) =
            *reinterpret_cast<const u16x8*>(&x[(int64_t)r * K + k]);
    __syncthreads();
    for (int j = threadIdx.x; j < N1; j += blockDim.x) {
        float acc = b1[j];
        const unsigned short* wr = w1 + (int64_t)j * K;
        for (int k = 0; k < K; k += 8) {
            u16x8 xv = *reinterpret_cast<const u16x8*>(&lds[k]);
            u16x8 wv = *reinterpret_cast<const u16x8*>(&wr[k]);
#pragma unroll
            for (int t = 0; t < 8; ++t) acc += bf2f(xv[t]) * bf2f(wv[t]);
        }
        acc = acc > 0.f ? acc : 0.f;                 // fused ReLU
        h1[(int64_t)r * N1 + j] = f2bf(acc);
        hbuf[j] = acc;
    }
    __syncthreads();
    for (int j = threadIdx.x; j < N2; j += blockDim.x) {
        float acc = b2[j];
        const unsigned short* wr = w2 + (int64_t)j * N1;
        for (int i = 0; i < N1; ++i) acc += hbuf[i] * bf2f(wr[i]);
        logits[(int64_t)r * N2 + j] = f2bf(acc);
    }
}


__global__ void __launch_bounds__(TPB)
dense_head2_bwd_kernel(const unsigned short* __restrict__ dl_g,  // [M, N2]
                       const unsigned short* __restrict__ x,     // [M, K]
                       const unsigned short* __restrict__ h1g,   // [M, N1]
                       const unsigned short* __restrict__ w1,    // [N1, K]
                       const unsigned short* __restrict__ w2,    // [N2, N1]
                       unsigned short* __restrict__ dx,          // [M, K]
                       float* __restrict__ dw1,                  // [N1, K]
                       float* __restrict__ db1,                  // [N1]
                       float* __restrict__ dw2,                  // [N2, N1]
                       float* __restrict__ db2,                  // [N2]
                       int M, int K, int N1, int N2) {
    constexpr int MP = 32;   // padded batch rows (mfma K-depth)
    constexpr int NP = 32;   // padded N2 (k-depth of the dh1 product)
    extern __shared__ unsigned short xs[];        // [MP][K]
    __shared__ unsigned short dl[MP][NP];         // dlogits, zero-padded
    __shared__ unsigned short h1s[MP][128];       // post-relu activations
    __shared__ unsigned short dh[MP][128];        // gated dh1
    __shared__ unsigned short w2s[NP][128];       // w2, zero-padded rows
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int half = lane >> 4, sub = lane & 15;
    // ---- stage (zero-padding all tails) ----
    for (int i = tid; i < MP * NP; i += TPB) {
        const int m = i / NP, c = i % NP;
        dl[m][c] = (m < M && c < N2) ? dl_g[m * N2 + c] : (unsigned short)0;
    }
    for (int i = tid; i < MP * N1; i += TPB) {
        const int m = i / N1, c = i % N1;
        h1s[m][c] = m < M ? h1g[m * N1 + c] : (unsigned short)0;
    }
    for (int i = tid; i < NP * N1; i += TPB) {
        const int r = i / N1, c = i % N1;
        w2s[r][c] = r < N2 ? w2[r * N1 + c] : (unsigned short)0;
    }
    for (int i = tid; i < MP * (K / 8); i += TPB) {
        const int m = i / (K / 8), c8 = (i % (K / 8)) * 8;
        u16x8 v = {};
        if (m < M) v = *reinterpret_cast<const u16x8*>(&x[(int64_t)m * K + c8]);
        *reinterpret_cast<u16x8*>(&xs[m * K + c8]) = v;
    }
    __syncthreads();
    // ---- phase A: dh1 = relu-gate(dlogits @ w2), into LDS ----
    {
        const int ntile = (MP / 16) * (N1 / 16);
        for (int t = wave; t < ntile; t += TPB / 64) {
            const int rb = (t / (N1 / 16)) * 16;   // m rows
            const int cb = (t % (N1 / 16)) * 16;   // n1 cols
            bf16x8 a, b;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                a[j] = *reinterpret_cast<const bf16_t*>(
                    &dl[rb + sub][half * 8 + j]);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                b[j] = *reinterpret_cast<const bf16_t*>(
                    &w2s[half * 8 + j][cb + sub]);
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = rb + half * 4 + r, c = cb + sub;
                const unsigned short hv = h1s[m][c];
                const bool act = (hv & 0x7fffu) != 0 && !(hv & 0x8000u);
                dh[m][c] = act ? f2bf(acc[r]) : (unsigned short)0;
            }
        }
    }
    __syncthreads();
    // ---- phase B: independent products, wave-strided tile list ----
    const int t_dw2 = N1 / 16;                    // dW2 tiles (N2 fits 16)
    const int t_dw1 = (N1 / 16) * (K / 16);
    const int t_dx = (MP / 16) * (K / 16);
    const int total = t_dw2 + t_dw1 + t_dx;
    for (int t = wave; t < total; t += TPB / 64) {
        bf16x8 a, b;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        if (t < t_dw2) {
            // dW2[n2, n1] = sum_m dl[m][n2] * h1[m][n1]
            const int cb = t * 16;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                a[j] = *reinterpret_cast<const bf16_t*>(
                    &dl[half * 8 + j][sub]);
                b[j] = *reinterpret_cast<const bf16_t*>(
                    &h1s[half * 8 + j][cb + sub]);
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int n2 = half * 4 + r;
                if (n2 < N2) dw2[n2 * N1 + cb + sub] = acc[r];
            }
        } else if (t < t_dw2 + t_dw1) {
            // dW1[n1, k] = sum_m dh[m][n1] * x[m][k]
            const int tt = t - t_dw2;
            const int rb = (tt / (K / 16)) * 16;
            const int cb = (tt % (K / 16)) * 16;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                a[j] = *reinterpret_cast<const bf16_t*>(
                    &dh[half * 8 + j][rb + sub]);
                b[j] = *reinterpret_cast<const bf16_t*>(
                    &xs[(half * 8 + j) * K + cb + sub]);
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
            for (int r = 0; r < 4; ++r)
                dw1[(int64_t)(rb + half * 4 + r) * K + cb + sub] = acc[r];
        } else {
            // dx[m, k] = sum_n1 dh[m][n1] * w1[n1][k]  (N1/32 k-steps)
            const int tt = t - t_dw2 - t_dw1;
            const int rb = (tt / (K / 16)) * 16;
            const int cb = (tt % (K / 16)) * 16;
            for (int kk = 0; kk < N1; kk += 32) {
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    a[j] = *reinterpret_cast<const bf16_t*>(
                        &dh[rb + sub][kk + half * 8 + j]);
                    b[j] = *reinterpret_cast<const bf16_t*>(
                        &w1[(int64_t)(kk + half * 8 + j) * K + cb + sub]);
                }
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc,
                                                              0, 0, 0);
            }
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int m = rb + half * 4 + r;
                if (m < M) dx[(int64_t)m * K + cb + sub] = f2bf(acc[r]);
            }
        }
    }
    // ---- bias grads (tiny column sums from LDS) ----
    if (tid < N1) {
        float s = 0.f;
        for (int m = 0; m < M; ++m) s += bf2f(dh[m][tid]);
        db1[tid] = s;
    } else if (tid >= 128 && tid - 128 < N2) {
        const int c = tid - 128;
        float s = 0.f;
        for (int m = 0; m < M; ++m) s += bf2f(dl[m][c]);
        db2[c] = s;
    }
}

// ---------------------------------------------------------------------------
// Fused softmax + categorical cross-entropy (mean), one wave per row.
// logits bf16 [M, C]; probs f32 out; loss = sum(-log p[label]) / M.
// ---------------------------------------------------------------------------

// Also fuses the training statistics: when acc_loss/acc_correct are given,
// the kernel accumulates mean loss and argmax==label counts into those
// persistent buffers — replacing a ~7-kernel torch chain per step.
// SINGLE_BLOCK: grid(1) with an intra-block reduction — loss is written
// directly (no zero-fill + atomics); rows strided over the block's waves.
template <bool SINGLE_BLOCK>
__global__ void softmax_xent_fwd_kernel(const unsigned short* __restrict__ logits,
                                        const int64_t* __restrict__ labels,
                                        float* __restrict__ probs,
                                        float* __restrict__ loss, int M, int C,
                                        float* __restrict__ acc_loss,
                                        float* __restrict__ acc_correct) {
    __shared__ float lred[2][8];
    const int lane = threadIdx.x & 63;
    const int nw = blockDim.x / 64;
    float my_loss = 0.f, my_corr = 0.f;
    for (int row = SINGLE_BLOCK ? (threadIdx.x >> 6)
                                : blockIdx.x * nw + (threadIdx.x >> 6);
         row < M; row += SINGLE_BLOCK ? nw : gridDim.x * nw) {
    const unsigned short* lr = logits + (int64_t)row * C;
    float mx = -3.4e38f;
    int arg = 0;
    for (int c = lane; c < C; c += 64) {
        float v = bf2f(lr[c]);
        if (v > mx) { mx = v; arg = c; }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        float omx = __shfl_xor(mx, off, 64);
        int oarg = __shfl_xor(arg, off, 64);
        if (omx > mx || (omx == mx && oarg < arg)) { mx = omx; arg = oarg; }
    }
    float sum = 0.f;
    for (int c = lane; c < C; c += 64) sum += __expf(bf2f(lr[c]) - mx);
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) sum += __shfl_xor(sum, off, 64);
    float inv = 1.f / sum;
    for (int c = lane; c < C; c += 64)
        probs[(int64_t)row * C + c] = __expf(bf2f(lr[c]) - mx) * inv;
    if (lane == 0) {
        int64_t lab = labels[row];
        float p = __expf(bf2f(lr[lab]) - mx) * inv;
        float l = -__logf(fmaxf(p, 1e-30f)) / M;
        if (SINGLE_BLOCK) {
            my_loss += l;
            if (arg == (int)lab) my_corr += 1.f;
        } else {
            atomicAdd(loss, l);
            if (acc_loss) atomicAdd(acc_loss, l);
            if (acc_correct && arg == (int)lab) atomicAdd(acc_correct, 1.f);
        }
    }
    }  // row loop
    if (SINGLE_BLOCK) {
        const int w = threadIdx.x >> 6;
        if (lane == 0) {
            lred[0][w] = my_loss;
            lred[1][w] = my_corr;
        }
        __syncthreads();
        if (threadIdx.x == 0) {
            float L = 0.f, Cc = 0.f;
            for (int i = 0; i < nw; ++i) {
                L += lred[0][i];
                Cc += lred[1][i];
            }
            loss[0] = L;                      // direct write: no zero-fill
            if (acc_loss) atomicAdd(acc_loss, L);
            if (acc_correct) atomicAdd(acc_correct, Cc);
        }
    }
}

// dloss is a device scalar (graph-capture safe: no host readback of the
// upstream gradient); scale = dloss / M.
template <bool OUT_BF16>
__global__ void softmax_xent_bwd_kernel(const float* __restrict__ probs,
                                        const int64_t* __restrict__ labels,
                                        const float* __restrict__ dloss,
                                        void* __restrict__ dlogits, int64_t M,
                                        int C) {
    const float scale = dloss[0] / (float)M;
    int64_t total = M * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        int64_t row = i / C;
        int c = i % C;
        float g = (probs[i] - (labels[row] == c ? 1.f : 0.f)) * scale;
        if (OUT_BF16)
            reinterpret_cast<unsigned short*>(dlogits)[i] = f2bf(g);
        else
            reinterpret_cast<float*>(dlogits)[i] = g;
    }
}

// ---------------------------------------------------------------------------
// Fused Adam (fp32 params/grads/states, Keras-decay lr passed pre-computed).
// ---------------------------------------------------------------------------

__global__ void fused_adam_kernel(float* __restrict__ p,
                                  const float* __restrict__ g,
                                  float* __restrict__ m, float* __restrict__ v,
                                  int64_t total, float lr, float b1, float b2,
                                  float eps, float bc1, float bc2) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        float gi = g[i];
        float mi = b1 * m[i] + (1.f - b1) * gi;
        float vi = b2 * v[i] + (1.f - b2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        p[i] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
    }
}

// hipGraph-capturable Adam: the per-step schedule {lr_t, bc1, bc2} lives in a
// device buffer advanced by adam_prep_kernel INSIDE the graph — replays need
// no host-side scalar updates.
__global__ void adam_prep_kernel(int64_t* __restrict__ step,
                                 float* __restrict__ sched,
                                 const float* __restrict__ hyper,  // {lr, decay}
                                 float b1, float b2) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        int64_t t = ++step[0];
        // hyper lives in a device buffer so lr changes (ReduceLROnPlateau)
        // reach a captured graph without re-capture
        sched[0] = hyper[0] / (1.f + hyper[1] * (float)(t - 1));  // Keras decay
        sched[1] = 1.f - powf(b1, (float)t);
        sched[2] = 1.f - powf(b2, (float)t);
    }
}

__global__ void adam_prep_epoch_kernel(int64_t* __restrict__ step,
                                       float* __restrict__ sched,
                                       const float* __restrict__ hyper,
                                       float b1, float b2, int S) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        int64_t base = step[0];
        float b1p = powf(b1, (float)base);
        float b2p = powf(b2, (float)base);
        for (int s = 0; s < S; ++s) {
            int64_t t = base + 1 + s;
            b1p *= b1;
            b2p *= b2;
            sched[s * 3 + 0] =
                hyper[0] / (1.f + hyper[1] * (float)(t - 1));  // Keras decay
            sched[s * 3 + 1] = 1.f - b1p;
            sched[s * 3 + 2] = 1.f - b2p;
        }
        step[0] = base + S;
    }
}


// ---------------------------------------------------------------------------
// Skinny linear fwd (M ~ batch 32): one wave per (16x16 tile, K-chunk),
// fragments loaded STRAIGHT from global (both operands contiguous along K,
// L2-resident at these sizes), fp32 atomics into y32, tiny epilogue kernel.
// Replaces the LDS-staged path that was latency-bound at 4 workgroups.
// ---------------------------------------------------------------------------

template <int DIRECT>  // DIRECT: one slab -> write bf16 y with bias/relu
__global__ void linear_splitk_kernel(const unsigned short* __restrict__ x,
                                     const unsigned short* __restrict__ w,
                                     float* __restrict__ y32, int M, int N,
                                     int K, int kc_len,
                                     const float* __restrict__ bias,
                                     unsigned short* __restrict__ yout,
                                     int relu) {
    const int lane = threadIdx.x & 63;
    const int m0 = blockIdx.x * 16;
    const int n0 = blockIdx.y * 16;
    const int k0 = blockIdx.z * kc_len;
    const int kend = min(k0 + kc_len, K);
    const int sub = lane & 15, half = lane >> 4;
    const int row = m0 + sub;         // A row
    const int col = n0 + sub;         // B col (w row)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    if ((K & 31) == 0 && m0 + 16 <= M && n0 + 16 <= N) {
        // aligned full-tile fast path: hoisted bounds, unrolled so the
        // compiler keeps several 16-B loads in flight ahead of the MFMAs
        // (the checked loop serialized ~300 ns global latency per k-step).
        // The guard is WAVE-UNIFORM (m0/n0, not row/col): MFMA is a
        // wave-level op, so mixed fast/slow lanes would execute separate
        // MFMAs with garbage operands in the inactive lanes.
        const unsigned short* xr = x + (int64_t)row * K + half * 8;
        const unsigned short* wr = w + (int64_t)col * K + half * 8;
#pragma unroll 4
        for (int k = k0; k < kend; k += 32) {
            bf16x8 a = *reinterpret_cast<const bf16x8*>(xr + k);
            bf16x8 b = *reinterpret_cast<const bf16x8*>(wr + k);
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
        }
    } else
    for (int k = k0; k < kend; k += 32) {
        bf16x8 a{}, b{};
        const int kk = k + half * 8;
        if (row < M) {
            if (kk + 8 <= K && (kk & 7) == 0)
                a = *reinterpret_cast<const bf16x8*>(&x[(int64_t)row * K + kk]);
            else
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    a[j] = (kk + j < K) ? *reinterpret_cast<const bf16_t*>(
                                              &x[(int64_t)row * K + kk + j])
                                        : (bf16_t)0.f;
        }
        if (col < N) {
            if (kk + 8 <= K && (kk & 7) == 0)
                b = *reinterpret_cast<const bf16x8*>(&w[(int64_t)col * K + kk]);
            else
#pragma unroll
                for (int j = 0; j < 8; ++j)
                    b[j] = (kk + j < K) ? *reinterpret_cast<const bf16_t*>(
                                              &w[(int64_t)col * K + kk + j])
                                        : (bf16_t)0.f;
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    if (DIRECT) {  // single slab: fused bias + relu + bf16 store, no epilogue
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            int orow = m0 + (lane >> 4) * 4 + r;
            int ocol = n0 + (lane & 15);
            if (orow < M && ocol < N) {
                float v = acc[r] + (bias ? bias[ocol] : 0.f);
                if (relu) v = v > 0.f ? v : 0.f;
                yout[(int64_t)orow * N + ocol] = f2bf(v);
            }
        }
        return;
    }
    // slab write [kc][M][N]: no zero-init, no atomics; epilogue sums slabs
    float* slab = y32 + (int64_t)blockIdx.z * M * N;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        int orow = m0 + (lane >> 4) * 4 + r;
        int ocol = n0 + (lane & 15);
        if (orow < M && ocol < N)
            slab[(int64_t)orow * N + ocol] = acc[r];
    }
}

// clear=1: consume-and-clear of a pooled split-K accumulator (see
// cast_f32_bf16_kernel / acc_pool).
__global__ void linear_epilogue_kernel(float* __restrict__ y32,
                                       const float* __restrict__ bias,
                                       unsigned short* __restrict__ y,
                                       int64_t total, int N, int relu,
                                       int slabs, int clear) {
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        float v = bias ? bias[i % N] : 0.f;
        for (int sl = 0; sl < slabs; ++sl) v += y32[sl * total + i];
        if (clear)
            for (int sl = 0; sl < slabs; ++sl) y32[sl * total + i] = 0.f;
        if (relu) v = v > 0.f ? v : 0.f;
        y[i] = f2bf(v);
    }
}

// ---------------------------------------------------------------------------
// Multi-tensor fused Adam + multi-tensor grad zero: ONE launch over all
// parameter tensors via a precomputed chunk table (built once per optimizer;
// hipGraph replays touch no host state).
// meta: int64 [nchunks][2] = {tensor index, element offset};
// ptrs: int64 [T][4] = {p, g, m, v} device addresses; sizes: int64 [T].
// ---------------------------------------------------------------------------

constexpr int MT_CHUNK = 2048;

__global__ void fused_adam_mt_kernel(const int64_t* __restrict__ meta,
                                     const int64_t* __restrict__ ptrs,
                                     const int64_t* __restrict__ sizes,
                                     const float* __restrict__ sched, float b1,
                                     float b2, float eps, int zero_g,
                                     int sched_off) {
    sched += 3 * sched_off;
    const int c = blockIdx.x;
    const int t = (int)meta[c * 2];
    const int64_t off = meta[c * 2 + 1];
    float* p = reinterpret_cast<float*>(ptrs[t * 5 + 0]);
    float* g = reinterpret_cast<float*>(ptrs[t * 5 + 1]);
    float* m = reinterpret_cast<float*>(ptrs[t * 5 + 2]);
    float* v = reinterpret_cast<float*>(ptrs[t * 5 + 3]);
    // bf16 shadow of the fp32 master (what conv/linear forward reads):
    // refreshed here so no per-step cast kernels exist
    unsigned short* sh = reinterpret_cast<unsigned short*>(ptrs[t * 5 + 4]);
    const int64_t n = min(off + (int64_t)MT_CHUNK, sizes[t]);
    const float lr = sched[0], bc1 = sched[1], bc2 = sched[2];
    for (int64_t i = off + threadIdx.x; i < n; i += blockDim.x) {
        float gi = g[i];
        float mi = b1 * m[i] + (1.f - b1) * gi;
        float vi = b2 * v[i] + (1.f - b2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        float pv = p[i] - lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
        p[i] = pv;
        if (sh) sh[i] = f2bf(pv);
        // zero_g: consume-and-clear so the next captured step's backward
        // accumulates into zeroed buffers (epoch-graph mode: one pointer
        // table serves every step in the graph)
        if (zero_g) g[i] = 0.f;
    }
}

// Multi-tensor weight pack/unpack: the per-round FedAvg weight transport
// (flat fp32 vector <-> parameter tensors + bf16 shadows) as ONE kernel
// instead of ~3 per parameter (cat / per-param copy_ / shadow cast).
// meta: [nchunk][2] = {tensor idx, elem offset}; offs[t] = flat offset.
__global__ void pack_mt_kernel(const int64_t* __restrict__ meta,
                               const int64_t* __restrict__ ptrs,
                               const int64_t* __restrict__ sizes,
                               const int64_t* __restrict__ offs,
                               float* __restrict__ flat) {
    const int c = blockIdx.x;
    const int t = (int)meta[c * 2];
    const int64_t off = meta[c * 2 + 1];
    const float* p = reinterpret_cast<const float*>(ptrs[t]);
    const int64_t n = min(off + (int64_t)MT_CHUNK, sizes[t]);
    const int64_t base = offs[t];
    for (int64_t i = off + threadIdx.x; i < n; i += blockDim.x)
        flat[base + i] = p[i];
}

__global__ void unpack_mt_kernel(const float* __restrict__ flat,
                                 const int64_t* __restrict__ meta,
                                 const int64_t* __restrict__ ptrs,
                                 const int64_t* __restrict__ shptrs,
                                 const int64_t* __restrict__ sizes,
                                 const int64_t* __restrict__ offs) {
    const int c = blockIdx.x;
    const int t = (int)meta[c * 2];
    const int64_t off = meta[c * 2 + 1];
    float* p = reinterpret_cast<float*>(ptrs[t]);
    unsigned short* sh = reinterpret_cast<unsigned short*>(shptrs[t]);
    const int64_t n = min(off + (int64_t)MT_CHUNK, sizes[t]);
    const int64_t base = offs[t];
    for (int64_t i = off + threadIdx.x; i < n; i += blockDim.x) {
        float v = flat[base + i];
        p[i] = v;
        if (sh) sh[i] = f2bf(v);  // bf16 shadow refreshed in the same pass
    }
}


// ---------------------------------------------------------------------------
// BatchNorm (NHWC, per-channel over N*H*W) — ResNet-18 (config #5) support.
// ---------------------------------------------------------------------------

// Coalesced two-stage reduction: stage 1 blocks each cover a row-chunk x
// all C channels (consecutive threads -> consecutive channels), partials
// atomically added into fp32 accumulators; stage 2 finalizes mean/invstd.
// writes per-block partial sums into slab[b][2][C] (torch::empty — every
// cell is stored exactly once, so no zero-fill or atomics are needed)
// ATOMIC=true: `slab` is a single [2*C] sums buffer (zero at entry) and
// every per-block partial is atomicAdd'ed into it — no finalize launch;
// the consumer (bn_apply_stats_kernel) derives mean/invstd from the raw
// sums and the NEXT bn_bwd_partial<true> clears the buffer (stream-order
// consume-and-clear, same contract as acc_pool).
#define BN_ST(arr, i, v) \
    do { if (ATOMIC) atomicAdd(&(arr)[i], (v)); else (arr)[i] = (v); } while (0)

template <bool ATOMIC>
__global__ void bn_partial_kernel(const unsigned short* __restrict__ x,
                                  float* __restrict__ slab, int64_t M, int C,
                                  int rows_per_block) {
    float* gsum = ATOMIC ? slab : slab + (int64_t)blockIdx.x * 2 * C;
    float* gsq = gsum + C;
    __shared__ float red[2][256];
    const int64_t r0 = (int64_t)blockIdx.x * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    if ((C & 7) == 0 && C <= 2048) {
        // vectorized: thread = (channel octet, row-lane), 16-byte loads,
        // 8 per-channel partials in registers (scalar 2-byte loads left
        // this kernel ~10x off the HBM roofline)
        __shared__ float red8[2][2048];
        const int noct = C >> 3;
        int lanes = (int)blockDim.x / noct;
        lanes = lanes ? (1 << (31 - __clz(lanes))) : 0;
        const int oct = threadIdx.x % noct;
        const int rl = threadIdx.x / noct;
        float acc[8] = {0.f}, acc2[8] = {0.f};
        if (lanes == 0) {  // C > 8*blockDim: strided octets, no lane reduce
            for (int o = threadIdx.x; o < noct; o += blockDim.x) {
                float a[8] = {0.f}, a2[8] = {0.f};
                for (int64_t r = r0; r < r1; ++r) {
                    u16x8 v8 = *reinterpret_cast<const u16x8*>(
                        &x[r * C + o * 8]);
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        float v = bf2f(v8[j]);
                        a[j] += v;
                        a2[j] += v * v;
                    }
                }
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    BN_ST(gsum, o * 8 + j, a[j]);
                    BN_ST(gsq, o * 8 + j, a2[j]);
                }
            }
            return;
        }
        if (rl < lanes) {
            for (int64_t r = r0 + rl; r < r1; r += lanes) {
                u16x8 v8 = *reinterpret_cast<const u16x8*>(&x[r * C + oct * 8]);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float v = bf2f(v8[j]);
                    acc[j] += v;
                    acc2[j] += v * v;
                }
            }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            red8[0][threadIdx.x * 8 + j] = acc[j];
            red8[1][threadIdx.x * 8 + j] = acc2[j];
        }
        __syncthreads();
        for (int off = lanes >> 1; off >= 1; off >>= 1) {
            if (rl < off) {
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    red8[0][threadIdx.x * 8 + j] +=
                        red8[0][(threadIdx.x + off * noct) * 8 + j];
                    red8[1][threadIdx.x * 8 + j] +=
                        red8[1][(threadIdx.x + off * noct) * 8 + j];
                }
            }
            __syncthreads();
        }
        if (rl == 0) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                BN_ST(gsum, oct * 8 + j, red8[0][threadIdx.x * 8 + j]);
                BN_ST(gsq, oct * 8 + j, red8[1][threadIdx.x * 8 + j]);
            }
        }
        return;
    }
    if (C >= (int)blockDim.x) {  // one channel per thread, strided
        for (int c = threadIdx.x; c < C; c += blockDim.x) {
            float acc = 0.f, acc2 = 0.f;
            for (int64_t r = r0; r < r1; ++r) {
                float v = bf2f(x[r * C + c]);
                acc += v;
                acc2 += v * v;
            }
            BN_ST(gsum, c, acc);
            BN_ST(gsq, c, acc2);
        }
        return;
    }
    // thread = (channel, row-lane); LDS-reduce the row-lanes so each block
    // issues ONE atomic per channel (hot-word atomics dominated before).
    // lanes rounded down to a power of two so the tree reduction is exact.
    int lanes = (int)blockDim.x / C;                 // row-parallel lanes
    lanes = 1 << (31 - __clz(lanes));
    const int c = threadIdx.x % C;
    const int rl = threadIdx.x / C;
    float acc = 0.f, acc2 = 0.f;
    if (rl < lanes) {
        for (int64_t r = r0 + rl; r < r1; r += lanes) {
            float v = bf2f(x[r * C + c]);
            acc += v;
            acc2 += v * v;
        }
    }
    red[0][threadIdx.x] = acc;
    red[1][threadIdx.x] = acc2;
    __syncthreads();
    for (int off = lanes >> 1; off >= 1; off >>= 1) {
        if (rl < off) {
            red[0][threadIdx.x] += red[0][threadIdx.x + off * C];
            red[1][threadIdx.x] += red[1][threadIdx.x + off * C];
        }
        __syncthreads();
    }
    if (rl == 0) {
        BN_ST(gsum, c, red[0][threadIdx.x]);
        BN_ST(gsq, c, red[1][threadIdx.x]);
    }
}

// one block per channel: 256 threads tree-reduce the nblk slab rows
// bn_partial octet path + counter-gated LAST-BLOCK finalize: saves the
// separate bn_finalize launch (~4.7 us each, 20 BN layers x fwd/bwd per
// ResNet step). The finalize tail runs in ONE block after a fence; every
// other block pays only fence + one atomic (cheap — unlike the pool
// backward, whose EVERY block paid a barrier tail). counter is a
// persistent zeroed int the tail resets for the next launch.
__global__ void bn_partial_fused_kernel(const unsigned short* __restrict__ x,
                                        float* __restrict__ slab, int64_t M,
                                        int C, int rows_per_block,
                                        float* __restrict__ mean,
                                        float* __restrict__ invstd,
                                        float* __restrict__ running_mean,
                                        float* __restrict__ running_var,
                                        float eps, float momentum,
                                        int* __restrict__ counter) {
    float* gsum = slab + (int64_t)blockIdx.x * 2 * C;
    float* gsq = gsum + C;
    __shared__ float red8[2][2048];
    const int64_t r0 = (int64_t)blockIdx.x * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    const int noct = C >> 3;
    int lanes = (int)blockDim.x / noct;
    lanes = lanes ? (1 << (31 - __clz(lanes))) : 1;
    const int oct = threadIdx.x % noct;
    const int rl = threadIdx.x / noct;
    float acc[8] = {0.f}, acc2[8] = {0.f};
    if (rl < lanes) {
        for (int64_t r = r0 + rl; r < r1; r += lanes) {
            u16x8 v8 = *reinterpret_cast<const u16x8*>(&x[r * C + oct * 8]);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float v = bf2f(v8[j]);
                acc[j] += v;
                acc2[j] += v * v;
            }
        }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
        red8[0][threadIdx.x * 8 + j] = acc[j];
        red8[1][threadIdx.x * 8 + j] = acc2[j];
    }
    __syncthreads();
    for (int off = lanes >> 1; off >= 1; off >>= 1) {
        if (rl < off) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                red8[0][threadIdx.x * 8 + j] +=
                    red8[0][(threadIdx.x + off * noct) * 8 + j];
                red8[1][threadIdx.x * 8 + j] +=
                    red8[1][(threadIdx.x + off * noct) * 8 + j];
            }
        }
        __syncthreads();
    }
    if (rl == 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            gsum[oct * 8 + j] = red8[0][threadIdx.x * 8 + j];
            gsq[oct * 8 + j] = red8[1][threadIdx.x * 8 + j];
        }
    }
    // last arriving block finalizes (slab rows are visible via the fence)
    __shared__ int is_last;
    __threadfence();
    if (threadIdx.x == 0)
        is_last = (atomicAdd(counter, 1) == (int)gridDim.x - 1) ? 1 : 0;
    __syncthreads();
    if (!is_last) return;
    for (int c = threadIdx.x; c < C; c += blockDim.x) {
        float s1 = 0.f, s2 = 0.f;
        for (int b = 0; b < (int)gridDim.x; ++b) {
            s1 += slab[(int64_t)b * 2 * C + c];
            s2 += slab[(int64_t)b * 2 * C + C + c];
        }
        const float mu = s1 / (float)M;
        const float var = fmaxf(s2 / (float)M - mu * mu, 0.f);
        mean[c] = mu;
        invstd[c] = rsqrtf(var + eps);
        if (running_mean) {
            float unb = var * ((float)M / (float)max(M - 1, (int64_t)1));
            running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
            running_var[c] = (1.f - momentum) * running_var[c] + momentum * unb;
        }
    }
    if (threadIdx.x == 0) *counter = 0;  // self-reset for the next launch
}

__global__ void bn_finalize_kernel(const float* __restrict__ slab, int nblk,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var, int64_t M,
                                   int C, float eps, float momentum) {
    __shared__ float red[2][256];
    const int c = blockIdx.x;
    float s1 = 0.f, s2 = 0.f;
    for (int b = threadIdx.x; b < nblk; b += blockDim.x) {
        s1 += slab[(int64_t)b * 2 * C + c];
        s2 += slab[(int64_t)b * 2 * C + C + c];
    }
    red[0][threadIdx.x] = s1;
    red[1][threadIdx.x] = s2;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
        if ((int)threadIdx.x < off) {
            red[0][threadIdx.x] += red[0][threadIdx.x + off];
            red[1][threadIdx.x] += red[1][threadIdx.x + off];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        float mu = red[0][0] / (float)M;
        float var = fmaxf(red[1][0] / (float)M - mu * mu, 0.f);
        mean[c] = mu;
        invstd[c] = rsqrtf(var + eps);
        if (running_mean) {  // torch semantics: UNBIASED var in running_var
            float unb = var * ((float)M / (float)max(M - 1, (int64_t)1));
            running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
            running_var[c] = (1.f - momentum) * running_var[c] + momentum * unb;
        }
    }
}

__global__ void bn_apply_kernel(const unsigned short* __restrict__ x,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                unsigned short* __restrict__ y, int64_t total,
                                int C, int relu, FastDiv fC) {
    if ((C & 7) == 0) {  // octet path: one 16-B load/store per thread.
        // NOTE: the host passes fC built over C/8 (octet count) here.
        const int64_t t8 = total >> 3;
        const int noct = C >> 3;
        for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
             i < t8; i += (int64_t)gridDim.x * blockDim.x) {
            const int c0 = 8 * (int)((unsigned)i - fdiv((unsigned)i, fC) * noct);
            u16x8 x8 = *reinterpret_cast<const u16x8*>(&x[i * 8]);
            u16x8 y8;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = c0 + j;
                float v = (bf2f(x8[j]) - mean[c]) * invstd[c] * gamma[c] +
                          beta[c];
                if (relu) v = v > 0.f ? v : 0.f;
                y8[j] = f2bf(v);
            }
            *reinterpret_cast<u16x8*>(&y[i * 8]) = y8;
        }
        return;
    }
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        int c = (int)((unsigned)i - fdiv((unsigned)i, fC) * C);
        float v = (bf2f(x[i]) - mean[c]) * invstd[c] * gamma[c] + beta[c];
        if (relu) v = v > 0.f ? v : 0.f;
        y[i] = f2bf(v);
    }
}

// bn_apply over ATOMIC raw sums: mean/invstd are derived per element from
// the [2*C] sums written by bn_partial_kernel<true> (2 extra L2-resident
// loads + a rsqrt on a memory-bound pass); block 0 additionally persists
// mean/invstd for the backward and updates the running stats — this
// replaces the bn_finalize launch entirely. Unlike the counter-gated
// last-block probe (bn_partial_fused_kernel, measured -15%), no block
// waits on any other.
__global__ void bn_apply_stats_kernel(const unsigned short* __restrict__ x,
                                      const float* __restrict__ sums,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      unsigned short* __restrict__ y,
                                      int64_t total, int C, int64_t M,
                                      float eps, float momentum,
                                      float* __restrict__ mean_out,
                                      float* __restrict__ invstd_out,
                                      float* __restrict__ running_mean,
                                      float* __restrict__ running_var,
                                      int relu, FastDiv fC) {
    const float invM = 1.f / (float)M;
    if (blockIdx.x == 0) {
        for (int c = threadIdx.x; c < C; c += blockDim.x) {
            const float mu = sums[c] * invM;
            const float var = fmaxf(sums[C + c] * invM - mu * mu, 0.f);
            mean_out[c] = mu;
            invstd_out[c] = rsqrtf(var + eps);
            if (running_mean) {  // torch semantics: UNBIASED running var
                float unb = var * ((float)M / (float)max(M - 1, (int64_t)1));
                running_mean[c] =
                    (1.f - momentum) * running_mean[c] + momentum * mu;
                running_var[c] =
                    (1.f - momentum) * running_var[c] + momentum * unb;
            }
        }
    }
    if ((C & 7) == 0) {  // octet path (fC built over C/8 by the host)
        const int64_t t8 = total >> 3;
        const int noct = C >> 3;
        for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
             i < t8; i += (int64_t)gridDim.x * blockDim.x) {
            const int c0 = 8 * (int)((unsigned)i - fdiv((unsigned)i, fC) * noct);
            u16x8 x8 = *reinterpret_cast<const u16x8*>(&x[i * 8]);
            u16x8 y8;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = c0 + j;
                const float mu = sums[c] * invM;
                const float is =
                    rsqrtf(fmaxf(sums[C + c] * invM - mu * mu, 0.f) + eps);
                float v = (bf2f(x8[j]) - mu) * is * gamma[c] + beta[c];
                if (relu) v = v > 0.f ? v : 0.f;
                y8[j] = f2bf(v);
            }
            *reinterpret_cast<u16x8*>(&y[i * 8]) = y8;
        }
        return;
    }
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        int c = (int)((unsigned)i - fdiv((unsigned)i, fC) * C);
        const float mu = sums[c] * invM;
        const float is =
            rsqrtf(fmaxf(sums[C + c] * invM - mu * mu, 0.f) + eps);
        float v = (bf2f(x[i]) - mu) * is * gamma[c] + beta[c];
        if (relu) v = v > 0.f ? v : 0.f;
        y[i] = f2bf(v);
    }
}

// dgamma[c] = sum dy*xhat; dbeta[c] = sum dy — same coalesced two-stage
// shape (partials straight into dgamma/dbeta, zeroed by the wrapper).
// relu_y: optional post-ReLU BN output — gates dy in place of a separate
// relu_bwd pre-pass (one fewer full read+write of the activation grad)
__device__ __forceinline__ float bf_gated(const unsigned short* dy,
                                          const unsigned short* relu_y,
                                          int64_t i) {
    if (relu_y) {
        unsigned short yv = relu_y[i];
        if ((yv & 0x7fffu) == 0 || (yv & 0x8000u)) return 0.f;
    }
    return bf2f(dy[i]);
}

// ATOMIC=true: `slab` IS the {dgamma;dbeta} [2*C] output (zero at entry —
// the captured Adam consume-and-cleared it last step under the epoch-graph
// contract); block 0 also zeroes this layer's forward sums buffer for the
// next replay's bn_partial<true> (the forward consumer already ran, in
// stream order).
template <bool ATOMIC>
__global__ void bn_bwd_partial_kernel(const unsigned short* __restrict__ dy,
                                      const unsigned short* __restrict__ x,
                                      const unsigned short* __restrict__ relu_y,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ invstd,
                                      float* __restrict__ slab, int64_t M,
                                      int C, int rows_per_block,
                                      float* __restrict__ dbeta_buf,
                                      float* __restrict__ fwd_sums) {
    // ATOMIC: slab IS dgamma [C] and dbeta_buf IS dbeta [C] (two separate
    // tensors — each is stolen into its own p.grad, so they cannot share
    // one allocation). Classic: per-block slab rows, dbeta_buf unused.
    float* dgamma = ATOMIC ? slab : slab + (int64_t)blockIdx.x * 2 * C;
    float* dbeta = ATOMIC ? dbeta_buf : dgamma + C;
    if (ATOMIC && fwd_sums && blockIdx.x == 0)
        for (int i = threadIdx.x; i < 2 * C; i += blockDim.x)
            fwd_sums[i] = 0.f;
    __shared__ float red[2][256];
    const int64_t r0 = (int64_t)blockIdx.x * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    if ((C & 7) == 0 && C <= 2048) {  // vectorized octet path (see fwd)
        __shared__ float red8[2][2048];
        const int noct = C >> 3;
        int lanes = (int)blockDim.x / noct;
        lanes = lanes ? (1 << (31 - __clz(lanes))) : 0;
        const int oct = threadIdx.x % noct;
        const int rl = threadIdx.x / noct;
        if (lanes == 0) {
            for (int o = threadIdx.x; o < noct; o += blockDim.x) {
                float dg[8] = {0.f}, db[8] = {0.f}, mu[8], is[8];
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    mu[j] = mean[o * 8 + j];
                    is[j] = invstd[o * 8 + j];
                }
                for (int64_t r = r0; r < r1; ++r) {
                    u16x8 g8 = *reinterpret_cast<const u16x8*>(
                        &dy[r * C + o * 8]);
                    u16x8 x8 = *reinterpret_cast<const u16x8*>(
                        &x[r * C + o * 8]);
                    u16x8 y8{};
                    if (relu_y)
                        y8 = *reinterpret_cast<const u16x8*>(
                            &relu_y[r * C + o * 8]);
#pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        float g = bf2f(g8[j]);
                        if (relu_y &&
                            ((y8[j] & 0x7fffu) == 0 || (y8[j] & 0x8000u)))
                            g = 0.f;
                        dg[j] += g * (bf2f(x8[j]) - mu[j]) * is[j];
                        db[j] += g;
                    }
                }
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    BN_ST(dgamma, o * 8 + j, dg[j]);
                    BN_ST(dbeta, o * 8 + j, db[j]);
                }
            }
            return;
        }
        float dg[8] = {0.f}, db[8] = {0.f};
        if (rl < lanes) {
            float mu[8], is[8];
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                mu[j] = mean[oct * 8 + j];
                is[j] = invstd[oct * 8 + j];
            }
            for (int64_t r = r0 + rl; r < r1; r += lanes) {
                u16x8 g8 = *reinterpret_cast<const u16x8*>(&dy[r * C + oct * 8]);
                u16x8 x8 = *reinterpret_cast<const u16x8*>(&x[r * C + oct * 8]);
                u16x8 y8{};
                if (relu_y)
                    y8 = *reinterpret_cast<const u16x8*>(
                        &relu_y[r * C + oct * 8]);
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    float g = bf2f(g8[j]);
                    if (relu_y && ((y8[j] & 0x7fffu) == 0 || (y8[j] & 0x8000u)))
                        g = 0.f;
                    dg[j] += g * (bf2f(x8[j]) - mu[j]) * is[j];
                    db[j] += g;
                }
            }
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            red8[0][threadIdx.x * 8 + j] = dg[j];
            red8[1][threadIdx.x * 8 + j] = db[j];
        }
        __syncthreads();
        for (int off = lanes >> 1; off >= 1; off >>= 1) {
            if (rl < off) {
#pragma unroll
                for (int j = 0; j < 8; ++j) {
                    red8[0][threadIdx.x * 8 + j] +=
                        red8[0][(threadIdx.x + off * noct) * 8 + j];
                    red8[1][threadIdx.x * 8 + j] +=
                        red8[1][(threadIdx.x + off * noct) * 8 + j];
                }
            }
            __syncthreads();
        }
        if (rl == 0) {
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                BN_ST(dgamma, oct * 8 + j, red8[0][threadIdx.x * 8 + j]);
                BN_ST(dbeta, oct * 8 + j, red8[1][threadIdx.x * 8 + j]);
            }
        }
        return;
    }
    if (C >= (int)blockDim.x) {
        for (int c = threadIdx.x; c < C; c += blockDim.x) {
            float dg = 0.f, db = 0.f;
            const float mu = mean[c], is = invstd[c];
            for (int64_t r = r0; r < r1; ++r) {
                float g = bf_gated(dy, relu_y, r * C + c);
                dg += g * (bf2f(x[r * C + c]) - mu) * is;
                db += g;
            }
            BN_ST(dgamma, c, dg);
            BN_ST(dbeta, c, db);
        }
        return;
    }
    int lanes = (int)blockDim.x / C;
    lanes = 1 << (31 - __clz(lanes));
    const int c = threadIdx.x % C;
    const int rl = threadIdx.x / C;
    float dg = 0.f, db = 0.f;
    if (rl < lanes) {
        const float mu = mean[c], is = invstd[c];
        for (int64_t r = r0 + rl; r < r1; r += lanes) {
            float g = bf_gated(dy, relu_y, r * C + c);
            dg += g * (bf2f(x[r * C + c]) - mu) * is;
            db += g;
        }
    }
    red[0][threadIdx.x] = dg;
    red[1][threadIdx.x] = db;
    __syncthreads();
    for (int off = lanes >> 1; off >= 1; off >>= 1) {
        if (rl < off) {
            red[0][threadIdx.x] += red[0][threadIdx.x + off * C];
            red[1][threadIdx.x] += red[1][threadIdx.x + off * C];
        }
        __syncthreads();
    }
    if (rl == 0) {
        BN_ST(dgamma, c, red[0][threadIdx.x]);
        BN_ST(dbeta, c, red[1][threadIdx.x]);
    }
}

__global__ void bn_bwd_finalize_kernel(const float* __restrict__ slab,
                                       int nblk, float* __restrict__ dgamma,
                                       float* __restrict__ dbeta, int C) {
    __shared__ float red[2][256];
    const int c = blockIdx.x;
    float dg = 0.f, db = 0.f;
    for (int b = threadIdx.x; b < nblk; b += blockDim.x) {
        dg += slab[(int64_t)b * 2 * C + c];
        db += slab[(int64_t)b * 2 * C + C + c];
    }
    red[0][threadIdx.x] = dg;
    red[1][threadIdx.x] = db;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
        if ((int)threadIdx.x < off) {
            red[0][threadIdx.x] += red[0][threadIdx.x + off];
            red[1][threadIdx.x] += red[1][threadIdx.x + off];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        dgamma[c] = red[0][0];
        dbeta[c] = red[1][0];
    }
}

__global__ void bn_dx_kernel(const unsigned short* __restrict__ dy,
                             const unsigned short* __restrict__ x,
                             const unsigned short* __restrict__ relu_y,
                             const float* __restrict__ mean,
                             const float* __restrict__ invstd,
                             const float* __restrict__ gamma,
                             const float* __restrict__ dgamma,
                             const float* __restrict__ dbeta,
                             unsigned short* __restrict__ dx, int64_t total,
                             int C, int64_t M, int train,
                             FastDiv fC) {
    const float invM = 1.f / (float)M;
    if ((C & 7) == 0) {  // octet path (fC built over C/8 by the host)
        const int64_t t8 = total >> 3;
        const int noct = C >> 3;
        for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
             i < t8; i += (int64_t)gridDim.x * blockDim.x) {
            const int c0 = 8 * (int)((unsigned)i - fdiv((unsigned)i, fC) * noct);
            u16x8 g8 = *reinterpret_cast<const u16x8*>(&dy[i * 8]);
            u16x8 x8 = *reinterpret_cast<const u16x8*>(&x[i * 8]);
            u16x8 y8{};
            if (relu_y) y8 = *reinterpret_cast<const u16x8*>(&relu_y[i * 8]);
            u16x8 o8;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const int c = c0 + j;
                float g = bf2f(g8[j]);
                if (relu_y && ((y8[j] & 0x7fffu) == 0 || (y8[j] & 0x8000u)))
                    g = 0.f;
                float v;
                if (train) {
                    float xh = (bf2f(x8[j]) - mean[c]) * invstd[c];
                    v = gamma[c] * invstd[c] *
                        (g - dbeta[c] * invM - xh * dgamma[c] * invM);
                } else {
                    v = gamma[c] * invstd[c] * g;
                }
                o8[j] = f2bf(v);
            }
            *reinterpret_cast<u16x8*>(&dx[i * 8]) = o8;
        }
        return;
    }
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        int c = (int)((unsigned)i - fdiv((unsigned)i, fC) * C);
        float g = bf_gated(dy, relu_y, i);
        float v;
        if (train) {
            float xh = (bf2f(x[i]) - mean[c]) * invstd[c];
            v = gamma[c] * invstd[c] *
                (g - dbeta[c] * invM - xh * dgamma[c] * invM);
        } else {
            v = gamma[c] * invstd[c] * g;  // frozen stats
        }
        dx[i] = f2bf(v);
    }
}

// ---------------------------------------------------------------------------
// Generic MaxPool k x k, stride s, padding p (ResNet stem: 3x3 s2 p1).
// Overlapping windows => backward accumulates into fp32 with atomics.
// ---------------------------------------------------------------------------

__global__ void maxpool_gen_fwd_kernel(const unsigned short* __restrict__ x,
                                       unsigned short* __restrict__ y,
                                       uint8_t* __restrict__ idx, int N, int H,
                                       int W, int C, int OH, int OW, int k,
                                       int s, int p, FastDiv fC, FastDiv fOW,
                                       FastDiv fOH) {
    int64_t total = (int64_t)N * OH * OW * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t1 = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t1 * C);
        unsigned t2 = fdiv(t1, fOW);
        int ow = (int)(t1 - t2 * OW);
        unsigned t3 = fdiv(t2, fOH);
        int oh = (int)(t2 - t3 * OH);
        int n = (int)t3;
        float best = -3.4e38f;
        int bi = 0;
        int dh = -1, dwc = k;  // incremental d/k, d%k
        for (int d = 0; d < k * k; ++d) {
            if (++dwc >= k) { dwc = 0; ++dh; }
            int ih = oh * s + dh - p, iw = ow * s + dwc - p;
            if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
            float v = bf2f(x[(((int64_t)n * H + ih) * W + iw) * C + c]);
            if (v > best) { best = v; bi = d; }
        }
        y[i] = f2bf(best);
        idx[i] = (uint8_t)bi;
    }
}

__global__ void maxpool_gen_bwd_kernel(const unsigned short* __restrict__ dy,
                                       const uint8_t* __restrict__ idx,
                                       float* __restrict__ dx32, int N, int H,
                                       int W, int C, int OH, int OW, int k,
                                       int s, int p, FastDiv fC, FastDiv fOW,
                                       FastDiv fOH, FastDiv fk) {
    int64_t total = (int64_t)N * OH * OW * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t1 = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t1 * C);
        unsigned t2 = fdiv(t1, fOW);
        int ow = (int)(t1 - t2 * OW);
        unsigned t3 = fdiv(t2, fOH);
        int oh = (int)(t2 - t3 * OH);
        int n = (int)t3;
        int d = idx[i];
        int dk = (int)fdiv((unsigned)d, fk);
        int ih = oh * s + dk - p, iw = ow * s + (d - dk * k) - p;
        if (ih < 0 || ih >= H || iw < 0 || iw >= W) continue;
        atomicAdd(dx32 + (((int64_t)n * H + ih) * W + iw) * C + c,
                  bf2f(dy[i]));
    }
}

// ---------------------------------------------------------------------------
// Global average pool [N,H,W,C] -> [N,C] + backward broadcast.
// ---------------------------------------------------------------------------

__global__ void avgpool_global_fwd_kernel(const unsigned short* __restrict__ x,
                                          unsigned short* __restrict__ y,
                                          int N, int HW, int C, FastDiv fC) {
    int64_t total = (int64_t)N * C;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t * C);
        int n = (int)t;
        float acc = 0.f;
        const unsigned short* base = x + (int64_t)n * HW * C + c;
        for (int r = 0; r < HW; ++r) acc += bf2f(base[(int64_t)r * C]);
        y[i] = f2bf(acc / (float)HW);
    }
}

__global__ void avgpool_global_bwd_kernel(const unsigned short* __restrict__ dy,
                                          unsigned short* __restrict__ dx,
                                          int N, int HW, int C, FastDiv fC, FastDiv fHW) {
    int64_t total = (int64_t)N * HW * C;
    float inv;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
         i += (int64_t)gridDim.x * blockDim.x) {
        unsigned t0 = fdiv((unsigned)i, fC);
        int c = (int)((unsigned)i - t0 * C);
        int n = (int)fdiv(t0, fHW);
        inv = 1.f / (float)HW;
        dx[i] = f2bf(bf2f(dy[(int64_t)n * C + c]) * inv);
    }
}

// ---------------------------------------------------------------------------
// Fused residual add + ReLU.
// ---------------------------------------------------------------------------

__global__ void add_relu_kernel(const unsigned short* __restrict__ a,
                                const unsigned short* __restrict__ b,
                                unsigned short* __restrict__ y, int64_t total) {
    const int64_t t8 = total >> 3;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < t8;
         i += (int64_t)gridDim.x * blockDim.x) {
        u16x8 a8 = *reinterpret_cast<const u16x8*>(&a[i * 8]);
        u16x8 b8 = *reinterpret_cast<const u16x8*>(&b[i * 8]);
        u16x8 y8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            float v = bf2f(a8[j]) + bf2f(b8[j]);
            y8[j] = f2bf(v > 0.f ? v : 0.f);
        }
        *reinterpret_cast<u16x8*>(&y[i * 8]) = y8;
    }
    for (int64_t i = (t8 << 3) + blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total; i += (int64_t)gridDim.x * blockDim.x) {
        float v = bf2f(a[i]) + bf2f(b[i]);
        y[i] = f2bf(v > 0.f ? v : 0.f);
    }
}

// ---------------------------------------------------------------------------
// ReLU backward + bias grad
// ---------------------------------------------------------------------------

__global__ void relu_bwd_kernel(const unsigned short* __restrict__ dy,
                                const unsigned short* __restrict__ y,
                                unsigned short* __restrict__ dx, int64_t total) {
    const int64_t t8 = total >> 3;
    for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < t8;
         i += (int64_t)gridDim.x * blockDim.x) {
        u16x8 d8 = *reinterpret_cast<const u16x8*>(&dy[i * 8]);
        u16x8 y8 = *reinterpret_cast<const u16x8*>(&y[i * 8]);
        u16x8 o8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
            o8[j] = (y8[j] & 0x7fffu) != 0 && !(y8[j] & 0x8000u) ? d8[j] : 0;
        *reinterpret_cast<u16x8*>(&dx[i * 8]) = o8;
    }
    for (int64_t i = (t8 << 3) + blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
         i < total; i += (int64_t)gridDim.x * blockDim.x)
        dx[i] = (y[i] & 0x7fffu) != 0 && !(y[i] & 0x8000u) ? dy[i] : 0;
}

// db[k] = sum over rows of dy[., k]: 2-D grid (k x row-chunks), coalesced
// row sweeps, fp32 atomics into db (zeroed by the wrapper).
__global__ void bias_grad_kernel(const unsigned short* __restrict__ dy,
                                 float* __restrict__ db, int64_t M, int K,
                                 int rows_per_block) {
    __shared__ float red[256];
    const int k = blockIdx.x;
    const int64_t r0 = (int64_t)blockIdx.y * rows_per_block;
    const int64_t r1 = min(r0 + rows_per_block, M);
    float acc = 0.f;
    for (int64_t r = r0 + threadIdx.x; r < r1; r += blockDim.x)
        acc += bf2f(dy[r * K + k]);
    red[threadIdx.x] = acc;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
        if (threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        if (gridDim.y == 1) db[k] = red[0];  // single chunk: direct store
        else atomicAdd(db + k, red[0]);
    }
}

__global__ void sum_slabs_f32_kernel(const float* __restrict__ slab,
                                     float* __restrict__ out, int64_t total,
                                     int nslab) {
    // thread = (slab lane, elem): 16 lanes sum strided slabs in parallel,
    // LDS tree-reduce (a serial slab loop left ~90% of the block idle for
    // the small dw tensors this sums)
    __shared__ float red[256];
    const int lane = threadIdx.x & 15;
    const int eloc = threadIdx.x >> 4;
    const int64_t e = (int64_t)blockIdx.x * 16 + eloc;
    float a = 0.f;
    if (e < total)
        for (int sl = lane; sl < nslab; sl += 16)
            a += slab[(int64_t)sl * total + e];
    red[threadIdx.x] = a;
    __syncthreads();
    for (int off = 8; off >= 1; off >>= 1) {
        if (lane < off) red[threadIdx.x] += red[threadIdx.x + off];
        __syncthreads();
    }
    if (lane == 0 && e < total) out[e] = red[threadIdx.x];
}

inline int ceildiv(int64_t a, int64_t b) { return (int)((a + b - 1) / b); }

inline FastDiv fdiv_make(unsigned d) {
    int sh = 0;
    while ((1ull << sh) < d) ++sh;  // ceil log2(d)
    FastDiv f;
    f.p = 32 + sh;
    f.m = (unsigned long long)(((static_cast<unsigned __int128>(1) << f.p) +
                                d - 1) / d);
    return f;
}

inline void fill_magic(ConvShape& s) {
    s.fC = fdiv_make((unsigned)s.C);
    s.fS = fdiv_make((unsigned)s.S);
    s.fOW = fdiv_make((unsigned)s.OW);
    s.fOHOW = fdiv_make((unsigned)(s.OH * s.OW));
    s.fKout = fdiv_make((unsigned)s.Kout);
}

inline ConvShape make_shape(const torch::Tensor& x, const torch::Tensor& w,
                            int stride, int pad) {
    ConvShape s;
    s.N = (int)x.size(0);
    s.H = (int)x.size(1);
    s.W = (int)x.size(2);
    s.C = (int)x.size(3);
    s.Kout = (int)w.size(0);
    s.R = (int)w.size(1);
    s.S = (int)w.size(2);
    s.stride = stride;
    s.pad = pad;
    s.OH = (s.H + 2 * pad - s.R) / stride + 1;
    s.OW = (s.W + 2 * pad - s.S) / stride + 1;
    fill_magic(s);
    return s;
}

const unsigned short* bf_ptr(const torch::Tensor& t) {
    return reinterpret_cast<const unsigned short*>(t.data_ptr<at::BFloat16>());
}
unsigned short* bf_ptr_mut(torch::Tensor& t) {
    return reinterpret_cast<unsigned short*>(t.data_ptr<at::BFloat16>());
}

// Persistent fp32 split-K accumulator pool. Invariant: every element is
// zero BETWEEN ops on the compute stream — each producer atomically
// accumulates into it and its consumer kernel (linear_epilogue /
// cast_f32_bf16 with clear=1) zeroes what it read, the same
// consume-and-clear idiom fused_adam_mt_kernel uses for grads. This
// replaces a per-call torch::zeros whose fill launches were 11.5% of
// config #2 kernel time (4 us launch-bound fills,
// profiles/r02_config2_final_kernel_stats.csv). Valid because the engine
// runs all conv/pool work on one stream per rank, and each host wrapper
// enqueues accumulate + consume before returning. Growth while a hipGraph
// capture is active would allocate from the capture mempool (whose blocks
// die with the graph), so it falls back to a per-call zeros tensor then —
// eager warmup before capture normally sizes the pool first.
// Sound zero-init-free grad buffers for the epoch-graph fill-skip
// (fl/client.py, HEFL_GRAPH_NO_ZERO): raw hipMalloc allocations keyed by
// the caller's param identity, wrapped per call in from_blob — a fresh
// TensorImpl with use_count 1, so AccumulateGrad STEALS it into p.grad
// and the captured multi-tensor Adam consume-and-clears the memory each
// step. Because the memory never belongs to the caching allocator, no
// tensor allocated during a hipGraph capture can alias it — the
// capture-pool block-reuse hazard that sank the first fill-skip design
// (a freed block's previous owner re-polluted the "zero" buffer every
// replay; observed as resnet18 divergence) cannot occur. Zeroed once at
// creation; the Adam clear maintains the invariant thereafter.
// Caveats (checked/benign): a key whose numel changed (CPython id reuse
// across client lifetimes) falls back to the classic zeroed path; two
// clients' graphs may share a buffer on id reuse with equal shapes,
// which is safe because every accumulate->consume pair completes within
// one captured step. Weight sharing (one param receiving two grads in a
// step) is NOT supported on this path — no model here shares weights.
torch::Tensor grad_buf(int64_t key, at::IntArrayRef sizes,
                       const torch::TensorOptions& opts, hipStream_t stream,
                       bool& ok) {
    static std::unordered_map<int64_t, std::pair<void*, int64_t>> bufs;
    int64_t numel = 1;
    for (auto s : sizes) numel *= s;
    auto it = bufs.find(key);
    if (it == bufs.end()) {
        hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
        (void)hipStreamIsCapturing(stream, &st);
        if (st != hipStreamCaptureStatusNone) {
            ok = false;  // cannot hipMalloc during capture: classic path
            return {};
        }
        void* p = nullptr;
        if (hipMalloc(&p, numel * sizeof(float)) != hipSuccess) {
            ok = false;
            return {};
        }
        (void)hipMemsetAsync(p, 0, numel * sizeof(float), stream);
        it = bufs.emplace(key, std::make_pair(p, numel)).first;
    }
    if (it->second.second != numel) {
        ok = false;
        return {};
    }
    ok = true;
    return torch::from_blob(it->second.first, sizes,
                            opts.dtype(torch::kFloat32));
}

float* acc_pool(int64_t n, const torch::TensorOptions& opts,
                hipStream_t stream, torch::Tensor& holder) {
    static torch::Tensor pool;
    auto f32 = opts.dtype(torch::kFloat32);
    if (!pool.defined() || pool.device() != f32.device_opt().value() ||
        pool.numel() < n) {
        hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
        (void)hipStreamIsCapturing(stream, &st);
        if (st != hipStreamCaptureStatusNone) {
            holder = torch::zeros({n}, f32);
            return holder.data_ptr<float>();
        }
        pool = torch::zeros({n}, f32);
    }
    return pool.data_ptr<float>();
}

}  // namespace

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

torch::Tensor conv2d_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         int64_t stride, bool relu, int64_t pad) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
    TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16);
    ConvShape s = make_shape(x, w, (int)stride, (int)pad);
    auto y = torch::empty({s.N, s.OH, s.OW, s.Kout}, x.options());
    const int M = s.N * s.OH * s.OW;
    const float* bias = b.numel() ? b.data_ptr<float>() : nullptr;
    auto stream = at::cuda::getCurrentCUDAStream();
    const int KKf = s.R * s.S * s.C;
    // Direct tiled kernel for 3x3 s1 convs with C % 32 == 0: the input
    // tile is staged once per channel slab and all nine taps compute from
    // it (vs 9x re-gather + 4 MFMAs/barrier in the implicit-GEMM path).
    // HEFL_TILE3: 0 = off (A/B), 2 = force even on tiny grids (tests);
    // read per call so tests can toggle it.
    const char* t3env = getenv("HEFL_TILE3");
    const int t3mode = t3env ? t3env[0] - '0' : 1;
    // A/B (profiles/r02_bench_conv_tile3 vs _notile): the 16x16/BN=32 variant
    // wins +34% on Kout<=32 layers (their BN=64 implicit-GEMM tile wasted
    // half its columns AND re-gathered 9x); the BN=64 and 8x16 variants
    // LOSE 10-45% to the glds pipeline (fewer resident blocks, conflictier
    // B reads) — auto (1) dispatches only the measured winner; 2 forces any
    // grid (tests); 3 probes BN=32 column-tiles for EVERY Kout.
    if (t3mode != 0 && s.R == 3 && s.S == 3 && s.stride == 1 && s.pad <= 1 &&
        s.C % 32 == 0 && s.Kout > 16 &&
        (t3mode >= 2 || s.Kout <= 32)) {
        const int BN3 = (t3mode == 3 || s.Kout <= 32) ? 32 : 64;
        const int kt = ceildiv(s.Kout, BN3);
        const int64_t min_tiles = t3mode == 2 ? 1 : 256;
        auto tiles = [&](int th, int tw) {
            return (int64_t)s.N * ceildiv(s.OH, th) * ceildiv(s.OW, tw) * kt;
        };
        int TH = 0;
        if (s.OH >= 12 && tiles(16, 16) >= min_tiles) TH = 16;
        else if (t3mode >= 2 && s.OH >= 6 && tiles(8, 16) >= min_tiles) TH = 8;
        if (TH) {
            const int th_ = ceildiv(s.OH, TH), tw_ = ceildiv(s.OW, 16);
            dim3 grid((unsigned)(s.N * th_ * tw_), (unsigned)kt);
            #define LAUNCH_T3(TH_, TW_, BN_, WM_, WN_, FM_, FN_)              \
                hipLaunchKernelGGL(                                           \
                    (conv_fwd_tile3_kernel<TH_, TW_, BN_, WM_, WN_, FM_,      \
                                           FN_>),                             \
                    grid, dim3(TPB), 0, stream, bf_ptr(x), bf_ptr(w), bias,   \
                    bf_ptr_mut(y), s, relu ? 1 : 0, th_, tw_)
            if (TH == 16 && BN3 == 64)      LAUNCH_T3(16, 16, 64, 2, 2, 8, 2);
            else if (TH == 16)              LAUNCH_T3(16, 16, 32, 4, 1, 4, 2);
            else if (BN3 == 64)             LAUNCH_T3(8, 16, 64, 2, 2, 4, 2);
            else                            LAUNCH_T3(8, 16, 32, 2, 2, 4, 1);
            #undef LAUNCH_T3
            return y;
        }
    }
    // Measured on MI355X (profiles/r02_bench_conv_glds64 vs _glds32): the
    // BK=64 two-buffer kernel LOSES 8-30% to the BK=32 three-buffer one at
    // every conv shape here — K loops are 2-18 steps, so the prologue DMA
    // latency (3-buf pre-stages two tiles, 2-buf one) outweighs the halved
    // barrier count; the shapes are gather-bound, not MFMA-issue-bound.
    // Kept behind HEFL_GLDS64=1 as the documented A/B.
    static const bool use64 = [] {
        const char* e = getenv("HEFL_GLDS64");
        return e && e[0] == '1';
    }();
    if (use64 && s.Kout > 16 && s.C % 8 == 0 && KKf % 8 == 0) {
        static torch::Tensor zbuf64;
        if (!zbuf64.defined() || zbuf64.device() != x.device())
            zbuf64 = torch::zeros({8}, x.options());
        const bool narrow = s.Kout <= 32;  // BN=32 tile: no wasted columns
        const int BM = narrow ? 256 : 128, BN = narrow ? 32 : 64;
        int tiles = ceildiv(M, BM) * ceildiv(s.Kout, BN);
        int ksteps = ceildiv(KKf, 64);
        int k_chunks = 1;
        if (tiles < 256 && ksteps >= 8)
            k_chunks = std::max(1, std::min(ksteps / 2, 512 / std::max(tiles, 1)));
        dim3 grid(ceildiv(M, BM), ceildiv(s.Kout, BN), k_chunks);
        torch::Tensor y32;
        float* y32p = nullptr;
        if (k_chunks > 1)
            y32p = acc_pool((int64_t)M * s.Kout, x.options(), stream, y32);
        #define LAUNCH_F64(BM_, BN_, WM_, WN_, FM_, FN_, P0)                  \
            hipLaunchKernelGGL((conv_fwd_glds64_kernel<BM_, BN_, WM_, WN_,    \
                                                       FM_, FN_, P0>),       \
                               grid, dim3(TPB), 0, stream, bf_ptr(x),         \
                               bf_ptr(w), bias, bf_ptr_mut(y), y32p,          \
                               bf_ptr(zbuf64), s, relu ? 1 : 0, k_chunks)
        if (narrow) {
            if (s.pad == 0) LAUNCH_F64(256, 32, 4, 1, 4, 2, true);
            else            LAUNCH_F64(256, 32, 4, 1, 4, 2, false);
        } else {
            if (s.pad == 0) LAUNCH_F64(128, 64, 2, 2, 4, 2, true);
            else            LAUNCH_F64(128, 64, 2, 2, 4, 2, false);
        }
        #undef LAUNCH_F64
        if (k_chunks > 1) {
            int64_t total = (int64_t)M * s.Kout;
            hipLaunchKernelGGL(linear_epilogue_kernel,
                               dim3((int)std::min<int64_t>(ceildiv(total, 256), 2048)),
                               dim3(256), 0, stream, y32p, bias,
                               bf_ptr_mut(y), total, s.Kout, relu ? 1 : 0, 1, 1);
        }
        return y;
    }
    if (s.Kout > 16 && s.C % 8 == 0 && KKf % 8 == 0) {
        // glds double-buffered pipeline (DMA flight hides under MFMA);
        // small-M/N late layers split the K loop over grid.z to fill the chip
        static torch::Tensor zbuf;
        if (!zbuf.defined() || zbuf.device() != x.device())
            zbuf = torch::zeros({8}, x.options());
        int tiles = ceildiv(M, 64) * ceildiv(s.Kout, 64);
        int ksteps = ceildiv(KKf, 32);
        int k_chunks = 1;
        if (tiles < 256 && ksteps >= 16)
            k_chunks = std::max(1, std::min(ksteps / 4, 512 / tiles));
        dim3 grid(ceildiv(M, 64), ceildiv(s.Kout, 64), k_chunks);
        if (k_chunks > 1) {
            torch::Tensor y32h;
            float* y32p = acc_pool((int64_t)M * s.Kout, x.options(), stream,
                                   y32h);
            hipLaunchKernelGGL((conv_fwd_glds_kernel<64, 64, 2, 2, 2, 2, 32>),
                               grid, dim3(TPB), 0, stream, bf_ptr(x), bf_ptr(w),
                               bias, bf_ptr_mut(y), y32p,
                               bf_ptr(zbuf), s, relu ? 1 : 0, k_chunks);
            int64_t total = (int64_t)M * s.Kout;
            hipLaunchKernelGGL(linear_epilogue_kernel,
                               dim3((int)std::min<int64_t>(ceildiv(total, 256), 2048)),
                               dim3(256), 0, stream, y32p, bias,
                               bf_ptr_mut(y), total, s.Kout, relu ? 1 : 0, 1, 1);
        } else {
            hipLaunchKernelGGL((conv_fwd_glds_kernel<64, 64, 2, 2, 2, 2, 32>),
                               grid, dim3(TPB), 0, stream, bf_ptr(x), bf_ptr(w),
                               bias, bf_ptr_mut(y), (float*)nullptr,
                               bf_ptr(zbuf), s, relu ? 1 : 0, 1);
        }
    } else if (s.Kout > 16) {
        dim3 grid(ceildiv(M, 64), ceildiv(s.Kout, 64));
        hipLaunchKernelGGL((conv_fwd_kernel<64, 64, 2, 2, 2, 2>), grid,
                           dim3(TPB), 0, stream, bf_ptr(x), bf_ptr(w), bias,
                           bf_ptr_mut(y), s, relu ? 1 : 0);
    } else {
        dim3 grid(ceildiv(M, 64), ceildiv(s.Kout, 16));
        hipLaunchKernelGGL((conv_fwd_kernel<64, 16, 4, 1, 1, 1>), grid,
                           dim3(TPB), 0, stream, bf_ptr(x), bf_ptr(w), bias,
                           bf_ptr_mut(y), s, relu ? 1 : 0);
    }
    return y;
}

torch::Tensor conv2d_dgrad(torch::Tensor dy, torch::Tensor w, int64_t stride,
                           int64_t H, int64_t W, int64_t pad) {
    CHECK_GPU(dy);
    TORCH_CHECK(dy.is_contiguous() && w.is_contiguous());
    const int N = (int)dy.size(0);
    const int C = (int)w.size(3);
    ConvShape s;
    s.N = N; s.H = (int)H; s.W = (int)W; s.C = C;
    s.Kout = (int)w.size(0); s.R = (int)w.size(1); s.S = (int)w.size(2);
    s.stride = (int)stride; s.pad = (int)pad;
    s.OH = (int)dy.size(1); s.OW = (int)dy.size(2);
    fill_magic(s);
    auto dx = torch::empty({N, (int64_t)H, (int64_t)W, C}, dy.options());
    const int M = N * (int)H * (int)W;
    auto stream = at::cuda::getCurrentCUDAStream();
    // dgrad as a direct tiled conv of dy with the 180-rotated transposed
    // weights: dx = conv3x3(dy, wT, pad' = 2 - pad). Same winner shape
    // class as forward (output channels <= 32 -> the BN=32 16x16 tile).
    const char* t3env_d = getenv("HEFL_TILE3");
    const int t3mode_d = t3env_d ? t3env_d[0] - '0' : 1;
    if (t3mode_d != 0 && s.R == 3 && s.S == 3 && stride == 1 && pad <= 1 &&
        s.Kout % 32 == 0 && C > 16 && (t3mode_d == 2 || C <= 32)) {
        ConvShape s3;
        s3.N = N; s3.H = s.OH; s3.W = s.OW; s3.C = s.Kout;
        s3.Kout = C; s3.R = 3; s3.S = 3; s3.stride = 1; s3.pad = 2 - (int)pad;
        s3.OH = (int)H; s3.OW = (int)W;
        fill_magic(s3);
        const int BN3 = C <= 32 ? 32 : 64;
        const int kt = ceildiv(C, BN3);
        const int64_t min_tiles = t3mode_d == 2 ? 1 : 256;
        const int64_t t16 = (int64_t)N * ceildiv(s3.OH, 16)
                            * ceildiv(s3.OW, 16) * kt;
        if (s3.OH >= 12 && t16 >= min_tiles) {
            auto wT = torch::empty({C, 3, 3, s.Kout}, w.options());
            const int64_t tot = (int64_t)C * 9 * s.Kout;
            hipLaunchKernelGGL(rot180_transpose_w_kernel,
                               dim3((int)std::min<int64_t>(ceildiv(tot, 256), 1024)),
                               dim3(256), 0, stream, bf_ptr(w), bf_ptr_mut(wT),
                               s.Kout, C, tot);
            const int th_ = ceildiv(s3.OH, 16), tw_ = ceildiv(s3.OW, 16);
            dim3 grid((unsigned)(N * th_ * tw_), (unsigned)kt);
            if (BN3 == 32)
                hipLaunchKernelGGL((conv_fwd_tile3_kernel<16, 16, 32, 4, 1, 4, 2>),
                                   grid, dim3(TPB), 0, stream, bf_ptr(dy),
                                   bf_ptr(wT), (const float*)nullptr,
                                   bf_ptr_mut(dx), s3, 0, th_, tw_);
            else
                hipLaunchKernelGGL((conv_fwd_tile3_kernel<16, 16, 64, 4, 1, 4, 4>),
                                   grid, dim3(TPB), 0, stream, bf_ptr(dy),
                                   bf_ptr(wT), (const float*)nullptr,
                                   bf_ptr_mut(dx), s3, 0, th_, tw_);
            return dx;
        }
    }
    // C == 16 included (cnn2 conv2 dgrad): measured a WASH end-to-end on
    // config #2 (38.45 vs 38.44 rounds/s) — the BN=64 tile wastes 3/4 of
    // its columns, which cancels the 16-B-vector advantage over the
    // generic gather at this width. Kept for the uniform path.
    if (C >= 16 && C % 8 == 0 && s.Kout % 8 == 0) {
        static torch::Tensor zbuf;
        if (!zbuf.defined() || zbuf.device() != dy.device())
            zbuf = torch::zeros({8}, dy.options());
        const int KKd = s.Kout * s.R * s.S;
        int tiles = ceildiv(M, 64) * ceildiv(C, 64);
        int ksteps = ceildiv(KKd, 32);
        int k_chunks = 1;
        if (tiles < 256 && ksteps >= 16)
            k_chunks = std::max(1, std::min(ksteps / 4, 512 / tiles));
        dim3 grid(ceildiv(M, 64), ceildiv(C, 64), k_chunks);
        torch::Tensor dx32;
        float* dx32p = nullptr;
        if (k_chunks > 1)
            dx32p = acc_pool((int64_t)M * C, dy.options(), stream, dx32);
        if (stride == 1)
            hipLaunchKernelGGL((conv_dgrad_glds_kernel<64, 64, 2, 2, 2, 2, true>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), dx32p, bf_ptr(zbuf),
                               s, k_chunks);
        else
            hipLaunchKernelGGL((conv_dgrad_glds_kernel<64, 64, 2, 2, 2, 2, false>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), dx32p, bf_ptr(zbuf),
                               s, k_chunks);
        if (k_chunks > 1) {
            int64_t total = (int64_t)M * C;
            hipLaunchKernelGGL(cast_f32_bf16_kernel,
                               dim3((int)std::min<int64_t>(ceildiv(total, 256), 2048)),
                               dim3(256), 0, stream, dx32p, bf_ptr_mut(dx),
                               total, 1);
        }
    } else if (false) {  // 128-tile: measured neutral-to-worse (barrier-bound)
        dim3 grid(ceildiv(M, 128), ceildiv(C, 64));
        if (stride == 1)
            hipLaunchKernelGGL((conv_dgrad_kernel<128, 64, 2, 2, 4, 2, true>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
        else
            hipLaunchKernelGGL((conv_dgrad_kernel<128, 64, 2, 2, 4, 2, false>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
    } else if (C > 16) {
        dim3 grid(ceildiv(M, 64), ceildiv(C, 64));
        if (stride == 1)
            hipLaunchKernelGGL((conv_dgrad_kernel<64, 64, 2, 2, 2, 2, true>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
        else
            hipLaunchKernelGGL((conv_dgrad_kernel<64, 64, 2, 2, 2, 2, false>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
    } else {
        dim3 grid(ceildiv(M, 64), ceildiv(C, 16));
        if (stride == 1)
            hipLaunchKernelGGL((conv_dgrad_kernel<64, 16, 4, 1, 1, 1, true>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
        else
            hipLaunchKernelGGL((conv_dgrad_kernel<64, 16, 4, 1, 1, 1, false>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(w), bf_ptr_mut(dx), s);
    }
    return dx;
}

torch::Tensor conv2d_wgrad(torch::Tensor dy, torch::Tensor x, int64_t stride,
                           int64_t R, int64_t S, int64_t pad,
                           int64_t gkey) {
    // gkey != 0: epoch-graph fill-skip (ops/functional.py GRAPH_NO_ZERO) —
    // the atomics accumulate into the process-lifetime grad_buf for this
    // param, which the captured Adam consume-and-clears each step; no
    // zero-fill launch. gkey == 0 (or grad_buf unavailable): classic
    // zeroed allocation.
    CHECK_GPU(dy);
    TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
    ConvShape s;
    s.N = (int)x.size(0); s.H = (int)x.size(1); s.W = (int)x.size(2);
    s.C = (int)x.size(3);
    s.Kout = (int)dy.size(3); s.R = (int)R; s.S = (int)S;
    s.stride = (int)stride; s.pad = (int)pad;
    s.OH = (int)dy.size(1); s.OW = (int)dy.size(2);
    fill_magic(s);
    const int NN = s.R * s.S * s.C;
    const int KK = s.N * s.OH * s.OW;
    const bool big = NN > 16;
    const bool small_fast = (!big) && s.Kout <= 64;
    const bool glds_ok = big && (s.C % 8 == 0) && (s.Kout % 8 == 0);
    const int bn = big ? 64 : 16;
    // split-K to fill the chip: target >= 512 blocks (2 per CU)
    int tiles = ceildiv(s.Kout, 64) * ceildiv(NN, bn);
    int k_chunks = std::max(1, std::min(ceildiv(KK, 32 * 2),
                                        512 / std::max(tiles, 1)));
    auto stream = at::cuda::getCurrentCUDAStream();
    if (small_fast) {
        auto dw = torch::empty({s.Kout, R, S, s.C},
                               x.options().dtype(torch::kFloat32));
        // measured: KK/64 slabs (338 blocks) LOST to KK/256 (84) — the
        // slab reduce outgrows the parallel gain; this path is latency-
        // not occupancy-bound
        int kc = std::max(1, std::min(ceildiv(KK, 256), 512));
        if (kc == 1) {
            hipLaunchKernelGGL(conv_wgrad_small_kernel, dim3(1, 1, 1),
                               dim3(TPB), 0, stream, bf_ptr(dy), bf_ptr(x),
                               dw.data_ptr<float>(), s, 1);
            return dw;
        }
        const int64_t total = (int64_t)s.Kout * R * S * s.C;
        auto slab = torch::empty({kc, total},
                                 x.options().dtype(torch::kFloat32));
        hipLaunchKernelGGL(conv_wgrad_small_kernel, dim3(1, 1, kc), dim3(TPB),
                           0, stream, bf_ptr(dy), bf_ptr(x),
                           slab.data_ptr<float>(), s, kc);
        hipLaunchKernelGGL(sum_slabs_f32_kernel,
                           dim3((int)((total + 15) / 16)), dim3(256), 0,
                           stream, slab.data_ptr<float>(),
                           dw.data_ptr<float>(), total, kc);
        return dw;
    }
    if (glds_ok && s.Kout <= 32) {
        // 32-ko tile, 64-pixel K-step: no ko waste on 32-filter layers
        static torch::Tensor zbuf32;
        if (!zbuf32.defined() || zbuf32.device() != dy.device())
            zbuf32 = torch::zeros({8}, dy.options());
        // 128-pixel K-steps measured +3-5% over 64 after the staging fix
        // (the first +15% probe was staging only half the Xs tile — a
        // timing-only A/B hides correctness bugs); HEFL_K32B=0 for A/B
        static const bool bkp128 = [] {
            const char* e = getenv("HEFL_K32B");
            return !e || e[0] != '0';
        }();
        int tiles32 = ceildiv(s.Kout, 32) * ceildiv(NN, 64);
        int kc = std::max(1, std::min(ceildiv(KK, 128),
                                      512 / std::max(tiles32, 1)));
        torch::Tensor dw32;
        bool pooled = false;
        if (kc > 1 && gkey)
            dw32 = grad_buf(gkey, {s.Kout, R, S, s.C}, x.options(), stream,
                            pooled);
        if (!pooled)
            dw32 = kc > 1 ? torch::zeros({s.Kout, R, S, s.C},
                                         x.options().dtype(torch::kFloat32))
                          : torch::empty({s.Kout, R, S, s.C},
                                         x.options().dtype(torch::kFloat32));
        dim3 g32(ceildiv(s.Kout, 32), ceildiv(NN, 64), kc);
        if (bkp128)
            hipLaunchKernelGGL((conv_wgrad_glds_k32_kernel<64, 2, 2, 1, 2, 128>),
                               g32, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(x), dw32.data_ptr<float>(),
                               bf_ptr(zbuf32), s, kc);
        else
            hipLaunchKernelGGL((conv_wgrad_glds_k32_kernel<64, 2, 2, 1, 2>),
                               g32, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(x), dw32.data_ptr<float>(),
                               bf_ptr(zbuf32), s, kc);
        return dw32;
    }
    // measured: slab-rows + reduce LOSES to fp32 atomics here (CDNA4 L2
    // atomics absorb the z-chunk contention; the slab variant paid extra
    // write+read traffic) — profiles/r02_bench_conv logs
    torch::Tensor dw;
    bool dw_pooled = false;
    if (k_chunks > 1 && gkey)
        dw = grad_buf(gkey, {s.Kout, R, S, s.C}, x.options(), stream,
                      dw_pooled);
    if (!dw_pooled)
        dw = k_chunks > 1 ? torch::zeros({s.Kout, R, S, s.C},
                                         x.options().dtype(torch::kFloat32))
                          : torch::empty({s.Kout, R, S, s.C},
                                         x.options().dtype(torch::kFloat32));
    dim3 grid(ceildiv(s.Kout, 64), ceildiv(NN, bn), k_chunks);
    if (glds_ok) {
        static torch::Tensor zbuf;
        if (!zbuf.defined() || zbuf.device() != dy.device())
            zbuf = torch::zeros({8}, dy.options());
        // 64-pixel K-steps: +3-5% at every measured shape; HEFL_WGB=0 A/B
        static const bool wgb64 = [] {
            const char* e = getenv("HEFL_WGB");
            return !e || e[0] != '0';
        }();
        if (wgb64)
            hipLaunchKernelGGL((conv_wgrad_glds_kernel<64, 2, 2, 2, 2, 64>),
                               grid, dim3(TPB), 0, stream, bf_ptr(dy),
                               bf_ptr(x), dw.data_ptr<float>(), bf_ptr(zbuf),
                               s, k_chunks);
        else
            hipLaunchKernelGGL((conv_wgrad_glds_kernel<64, 2, 2, 2, 2>), grid,
                               dim3(TPB), 0, stream, bf_ptr(dy), bf_ptr(x),
                               dw.data_ptr<float>(), bf_ptr(zbuf), s,
                               k_chunks);
    } else if (big) {
        hipLaunchKernelGGL((conv_wgrad_kernel<64, 2, 2, 2, 2>), grid, dim3(TPB),
                           0, stream, bf_ptr(dy), bf_ptr(x),
                           dw.data_ptr<float>(), s, k_chunks);
    } else {
        hipLaunchKernelGGL((conv_wgrad_kernel<16, 4, 1, 1, 1>), grid, dim3(TPB),
                           0, stream, bf_ptr(dy), bf_ptr(x),
                           dw.data_ptr<float>(), s, k_chunks);
    }
    return dw;
}

torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor b,
                         bool relu) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous() && w.is_contiguous());
    const int M = (int)x.size(0), K = (int)x.size(1), N = (int)w.size(0);
    auto stream = at::cuda::getCurrentCUDAStream();
    const float* bias = b.numel() ? b.data_ptr<float>() : nullptr;
    auto y = torch::empty({M, N}, x.options());
    if (M <= 64) {
        // slab count targets >= 256 blocks: K/4 left the cnn4 flatten
        // layer (K = 18432) at 64 blocks / 25% of the CUs (~440 GB/s)
        const int gmn = ceildiv(M, 16) * ceildiv(N, 16);
        const int target = std::max(4, 256 / std::max(gmn, 1));
        int kc = std::max(256, ceildiv(K, target));
        kc = ((kc + 31) / 32) * 32;
        if (K <= 1024) kc = ((K + 31) / 32) * 32;  // one slab: direct write
        const int slabs = ceildiv(K, kc);
        if (slabs == 1) {  // whole K in one pass: direct bf16 write
            dim3 grid(ceildiv(M, 16), ceildiv(N, 16), 1);
            hipLaunchKernelGGL((linear_splitk_kernel<1>), grid, dim3(64, 1, 1),
                               0, stream, bf_ptr(x), bf_ptr(w), nullptr, M, N,
                               K, kc, bias, bf_ptr_mut(y), relu ? 1 : 0);
            return y;
        }
        auto y32 = torch::empty({slabs, M, N},
                                x.options().dtype(torch::kFloat32));
        dim3 grid(ceildiv(M, 16), ceildiv(N, 16), slabs);
        hipLaunchKernelGGL((linear_splitk_kernel<0>), grid, dim3(64), 0, stream,
                           bf_ptr(x), bf_ptr(w), y32.data_ptr<float>(), M, N,
                           K, kc, nullptr, nullptr, 0);
        int64_t total = (int64_t)M * N;
        hipLaunchKernelGGL(linear_epilogue_kernel,
                           dim3((int)std::min<int64_t>(ceildiv(total, 256), 2048)),
                           dim3(256), 0, stream, y32.data_ptr<float>(), bias,
                           bf_ptr_mut(y), total, N, relu ? 1 : 0, slabs, 0);
        return y;
    }
    dim3 grid(ceildiv(M, BM), ceildiv(N, BN));
    hipLaunchKernelGGL((gemm_kernel<false, true, true, false>), grid, dim3(TPB),
                       0, stream, bf_ptr(x), bf_ptr(w),
                       bf_ptr_mut(y), bias, M, N, K, relu ? 1 : 0);
    return y;
}

torch::Tensor linear_dgrad(torch::Tensor dy, torch::Tensor w) {
    CHECK_GPU(dy);
    const int M = (int)dy.size(0), N = (int)dy.size(1), K = (int)w.size(1);
    auto dx = torch::empty({M, K}, dy.options());
    dim3 grid(ceildiv(M, BM), ceildiv(K, BN));
    // dx[M,K] = dy[M,N] @ w[N,K]: A = dy (plain), B = w (plain [N->K])
    hipLaunchKernelGGL((gemm_kernel<false, false, true, false>), grid, dim3(TPB),
                       0, at::cuda::getCurrentCUDAStream(), bf_ptr(dy), bf_ptr(w),
                       bf_ptr_mut(dx), nullptr, M, K, N, 0);
    return dx;
}

torch::Tensor linear_wgrad(torch::Tensor dy, torch::Tensor x) {
    CHECK_GPU(dy);
    const int M = (int)dy.size(0), N = (int)dy.size(1), K = (int)x.size(1);
    auto dw = torch::empty({N, K}, dy.options().dtype(torch::kFloat32));
    dim3 grid(ceildiv(N, BM), ceildiv(K, BN));
    // dw[N,K] = dy^T[N,M] @ x[M,K]: A transposed (stored [M,N]), B plain
    hipLaunchKernelGGL((gemm_kernel<true, false, false, false>), grid, dim3(TPB),
                       0, at::cuda::getCurrentCUDAStream(), bf_ptr(dy), bf_ptr(x),
                       dw.data_ptr<float>(), nullptr, N, K, M, 0);
    return dw;
}

static torch::Tensor synth_launch(torch::Tensor templates,
                                  torch::Tensor labels, int64_t seed,
                                  const long long* seed_buf, double zoom,
                                  double shear, int64_t flip) {
    CHECK_GPU(templates);
    TORCH_CHECK(templates.is_contiguous() && labels.is_contiguous());
    const int64_t n = labels.size(0);
    const int H = (int)templates.size(1), W = (int)templates.size(2),
              C = (int)templates.size(3);
    const int64_t per = templates.numel() / templates.size(0);
    auto out = torch::empty({n, templates.size(1), templates.size(2),
                             templates.size(3)},
                            templates.options().dtype(torch::kBFloat16));
    const int64_t total = n * per;
    int blocks = (int)std::min<int64_t>((total + 255) / 256, 8192);
    const bool aug = zoom != 0.0 || shear != 0.0 || flip != 0;
    auto stream = at::cuda::getCurrentCUDAStream();
    if (aug)
        hipLaunchKernelGGL(synth_batch_kernel<true>, dim3(blocks), dim3(256),
                           0, stream, templates.data_ptr<float>(),
                           labels.data_ptr<int64_t>(),
                           reinterpret_cast<unsigned short*>(out.data_ptr()),
                           per, total, (unsigned long long)seed,
                           fdiv_make((unsigned)per), seed_buf, H, W, C,
                           fdiv_make((unsigned)C), fdiv_make((unsigned)W),
                           (float)zoom, (float)shear, (int)flip);
    else
        hipLaunchKernelGGL(synth_batch_kernel<false>, dim3(blocks), dim3(256),
                           0, stream, templates.data_ptr<float>(),
                           labels.data_ptr<int64_t>(),
                           reinterpret_cast<unsigned short*>(out.data_ptr()),
                           per, total, (unsigned long long)seed,
                           fdiv_make((unsigned)per), seed_buf, H, W, C,
                           fdiv_make((unsigned)C), fdiv_make((unsigned)W),
                           0.f, 0.f, 0);
    return out;
}

torch::Tensor synth_batch(torch::Tensor templates, torch::Tensor labels,
                          int64_t seed, double zoom, double shear,
                          int64_t flip) {
    return synth_launch(templates, labels, seed, nullptr, zoom, shear, flip);
}

torch::Tensor synth_batch_g(torch::Tensor templates, torch::Tensor labels,
                            torch::Tensor seed_buf, int64_t salt, double zoom,
                            double shear, int64_t flip) {
    // graph-capturable variant: base seed read from a device buffer the
    // host rewrites before each replay; salt distinguishes captured calls
    return synth_launch(templates, labels, salt,
                        (const long long*)seed_buf.data_ptr<int64_t>(), zoom,
                        shear, flip);
}

std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous());
    const int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
              C = (int)x.size(3);
    const int OH = H / 2, OW = W / 2;
    auto y = torch::empty({N, OH, OW, C}, x.options());
    auto idx = torch::empty({N, OH, OW, C}, x.options().dtype(torch::kUInt8));
    int64_t total = (int64_t)N * OH * OW * C;
    if ((C & 7) == 0) {
        int64_t total8 = total >> 3;
        int blocks = (int)std::min<int64_t>(ceildiv(total8, 256), 4096);
        hipLaunchKernelGGL(maxpool_fwd_oct_kernel, dim3(blocks), dim3(256), 0,
                           at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                           bf_ptr_mut(y), idx.data_ptr<uint8_t>(), total8, H,
                           W, C, OH, OW, fdiv_make((unsigned)(C >> 3)),
                           fdiv_make((unsigned)OW), fdiv_make((unsigned)OH));
        return {y, idx};
    }
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                       bf_ptr_mut(y), idx.data_ptr<uint8_t>(), N, H, W, C, OH,
                       OW, fdiv_make((unsigned)C), fdiv_make((unsigned)OW),
                       fdiv_make((unsigned)OH));
    return {y, idx};
}

torch::Tensor maxpool2x2_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                             int64_t W) {
    CHECK_GPU(dy);
    const int N = (int)dy.size(0), OH = (int)dy.size(1), OW = (int)dy.size(2),
              C = (int)dy.size(3);
    auto dx = torch::empty({N, H, W, C}, dy.options());
    int64_t total = (int64_t)N * H * W * C;
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(maxpool2x2_bwd_gather_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dy),
                       idx.data_ptr<uint8_t>(), bf_ptr_mut(dx), N, (int)H,
                       (int)W, C, OH, OW, fdiv_make((unsigned)C),
                       fdiv_make((unsigned)W), fdiv_make((unsigned)H));
    return dx;
}

std::vector<torch::Tensor> relu_bias_bwd(torch::Tensor dy, torch::Tensor y,
                                         int64_t gkey) {
    CHECK_GPU(dy);
    auto dyc = dy.contiguous();
    const int K = (int)dyc.size(-1);
    const int64_t M = dyc.numel() / K;
    auto dym = torch::empty_like(dyc);
    int rpb = (int)std::max<int64_t>(64, (M + 511) / 512);
    int nblk = (int)((M + rpb - 1) / rpb);
    auto stream0 = at::cuda::getCurrentCUDAStream();
    torch::Tensor db;
    bool db_pooled = false;
    if (nblk > 1 && gkey)
        db = grad_buf(gkey, {K}, dyc.options(), stream0, db_pooled);
    if (!db_pooled)
        db = nblk == 1
                 ? torch::empty({K}, dyc.options().dtype(torch::kFloat32))
                 : torch::zeros({K}, dyc.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(relu_bias_bwd_kernel, dim3(nblk), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dyc), bf_ptr(y),
                       bf_ptr_mut(dym), db.data_ptr<float>(), M, K, rpb);
    return {dym, db};
}

std::vector<torch::Tensor> pool_relu_bias_bwd(torch::Tensor dy,
                                              torch::Tensor idx,
                                              torch::Tensor p, int64_t H,
                                              int64_t W, int64_t gkey) {
    CHECK_GPU(dy);
    auto dyc = dy.contiguous();
    const int N = (int)dyc.size(0), OH = (int)dyc.size(1),
              OW = (int)dyc.size(2), K = (int)dyc.size(3);
    auto dym = torch::empty({N, H, W, (int64_t)K}, dyc.options());
    auto stream = at::cuda::getCurrentCUDAStream();
    const int noct = K >> 3;
    if ((K & 7) == 0 && (noct & (noct - 1)) == 0 && noct <= 256) {
        // 2x2-block octet kernel: dy/idx/p read once per output cell
        const int HB = (int)((H + 1) / 2), WB = (int)((W + 1) / 2);
        const int64_t total8 = (int64_t)N * HB * WB * noct;
        // measured: forcing ONE block on small layers to skip the db
        // zero-fill serialized the whole layer onto one CU (-30% on the
        // headline config) — keep the parallel grid and pay the 4.7 us fill
        int blocks = (int)std::min<int64_t>(ceildiv(total8, 256), 4096);
        torch::Tensor db;
        bool db_pooled = false;
        if (blocks > 1 && gkey)
            db = grad_buf(gkey, {K}, dyc.options(), stream, db_pooled);
        if (!db_pooled)
            db = blocks == 1
                     ? torch::empty({K},
                                    dyc.options().dtype(torch::kFloat32))
                     : torch::zeros({K},
                                    dyc.options().dtype(torch::kFloat32));
        hipLaunchKernelGGL(pool_relu_bias_bwd2_kernel, dim3(blocks),
                           dim3(256), K * sizeof(float), stream, bf_ptr(dyc),
                           idx.data_ptr<uint8_t>(), bf_ptr(p),
                           bf_ptr_mut(dym), db.data_ptr<float>(), total8, K,
                           (int)H, (int)W, OH, OW, HB, WB,
                           fdiv_make((unsigned)noct), fdiv_make((unsigned)WB),
                           fdiv_make((unsigned)HB));
        return {dym, db};
    }
    TORCH_CHECK(K <= 256, "scalar pool-backward fallback expects K <= 256");
    const int64_t M = (int64_t)N * H * W;
    int rpb = (int)std::max<int64_t>(64, (M + 511) / 512);
    int lanes = 256 / K;
    lanes = 1 << (31 - __builtin_clz(lanes));
    rpb = std::max(rpb, 2 * lanes);
    int nblk = (int)((M + rpb - 1) / rpb);
    torch::Tensor db;
    bool db_pooled = false;
    if (nblk > 1 && gkey)
        db = grad_buf(gkey, {K}, dyc.options(), stream, db_pooled);
    if (!db_pooled)
        db = nblk == 1
                 ? torch::empty({K}, dyc.options().dtype(torch::kFloat32))
                 : torch::zeros({K}, dyc.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(pool_relu_bias_bwd_scalar_kernel, dim3(nblk),
                       dim3(256), 0, stream, bf_ptr(dyc),
                       idx.data_ptr<uint8_t>(), bf_ptr(p), bf_ptr_mut(dym),
                       db.data_ptr<float>(), M, K, rpb, (int)H, (int)W, OH,
                       OW, fdiv_make((unsigned)W), fdiv_make((unsigned)H));
    return {dym, db};
}

std::vector<torch::Tensor> dense_head2_bwd(torch::Tensor dlogits,
                                           torch::Tensor x, torch::Tensor h1,
                                           torch::Tensor w1,
                                           torch::Tensor w2) {
    CHECK_GPU(dlogits);
    const int M = (int)dlogits.size(0), N2 = (int)dlogits.size(1);
    const int N1 = (int)w1.size(0), K = (int)w1.size(1);
    TORCH_CHECK(M <= 32 && N2 <= 16 && N1 <= 128 && N1 % 16 == 0 &&
                K % 16 == 0 && K <= 1024,
                "dense_head2_bwd shape envelope");
    auto f32 = x.options().dtype(torch::kFloat32);
    auto dx = torch::empty({M, (int64_t)K}, x.options());
    auto dw1 = torch::empty({N1, (int64_t)K}, f32);
    auto db1 = torch::empty({N1}, f32);
    auto dw2 = torch::empty({N2, (int64_t)N1}, f32);
    auto db2 = torch::empty({N2}, f32);
    hipLaunchKernelGGL(dense_head2_bwd_kernel, dim3(1), dim3(TPB),
                       32 * K * sizeof(unsigned short),
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dlogits),
                       bf_ptr(x), bf_ptr(h1), bf_ptr(w1), bf_ptr(w2),
                       bf_ptr_mut(dx), dw1.data_ptr<float>(),
                       db1.data_ptr<float>(), dw2.data_ptr<float>(),
                       db2.data_ptr<float>(), M, K, N1, N2);
    return {dx, dw1, db1, dw2, db2};
}

std::vector<torch::Tensor> dense_head2_fwd(torch::Tensor x,
                                           torch::Tensor w1, torch::Tensor b1,
                                           torch::Tensor w2,
                                           torch::Tensor b2) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous());
    const int M = (int)x.size(0), K = (int)x.size(1);
    const int N1 = (int)w1.size(0), N2 = (int)w2.size(0);
    TORCH_CHECK(K % 8 == 0 && K <= 4096 && N1 <= 256 && N2 <= 256 &&
                M <= 4096, "dense_head2_fwd shape envelope");
    auto h1 = torch::empty({M, (int64_t)N1}, x.options());
    auto logits = torch::empty({M, (int64_t)N2}, x.options());
    const size_t lds = ((K + 7) & ~7) * sizeof(unsigned short) +
                       N1 * sizeof(float);
    hipLaunchKernelGGL(dense_head2_fwd_kernel, dim3(M), dim3(256), lds,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                       bf_ptr(w1), b1.data_ptr<float>(), bf_ptr(w2),
                       b2.data_ptr<float>(), bf_ptr_mut(h1),
                       bf_ptr_mut(logits), M, K, N1, N2);
    return {h1, logits};
}

std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor labels,
                                            torch::Tensor acc_loss,
                                            torch::Tensor acc_correct) {
    CHECK_GPU(logits);
    TORCH_CHECK(logits.is_contiguous());
    const int M = (int)logits.size(0), C = (int)logits.size(1);
    auto probs = torch::empty({M, C}, logits.options().dtype(torch::kFloat32));
    const int waves_per_block = 8;
    auto stream = at::cuda::getCurrentCUDAStream();
    if (M <= 1024) {
        // single block: loss written directly, no zero-fill kernel
        auto loss = torch::empty({}, logits.options().dtype(torch::kFloat32));
        hipLaunchKernelGGL((softmax_xent_fwd_kernel<true>), dim3(1),
                           dim3(64 * waves_per_block), 0, stream,
                           bf_ptr(logits), labels.data_ptr<int64_t>(),
                           probs.data_ptr<float>(), loss.data_ptr<float>(), M,
                           C,
                           acc_loss.numel() ? acc_loss.data_ptr<float>() : nullptr,
                           acc_correct.numel() ? acc_correct.data_ptr<float>()
                                               : nullptr);
        return {loss, probs};
    }
    auto loss = torch::zeros({}, logits.options().dtype(torch::kFloat32));
    int blocks = ceildiv(M, waves_per_block);
    hipLaunchKernelGGL((softmax_xent_fwd_kernel<false>), dim3(blocks),
                       dim3(64 * waves_per_block), 0, stream, bf_ptr(logits),
                       labels.data_ptr<int64_t>(), probs.data_ptr<float>(),
                       loss.data_ptr<float>(), M, C,
                       acc_loss.numel() ? acc_loss.data_ptr<float>() : nullptr,
                       acc_correct.numel() ? acc_correct.data_ptr<float>()
                                           : nullptr);
    return {loss, probs};
}

torch::Tensor softmax_xent_bwd(torch::Tensor probs, torch::Tensor labels,
                               torch::Tensor dloss, bool out_bf16) {
    CHECK_GPU(probs);
    const int64_t M = probs.size(0);
    const int C = (int)probs.size(1);
    auto dlogits = torch::empty({M, C}, probs.options().dtype(
                                    out_bf16 ? torch::kBFloat16 : torch::kFloat32));
    int64_t total = M * C;
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    if (out_bf16)
        hipLaunchKernelGGL((softmax_xent_bwd_kernel<true>), dim3(blocks),
                           dim3(256), 0, at::cuda::getCurrentCUDAStream(),
                           probs.data_ptr<float>(), labels.data_ptr<int64_t>(),
                           dloss.data_ptr<float>(),
                           (void*)bf_ptr_mut(dlogits), M, C);
    else
        hipLaunchKernelGGL((softmax_xent_bwd_kernel<false>), dim3(blocks),
                           dim3(256), 0, at::cuda::getCurrentCUDAStream(),
                           probs.data_ptr<float>(), labels.data_ptr<int64_t>(),
                           dloss.data_ptr<float>(),
                           (void*)dlogits.data_ptr<float>(), M, C);
    return dlogits;
}

void fused_adam(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, double lr, double b1, double b2, double eps,
                double bc1, double bc2) {
    CHECK_GPU(p);
    auto gc = g.contiguous();
    int64_t total = p.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(fused_adam_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), p.data_ptr<float>(),
                       gc.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), total, (float)lr, (float)b1,
                       (float)b2, (float)eps, (float)bc1, (float)bc2);
}

void adam_prep(torch::Tensor step, torch::Tensor sched, torch::Tensor hyper,
               double b1, double b2) {
    CHECK_GPU(step);
    hipLaunchKernelGGL(adam_prep_kernel, dim3(1), dim3(64), 0,
                       at::cuda::getCurrentCUDAStream(),
                       step.data_ptr<int64_t>(), sched.data_ptr<float>(),
                       hyper.data_ptr<float>(), (float)b1, (float)b2);
}


torch::Tensor relu_bwd(torch::Tensor dy, torch::Tensor y) {
    CHECK_GPU(dy);
    auto dx = torch::empty_like(dy);
    int64_t total = dy.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(relu_bwd_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dy), bf_ptr(y),
                       bf_ptr_mut(dx), total);
    return dx;
}

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta, torch::Tensor rmean,
                                  torch::Tensor rvar, double eps,
                                  double momentum, bool relu,
                                  torch::Tensor sums) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous());
    const int C = (int)x.size(-1);
    const int64_t M = x.numel() / C;
    auto f32 = x.options().dtype(torch::kFloat32);
    auto mean = torch::empty({C}, f32);
    auto invstd = torch::empty({C}, f32);
    auto y = torch::empty_like(x);
    auto stream = at::cuda::getCurrentCUDAStream();
    int rpb = (int)std::max<int64_t>(64, (M + 511) / 512);  // fill 2 blocks/CU
    int nblk = (int)((M + rpb - 1) / rpb);
    if (sums.numel()) {
        // atomic path (epoch-graph contract, ops/functional.py): partials
        // straight into the caller's [2*C] sums buffer (zero at entry),
        // stats derived inline by the apply kernel — no finalize launch.
        TORCH_CHECK(sums.numel() == 2 * C);
        hipLaunchKernelGGL(bn_partial_kernel<true>, dim3(nblk), dim3(256), 0,
                           stream, bf_ptr(x), sums.data_ptr<float>(), M, C,
                           rpb);
        int64_t total = x.numel();
        int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
        hipLaunchKernelGGL(
            bn_apply_stats_kernel, dim3(blocks), dim3(256), 0, stream,
            bf_ptr(x), sums.data_ptr<float>(), gamma.data_ptr<float>(),
            beta.data_ptr<float>(), bf_ptr_mut(y), total, C, M, (float)eps,
            (float)momentum, mean.data_ptr<float>(),
            invstd.data_ptr<float>(),
            rmean.numel() ? rmean.data_ptr<float>() : nullptr,
            rvar.numel() ? rvar.data_ptr<float>() : nullptr, relu ? 1 : 0,
            fdiv_make((unsigned)((C & 7) == 0 ? C / 8 : C)));
        return {y, mean, invstd};
    }
    auto slab = torch::empty({nblk, 2, C}, f32);
    static const bool bn_fuse = [] {
        const char* e = getenv("HEFL_BN_FUSE");
        return e && e[0] == '1';  // probe: last-block finalize
    }();
    if (bn_fuse && (C & 7) == 0 && C <= 2048) {
        static torch::Tensor ctr;
        if (!ctr.defined() || ctr.device() != x.device())
            ctr = torch::zeros({1}, x.options().dtype(torch::kInt32));
        hipLaunchKernelGGL(bn_partial_fused_kernel, dim3(nblk), dim3(256), 0,
                           stream, bf_ptr(x), slab.data_ptr<float>(), M, C,
                           rpb, mean.data_ptr<float>(),
                           invstd.data_ptr<float>(),
                           rmean.numel() ? rmean.data_ptr<float>() : nullptr,
                           rvar.numel() ? rvar.data_ptr<float>() : nullptr,
                           (float)eps, (float)momentum,
                           ctr.data_ptr<int>());
    } else {
        hipLaunchKernelGGL(bn_partial_kernel<false>, dim3(nblk), dim3(256), 0,
                           stream, bf_ptr(x), slab.data_ptr<float>(), M, C,
                           rpb);
        hipLaunchKernelGGL(bn_finalize_kernel, dim3(C), dim3(256), 0,
                           stream, slab.data_ptr<float>(), nblk,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           rmean.numel() ? rmean.data_ptr<float>() : nullptr,
                           rvar.numel() ? rvar.data_ptr<float>() : nullptr, M,
                           C, (float)eps, (float)momentum);
    }
    int64_t total = x.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(bn_apply_kernel, dim3(blocks), dim3(256), 0, stream,
                       bf_ptr(x), mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), bf_ptr_mut(y), total, C,
                       relu ? 1 : 0, fdiv_make((unsigned)((C & 7) == 0 ? C / 8 : C)));
    return {y, mean, invstd};
}

torch::Tensor bn_apply(torch::Tensor x, torch::Tensor mean,
                       torch::Tensor invstd, torch::Tensor gamma,
                       torch::Tensor beta, bool relu) {
    CHECK_GPU(x);
    const int C = (int)x.size(-1);
    auto y = torch::empty_like(x);
    int64_t total = x.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(bn_apply_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                       mean.data_ptr<float>(), invstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       bf_ptr_mut(y), total, C, relu ? 1 : 0, fdiv_make((unsigned)((C & 7) == 0 ? C / 8 : C)));
    return y;
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor mean, torch::Tensor invstd,
                                  torch::Tensor gamma, bool train,
                                  torch::Tensor relu_y,
                                  torch::Tensor fwd_sums) {
    CHECK_GPU(dy);
    auto dyc = dy.contiguous();
    const unsigned short* ry =
        relu_y.numel() ? bf_ptr(relu_y) : nullptr;
    const int C = (int)x.size(-1);
    const int64_t M = x.numel() / C;
    auto f32 = x.options().dtype(torch::kFloat32);
    auto dx = torch::empty_like(x);
    auto stream = at::cuda::getCurrentCUDAStream();
    int rpb = (int)std::max<int64_t>(64, (M + 511) / 512);  // fill 2 blocks/CU
    int nblk = (int)((M + rpb - 1) / rpb);
    torch::Tensor dgamma, dbeta;
    if (fwd_sums.numel()) {
        // atomic path (HEFL_BN_ATOMIC probe): partials straight into
        // {dgamma;dbeta}; zeroed allocations keep the probe correct
        // standalone (the two fills cost what the finalize launch saved —
        // the probe's measured loss is the hot-word atomic contention,
        // see ops/functional.py); block 0 zeroes this layer's forward
        // sums for the next replay.
        dgamma = torch::zeros({C}, f32);
        dbeta = torch::zeros({C}, f32);
        hipLaunchKernelGGL(bn_bwd_partial_kernel<true>, dim3(nblk), dim3(256),
                           0, stream, bf_ptr(dyc), bf_ptr(x), ry,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           dgamma.data_ptr<float>(), M, C, rpb,
                           dbeta.data_ptr<float>(),
                           fwd_sums.data_ptr<float>());
    } else {
        dgamma = torch::empty({C}, f32);
        dbeta = torch::empty({C}, f32);
        auto slab = torch::empty({nblk, 2, C}, f32);
        hipLaunchKernelGGL(bn_bwd_partial_kernel<false>, dim3(nblk), dim3(256),
                           0, stream, bf_ptr(dyc), bf_ptr(x), ry,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           slab.data_ptr<float>(), M, C, rpb, nullptr,
                           nullptr);
        hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(C), dim3(256), 0,
                           stream, slab.data_ptr<float>(), nblk,
                           dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                           C);
    }
    int64_t total = x.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(bn_dx_kernel, dim3(blocks), dim3(256), 0, stream,
                       bf_ptr(dyc), bf_ptr(x), ry, mean.data_ptr<float>(),
                       invstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                       bf_ptr_mut(dx), total, C, M, train ? 1 : 0, fdiv_make((unsigned)((C & 7) == 0 ? C / 8 : C)));
    return {dx, dgamma, dbeta};
}

std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k, int64_t s,
                                       int64_t p) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous());
    const int N = (int)x.size(0), H = (int)x.size(1), W = (int)x.size(2),
              C = (int)x.size(3);
    const int OH = (H + 2 * (int)p - (int)k) / (int)s + 1;
    const int OW = (W + 2 * (int)p - (int)k) / (int)s + 1;
    auto y = torch::empty({N, OH, OW, C}, x.options());
    auto idx = torch::empty({N, OH, OW, C}, x.options().dtype(torch::kUInt8));
    int64_t total = (int64_t)N * OH * OW * C;
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(maxpool_gen_fwd_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                       bf_ptr_mut(y), idx.data_ptr<uint8_t>(), N, H, W, C, OH,
                       OW, (int)k, (int)s, (int)p,
                       fdiv_make((unsigned)C), fdiv_make((unsigned)OW),
                       fdiv_make((unsigned)OH));
    return {y, idx};
}

torch::Tensor maxpool_bwd(torch::Tensor dy, torch::Tensor idx, int64_t H,
                          int64_t W, int64_t k, int64_t s, int64_t p) {
    CHECK_GPU(dy);
    const int N = (int)dy.size(0), OH = (int)dy.size(1), OW = (int)dy.size(2),
              C = (int)dy.size(3);
    auto stream = at::cuda::getCurrentCUDAStream();
    const int64_t in_total = (int64_t)N * H * W * C;
    torch::Tensor dx32h;
    float* dx32p = acc_pool(in_total, dy.options(), stream, dx32h);
    int64_t total = (int64_t)N * OH * OW * C;
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(maxpool_gen_bwd_kernel, dim3(blocks), dim3(256), 0,
                       stream, bf_ptr(dy),
                       idx.data_ptr<uint8_t>(), dx32p, N,
                       (int)H, (int)W, C, OH, OW, (int)k, (int)s, (int)p,
                       fdiv_make((unsigned)C), fdiv_make((unsigned)OW),
                       fdiv_make((unsigned)OH), fdiv_make((unsigned)k));
    auto dx = torch::empty({N, H, W, C}, dy.options());
    hipLaunchKernelGGL(cast_f32_bf16_kernel,
                       dim3((int)std::min<int64_t>(ceildiv(in_total, 256), 2048)),
                       dim3(256), 0, stream, dx32p, bf_ptr_mut(dx), in_total,
                       1);
    return dx;
}

torch::Tensor avgpool_global_fwd(torch::Tensor x) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous());
    const int N = (int)x.size(0), HW = (int)(x.size(1) * x.size(2)),
              C = (int)x.size(3);
    auto y = torch::empty({N, C}, x.options());
    int64_t total = (int64_t)N * C;
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(avgpool_global_fwd_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(x),
                       bf_ptr_mut(y), N, HW, C, fdiv_make((unsigned)C));
    return y;
}

torch::Tensor avgpool_global_bwd(torch::Tensor dy, int64_t H, int64_t W) {
    CHECK_GPU(dy);
    const int N = (int)dy.size(0), C = (int)dy.size(1);
    auto dx = torch::empty({N, H, W, C}, dy.options());
    int64_t total = dx.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(avgpool_global_bwd_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dy),
                       bf_ptr_mut(dx), N, (int)(H * W), C,
                       fdiv_make((unsigned)C),
                       fdiv_make((unsigned)(H * W)));
    return dx;
}

torch::Tensor pad_channels(torch::Tensor x, int64_t C8) {
    CHECK_GPU(x);
    TORCH_CHECK(x.is_contiguous() && x.dtype() == torch::kBFloat16);
    const int C = (int)x.size(-1);
    TORCH_CHECK(C <= C8);
    auto sizes = x.sizes().vec();
    sizes.back() = C8;
    auto out = torch::empty(sizes, x.options());
    int64_t total = out.numel();
    auto stream = at::cuda::getCurrentCUDAStream();
    int blocks = (int)std::min<int64_t>(ceildiv(total, 256), 4096);
    hipLaunchKernelGGL(pad_channels_kernel, dim3(blocks), dim3(256), 0, stream,
                       bf_ptr(x), bf_ptr_mut(out), total, C, (int)C8);
    return out;
}

torch::Tensor add_relu(torch::Tensor a, torch::Tensor b) {
    CHECK_GPU(a);
    auto y = torch::empty_like(a);
    int64_t total = a.numel();
    int blocks = std::min<int64_t>(ceildiv(total, 256), 2048);
    hipLaunchKernelGGL(add_relu_kernel, dim3(blocks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(a), bf_ptr(b),
                       bf_ptr_mut(y), total);
    return y;
}

torch::Tensor bias_grad(torch::Tensor dy, int64_t gkey) {
    CHECK_GPU(dy);
    auto dyc = dy.contiguous();
    const int K = (int)dyc.size(-1);
    const int64_t M = dyc.numel() / K;
    int rpb = (int)std::max<int64_t>(2048, (M + 63) / 64);
    dim3 grid(K, (unsigned)((M + rpb - 1) / rpb));
    auto stream0 = at::cuda::getCurrentCUDAStream();
    torch::Tensor db;
    bool db_pooled = false;
    if (grid.y > 1 && gkey)
        db = grad_buf(gkey, {K}, dyc.options(), stream0, db_pooled);
    if (!db_pooled)
        db = grid.y == 1
                 ? torch::empty({K}, dyc.options().dtype(torch::kFloat32))
                 : torch::zeros({K}, dyc.options().dtype(torch::kFloat32));
    hipLaunchKernelGGL(bias_grad_kernel, grid, dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(), bf_ptr(dyc),
                       db.data_ptr<float>(), M, K, rpb);
    return db;
}

void fused_adam_mt(torch::Tensor meta, torch::Tensor ptrs, torch::Tensor sizes,
                   int64_t nchunks, torch::Tensor sched, double b1, double b2,
                   double eps, int64_t zero_g, int64_t sched_off) {
    CHECK_GPU(meta);
    hipLaunchKernelGGL(fused_adam_mt_kernel, dim3((unsigned)nchunks), dim3(256),
                       0, at::cuda::getCurrentCUDAStream(),
                       meta.data_ptr<int64_t>(), ptrs.data_ptr<int64_t>(),
                       sizes.data_ptr<int64_t>(), sched.data_ptr<float>(),
                       (float)b1, (float)b2, (float)eps, (int)zero_g,
                       (int)sched_off);
}

void adam_prep_epoch(torch::Tensor step, torch::Tensor sched,
                     torch::Tensor hyper, double b1, double b2, int64_t S) {
    CHECK_GPU(step);
    hipLaunchKernelGGL(adam_prep_epoch_kernel, dim3(1), dim3(64), 0,
                       at::cuda::getCurrentCUDAStream(),
                       step.data_ptr<int64_t>(), sched.data_ptr<float>(),
                       hyper.data_ptr<float>(), (float)b1, (float)b2, (int)S);
}

void pack_mt(torch::Tensor meta, torch::Tensor ptrs, torch::Tensor sizes,
             torch::Tensor offs, int64_t nchunks, torch::Tensor flat) {
    CHECK_GPU(meta);
    hipLaunchKernelGGL(pack_mt_kernel, dim3((unsigned)nchunks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(),
                       meta.data_ptr<int64_t>(), ptrs.data_ptr<int64_t>(),
                       sizes.data_ptr<int64_t>(), offs.data_ptr<int64_t>(),
                       flat.data_ptr<float>());
}

void unpack_mt(torch::Tensor flat, torch::Tensor meta, torch::Tensor ptrs,
               torch::Tensor shptrs, torch::Tensor sizes, torch::Tensor offs,
               int64_t nchunks) {
    CHECK_GPU(meta);
    hipLaunchKernelGGL(unpack_mt_kernel, dim3((unsigned)nchunks), dim3(256), 0,
                       at::cuda::getCurrentCUDAStream(),
                       flat.data_ptr<float>(), meta.data_ptr<int64_t>(),
                       ptrs.data_ptr<int64_t>(), shptrs.data_ptr<int64_t>(),
                       sizes.data_ptr<int64_t>(), offs.data_ptr<int64_t>());
}

