// CKKS canonical-embedding special FFT (encode/decode) on gfx950 — the
// hand-written replacement for the torch complex128 at::native path that
// round-1 profiles showed as the last generic-kernel compute block (~6% of
// the config #5 round). Reference surface: Pyfhel encryptFrac/decryptFrac
// encode/decode (FLPyfhelin.py:217,295); algorithm parity with the CPU
// oracle hefl/he/encoder.py (HEAAN-style butterflies over the 5^j group).
//
// Layout: one workgroup owns a CONTIGUOUS chunk of <= kFftNblk slots of one
// row, SoA re/im doubles in LDS (2 * 8192 * 8 B = 128 KiB of the CU's
// 160 KiB at the largest block). Stages whose butterfly span exceeds the
// chunk run in strided global kernels (only slots = 2^14, i.e. n = 2^15,
// needs one). Twiddles are host-built (float64, interleaved re/im, stage-
// major DIF/DIT order) and shared by every row — they live in L2.
//
// Encode (DIF, large strides first):  u = a + b; t = (a - b) * w
//   then bit-reverse permute, * scale / slots, round to int64 coeffs
//   [re(0..slots) | im(0..slots)] — exactly encoder.encode's layout.
// Decode (DIT): bit-reverse gather of (re, im) = centered coeffs / scale,
//   then u + t*w / u - t*w; real part of the first k slots is the result.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>

#define CHECK_CUDA_OK(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")

namespace {

constexpr int kFftThreads = 256;
constexpr int kFftNblk = 8192;  // slots per LDS block (128 KiB SoA re+im)

__device__ __forceinline__ unsigned brev_bits(unsigned i, int bits) {
    return __brev(i) >> (32 - bits);
}

// ---- encode: one DIF global stage (span lenh > nblk), strided in HBM ----
__global__ void fft_enc_global_kernel(double* __restrict__ re,
                                      double* __restrict__ im,
                                      const double* __restrict__ tw,
                                      int twoff, int slots, int length) {
    const int lenh = length >> 1;
    const int64_t base = (int64_t)blockIdx.y * slots;
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < slots / 2;
         p += gridDim.x * blockDim.x) {
        const int g = p / lenh, j = p % lenh;
        const int64_t i0 = base + (int64_t)g * length + j;
        const int64_t i1 = i0 + lenh;
        const double wr = tw[2 * (twoff + j)], wi = tw[2 * (twoff + j) + 1];
        const double ur = re[i0], ui = im[i0];
        const double vr = re[i1], vi = im[i1];
        re[i0] = ur + vr;
        im[i0] = ui + vi;
        const double dr = ur - vr, di = ui - vi;
        re[i1] = dr * wr - di * wi;
        im[i1] = dr * wi + di * wr;
    }
}

// ---- encode: LDS phase (all stages with length <= nblk) + bitrev store.
// vals != nullptr: first call, load reals from vals and zero imag;
// vals == nullptr: continue from the re/im scratch written by the global
// stage. Output int64 coeffs with the global bit-reverse permutation. ----
__global__ void __launch_bounds__(kFftThreads)
fft_enc_lds_kernel(const double* __restrict__ vals,
                   const double* __restrict__ reg,
                   const double* __restrict__ img, int64_t* __restrict__ out,
                   const double* __restrict__ tw, int twoff0, int slots,
                   int log_slots, int nblk, double scale) {
    extern __shared__ double smem[];
    double* re = smem;
    double* im = smem + nblk;
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    const int64_t row_off = (int64_t)blockIdx.y * slots;
    const int64_t chunk = row_off + (int64_t)blk * nblk;
    if (vals) {
        for (int i = tid; i < nblk; i += kFftThreads) {
            re[i] = vals[chunk + i];
            im[i] = 0.0;
        }
    } else {
        for (int i = tid; i < nblk; i += kFftThreads) {
            re[i] = reg[chunk + i];
            im[i] = img[chunk + i];
        }
    }
    __syncthreads();
    int twoff = twoff0;
    const int nb2 = nblk >> 1;
    for (int length = nblk; length >= 2; length >>= 1) {
        const int lenh = length >> 1;
        for (int p = tid; p < nb2; p += kFftThreads) {
            const int g = p / lenh, j = p % lenh;
            const int i0 = g * length + j;
            const int i1 = i0 + lenh;
            const double wr = tw[2 * (twoff + j)];
            const double wi = tw[2 * (twoff + j) + 1];
            const double ur = re[i0], ui = im[i0];
            const double vr = re[i1], vi = im[i1];
            re[i0] = ur + vr;
            im[i0] = ui + vi;
            const double dr = ur - vr, di = ui - vi;
            re[i1] = dr * wr - di * wi;
            im[i1] = dr * wi + di * wr;
        }
        twoff += lenh;
        __syncthreads();
    }
    // value at global position j lands at out index brev(j); fold in the
    // 1/slots of the inverse embedding and the CKKS Delta scale
    const double s = scale / (double)slots;
    int64_t* orow = out + (int64_t)blockIdx.y * 2 * slots;
    for (int i = tid; i < nblk; i += kFftThreads) {
        const unsigned jglob = (unsigned)(blk * nblk + i);
        const unsigned dst = brev_bits(jglob, log_slots);
        orow[dst] = (int64_t)llrint(re[i] * s);
        orow[dst + slots] = (int64_t)llrint(im[i] * s);
    }
}

// ---- decode: LDS phase (bitrev gather load of coeffs/scale, stages with
// length <= nblk), writing re/im back to scratch (or the final f32 output
// when no global stage follows). ----
__global__ void __launch_bounds__(kFftThreads)
fft_dec_lds_kernel(const int64_t* __restrict__ coeffs,
                   double* __restrict__ reg, double* __restrict__ img,
                   float* __restrict__ outk, const double* __restrict__ tw,
                   int slots, int log_slots, int nblk, double inv_scale,
                   int k) {
    extern __shared__ double smem[];
    double* re = smem;
    double* im = smem + nblk;
    const int tid = threadIdx.x;
    const int blk = blockIdx.x;
    const int64_t row_off = (int64_t)blockIdx.y * slots;
    const int64_t* crow = coeffs + (int64_t)blockIdx.y * 2 * slots;
    for (int i = tid; i < nblk; i += kFftThreads) {
        const unsigned iglob = (unsigned)(blk * nblk + i);
        const unsigned src = brev_bits(iglob, log_slots);
        re[i] = (double)crow[src] * inv_scale;
        im[i] = (double)crow[src + slots] * inv_scale;
    }
    __syncthreads();
    int twoff = 0;
    const int nb2 = nblk >> 1;
    for (int length = 2; length <= nblk; length <<= 1) {
        const int lenh = length >> 1;
        for (int p = tid; p < nb2; p += kFftThreads) {
            const int g = p / lenh, j = p % lenh;
            const int i0 = g * length + j;
            const int i1 = i0 + lenh;
            const double wr = tw[2 * (twoff + j)];
            const double wi = tw[2 * (twoff + j) + 1];
            const double ur = re[i0], ui = im[i0];
            const double tr = re[i1] * wr - im[i1] * wi;
            const double ti = re[i1] * wi + im[i1] * wr;
            re[i0] = ur + tr;
            im[i0] = ui + ti;
            re[i1] = ur - tr;
            im[i1] = ui - ti;
        }
        twoff += lenh;
        __syncthreads();
    }
    if (outk) {  // no global stage follows: emit the first k real slots
        float* orow = outk + (int64_t)blockIdx.y * k;
        for (int i = tid; i < nblk; i += kFftThreads) {
            const int jglob = blk * nblk + i;
            if (jglob < k) orow[jglob] = (float)re[i];
        }
    } else {
        for (int i = tid; i < nblk; i += kFftThreads) {
            reg[row_off + blk * nblk + i] = re[i];
            img[row_off + blk * nblk + i] = im[i];
        }
    }
}

// ---- decode: one DIT global stage (length > nblk) + final f32 emit ----
__global__ void fft_dec_global_kernel(const double* __restrict__ re,
                                      const double* __restrict__ im,
                                      float* __restrict__ outk,
                                      const double* __restrict__ tw,
                                      int twoff, int slots, int length,
                                      int k) {
    const int lenh = length >> 1;
    const int64_t base = (int64_t)blockIdx.y * slots;
    float* orow = outk + (int64_t)blockIdx.y * k;
    for (int p = blockIdx.x * blockDim.x + threadIdx.x; p < slots / 2;
         p += gridDim.x * blockDim.x) {
        const int g = p / lenh, j = p % lenh;
        const int i0g = g * length + j;
        const int i1g = i0g + lenh;
        const int64_t i0 = base + i0g, i1 = base + i1g;
        const double wr = tw[2 * (twoff + j)], wi = tw[2 * (twoff + j) + 1];
        const double ur = re[i0];
        const double tr = re[i1] * wr - im[i1] * wi;
        if (i0g < k) orow[i0g] = (float)(ur + tr);
        if (i1g < k) orow[i1g] = (float)(ur - tr);
    }
}

}  // namespace

// vals f64 [.., slots] -> int64 coeffs [.., 2*slots]. tw_enc: f64
// [slots-1, 2] twiddles in DIF stage order (length = slots down to 2).
torch::Tensor fft_encode(torch::Tensor vals, torch::Tensor tw_enc,
                         double scale) {
    CHECK_CUDA_OK(vals);
    TORCH_CHECK(vals.is_contiguous() && vals.dtype() == torch::kFloat64);
    const int slots = (int)vals.size(-1);
    const int64_t rows = vals.numel() / slots;
    TORCH_CHECK(rows <= 65535, "row count exceeds gridDim.y");
    const int log_slots = 31 - __builtin_clz((unsigned)slots);
    const int nblk = std::min(slots, kFftNblk);
    auto sizes = vals.sizes().vec();
    sizes.back() = 2 * slots;
    auto out = torch::empty(sizes, vals.options().dtype(torch::kInt64));
    auto stream = at::cuda::getCurrentCUDAStream();
    const double* tw = tw_enc.data_ptr<double>();
    if (nblk == slots) {
        dim3 grid(1, (unsigned)rows);
        hipLaunchKernelGGL(fft_enc_lds_kernel, grid, dim3(kFftThreads),
                           2 * nblk * sizeof(double), stream,
                           vals.data_ptr<double>(), nullptr, nullptr,
                           out.data_ptr<int64_t>(), tw, 0, slots, log_slots,
                           nblk, scale);
        return out;
    }
    // global DIF stages for length > nblk, staged through f64 scratch
    auto re = torch::empty_like(vals);
    auto im = torch::zeros_like(vals);
    re.copy_(vals);
    int twoff = 0;
    int blocks = (int)std::min<int64_t>((slots / 2 + kFftThreads - 1) / kFftThreads, 1024);
    for (int length = slots; length > nblk; length >>= 1) {
        hipLaunchKernelGGL(fft_enc_global_kernel, dim3(blocks, (unsigned)rows),
                           dim3(kFftThreads), 0, stream,
                           re.data_ptr<double>(), im.data_ptr<double>(), tw,
                           twoff, slots, length);
        twoff += length >> 1;
    }
    dim3 grid((unsigned)(slots / nblk), (unsigned)rows);
    hipLaunchKernelGGL(fft_enc_lds_kernel, grid, dim3(kFftThreads),
                       2 * nblk * sizeof(double), stream, nullptr,
                       re.data_ptr<double>(), im.data_ptr<double>(),
                       out.data_ptr<int64_t>(), tw, twoff, slots, log_slots,
                       nblk, scale);
    return out;
}

// coeffs int64 (centered) [.., 2*slots] -> f32 [.., k]. tw_dec: f64
// [slots-1, 2] twiddles in DIT stage order (length = 2 up to slots).
torch::Tensor fft_decode(torch::Tensor coeffs, torch::Tensor tw_dec,
                         double scale, int64_t k) {
    CHECK_CUDA_OK(coeffs);
    TORCH_CHECK(coeffs.is_contiguous() && coeffs.dtype() == torch::kInt64);
    const int slots = (int)coeffs.size(-1) / 2;
    const int64_t rows = coeffs.numel() / (2 * slots);
    TORCH_CHECK(rows <= 65535, "row count exceeds gridDim.y");
    const int log_slots = 31 - __builtin_clz((unsigned)slots);
    const int nblk = std::min(slots, kFftNblk);
    auto sizes = coeffs.sizes().vec();
    sizes.back() = k;
    auto out = torch::empty(sizes, coeffs.options().dtype(torch::kFloat32));
    auto stream = at::cuda::getCurrentCUDAStream();
    const double* tw = tw_dec.data_ptr<double>();
    const double inv_scale = 1.0 / scale;
    if (nblk == slots) {
        dim3 grid(1, (unsigned)rows);
        hipLaunchKernelGGL(fft_dec_lds_kernel, grid, dim3(kFftThreads),
                           2 * nblk * sizeof(double), stream,
                           coeffs.data_ptr<int64_t>(), nullptr, nullptr,
                           out.data_ptr<float>(), tw, slots, log_slots, nblk,
                           inv_scale, (int)k);
        return out;
    }
    auto fsz = coeffs.sizes().vec();
    fsz.back() = slots;
    auto re = torch::empty(fsz, coeffs.options().dtype(torch::kFloat64));
    auto im = torch::empty_like(re);
    dim3 grid((unsigned)(slots / nblk), (unsigned)rows);
    hipLaunchKernelGGL(fft_dec_lds_kernel, grid, dim3(kFftThreads),
                       2 * nblk * sizeof(double), stream,
                       coeffs.data_ptr<int64_t>(), re.data_ptr<double>(),
                       im.data_ptr<double>(), nullptr, tw, slots, log_slots,
                       nblk, inv_scale, (int)k);
    int twoff = nblk - 1;  // stages 2..nblk consumed nblk-1 twiddles
    int blocks = (int)std::min<int64_t>((slots / 2 + kFftThreads - 1) / kFftThreads, 1024);
    for (int length = 2 * nblk; length <= slots; length <<= 1) {
        const bool last = (2 * length > slots);
        TORCH_CHECK(last, "decode >1 global stage needs scratch ping-pong");
        hipLaunchKernelGGL(fft_dec_global_kernel, dim3(blocks, (unsigned)rows),
                           dim3(kFftThreads), 0, stream,
                           re.data_ptr<double>(), im.data_ptr<double>(),
                           out.data_ptr<float>(), tw, twoff, slots, length,
                           (int)k);
        twoff += length >> 1;
    }
    return out;
}
