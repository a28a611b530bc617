// Image-normalization kernels: per-section histogram + LUT apply for the
// normalize-contrast operator (reference chunk/image/base.py:93-132,
// flow/flow.py:1672-1710). The LUT itself is computed on the host from the
// device histograms (a few KB; the reference's exact CDF/clamp/round math
// lives in chunkflow_amd/contrast.py).
#include <hip/hip_runtime.h>

#include "cfx_internal.h"

namespace {

// LDS-accumulated 256-bin histogram per section; grid (nwg_x, nsec).
__global__ void k_hist_u8(const unsigned char* __restrict__ in,
                          long long n_per_sec, unsigned int* __restrict__
                          hist /* nsec x 256 */) {
    __shared__ unsigned int lh[256];
    for (int i = threadIdx.x; i < 256; i += blockDim.x) lh[i] = 0;
    __syncthreads();
    int sec = blockIdx.y;
    const unsigned char* p = in + (long long)sec * n_per_sec;
    long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (; i < n_per_sec; i += stride) atomicAdd(&lh[p[i]], 1u);
    __syncthreads();
    for (int b = threadIdx.x; b < 256; b += blockDim.x)
        if (lh[b]) atomicAdd(&hist[sec * 256 + b], lh[b]);
}

// out-of-place LUT gather per section (LDS-staged LUT row)
__global__ void k_lut_apply_u8(unsigned char* __restrict__ buf,
                               long long n_per_sec,
                               const unsigned char* __restrict__
                               lut /* nsec x 256 */) {
    __shared__ unsigned char llut[256];
    int sec = blockIdx.y;
    for (int i = threadIdx.x; i < 256; i += blockDim.x)
        llut[i] = lut[sec * 256 + i];
    __syncthreads();
    unsigned char* p = buf + (long long)sec * n_per_sec;
    long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    long long stride = (long long)gridDim.x * blockDim.x;
    for (; i < n_per_sec; i += stride) p[i] = llut[p[i]];
}

inline int grid_x(long long work) {
    long long want = (work + 255) / 256;
    return (int)(want < 1024 ? (want > 0 ? want : 1) : 1024);
}

}  // namespace

extern "C" int cfx_hist_u8(cfx_ctx* ctx, const unsigned char* in,
                           long long n_per_sec, int nsec,
                           unsigned int* hist) {
    CFX_CHECK(hipMemsetAsync(hist, 0,
                             (size_t)nsec * 256 * sizeof(unsigned int),
                             ctx->stream));
    dim3 grid(grid_x(n_per_sec), nsec);
    hipLaunchKernelGGL(k_hist_u8, grid, dim3(256), 0, ctx->stream, in,
                       n_per_sec, hist);
    CFX_CHECK(hipGetLastError());
    return 0;
}

extern "C" int cfx_lut_apply_u8(cfx_ctx* ctx, unsigned char* buf,
                                long long n_per_sec, int nsec,
                                const unsigned char* lut) {
    dim3 grid(grid_x(n_per_sec), nsec);
    hipLaunchKernelGGL(k_lut_apply_u8, grid, dim3(256), 0, ctx->stream, buf,
                       n_per_sec, lut);
    CFX_CHECK(hipGetLastError());
    return 0;
}
