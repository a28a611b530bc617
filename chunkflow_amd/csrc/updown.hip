// Up/down-sampling convolutions of the RSUNet encoder/decoder on gfx950.
//
// The reference runs these through torch -> MIOpen (ConvTranspose3d /
// strided Conv3d). On MI355X the (1,2,2)-kernel, (1,2,2)-stride shapes are
// pure HBM-streaming work (one input vector read + K-vector write per
// spatial position; zero tap overlap), but MIOpen's bf16 bwd_data
// implicit-GEMM runs them ~20x slower than the roofline. These kernels are
// plain VALU gather/scatter streams: weights LDS-resident, input rows
// served from L1 (threads of one workgroup share rows), f32 accumulation.
//
// Layouts: NDHWC (channels_last_3d); transposed-conv weights packed
// [parity q=(py<<1)|px][C][K]; down-conv weights packed the same.
// bias f32 (may be NULL). No activation (RSUNet applies none here).
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "cfx_internal.h"

namespace {

typedef __hip_bfloat16 cfx_bf16;

template <typename T>
__device__ __forceinline__ float ld(const T* p);
template <>
__device__ __forceinline__ float ld<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float ld<cfx_bf16>(const cfx_bf16* p) {
    return __bfloat162float(*p);
}
template <typename T>
__device__ __forceinline__ void st(T* p, float v);
template <>
__device__ __forceinline__ void st<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void st<cfx_bf16>(cfx_bf16* p, float v) {
    *p = __float2bfloat16(v);
}

// ---- transposed conv (1,2,2), stride (1,2,2) ------------------------------
// out[n, z, 2y+py, 2x+px, k] = bias[k] + sum_c in[n, z, y, x, c] * w[q][c][k]
// One workgroup: one (n, z, y) input row, XI input x positions, all K.
// threads 256 = 8 pos-slots x 32 k-slots; weights staged to LDS as
// [c][k][q] float4 so one ds_read_b128 serves all four parities.
template <typename T, int XI>
__global__ __launch_bounds__(256, 1) void k_upconv2(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W, int C, int K) {
    extern __shared__ float s_w[];  // [C][K][4], K <= 64
    const int tid = threadIdx.x;
    const int kslot = tid & 31;
    const int pslot = tid >> 5;

    // stage weights: wgt is [4][C][K] -> LDS [c][k][q]
    for (int idx = tid; idx < C * K * 4; idx += 256) {
        const int q = idx & 3;
        const int k = (idx >> 2) % K;
        const int c = (idx >> 2) / K;
        s_w[(c * K + k) * 4 + q] = ld(&wgt[(q * C + c) * K + k]);
    }
    __syncthreads();

    const int nz = blockIdx.z;           // n * D + z
    const int y = blockIdx.y;
    const int x0 = blockIdx.x * XI;
    const long long in_row =
        (((long long)nz * H) + y) * W * C;
    const long long out_base = ((long long)nz * 2 * H) * 2 * W * K;

    for (int xi = x0 + pslot; xi < min(x0 + XI, W); xi += 8) {
        const T* ip = in + in_row + (long long)xi * C;
        for (int k = kslot; k < K; k += 32) {
            const float bj = bias ? bias[k] : 0.f;
            float a0 = bj, a1 = bj, a2 = bj, a3 = bj;
            for (int c = 0; c < C; ++c) {
                const float v = ld(&ip[c]);  // same addr across k: broadcast
                const float4 w4 =
                    *reinterpret_cast<const float4*>(&s_w[(c * K + k) * 4]);
                a0 += v * w4.x;
                a1 += v * w4.y;
                a2 += v * w4.z;
                a3 += v * w4.w;
            }
            T* op = out + out_base + ((long long)2 * y * 2 * W +
                                      2 * xi) * K + k;
            st(op, a0);                       // (py=0, px=0)
            st(op + K, a1);                   // (0, 1)
            st(op + (long long)2 * W * K, a2);      // (1, 0)
            st(op + (long long)2 * W * K + K, a3);  // (1, 1)
        }
    }
}

// ---- down conv (1,2,2), stride (1,2,2) ------------------------------------
// out[n, z, y, x, k] = bias[k] + sum_q sum_c in[n, z, 2y+qy, 2x+qx, c] *
//                      w[q][c][k];  H, W are INPUT dims (out H/2 x W/2).
template <typename T, int XO>
__global__ __launch_bounds__(256, 1) void k_downconv2(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W, int C, int K) {
    extern __shared__ float s_w[];  // [c][k][q]
    const int tid = threadIdx.x;
    const int kslot = tid & 31;
    const int pslot = tid >> 5;
    const int HO = H >> 1, WO = W >> 1;

    for (int idx = tid; idx < C * K * 4; idx += 256) {
        const int q = idx & 3;
        const int k = (idx >> 2) % K;
        const int c = (idx >> 2) / K;
        s_w[(c * K + k) * 4 + q] = ld(&wgt[(q * C + c) * K + k]);
    }
    __syncthreads();

    const int nz = blockIdx.z;
    const int y = blockIdx.y;            // output row
    const int x0 = blockIdx.x * XO;
    const long long in_row0 =
        (((long long)nz * H) + 2 * y) * W * C;
    const long long out_row = (((long long)nz * HO) + y) * WO * K;

    for (int xo = x0 + pslot; xo < min(x0 + XO, WO); xo += 8) {
        const T* i00 = in + in_row0 + (long long)2 * xo * C;
        const T* i01 = i00 + C;
        const T* i10 = i00 + (long long)W * C;
        const T* i11 = i10 + C;
        for (int k = kslot; k < K; k += 32) {
            float acc = bias ? bias[k] : 0.f;
            for (int c = 0; c < C; ++c) {
                const float4 w4 =
                    *reinterpret_cast<const float4*>(&s_w[(c * K + k) * 4]);
                acc += ld(&i00[c]) * w4.x;
                acc += ld(&i01[c]) * w4.y;
                acc += ld(&i10[c]) * w4.z;
                acc += ld(&i11[c]) * w4.w;
            }
            st(&out[out_row + (long long)xo * K + k], acc);
        }
    }
}

}  // namespace

extern "C" int cfx_upconv_2x2(cfx_ctx* ctx, const void* in, const void* wgt,
                              const float* bias, void* out, int N, int D,
                              int H, int W, int C, int K, int is_bf16) {
    if (K > 64 || C > 64) {
        g_err = "cfx_upconv_2x2: C, K <= 64 supported";
        return -1;
    }
    constexpr int XI = 64;
    dim3 grid((W + XI - 1) / XI, H, (unsigned)(N * D));
    const size_t shmem = (size_t)C * K * 4 * sizeof(float);
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    if (is_bf16)
        hipLaunchKernelGGL((k_upconv2<cfx_bf16, XI>), grid, dim3(256),
                           shmem, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N, D,
                           H, W, C, K);
    else
        hipLaunchKernelGGL((k_upconv2<float, XI>), grid, dim3(256), shmem,
                           ctx->stream, (const float*)in, (const float*)wgt,
                           bias, (float*)out, N, D, H, W, C, K);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 4.0 * C * K * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
    return 0;
}

extern "C" int cfx_downconv_2x2(cfx_ctx* ctx, const void* in,
                                const void* wgt, const float* bias,
                                void* out, int N, int D, int H, int W,
                                int C, int K, int is_bf16) {
    if (K > 64 || C > 64) {
        g_err = "cfx_downconv_2x2: C, K <= 64 supported";
        return -1;
    }
    if ((H | W) & 1) {
        g_err = "cfx_downconv_2x2: H, W must be even";
        return -1;
    }
    constexpr int XO = 64;
    dim3 grid((W / 2 + XO - 1) / XO, H / 2, (unsigned)(N * D));
    const size_t shmem = (size_t)C * K * 4 * sizeof(float);
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    if (is_bf16)
        hipLaunchKernelGGL((k_downconv2<cfx_bf16, XO>), grid, dim3(256),
                           shmem, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N, D,
                           H, W, C, K);
    else
        hipLaunchKernelGGL((k_downconv2<float, XO>), grid, dim3(256), shmem,
                           ctx->stream, (const float*)in, (const float*)wgt,
                           bias, (float*)out, N, D, H, W, C, K);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 4.0 * C * K * (double)N * D * (H / 2) * (W / 2);
    if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
    return 0;
}
