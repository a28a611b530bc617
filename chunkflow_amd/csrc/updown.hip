// Up/down-sampling convolutions of the RSUNet encoder/decoder on gfx950.
//
// The reference runs these through torch -> MIOpen (ConvTranspose3d /
// strided Conv3d). The (1,2,2)-kernel, (1,2,2)-stride shapes are pure
// HBM-streaming work (a C-vector read + K-vector write per spatial
// position; no tap overlap), and MIOpen's bf16 bwd_data implicit GEMM is
// far off that roofline on the big up-conv (measured 8.0 ms vs ~0.2 ms
// algorithmic at 36->28 x 128^2, profiles/updown_probe_r02.json). These
// kernels are VALU streams: input row and weights staged to LDS in
// batched coalesced loads (a per-load global read next to its use costs a
// full vmcnt drain — the round-1 staging lesson), per-thread accumulators
// over 8 x-positions so each weight vector is read once per c.
//
// Layouts: NDHWC (channels_last_3d); weights packed
// [parity q=(py<<1)|px][C][K] (stored in LDS as [c][k][q] float4 so one
// ds_read_b128 serves all four parities). bias f32 (may be NULL).
// No activation (RSUNet applies none here). f32 accumulation.
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "cfx_internal.h"

namespace {

typedef __hip_bfloat16 cfx_bf16;

template <typename T>
__device__ __forceinline__ float ldf(const T* p);
template <>
__device__ __forceinline__ float ldf<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float ldf<cfx_bf16>(const cfx_bf16* p) {
    return __bfloat162float(*p);
}
__device__ __forceinline__ float cvf(float v) { return v; }
__device__ __forceinline__ float cvf(cfx_bf16 v) {
    return __bfloat162float(v);
}

// 4-wide raw vector per dtype (__hip_bfloat16 is a struct, so its raw
// bits travel as ushort lanes)
template <typename T>
struct vec4;
template <>
struct vec4<float> {
    typedef float type __attribute__((ext_vector_type(4)));
    static __device__ __forceinline__ float get(type v, int i) {
        return v[i];
    }
};
template <>
struct vec4<cfx_bf16> {
    typedef unsigned short type __attribute__((ext_vector_type(4)));
    static __device__ __forceinline__ float get(type v, int i) {
        return __bfloat162float(
            __hip_bfloat16(__hip_bfloat16_raw{(unsigned short)v[i]}));
    }
};
template <typename T>
__device__ __forceinline__ void stf(T* p, float v);
template <>
__device__ __forceinline__ void stf<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void stf<cfx_bf16>(cfx_bf16* p, float v) {
    *p = __float2bfloat16(v);
}

// cooperative stage of `rows` input rows (each `xn` positions x C
// channels, row r at in + row_off[r]) into s_in[r * xn * CPAD + x * CPAD
// + c] as f32, batched so the loads pipeline (no per-load drain).
template <typename T>
__device__ __forceinline__ void stage_rows(
    float* s_in, const T* in, const long long* row_off, int rows, int xn,
    int xvalid, int C, int CPAD, int tid, int nthreads) {
    const int total = rows * xn * C;
    for (int idx = tid; idx < total; idx += nthreads) {
        const int c = idx % C;
        const int x = (idx / C) % xn;
        const int r = idx / (C * xn);
        const float v = x < xvalid
                            ? ldf(&in[row_off[r] + (long long)x * C + c])
                            : 0.f;
        s_in[(r * xn + x) * CPAD + c] = v;
    }
}

// ---- transposed conv (1,2,2), stride (1,2,2) ------------------------------
// out[n, z, 2y+py, 2x+px, k] = bias[k] + sum_c in[n, z, y, x, c] * w[q][c][k]
// One workgroup: one (n, z, y) input row, XI input x positions, all K.
// 256 threads = 8 pos-slots x 32 k-slots; each thread accumulates XI/8
// x-positions x 4 parities so w4 is read once per (c, k).
template <typename T, int XI>
__global__ __launch_bounds__(256, 2) void k_upconv2(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W, int C, int K) {
    constexpr int PX = XI / 8;  // x-positions per thread
    extern __shared__ float smem[];
    float* s_w = smem;                    // [C][K][4]
    float* s_in = smem + C * K * 4;       // [XI][CPAD]
    const int CPAD = C | 1;
    const int tid = threadIdx.x;
    const int kslot = tid & 31;
    const int pslot = tid >> 5;

    for (int idx = tid; idx < C * K * 4; idx += 256) {
        const int q = idx & 3;
        const int k = (idx >> 2) % K;
        const int c = (idx >> 2) / K;
        s_w[(c * K + k) * 4 + q] = ldf(&wgt[(q * C + c) * K + k]);
    }

    const int nz = blockIdx.z;           // n * D + z
    const int y = blockIdx.y;
    const int x0 = blockIdx.x * XI;
    const int xvalid = min(XI, W - x0);
    long long row_off[1] = {((((long long)nz * H) + y) * W + x0) * C};
    stage_rows(s_in, in, row_off, 1, XI, xvalid, C, CPAD, tid, 256);
    __syncthreads();

    const long long out_base =
        ((((long long)nz * 2 * H) + 2 * y) * 2 * W) * K;

    for (int k = kslot; k < K; k += 32) {
        const float bj = bias ? bias[k] : 0.f;
        float acc[PX][4];
#pragma unroll
        for (int p = 0; p < PX; ++p)
#pragma unroll
            for (int q = 0; q < 4; ++q) acc[p][q] = bj;
        for (int c = 0; c < C; ++c) {
            const float4 w4 =
                *reinterpret_cast<const float4*>(&s_w[(c * K + k) * 4]);
#pragma unroll
            for (int p = 0; p < PX; ++p) {
                const float v = s_in[(pslot + p * 8) * CPAD + c];
                acc[p][0] += v * w4.x;
                acc[p][1] += v * w4.y;
                acc[p][2] += v * w4.z;
                acc[p][3] += v * w4.w;
            }
        }
#pragma unroll
        for (int p = 0; p < PX; ++p) {
            const int xi = x0 + pslot + p * 8;
            if (xi >= W) continue;
            T* op = out + out_base + (long long)2 * xi * K + k;
            stf(op, acc[p][0]);
            stf(op + K, acc[p][1]);
            stf(op + (long long)2 * W * K, acc[p][2]);
            stf(op + (long long)2 * W * K + K, acc[p][3]);
        }
    }
}

// ---- down conv (1,2,2), stride (1,2,2) ------------------------------------
// out[n, z, y, x, k] = bias[k] + sum_q sum_c in[n, z, 2y+qy, 2x+qx, c] *
//                      w[q][c][k];  H, W are INPUT dims (out H/2 x W/2).
template <typename T, int XO>
__global__ __launch_bounds__(256, 2) void k_downconv2(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W, int C, int K) {
    constexpr int PX = XO / 8;
    extern __shared__ float smem[];
    float* s_w = smem;                       // [C][K][4]
    float* s_in = smem + C * K * 4;          // [2 rows][2*XO][CPAD]
    const int CPAD = C | 1;
    const int tid = threadIdx.x;
    const int kslot = tid & 31;
    const int pslot = tid >> 5;
    const int HO = H >> 1, WO = W >> 1;

    for (int idx = tid; idx < C * K * 4; idx += 256) {
        const int q = idx & 3;
        const int k = (idx >> 2) % K;
        const int c = (idx >> 2) / K;
        s_w[(c * K + k) * 4 + q] = ldf(&wgt[(q * C + c) * K + k]);
    }

    const int nz = blockIdx.z;
    const int y = blockIdx.y;            // output row
    const int x0 = blockIdx.x * XO;      // output x base
    const int xvalid = min(2 * XO, W - 2 * x0);
    long long row_off[2] = {
        ((((long long)nz * H) + 2 * y) * W + 2 * x0) * C,
        ((((long long)nz * H) + 2 * y + 1) * W + 2 * x0) * C};
    stage_rows(s_in, in, row_off, 2, 2 * XO, xvalid, C, CPAD, tid, 256);
    __syncthreads();

    const long long out_row = (((long long)nz * HO) + y) * WO * K;

    for (int k = kslot; k < K; k += 32) {
        const float bj = bias ? bias[k] : 0.f;
        float acc[PX];
#pragma unroll
        for (int p = 0; p < PX; ++p) acc[p] = bj;
        for (int c = 0; c < C; ++c) {
            const float4 w4 =
                *reinterpret_cast<const float4*>(&s_w[(c * K + k) * 4]);
#pragma unroll
            for (int p = 0; p < PX; ++p) {
                const int xl = 2 * (pslot + p * 8);  // local input x
                acc[p] += s_in[xl * CPAD + c] * w4.x;
                acc[p] += s_in[(xl + 1) * CPAD + c] * w4.y;
                acc[p] += s_in[(2 * XO + xl) * CPAD + c] * w4.z;
                acc[p] += s_in[(2 * XO + xl + 1) * CPAD + c] * w4.w;
            }
        }
#pragma unroll
        for (int p = 0; p < PX; ++p) {
            const int xo = x0 + pslot + p * 8;
            if (xo >= WO) continue;
            stf(&out[out_row + (long long)xo * K + k], acc[p]);
        }
    }
}

// ---- input conv (1,5,5), pad (0,2,2), single input channel ---------------
// RSUNet's conv_in (1 -> 28). MIOpen's implicit GEMM collapses to a
// degenerate K-dim=25 GEMM here and runs ~39 ms (bf16) / ~7 ms (f32) per
// batch-24 launch vs ~0.5 ms of algorithmic traffic. Plain 2-D stencil:
// 5 input rows staged to LDS, per-thread weights in registers, PX
// contiguous x positions per thread so each input value loads once.
// out[n,z,y,x,k] = bias[k] + sum_{dy,dx} in[n,z,y+dy-2,x+dx-2] * w[k][tap]
template <typename T, int XI, int KC>  // KC: compile-time K (0 = runtime)
__global__ __launch_bounds__(256, 2) void k_conv155_c1(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W, int K_rt) {
    const int K = KC > 0 ? KC : K_rt;
    // thread = one output position, computing ALL K: the K outputs of one
    // position are contiguous in NDHWC, so each thread's stores are a
    // K-element run and a wave's stores coalesce (the k-parallel variant
    // wrote 2-byte strided singles and was store-issue-bound, 5.3 ms)
    __shared__ float s_in[5][XI + 4];
    __shared__ float s_w[32 * 28];        // rows padded 25 -> 28 (16 B)
    const int tid = threadIdx.x;          // position slot (XI == 256)
    const int nz = blockIdx.z;
    const int y = blockIdx.y;
    const int x0 = blockIdx.x * XI;

    const long long plane = (long long)nz * H;
    for (int idx = tid; idx < 5 * (XI + 4); idx += XI) {
        const int dy = idx / (XI + 4);
        const int xl = idx % (XI + 4);
        const int gy = y + dy - 2;
        const int gx = x0 + xl - 2;
        const bool ok = gy >= 0 && gy < H && gx >= 0 && gx < W;
        s_in[dy][xl] = ok ? ldf(&in[(plane + gy) * W + gx]) : 0.f;
    }
    for (int idx = tid; idx < K * 28; idx += XI) {
        const int t = idx % 28;
        const int k = idx / 28;
        s_w[idx] = t < 25 ? ldf(&wgt[k * 25 + t]) : 0.f;
    }
    __syncthreads();

    const int gx = x0 + tid;
    if (gx >= W) return;
    // the 5x5 window of this position, registers
    float v[25];
#pragma unroll
    for (int dy = 0; dy < 5; ++dy)
#pragma unroll
        for (int dx = 0; dx < 5; ++dx)
            v[dy * 5 + dx] = s_in[dy][tid + dx];
    T* op = out + (plane + y) * (long long)W * K + (long long)gx * K;
    if (KC > 0 && KC % 4 == 0) {
        // compile-time K in groups of 4: scalar accumulators packed into
        // 8-byte vector stores. No register array address-taking (spills
        // to scratch) and no full k unroll (28x7 hoisted float4 weight
        // reads blow the register file) — both measured ~10x slower.
        typedef unsigned int uint2v __attribute__((ext_vector_type(2)));
        typedef float float2v __attribute__((ext_vector_type(2)));
#pragma unroll 1
        for (int k = 0; k < KC; k += 4) {
            float a0 = bias ? bias[k] : 0.f;
            float a1 = bias ? bias[k + 1] : 0.f;
            float a2 = bias ? bias[k + 2] : 0.f;
            float a3 = bias ? bias[k + 3] : 0.f;
#pragma unroll
            for (int t4 = 0; t4 < 24; t4 += 4) {
                const float4 w0 = *reinterpret_cast<const float4*>(
                    &s_w[k * 28 + t4]);
                const float4 w1 = *reinterpret_cast<const float4*>(
                    &s_w[(k + 1) * 28 + t4]);
                const float4 w2 = *reinterpret_cast<const float4*>(
                    &s_w[(k + 2) * 28 + t4]);
                const float4 w3 = *reinterpret_cast<const float4*>(
                    &s_w[(k + 3) * 28 + t4]);
                a0 += v[t4] * w0.x + v[t4 + 1] * w0.y + v[t4 + 2] * w0.z +
                      v[t4 + 3] * w0.w;
                a1 += v[t4] * w1.x + v[t4 + 1] * w1.y + v[t4 + 2] * w1.z +
                      v[t4 + 3] * w1.w;
                a2 += v[t4] * w2.x + v[t4 + 1] * w2.y + v[t4 + 2] * w2.z +
                      v[t4 + 3] * w2.w;
                a3 += v[t4] * w3.x + v[t4 + 1] * w3.y + v[t4 + 2] * w3.z +
                      v[t4 + 3] * w3.w;
            }
            a0 += v[24] * s_w[k * 28 + 24];
            a1 += v[24] * s_w[(k + 1) * 28 + 24];
            a2 += v[24] * s_w[(k + 2) * 28 + 24];
            a3 += v[24] * s_w[(k + 3) * 28 + 24];
            if (sizeof(T) == 2) {
                uint2v u;
                u.x = (unsigned)((__hip_bfloat16_raw)__float2bfloat16(
                          a0)).x |
                      ((unsigned)((__hip_bfloat16_raw)__float2bfloat16(
                           a1)).x << 16);
                u.y = (unsigned)((__hip_bfloat16_raw)__float2bfloat16(
                          a2)).x |
                      ((unsigned)((__hip_bfloat16_raw)__float2bfloat16(
                           a3)).x << 16);
                *reinterpret_cast<uint2v*>(reinterpret_cast<char*>(op) +
                                           (size_t)k * 2) = u;
            } else {
                float2v f01, f23;
                f01.x = a0;
                f01.y = a1;
                f23.x = a2;
                f23.y = a3;
                float2v* d =
                    reinterpret_cast<float2v*>(reinterpret_cast<char*>(op) +
                                               (size_t)k * 4);
                d[0] = f01;
                d[1] = f23;
            }
        }
    } else {
        for (int k = 0; k < K; ++k) {
            float acc = bias ? bias[k] : 0.f;
#pragma unroll
            for (int t4 = 0; t4 < 24; t4 += 4) {
                const float4 w4 =
                    *reinterpret_cast<const float4*>(&s_w[k * 28 + t4]);
                acc += v[t4] * w4.x + v[t4 + 1] * w4.y +
                       v[t4 + 2] * w4.z + v[t4 + 3] * w4.w;
            }
            acc += v[24] * s_w[k * 28 + 24];
            stf(&op[k], acc);
        }
    }
}

// ---- output conv (1,5,5), pad (0,2,2), few output channels ---------------
// RSUNet's conv_out (28 -> 3). MIOpen's bf16 implicit GEMM degenerates on
// the K=3 output dim (measured 38.6 ms per batch-24 launch). Thread =
// one output position computing all K: 5 input rows staged to LDS in the
// raw dtype, weights [K][25][28-padded] f32 in LDS, f32 accumulation.
template <typename T, int XI, int CC, int KO, int MODE = 0>
__global__ __launch_bounds__(256, 2) void k_conv155_out(
    const T* __restrict__ in, const T* __restrict__ wgt,
    const float* __restrict__ bias, T* __restrict__ out, int N, int D,
    int H, int W) {
    // One WG = 4 output rows x XI x-positions: the 8 staged input rows
    // are shared by the 4 rows (2 LDS rows per output instead of 5 --
    // the single-row version was LDS-bound at 1 WG/CU and latency-bound).
    // 256 threads = 4 waves, wave = local y; lane covers 2 x positions.
    __shared__ T s_in[8][XI + 4][CC];
    __shared__ float s_w[KO][25][28];     // c rows padded to 28 (16 B)
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int nz = blockIdx.z;
    const int y0 = blockIdx.y * 4;
    const int x0 = blockIdx.x * XI;

    const long long plane = (long long)nz * H;
    const int C4 = CC / 4;
    typedef typename vec4<T>::type tx4;
    for (int idx = tid; MODE != 2 && idx < 8 * (XI + 4) * C4;
         idx += 256) {
        const int c4 = idx % C4;
        const int xl = (idx / C4) % (XI + 4);
        const int r = idx / (C4 * (XI + 4));
        const int gy = y0 + r - 2;
        const int gx = x0 + xl - 2;
        const bool ok = gy >= 0 && gy < H && gx >= 0 && gx < W;
        *reinterpret_cast<tx4*>(&s_in[r][xl][c4 * 4]) =
            ok ? *reinterpret_cast<const tx4*>(
                     &in[((plane + gy) * W + gx) * (long long)CC + c4 * 4])
               : tx4{};
    }
    for (int idx = tid; MODE != 2 && idx < KO * 25 * 28; idx += 256) {
        const int c = idx % 28;
        const int t = (idx / 28) % 25;
        const int k = idx / (28 * 25);
        s_w[k][t][c] = c < CC ? ldf(&wgt[(k * 25 + t) * CC + c]) : 0.f;
    }
    __syncthreads();

    const int y = y0 + wave;
    if (y >= H) return;
    float acc[2][KO];
#pragma unroll
    for (int p = 0; p < 2; ++p)
#pragma unroll
        for (int k = 0; k < KO; ++k) acc[p][k] = bias ? bias[k] : 0.f;
#pragma unroll 1
    for (int tap = 0; tap < 25 && MODE != 1; ++tap) {  // MODE 1: skip
        const int dy = tap / 5, dx = tap % 5;
        // unroll 2 gives the dependent LDS reads cross-iteration ILP
        // (full unroll hoists 7x3 float4 weight reads and spills)
#pragma unroll 2
        for (int c4 = 0; c4 < C4; ++c4) {
            float4 w4k[KO];
#pragma unroll
            for (int k = 0; k < KO; ++k)
                w4k[k] = *reinterpret_cast<const float4*>(
                    &s_w[k][tap][c4 * 4]);
#pragma unroll
            for (int p = 0; p < 2; ++p) {
                const tx4 v4 = *reinterpret_cast<const tx4*>(
                    &s_in[wave + dy][lane * 2 + p + dx][c4 * 4]);
                const float f0 = vec4<T>::get(v4, 0);
                const float f1 = vec4<T>::get(v4, 1);
                const float f2 = vec4<T>::get(v4, 2);
                const float f3 = vec4<T>::get(v4, 3);
#pragma unroll
                for (int k = 0; k < KO; ++k)
                    acc[p][k] += f0 * w4k[k].x + f1 * w4k[k].y +
                                 f2 * w4k[k].z + f3 * w4k[k].w;
            }
        }
    }
#pragma unroll
    for (int p = 0; p < 2; ++p) {
        const int gx = x0 + lane * 2 + p;
        if (gx >= W) continue;
        T* op =
            out + (plane + y) * (long long)W * KO + (long long)gx * KO;
#pragma unroll
        for (int k = 0; k < KO; ++k) stf(&op[k], acc[p][k]);
    }
}

}  // namespace

extern "C" int cfx_conv155_out(cfx_ctx* ctx, const void* in,
                               const void* wgt, const float* bias,
                               void* out, int N, int D, int H, int W,
                               int C, int K, int is_bf16) {
    if (C != 28 || K != 3) {
        g_err = "cfx_conv155_out: only C == 28, K == 3 instantiated";
        return -1;
    }
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    constexpr int XI = 128;  // wave covers 2*64 x; 4 rows per WG
    dim3 grid((W + XI - 1) / XI, (H + 3) / 4, (unsigned)(N * D));
    static const int om = [] {
        const char* e = getenv("CFX_CONVOUT_MODE");  // timing ablation
        return e ? atoi(e) : 0;
    }();
    if (is_bf16 && om == 1)
        hipLaunchKernelGGL((k_conv155_out<cfx_bf16, XI, 28, 3, 1>), grid,
                           dim3(256), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N,
                           D, H, W);
    else if (is_bf16 && om == 2)
        hipLaunchKernelGGL((k_conv155_out<cfx_bf16, XI, 28, 3, 2>), grid,
                           dim3(256), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N,
                           D, H, W);
    else if (is_bf16)
        hipLaunchKernelGGL((k_conv155_out<cfx_bf16, XI, 28, 3>), grid,
                           dim3(256), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N,
                           D, H, W);
    else
        hipLaunchKernelGGL((k_conv155_out<float, XI, 28, 3>), grid,
                           dim3(256), 0, ctx->stream, (const float*)in,
                           (const float*)wgt, bias, (float*)out, N, D, H,
                           W);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 25.0 * C * K * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV_STREAM, flops)) return -1;
    return 0;
}

extern "C" int cfx_conv155_c1(cfx_ctx* ctx, const void* in, const void* wgt,
                              const float* bias, void* out, int N, int D,
                              int H, int W, int K, int is_bf16) {
    if (K > 32) {
        g_err = "cfx_conv155_c1: K <= 32 supported";
        return -1;
    }
    constexpr int XI = 256;
    dim3 grid((W + XI - 1) / XI, H, (unsigned)(N * D));
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    if (is_bf16 && K == 28)
        hipLaunchKernelGGL((k_conv155_c1<cfx_bf16, XI, 28>), grid,
                           dim3(XI), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N,
                           D, H, W, K);
    else if (is_bf16)
        hipLaunchKernelGGL((k_conv155_c1<cfx_bf16, XI, 0>), grid,
                           dim3(XI), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias, (cfx_bf16*)out, N,
                           D, H, W, K);
    else if (K == 28)
        hipLaunchKernelGGL((k_conv155_c1<float, XI, 28>), grid, dim3(XI),
                           0, ctx->stream, (const float*)in,
                           (const float*)wgt, bias, (float*)out, N, D, H,
                           W, K);
    else
        hipLaunchKernelGGL((k_conv155_c1<float, XI, 0>), grid, dim3(XI),
                           0, ctx->stream, (const float*)in,
                           (const float*)wgt, bias, (float*)out, N, D, H,
                           W, K);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 25.0 * K * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV_STREAM, flops)) return -1;
    return 0;
}

extern "C" int cfx_upconv_2x2(cfx_ctx* ctx, const void* in, const void* wgt,
                              const float* bias, void* out, int N, int D,
                              int H, int W, int C, int K, int is_bf16) {
    if (K > 64 || C > 64) {
        g_err = "cfx_upconv_2x2: C, K <= 64 supported";
        return -1;
    }
    constexpr int XI = 64;
    dim3 grid((W + XI - 1) / XI, H, (unsigned)(N * D));
    const size_t shmem =
        ((size_t)C * K * 4 + (size_t)XI * (C | 1)) * sizeof(float);
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
    if (prof_end(ctx, e0, CFX_K_CONV_STREAM, flops)) return -1;
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
    const size_t shmem =
        ((size_t)C * K * 4 + (size_t)2 * 2 * XO * (C | 1)) * sizeof(float);
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
    if (prof_end(ctx, e0, CFX_K_CONV_STREAM, flops)) return -1;
    return 0;
}
