// Hand-written MFMA 3x3x3 convolution for gfx950 — the RSUNet ResBlock
// conv (C == K in {28, 36, 48, 64}, stride 1, pad 1, NDHWC f32).
//
// VARIANT INDEX (this file is a measured ladder, not dead code; every
// rejected variant stays callable behind its env switch and is covered
// by the sweep test in tests/test_gpu_parity.py; numbers in DESIGN.md
// §10/§10-r2):
//   f32: k_conv3 (slab), k_conv3_w32 (32x32x2), k_conv3_zring(+_pl)
//        [SHIPPED, 96.8 TF C=28], k_conv3_zring_dw (C=48, rejected 37 TF)
//   bf16 C=28 (CFX_ZRING_PL): 0 plain 458 | 1 _pl pipelined 528 |
//        2 _q 4-slot PCB36 104 (LDS conflicts) | 3 _pl2 prefetch 527 |
//        4 _w 2-row swizzled 280 (1 wave/SIMD) | 6 _a alt-acc 540 |
//        7 _t 16x16x32 444 | 8/9 PD/fence ablations | 10/12 timing-only
//        (WRONG results) | 11 _a+transposed epilogue 594-624 |
//        13 +iteration-ahead staging [SHIPPED DEFAULT, 610-646] |
//        14 _q2 4-slot swizzled 559 | 15 _q3 4-slot padded half-wall 622
//        (neutral) | 16/17 deferred epilogue 590/602 (rejected)
//   bf16 C=36/48: k_conv3_zring_bf16_s sliced c-half x j-tile schedule
//        [SHIPPED for 36 at 249 TF; 48 measured-rejected vs MIOpen]
//
// MIOpen's ck-xdlops kernels reach ~58 TF/s f32 on these shapes; the small
// channel depth (N-dim 28..64) starves generic implicit GEMM. This kernel
// exploits the structure directly:
//   * conv = sum over the 27 taps of GEMM( input-shifted [M x C],
//     W_tap [C x K] ) — computed per workgroup from ONE LDS-resident input
//     slab (output tile + 3x3x3 halo), so each input value is read from
//     HBM once per tile (x-halo 2/34, y-halo 2/TY+2, z-halo shared);
//   * v_mfma_f32_16x16x4_f32 (exact f32, 155 TF ceiling): D tile = 16
//     x-positions x 16 output channels, reduction 4 input channels per
//     instruction; A operand = one LDS dword per lane (bank-conflict-free
//     by construction: the lane address stride C mod 32 is kept out of
//     {0, 16} by padding the slab's x-stride), B operand = one weight
//     dword per lane, staged through LDS per tap and reused by every
//     M-tile of the wave;
//   * fused bias + optional ELU + optional residual add in the epilogue.
//
// Geometry per workgroup (template TY): output tile TZ=2 x TY x TX=32
// voxels, 4 waves; TY=4 for C<=36 (slab <= 118 KB), TY=2 for C in
// {48, 64} (slab <= 148 KB incl. padding). Weights are (27, C, K) f32
// (prepared host-side from torch's (K, C, 3, 3, 3)).
#include <hip/hip_runtime.h>

#include "cfx_internal.h"

namespace {

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int padc(int C) {  // slab voxel stride: keep (stride mod 32)
    return (C % 32 == 0 || C % 32 == 16) ? C + 4 : C;  // out of {0, 16}
}

// Generic wave->tile mapping: the workgroup's TZ x TY x (TX/16) 16-wide
// m-tiles are dealt round-robin to NW = NTHREADS/64 waves; NTHREADS=512
// puts 2 waves on each SIMD (the co-resident wave hides LDS latency the
// single wave of the 256-thread shape cannot). TAPG = taps whose weights
// are staged per barrier pair.
template <int C, int K, int TZv, int TY, int TX, int TAPG, int NTHREADS>
__global__ __launch_bounds__(NTHREADS, 1) void k_conv3(
    const float* __restrict__ in,    // (N, D, H, W, C) channels-last
    const float* __restrict__ wgt,   // (27, C, K)
    const float* __restrict__ bias,  // (K)
    const float* __restrict__ res,   // optional residual, same layout as out
    float* __restrict__ out,         // (N, D, H, W, K)
    int N, int D, int H, int W, int do_elu) {
    constexpr int PC = padc(C);
    constexpr int SX = TX + 2;            // slab x extent (halo)
    constexpr int SY = TY + 2;
    constexpr int SZ = TZv + 2;
    constexpr int KK = C / 4;             // reduction steps per tap
    constexpr int NT = (K + 15) / 16;     // 16-wide output-channel tiles
    constexpr int NW = NTHREADS / 64;     // waves per workgroup
    constexpr int XT = TX / 16;           // 16-wide x tiles
    constexpr int M_TILES = (TZv * TY * XT) / NW;  // m-tiles per wave
    static_assert((TZv * TY * XT) % NW == 0, "tiles must split evenly");

    __shared__ float slab[SZ * SY * SX * PC];
    // TAPG taps x NT 16-wide K tiles, each a [C][16] block (the 16-dword
    // row stride keeps the B-fragment reads bank-conflict-free)
    __shared__ float wtile[TAPG * NT * C * 16];

    const int bx = blockIdx.x;                  // x block
    const int by = blockIdx.y;                  // y block
    const int bzn = blockIdx.z;                 // fused (n, z-block)
    const int zblocks = (D + TZv - 1) / TZv;
    const int n = bzn / zblocks;
    const int z0 = (bzn % zblocks) * TZv;
    const int y0 = by * TY;
    const int x0 = bx * TX;

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    // ---- stage the input slab (zero-padded at volume borders) ----------
    // float4 runs along C (C % 4 == 0 for all widths). Loads are issued
    // UNCONDITIONALLY from clamped addresses and zeroed afterwards — a
    // per-element guarded load makes hipcc branch around each load and
    // drain vmcnt per element (cdna_hip_programming.md §5 trap (c)), which
    // was the dominant cost of the first version at 1 workgroup/CU.
    {
        const int c4n = C / 4;
        const int vox = SZ * SY * SX;
        const bool interior = z0 >= 1 && z0 + TZv + 1 <= D && y0 >= 1 &&
                              y0 + TY + 1 <= H && x0 >= 1 &&
                              x0 + TX + 1 <= W;
        const float* base =
            in + ((((long long)n * D + z0 - 1) * H + y0 - 1) * W + x0 - 1)
                 * C;
        if (interior) {
            for (int idx = tid; idx < vox * c4n; idx += NTHREADS) {
                const int c4 = idx % c4n;
                const int v = idx / c4n;
                const int sx = v % SX;
                const int sy = (v / SX) % SY;
                const int sz = v / (SX * SY);
                const f32x4 val = *reinterpret_cast<const f32x4*>(
                    base + (((long long)sz * H + sy) * W + sx) * C +
                    c4 * 4);
                *reinterpret_cast<f32x4*>(
                    &slab[((sz * SY + sy) * SX + sx) * PC + c4 * 4]) = val;
            }
        } else {
            for (int idx = tid; idx < vox * c4n; idx += NTHREADS) {
                const int c4 = idx % c4n;
                const int v = idx / c4n;
                const int sx = v % SX;
                const int sy = (v / SX) % SY;
                const int sz = v / (SX * SY);
                const int gz = z0 + sz - 1;
                const int gy = y0 + sy - 1;
                const int gx = x0 + sx - 1;
                const bool ok = gz >= 0 && gz < D && gy >= 0 && gy < H &&
                                gx >= 0 && gx < W;
                const int cz = ok ? gz : 0;
                const int cy = ok ? gy : 0;
                const int cx = ok ? gx : 0;
                f32x4 val = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + cz) * H + cy) * W + cx) * C +
                    c4 * 4);
                if (!ok) val = {0.f, 0.f, 0.f, 0.f};
                *reinterpret_cast<f32x4*>(
                    &slab[((sz * SY + sy) * SX + sx) * PC + c4 * 4]) = val;
            }
        }
    }
    __syncthreads();

    // ---- wave tile origins (generic round-robin deal) -------------------
    int tmz[M_TILES], tmy[M_TILES], tmx[M_TILES];
#pragma unroll
    for (int m = 0; m < M_TILES; ++m) {
        const int g = wave * M_TILES + m;
        tmx[m] = (g % XT) * 16;
        tmy[m] = (g / XT) % TY;
        tmz[m] = g / (XT * TY);
    }

    f32x4 acc[M_TILES][NT];
#pragma unroll
    for (int m = 0; m < M_TILES; ++m)
#pragma unroll
        for (int t = 0; t < NT; ++t)
            acc[m][t] = {0.f, 0.f, 0.f, 0.f};

    const int a_row = lane & 15;   // x within the m-tile (A row, B col)
    const int a_k = lane >> 4;     // reduction sub-index (0..3)
    const int col16 = lane & 15;

    // ---- 27 taps in groups of TAPG -------------------------------------
    static_assert(27 % TAPG == 0, "TAPG must divide 27");
    for (int g = 0; g < 27 / TAPG; ++g) {
        // stage the group's weights: TAPG x NT blocks of [C][16]
        for (int idx = tid; idx < TAPG * NT * C * 16;
             idx += NTHREADS) {
            const int j = idx & 15;
            const int c = (idx >> 4) % C;
            const int nt = (idx >> 4) / C % NT;
            const int tl = (idx >> 4) / C / NT;
            const int jg = nt * 16 + j;
            wtile[idx] = jg < K
                ? wgt[((long long)(g * TAPG + tl) * C + c) * K + jg]
                : 0.f;
        }
        __syncthreads();
#pragma unroll
        for (int tl = 0; tl < TAPG; ++tl) {
            const int tap = g * TAPG + tl;
            const int dz = tap / 9 - 1;
            const int dy = (tap / 3) % 3 - 1;
            const int dx = tap % 3 - 1;
            // kk outer / m inner: the accumulator chains interleave,
            // hiding the 40-cycle dependent-MFMA latency (the 16x16x4
            // issue interval is 32) at one wave per SIMD
            const float* arow[M_TILES];
#pragma unroll
            for (int m = 0; m < M_TILES; ++m) {
                arow[m] = &slab[(((1 + tmz[m] + dz) * SY +
                                  (1 + tmy[m] + dy)) * SX +
                                 (1 + tmx[m] + dx)) * PC + a_row * PC + a_k];
            }
            const float* wblk = &wtile[tl * NT * C * 16];
#pragma unroll
            for (int nt = 0; nt < NT; ++nt) {
#pragma unroll
                for (int kk = 0; kk < KK; ++kk) {
                    // A[i = l&15][k = l>>4], B[k = l>>4][j = l&15]
                    const float b =
                        wblk[(nt * C + kk * 4 + a_k) * 16 + col16];
#pragma unroll
                    for (int m = 0; m < M_TILES; ++m) {
                        const float a = arow[m][kk * 4];
                        acc[m][nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                            a, b, acc[m][nt], 0, 0, 0);
                    }
                }
            }
        }
        __syncthreads();
    }

    // ---- epilogue: D[row = (lane>>4)*4 + reg][col = lane&15] -------------
    const int col = lane & 15;
    const int rbase = (lane >> 4) * 4;
#pragma unroll
    for (int m = 0; m < M_TILES; ++m) {
        const int gz = z0 + tmz[m];
        const int gy = y0 + tmy[m];
        if (gz >= D || gy >= H) continue;
#pragma unroll
        for (int t = 0; t < NT; ++t) {
            const int j = t * 16 + col;
            if (j >= K) continue;
            const float bj = bias ? bias[j] : 0.f;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int gx = x0 + tmx[m] + rbase + r;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + gz) * H + gy) * W + gx) * K + j;
                float v = acc[m][t][r] + bj;
                if (res) v += res[o];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = v;
            }
        }
    }
}

}  // namespace

extern "C" int cfx_conv3_ndhwc(cfx_ctx* ctx, const float* in,
                               const float* wgt, const float* bias,
                               const float* residual, float* out, int N,
                               int D, int H, int W, int C, int K,
                               int do_elu) {
    if (C != K) {
        g_err = "cfx_conv3_ndhwc: only C == K widths are instantiated";
        return -1;
    }
    // LDS budget per width (slab + wtile <= 160 KiB); NTHREADS=512 puts
    // 2 waves per SIMD where the slab allows it
#define CFX_CONV_CASE(CW, TZV, TYV, TXV, TAPGV, NTH)                         \
    case CW: {                                                               \
        const int zb = (D + TZV - 1) / TZV;                                  \
        dim3 grid((W + TXV - 1) / TXV, (H + TYV - 1) / TYV,                  \
                  (unsigned)(N * zb));                                       \
        hipEvent_t e0;                                                       \
        if (prof_begin(ctx, &e0)) return -1;                                 \
        hipLaunchKernelGGL((k_conv3<CW, CW, TZV, TYV, TXV, TAPGV, NTH>),     \
                           grid, dim3(NTH), 0, ctx->stream, in, wgt, bias,   \
                           residual, out, N, D, H, W, do_elu);               \
        CFX_CHECK(hipGetLastError());                                        \
        double flops = 2.0 * 27.0 * CW * CW * (double)N * D * H * W;         \
        if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;                 \
        break;                                                               \
    }
    // CFX_CONV_V=1 selects the half-slab TX=16 shape for C=28: 71 KB of
    // LDS admits TWO 512-thread workgroups per CU (4 waves/SIMD), so one
    // workgroup's slab staging overlaps the other's MFMAs
    static int conv_v = [] {
        const char* s = getenv("CFX_CONV_V");
        return s ? atoi(s) : 0;
    }();
    if (C == 28 && conv_v == 2) {
        // 3 workgroups/CU: 47 KB LDS (256 thr, TZ1 TY4 TX16)
        dim3 grid((W + 15) / 16, (H + 3) / 4, (unsigned)(N * D));
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
        hipLaunchKernelGGL((k_conv3<28, 28, 1, 4, 16, 3, 256>), grid,
                           dim3(256), 0, ctx->stream, in, wgt, bias,
                           residual, out, N, D, H, W, do_elu);
        CFX_CHECK(hipGetLastError());
        double flops = 2.0 * 27.0 * 28 * 28 * (double)N * D * H * W;
        if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
        return 0;
    }
    if (C == 28 && conv_v == 3) {
        // 2 x 384-thread workgroups/CU (3 waves/SIMD), TZ2 TY6 TX16:
        // slab 4*8*18*28*4 = 64.5 KB + wtile 10.5 KB = 75 KB
        const int zb = (D + 1) / 2;
        dim3 grid((W + 15) / 16, (H + 5) / 6, (unsigned)(N * zb));
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
        hipLaunchKernelGGL((k_conv3<28, 28, 2, 6, 16, 3, 384>), grid,
                           dim3(384), 0, ctx->stream, in, wgt, bias,
                           residual, out, N, D, H, W, do_elu);
        CFX_CHECK(hipGetLastError());
        double flops = 2.0 * 27.0 * 28 * 28 * (double)N * D * H * W;
        if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
        return 0;
    }
    if (C == 28 && conv_v == 1) {
        // half-slab TX=16 shape: 71 KB LDS -> 2 workgroups per CU
        dim3 grid((W + 15) / 16, (H + 7) / 8, (unsigned)(N * D));
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
        hipLaunchKernelGGL((k_conv3<28, 28, 1, 8, 16, 3, 512>), grid,
                           dim3(512), 0, ctx->stream, in, wgt, bias,
                           residual, out, N, D, H, W, do_elu);
        CFX_CHECK(hipGetLastError());
        double flops = 2.0 * 27.0 * 28 * 28 * (double)N * D * H * W;
        if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
        return 0;
    }
    switch (C) {
        // 28: slab 3*10*34*28*4 = 114 KB + wtile 10.5 KB (TZ1 TY8, 8 waves)
        CFX_CONV_CASE(28, 1, 8, 32, 3, 512)
        // 36: slab 3*10*34*36*4 = 147 KB + wtile 4.5 KB (TAPG 1)
        CFX_CONV_CASE(36, 1, 8, 32, 1, 512)
        // 48: slab 3*6*34*52*4 = 127 KB + wtile 9.2 KB (TZ1 TY4, 8 waves
        // -> 1 m-tile per wave)
        CFX_CONV_CASE(48, 1, 4, 32, 1, 512)
        // 64: slab 3*6*18*68*4 = 86 KB + wtile 16.4 KB (TZ1 TY4 TX16;
        // 8 waves x ... 1*4*1 = 4 tiles < 8 waves -> use 256 threads)
        CFX_CONV_CASE(64, 2, 2, 16, 1, 256)
        default:
            g_err = "cfx_conv3_ndhwc: unsupported channel width";
            return -1;
    }
#undef CFX_CONV_CASE
    return 0;
}

namespace {

// 32x32x2 variant (C == K <= 32): one 32-x-position x 32-channel D tile
// per wave, single accumulator chain (the 32x32x2 dependent latency equals
// its 64-cycle issue interval, so one chain sustains the full rate), and
// one A + one B LDS dword per 4096 flops (3x fewer LDS reads per flop than
// the 16x16x4 path). Slab layout [z][y][c][x] (x fastest) makes the
// A-fragment read (lanes 0..31 = consecutive x) trivially conflict-free.
template <int C, int K, int TY, int TAPG>
__global__ __launch_bounds__(512, 1) void k_conv3_w32(
    const float* __restrict__ in, const float* __restrict__ wgt,
    const float* __restrict__ bias, const float* __restrict__ res,
    float* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int TX = 32;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int SZ = 3;                // TZ = 1
    constexpr int KK = C / 2;            // reduction pairs per tap
    static_assert(K <= 32 && C % 2 == 0, "");

    __shared__ float slab[SZ * SY * C * SX];
    __shared__ float wtile[TAPG * C * 32];

    const int zblocks = D;               // TZ = 1
    const int n = blockIdx.z / zblocks;
    const int z0 = blockIdx.z % zblocks;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;           // 0..7 -> y row
    const int lane = tid & 63;

    // ---- stage the slab: read NDHWC float4 runs, scatter to [c][x] ------
    {
        const int c4n = C / 4;
        const bool interior = z0 >= 1 && z0 + 2 <= D && y0 >= 1 &&
                              y0 + TY + 1 <= H && x0 >= 1 &&
                              x0 + TX + 1 <= W;
        for (int idx = tid; idx < SZ * SY * SX * c4n; idx += 512) {
            const int c4 = idx % c4n;
            const int v = idx / c4n;
            const int sx = v % SX;
            const int sy = (v / SX) % SY;
            const int sz = v / (SX * SY);
            const int gz = z0 + sz - 1;
            const int gy = y0 + sy - 1;
            const int gx = x0 + sx - 1;
            f32x4 val;
            if (interior) {
                val = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + gz) * H + gy) * W + gx) * C +
                    c4 * 4);
            } else {
                const bool ok = gz >= 0 && gz < D && gy >= 0 && gy < H &&
                                gx >= 0 && gx < W;
                val = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + (ok ? gz : 0)) * H +
                           (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                    c4 * 4);
                if (!ok) val = {0.f, 0.f, 0.f, 0.f};
            }
            float* base = &slab[((sz * SY + sy) * C + c4 * 4) * SX + sx];
            base[0 * SX] = val.x;
            base[1 * SX] = val.y;
            base[2 * SX] = val.z;
            base[3 * SX] = val.w;
        }
    }
    __syncthreads();

    typedef float f32x16 __attribute__((ext_vector_type(16)));
    f32x16 acc = {};
    const int ax = lane & 31;            // A row (x), B col (j)
    const int ak = lane >> 5;            // reduction sub-index (0..1)

    static_assert(27 % TAPG == 0, "");
    for (int g = 0; g < 27 / TAPG; ++g) {
        for (int idx = tid; idx < TAPG * C * 32; idx += 512) {
            const int j = idx & 31;
            const int c = (idx >> 5) % C;
            const int tl = (idx >> 5) / C;
            wtile[idx] = j < K
                ? wgt[((long long)(g * TAPG + tl) * C + c) * K + j]
                : 0.f;
        }
        __syncthreads();
#pragma unroll
        for (int tl = 0; tl < TAPG; ++tl) {
            const int tap = g * TAPG + tl;
            const int dz = tap / 9 - 1;
            const int dy = (tap / 3) % 3 - 1;
            const int dx = tap % 3 - 1;
            const float* arow =
                &slab[((1 + dz) * SY + (1 + wave + dy)) * C * SX +
                      (1 + dx) + ax];
            const float* wrow = &wtile[tl * C * 32 + ax];
#pragma unroll
            for (int kk = 0; kk < KK; ++kk) {
                const float a = arow[(kk * 2 + ak) * SX];
                const float b = wrow[(kk * 2 + ak) * 32];
                acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0,
                                                           0);
            }
        }
        __syncthreads();
    }

    // ---- epilogue: D[row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)][col] -----
    const int gz = z0;
    const int gy = y0 + wave;
    const int j = lane & 31;
    if (gy < H && j < K) {
        const float bj = bias ? bias[j] : 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
            const int gx = x0 + row;
            if (gx >= W) continue;
            long long o =
                ((((long long)n * D + gz) * H + gy) * W + gx) * K + j;
            float v = acc[r] + bj;
            if (res) v += res[o];
            if (do_elu) v = v > 0.f ? v : expm1f(v);
            out[o] = v;
        }
    }
}

}  // namespace

// 32x32x2 path for C == K <= 32 (currently instantiated for 28); falls
// back to the 16x16x4 path when CFX_CONV_W32=0
extern "C" int cfx_conv3_ndhwc_w32(cfx_ctx* ctx, const float* in,
                                   const float* wgt, const float* bias,
                                   const float* residual, float* out, int N,
                                   int D, int H, int W, int C, int K,
                                   int do_elu) {
    if (C != 28 || K != 28) {
        g_err = "cfx_conv3_ndhwc_w32: only C == K == 28 instantiated";
        return -1;
    }
    dim3 grid((W + 31) / 32, (H + 7) / 8, (unsigned)(N * D));
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    hipLaunchKernelGGL((k_conv3_w32<28, 28, 8, 3>), grid, dim3(512), 0,
                       ctx->stream, in, wgt, bias, residual, out, N, D, H,
                       W, do_elu);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 27.0 * 28 * 28 * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
    return 0;
}

namespace {

// Persistent-z variant: the workgroup walks ALL z planes of its (y, x)
// tile. Weights for all 27 taps stay LDS-RESIDENT for the whole workgroup
// (staged once, amortized over D planes — per-tap-group re-staging was
// most of the per-z overhead), and the input slab is a 3-plane ring with
// ONE ~20 KB plane staged per z. LDS: ring 3 x SY x SX x C (60.5 KB at
// C=28, TY=8, TX=16) + weights 27 x NT x C x 16 (96.8 KB) = 157 KB.
template <int C, int K, int TY, int TX, int NTK>
__global__ __launch_bounds__(512, 1) void k_conv3_zring(
    const float* __restrict__ in, const float* __restrict__ wgt,
    const float* __restrict__ bias, const float* __restrict__ res,
    float* __restrict__ out, int N, int D, int H, int W, int do_elu,
    int j0) {
    constexpr int PC = padc(C);
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = C / 4;
    constexpr int NT = NTK;               // K tiles computed this launch
    constexpr int NW = 8;                 // 512 threads
    constexpr int XT = TX / 16;
    constexpr int M_TILES = (TY * XT) / NW;
    static_assert((TY * XT) % NW == 0, "");

    __shared__ float ring[3 * SY * SX * PC];
    __shared__ float wall[27 * NT * C * 16];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    // ---- weights: all taps, staged once -------------------------------
    for (int idx = tid; idx < 27 * NT * C * 16; idx += 512) {
        const int j = idx & 15;
        const int c = (idx >> 4) % C;
        const int nt = (idx >> 4) / C % NT;
        const int tap = (idx >> 4) / C / NT;
        const int jg = j0 + nt * 16 + j;
        wall[idx] =
            jg < K ? wgt[((long long)tap * C + c) * K + jg] : 0.f;
    }

    // plane stager: global plane P (may be -1 or D: zero padding) into
    // ring slot (P + 1) % 3
    auto stage_plane = [&](int P) {
        const int slot = ((P + 1) % 3 + 3) % 3;
        const int c4n = C / 4;
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && y0 >= 1 && y0 + TY + 1 <= H &&
                              x0 >= 1 && x0 + TX + 1 <= W;
        for (int idx = tid; idx < SY * SX * c4n; idx += 512) {
            const int c4 = idx % c4n;
            const int v = idx / c4n;
            const int sx = v % SX;
            const int sy = v / SX;
            const int gy = y0 + sy - 1;
            const int gx = x0 + sx - 1;
            f32x4 val;
            if (interior) {
                val = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + P) * H + gy) * W + gx) * C +
                    c4 * 4);
            } else {
                const bool ok = zin && gy >= 0 && gy < H && gx >= 0 &&
                                gx < W;
                val = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + (zin ? P : 0)) * H +
                           (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                    c4 * 4);
                if (!ok) val = {0.f, 0.f, 0.f, 0.f};
            }
            *reinterpret_cast<f32x4*>(
                &ring[(slot * SY * SX + v) * PC + c4 * 4]) = val;
        }
    };

    stage_plane(-1);
    stage_plane(0);

    int tmy[M_TILES], tmx[M_TILES];
#pragma unroll
    for (int m = 0; m < M_TILES; ++m) {
        const int g = wave * M_TILES + m;
        tmx[m] = (g % XT) * 16;
        tmy[m] = g / XT;
    }
    const int a_row = lane & 15;
    const int a_k = lane >> 4;
    const int col16 = lane & 15;
    const int colj = lane & 15;
    const int rbase = (lane >> 4) * 4;

    for (int z = 0; z < D; ++z) {
        stage_plane(z + 1);
        __syncthreads();

        f32x4 acc[M_TILES][NT];
#pragma unroll
        for (int m = 0; m < M_TILES; ++m)
#pragma unroll
            for (int t = 0; t < NT; ++t)
                acc[m][t] = {0.f, 0.f, 0.f, 0.f};

        for (int dzi = 0; dzi < 3; ++dzi) {
            const int slot = ((z + dzi) % 3 + 3) % 3;  // plane z + dzi - 1
            const float* plane = &ring[slot * SY * SX * PC];
#pragma unroll
            for (int tl = 0; tl < 9; ++tl) {
                const int tap = dzi * 9 + tl;
                const int dy = tl / 3 - 1;
                const int dx = tl % 3 - 1;
                const float* arow[M_TILES];
#pragma unroll
                for (int m = 0; m < M_TILES; ++m) {
                    arow[m] = &plane[((1 + tmy[m] + dy) * SX +
                                      (1 + tmx[m] + dx)) * PC +
                                     a_row * PC + a_k];
                }
                const float* wblk = &wall[tap * NT * C * 16];
#pragma unroll
                for (int nt = 0; nt < NT; ++nt) {
#pragma unroll
                    for (int kk = 0; kk < KK; ++kk) {
                        const float b =
                            wblk[(nt * C + kk * 4 + a_k) * 16 + col16];
#pragma unroll
                        for (int m = 0; m < M_TILES; ++m) {
                            const float a = arow[m][kk * 4];
                            acc[m][nt] =
                                __builtin_amdgcn_mfma_f32_16x16x4f32(
                                    a, b, acc[m][nt], 0, 0, 0);
                        }
                    }
                }
            }
        }

        // epilogue for this z
        float rv[M_TILES][NT][4];
        if (res) {  // residual reads batched from clamped addresses (a
                    // load in the store loop costs a vmcnt(0) drain each)
#pragma unroll
            for (int m = 0; m < M_TILES; ++m) {
                const int gym = min(y0 + tmy[m], H - 1);
#pragma unroll
                for (int t = 0; t < NT; ++t) {
                    const int jm = min(j0 + t * 16 + colj, K - 1);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int gxm = min(x0 + tmx[m] + rbase + r, W - 1);
                        rv[m][t][r] = res[
                            ((((long long)n * D + z) * H + gym) * W + gxm) *
                                K + jm];
                    }
                }
            }
        }
#pragma unroll
        for (int m = 0; m < M_TILES; ++m) {
            const int gy = y0 + tmy[m];
            if (gy >= H) continue;
#pragma unroll
            for (int t = 0; t < NT; ++t) {
                const int j = j0 + t * 16 + colj;
                if (j >= K) continue;
                const float bj = bias ? bias[j] : 0.f;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int gx = x0 + tmx[m] + rbase + r;
                    if (gx >= W) continue;
                    long long o =
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j;
                    float v = acc[m][t][r] + bj;
                    if (res) v += rv[m][t][r];
                    if (do_elu) v = v > 0.f ? v : expm1f(v);
                    out[o] = v;
                }
            }
        }
        __syncthreads();  // before the next z overwrites the oldest plane
    }
}

// Software-pipelined variant of k_conv3_zring: the NEXT plane's global
// loads are issued into registers first, the 18 taps that only touch the
// two already-resident planes run while those loads are in flight, then
// the registers drain to the ring slot and ONE barrier releases the last
// 9 taps. One barrier per z (the plain kernel needs two): iteration z's
// store targets the slot plane z-2 occupied, whose last readers (dzi=0 of
// iteration z-1) sit before iteration z-1's barrier.
// MODE (ablation, CFX_F32_MODE): 0 full, 1 skip mainloop, 3 skip per-z
// staging (timing only, wrong results)
template <int C, int K, int TY, int TX, int NTK, int MODE = 0>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_pl(
    const float* __restrict__ in, const float* __restrict__ wgt,
    const float* __restrict__ bias, const float* __restrict__ res,
    float* __restrict__ out, int N, int D, int H, int W, int do_elu,
    int j0) {
    constexpr int PC = padc(C);
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = C / 4;
    constexpr int NT = NTK;
    constexpr int NW = 8;
    constexpr int XT = TX / 16;
    constexpr int M_TILES = (TY * XT) / NW;
    constexpr int C4N = C / 4;
    constexpr int LV = (SY * SX * C4N + 511) / 512;  // loads held per thread
    static_assert((TY * XT) % NW == 0, "");

    __shared__ float ring[3 * SY * SX * PC];
    __shared__ float wall[27 * NT * C * 16];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * NT * C * 16; idx += 512) {
        const int j = idx & 15;
        const int c = (idx >> 4) % C;
        const int nt = (idx >> 4) / C % NT;
        const int tap = (idx >> 4) / C / NT;
        const int jg = j0 + nt * 16 + j;
        wall[idx] =
            jg < K ? wgt[((long long)tap * C + c) * K + jg] : 0.f;
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    // Unconditional clamped loads + deferred zero-select (see the bf16
    // twin: a select attached to each load makes hipcc emit a full
    // vmcnt(0) drain per load, serializing the HBM latency)
    auto plane_load = [&](int P, f32x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4N - 1);
            const int c4 = idx % C4N;
            const int v = idx / C4N;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool ok = zin && (interior ||
                                    (gy >= 0 && gy < H && gx >= 0 &&
                                     gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const f32x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C + c4 * 4);
        }
    };
    auto plane_store = [&](int P, const f32x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4N) break;
            const int c4 = idx % C4N;
            const int v = idx / C4N;
            *reinterpret_cast<f32x4*>(
                &ring[(slot * SY * SX + v) * PC + c4 * 4]) =
                keep[li] ? vals[li] : f32x4{0.f, 0.f, 0.f, 0.f};
        }
    };

    {
        f32x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();

    int tmy[M_TILES], tmx[M_TILES];
#pragma unroll
    for (int m = 0; m < M_TILES; ++m) {
        const int g = wave * M_TILES + m;
        tmx[m] = (g % XT) * 16;
        tmy[m] = g / XT;
    }
    const int a_row = lane & 15;
    const int a_k = lane >> 4;
    const int col16 = lane & 15;
    const int colj = lane & 15;
    const int rbase = (lane >> 4) * 4;

    for (int z = 0; z < D; ++z) {
        // Two accumulator chains per m-tile: consecutive MFMAs on the
        // SAME accumulator pay the 40-cyc dependent latency against a
        // 32-cyc issue slot (PMC: 51.6% SQ_WAIT_INST_ANY with the naive
        // nt-outer order). kk-outer alternates the NT=2 accumulators
        // (bit-identical sums — each chain keeps its order); NT==1
        // splits by kk parity into acc/acc2 (a benign reassociation,
        // folded in the epilogue).
        f32x4 acc[M_TILES][NT];
        f32x4 acc2[M_TILES];
#pragma unroll
        for (int m = 0; m < M_TILES; ++m) {
            acc2[m] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int t = 0; t < NT; ++t)
                acc[m][t] = {0.f, 0.f, 0.f, 0.f};
        }

        auto compute_dzi = [&](int dzi) {
            const int slot = ((z + dzi) % 3 + 3) % 3;  // plane z + dzi - 1
            const float* plane = &ring[slot * SY * SX * PC];
#pragma unroll
            for (int tl = 0; tl < 9; ++tl) {
                const int tap = dzi * 9 + tl;
                const int dy = tl / 3 - 1;
                const int dx = tl % 3 - 1;
                const float* arow[M_TILES];
#pragma unroll
                for (int m = 0; m < M_TILES; ++m) {
                    arow[m] = &plane[((1 + tmy[m] + dy) * SX +
                                      (1 + tmx[m] + dx)) * PC +
                                     a_row * PC + a_k];
                }
                const float* wblk = &wall[tap * NT * C * 16];
#pragma unroll
                for (int kk = 0; kk < KK; ++kk) {
                    float a[M_TILES];
#pragma unroll
                    for (int m = 0; m < M_TILES; ++m)
                        a[m] = arow[m][kk * 4];
#pragma unroll
                    for (int nt = 0; nt < NT; ++nt) {
                        const float b =
                            wblk[(nt * C + kk * 4 + a_k) * 16 + col16];
#pragma unroll
                        for (int m = 0; m < M_TILES; ++m) {
                            f32x4& dst = (NT == 1 && (kk & 1))
                                             ? acc2[m] : acc[m][nt];
                            dst = __builtin_amdgcn_mfma_f32_16x16x4f32(
                                a[m], b, dst, 0, 0, 0);
                        }
                    }
                }
            }
        };

        f32x4 vals[LV];
        bool keep[LV];
        if (MODE != 3)  // flies over the next 18 taps
            plane_load(z + 1, vals, keep);
        if (MODE != 1) {
            compute_dzi(0);
            compute_dzi(1);
        }
        if (MODE != 3) plane_store(z + 1, vals, keep);
        __syncthreads();
        if (MODE != 1) compute_dzi(2);

        if (MODE == 4) {  // timing ablation: skip the epilogue
            float sink = 0.f;
#pragma unroll
            for (int mt = 0; mt < M_TILES; ++mt)
#pragma unroll
                for (int t = 0; t < NT; ++t) sink += acc[mt][t][0];
            if (sink == 1e30f) out[tid] = sink;  // keep acc alive
            continue;
        }
        float rv[M_TILES][NT][4];
        if (res) {  // residual reads batched from clamped addresses (a
                    // load in the store loop costs a vmcnt(0) drain each)
#pragma unroll
            for (int m = 0; m < M_TILES; ++m) {
                const int gym = min(y0 + tmy[m], H - 1);
#pragma unroll
                for (int t = 0; t < NT; ++t) {
                    const int jm = min(j0 + t * 16 + colj, K - 1);
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int gxm = min(x0 + tmx[m] + rbase + r, W - 1);
                        rv[m][t][r] = res[
                            ((((long long)n * D + z) * H + gym) * W + gxm) *
                                K + jm];
                    }
                }
            }
        }
#pragma unroll
        for (int m = 0; m < M_TILES; ++m) {
            const int gy = y0 + tmy[m];
            if (gy >= H) continue;
#pragma unroll
            for (int t = 0; t < NT; ++t) {
                const int j = j0 + t * 16 + colj;
                if (j >= K) continue;
                const float bj = bias ? bias[j] : 0.f;
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int gx = x0 + tmx[m] + rbase + r;
                    if (gx >= W) continue;
                    long long o =
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j;
                    float v = acc[m][t][r] + bj;
                    if (NT == 1) v += acc2[m][r];
                    if (res) v += rv[m][t][r];
                    if (do_elu) v = v > 0.f ? v : expm1f(v);
                    out[o] = v;
                }
            }
        }
    }
}

// C=48 persistent-z ring: the full 27-tap weight wall (82.9 KB per
// 16-wide K tile) no longer fits LDS beside the 3-plane ring (103.7 KB),
// so the wall is DOUBLE-BUFFERED per 9-tap dzi group (2 x 27.6 KB) and
// each group's stage overlaps the previous group's MFMAs. K covered by
// three j0 launches. Per z: 3 barriers; wall restage is ~3% of the z's
// MFMA time (weights come from L2 after the first z). Buffer parity is
// (3*z + dzi) % 2 — odd group count flips the pairing every z, and each
// write lands on a buffer whose last reader sat before the previous
// barrier (see the B1/B2/B3 comments).
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_dw(
    const float* __restrict__ in, const float* __restrict__ wgt,
    const float* __restrict__ bias, const float* __restrict__ res,
    float* __restrict__ out, int N, int D, int H, int W, int do_elu,
    int j0) {
    // ring stride: C+2 when C%32==16 (48 -> 50; mod 32 = 18 keeps the
    // A reads bank-conflict-free) — the +4 pad of padc() would push the
    // ring + double wall 3.7 KB past the 160 KB LDS. Stride 50 dwords is
    // only 8-byte aligned, so the ring staging stores are b64.
    constexpr int PC = (C % 32 == 16) ? C + 2 : padc(C);
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = C / 4;
    constexpr int NW = 8;
    constexpr int XT = TX / 16;
    constexpr int M_TILES = (TY * XT) / NW;
    constexpr int C4N = C / 4;
    constexpr int LV = (SY * SX * C4N + 511) / 512;
    static_assert((TY * XT) % NW == 0, "");

    __shared__ float ring[3 * SY * SX * PC];
    __shared__ float wall[2][9 * C * 16];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    // stage the 9-tap group g (taps 9g..9g+8) into wall[buf]
    auto wall_store = [&](int g, int buf) {
        for (int idx = tid; idx < 9 * C * 16; idx += 512) {
            const int j = idx & 15;
            const int c = (idx >> 4) % C;
            const int tl = (idx >> 4) / C;
            const int jg = j0 + j;
            wall[buf][idx] =
                jg < K ? wgt[((long long)(9 * g + tl) * C + c) * K + jg]
                       : 0.f;
        }
    };

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;
    auto plane_load = [&](int P, f32x4 (&vals)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4N) break;
            const int c4 = idx % C4N;
            const int v = idx / C4N;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            if (interior) {
                vals[li] = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + P) * H + gy) * W + gx) * C +
                    c4 * 4);
            } else {
                const bool ok = zin && gy >= 0 && gy < H && gx >= 0 &&
                                gx < W;
                vals[li] = *reinterpret_cast<const f32x4*>(
                    in + ((((long long)n * D + (zin ? P : 0)) * H +
                           (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                    c4 * 4);
                if (!ok) vals[li] = {0.f, 0.f, 0.f, 0.f};
            }
        }
    };
    typedef float f32x2 __attribute__((ext_vector_type(2)));
    auto plane_store = [&](int P, const f32x4 (&vals)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4N) break;
            const int c4 = idx % C4N;
            const int v = idx / C4N;
            float* dst = &ring[(slot * SY * SX + v) * PC + c4 * 4];
            *reinterpret_cast<f32x2*>(dst) = f32x2{vals[li][0], vals[li][1]};
            *reinterpret_cast<f32x2*>(dst + 2) =
                f32x2{vals[li][2], vals[li][3]};
        }
    };

    {
        f32x4 v0[LV], v1[LV];
        plane_load(-1, v0);
        plane_load(0, v1);
        plane_store(-1, v0);
        plane_store(0, v1);
        wall_store(0, 0);  // g0 of z=0 -> buf (3*0+0)%2 = 0
    }
    __syncthreads();

    int tmy[M_TILES], tmx[M_TILES];
#pragma unroll
    for (int m = 0; m < M_TILES; ++m) {
        const int g = wave * M_TILES + m;
        tmx[m] = (g % XT) * 16;
        tmy[m] = g / XT;
    }
    const int a_row = lane & 15;
    const int a_k = lane >> 4;
    const int col16 = lane & 15;
    const int colj = lane & 15;
    const int rbase = (lane >> 4) * 4;

    for (int z = 0; z < D; ++z) {
        f32x4 acc[M_TILES];
#pragma unroll
        for (int m = 0; m < M_TILES; ++m)
            acc[m] = {0.f, 0.f, 0.f, 0.f};

        auto compute_dzi = [&](int dzi, int buf) {
            const int slot = ((z + dzi) % 3 + 3) % 3;
            const float* plane = &ring[slot * SY * SX * PC];
            const float* wb = wall[buf];
#pragma unroll
            for (int tl = 0; tl < 9; ++tl) {
                const int dy = tl / 3 - 1;
                const int dx = tl % 3 - 1;
                const float* arow[M_TILES];
#pragma unroll
                for (int m = 0; m < M_TILES; ++m) {
                    arow[m] = &plane[((1 + tmy[m] + dy) * SX +
                                      (1 + tmx[m] + dx)) * PC +
                                     a_row * PC + a_k];
                }
                const float* wblk = &wb[tl * C * 16];
#pragma unroll
                for (int kk = 0; kk < KK; ++kk) {
                    const float b = wblk[(kk * 4 + a_k) * 16 + col16];
#pragma unroll
                    for (int m = 0; m < M_TILES; ++m) {
                        const float a = arow[m][kk * 4];
                        acc[m] = __builtin_amdgcn_mfma_f32_16x16x4f32(
                            a, b, acc[m], 0, 0, 0);
                    }
                }
            }
        };

        const int p = (3 * z) % 2;  // buffer of this z's g0
        f32x4 vals[LV];
        plane_load(z + 1, vals);
        wall_store(1, 1 - p);      // g1 -> other buf (last read: g2 of z-1)
        compute_dzi(0, p);
        __syncthreads();           // B1: g1 visible; g0 buf free
        wall_store(2, p);          // g2 overwrites g0's buf
        compute_dzi(1, 1 - p);
        plane_store(z + 1, vals);
        __syncthreads();           // B2: g2 + plane z+1 visible; g1 buf free
        if (z + 1 < D)
            wall_store(0, 1 - p);  // next z's g0 -> (3(z+1))%2 == 1-p
        compute_dzi(2, p);

#pragma unroll
        for (int m = 0; m < M_TILES; ++m) {
            const int gy = y0 + tmy[m];
            if (gy >= H) continue;
            const int j = j0 + colj;
            if (j >= K) continue;
            const float bj = bias ? bias[j] : 0.f;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int gx = x0 + tmx[m] + rbase + r;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[m][r] + bj;
                if (res) v += res[o];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = v;
            }
        }
        __syncthreads();           // B3: next z's g0 visible; g2 buf free
    }
}

}  // namespace

extern "C" int cfx_conv3_ndhwc_zring(cfx_ctx* ctx, const float* in,
                                     const float* wgt, const float* bias,
                                     const float* residual, float* out,
                                     int N, int D, int H, int W, int C,
                                     int K, int do_elu) {
    dim3 grid((W + 15) / 16, (H + 7) / 8, (unsigned)N);
    // CFX_ZRING_PL=0 falls back to the barrier-per-phase kernel (the
    // pipelined one measured faster on hardware and is the default)
    static const int use_pl = [] {
        const char* e = getenv("CFX_ZRING_PL");
        return e ? atoi(e) : 1;
    }();
    static const int f32mode = [] {
        const char* e = getenv("CFX_F32_MODE");  // phase ablation (timing)
        return e ? atoi(e) : 0;
    }();
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    if (C == 28 && K == 28) {
        // wall holds both K tiles (96.8 KB + 60.5 KB ring)
        if (f32mode == 1)
            hipLaunchKernelGGL((k_conv3_zring_pl<28, 28, 8, 16, 2, 1>),
                               grid, dim3(512), 0, ctx->stream, in, wgt,
                               bias, residual, out, N, D, H, W, do_elu, 0);
        else if (f32mode == 3)
            hipLaunchKernelGGL((k_conv3_zring_pl<28, 28, 8, 16, 2, 3>),
                               grid, dim3(512), 0, ctx->stream, in, wgt,
                               bias, residual, out, N, D, H, W, do_elu, 0);
        else if (f32mode == 4)  // epilogue-skip ablation (WRONG results)
            hipLaunchKernelGGL((k_conv3_zring_pl<28, 28, 8, 16, 2, 4>),
                               grid, dim3(512), 0, ctx->stream, in, wgt,
                               bias, residual, out, N, D, H, W, do_elu, 0);
        else if (use_pl)
            hipLaunchKernelGGL((k_conv3_zring_pl<28, 28, 8, 16, 2>), grid,
                               dim3(512), 0, ctx->stream, in, wgt, bias,
                               residual, out, N, D, H, W, do_elu, 0);
        else
            hipLaunchKernelGGL((k_conv3_zring<28, 28, 8, 16, 2>), grid,
                               dim3(512), 0, ctx->stream, in, wgt, bias,
                               residual, out, N, D, H, W, do_elu, 0);
    } else if (C == 36 && K == 36) {
        // one 16-wide K tile per launch (wall 62.2 KB + ring 77.8 KB);
        // the input re-read costs ~3x HBM traffic of a 113 MB activation
        // per conv — negligible next to the compute
        for (int j0 = 0; j0 < 36; j0 += 16) {
            if (use_pl)
                hipLaunchKernelGGL((k_conv3_zring_pl<36, 36, 8, 16, 1>),
                                   grid, dim3(512), 0, ctx->stream, in, wgt,
                                   bias, residual, out, N, D, H, W, do_elu,
                                   j0);
            else
                hipLaunchKernelGGL((k_conv3_zring<36, 36, 8, 16, 1>), grid,
                                   dim3(512), 0, ctx->stream, in, wgt, bias,
                                   residual, out, N, D, H, W, do_elu, j0);
        }
    } else if (C == 48 && K == 48) {
        // double-buffered per-dzi weight wall (full wall no longer fits
        // beside the ring); one 16-wide K tile per launch
        for (int j0 = 0; j0 < 48; j0 += 16) {
            hipLaunchKernelGGL((k_conv3_zring_dw<48, 48, 8, 16>), grid,
                               dim3(512), 0, ctx->stream, in, wgt, bias,
                               residual, out, N, D, H, W, do_elu, j0);
        }
    } else {
        g_err = "cfx_conv3_ndhwc_zring: width not instantiated";
        return -1;
    }
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 27.0 * C * K * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
    return 0;
}

namespace {

// bf16 persistent-z ring (config-5 path): v_mfma_f32_32x32x16_bf16 — one
// 32-x by 32-channel D tile per wave, K-reduction 16 input channels per
// instruction, fragment layout probe-verified on hardware
// (tools/mfma_layout_probe.py): A[i=l&31][k=8*(l>>5)+e], B likewise, D as
// the f32 32x32 map. Activations ride the 3-plane LDS ring in [x][c]
// order (lane reads its 8-channel A fragment as one 16-byte ds_read; the
// per-x stride is padded to 40 elements so the 16-lane b128 groups land
// on distinct banks); the bf16 weight wall (27 taps x 32 j x padded c)
// stays LDS-resident for the whole workgroup. C is zero-padded to 32.
typedef __bf16 cfx_bf16;
typedef cfx_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

// MODE: 0 = full; 1 = skip the MFMA mainloop; 2 = skip the epilogue
// stores/res; 3 = skip per-z plane staging (wrong results; timing only) —
// phase-ablation instrumentation for the stall hunt (tools/conv_probe
// --bf16-ablate).
template <int C, int K, int TY, int TX, int MODE = 0>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16(
    const cfx_bf16* __restrict__ in,    // (N, D, H, W, C)
    const cfx_bf16* __restrict__ wgt,   // (27, 32, 32): [tap][j][c], padded
    const float* __restrict__ bias,     // (K) f32
    const cfx_bf16* __restrict__ res,   // optional residual (out layout)
    cfx_bf16* __restrict__ out,         // (N, D, H, W, K)
    int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;               // padded channel count
    constexpr int PCB = CP + 8;          // slab per-x stride (bank spread)
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;          // 2 reduction steps per tap
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;           // 0..7 -> y row (TY == 8)
    const int lane = tid & 63;

    // weight wall, staged once: wall[(tap*32 + j)*PCB + c]
    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    // Staging is load-phase/store-phase with EXPLICIT register arrays and
    // unconditional clamped loads: a single dynamic loop makes the
    // compiler round-robin two registers and emit a full vmcnt(0) drain
    // per 8-byte load (~900 cy of HBM latency each, serialized — the
    // disassembly showed gload/waitcnt(0) pairs and the phase ablation
    // priced it at ~2.6 ms of the 5.9 ms launch).
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int SLV = (STOT + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    auto stage_plane = [&](int P) {
        const int slot = ((P + 1) % 3 + 3) % 3;
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && y0 >= 1 && y0 + TY + 1 <= H &&
                              x0 >= 1 && x0 + TX + 1 <= W;
        // Loads are UNCONDITIONAL from clamped addresses (any masked or
        // pad lane reads a safe in-bounds element); the zero-select runs
        // in the store phase, after the whole batch is in flight — a
        // select attached to each load forces a full vmcnt(0) drain per
        // load (exec-masked load + v_cndmask), which is the serialization
        // this replaces.
        bf16x4 vals[SLV];
        bool keep[SLV];
#pragma unroll
        for (int li = 0; li < SLV; ++li) {
            const int idx = min(tid + li * 512, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
#pragma unroll
        for (int li = 0; li < SLV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    stage_plane(-1);
    stage_plane(0);

    const int ax = lane & 31;            // A row (x), B col (j)
    const int khalf = (lane >> 5) * 8;   // this lane's k sub-range base

    for (int z = 0; z < D; ++z) {
        if (MODE != 3) stage_plane(z + 1);
        __syncthreads();

        f32x16 acc = {};
        if (MODE != 1) {
            // Explicit 4-pair (2-tap) fragment prefetch pipeline. The
            // compiler's natural schedule issues a pair's ds_reads only
            // ~1 MFMA before their s_waitcnt, so every tap parks for the
            // LDS latency (PMC: 67% SQ_WAIT_ANY). With distance 4 the
            // 4 MFMAs in flight (~128 cyc) cover the read latency and the
            // waitcnts become lgkmcnt(6) partial waits.
            const cfx_bf16* planes[3];
#pragma unroll
            for (int dzi = 0; dzi < 3; ++dzi)
                planes[dzi] =
                    &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
            auto addrA = [&](int p) {
                const int tap = p >> 1, kk = p & 1;
                const int dzi = tap / 9, tl = tap % 9;
                const int dy = tl / 3 - 1, dx = tl % 3 - 1;
                return reinterpret_cast<const bf16x8*>(
                    &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) *
                                     PCB + khalf + kk * 16]);
            };
            auto addrB = [&](int p) {
                const int tap = p >> 1, kk = p & 1;
                return reinterpret_cast<const bf16x8*>(
                    &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
            };
            constexpr int PD = 4;        // pairs in flight
            constexpr int NP = 27 * KK;  // 54 A*B pairs per z
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < NP; ++p) {
                const int s = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[s], bbuf[s], acc, 0, 0, 0);
                // sched_barriers pin the issue order (mfma p, then the
                // p+PD reads) — without them the scheduler collapses the
                // pipeline back to distance-1 and re-exposes the latency
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < NP) {
                    abuf[s] = *addrA(p + PD);
                    bbuf[s] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        const int gy = y0 + wave;
        const int j = lane & 31;
        if (MODE != 2 && gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            // residual reads batched up front from clamped addresses
            // (a load inside the store loop serializes: one vmcnt(0)
            // drain per 2-byte read)
            cfx_bf16 rv[16];
            if (res) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
        __syncthreads();
    }
}

// Software-pipelined bf16 ring (same one-barrier-per-z scheme as
// k_conv3_zring_pl: next plane's global loads fly over the first 18 taps,
// drain to LDS, one barrier, last 9 taps). The plain bf16 kernel is
// staging-bound — only 2 MFMAs per tap — so the overlap matters more here
// than in f32.
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_pl(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;
    constexpr int C4 = CP / 4;
    constexpr int LV = (SY * SX * C4 + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    // batched unconditional clamped loads + deferred zero-select (see
    // the plain kernel's stage_plane comment: a per-load select costs a
    // full vmcnt(0) drain per 8-byte load)
    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4 - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 acc = {};
        // fragment addressing for the flat pair pipeline (pair p: tap
        // p>>1, kk p&1); same PD=4 prefetch-distance scheme as the plain
        // kernel, split at pair 36 around the plane barrier
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
        };
        constexpr int PD = 4;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 1, vals, keep);
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 36; ++p) {  // dzi 0,1: planes z-1, z
                const int si = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 36) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            plane_store(z + 1, vals, keep);
            __syncthreads();
#pragma unroll
            for (int p = 36; p < 36 + PD; ++p) {
                abuf[p % PD] = *addrA(p);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = 36; p < 54; ++p) {  // dzi 2: plane z + 1
                const int si = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {  // batched residual reads (clamped addresses)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
    }
}


// accumulator-alternating variant (CFX_ZRING_PL=6): targets the
// same-accumulator issue cliff (see comment in the z loop)
template <int C, int K, int TY, int TX, int PDX = 4,
          int SB = 1, int EPI = 1, int STG = 0>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_a(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;
    constexpr int C4 = CP / 4;
    constexpr int LV = (SY * SX * C4 + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    typedef cfx_bf16 bf16x8w __attribute__((ext_vector_type(8)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];
    // wave-private transpose scratch for the EPI==2 epilogue:
    // [wave][x-row 16][j 32] bf16 (1 KB per wave; no cross-wave sync)
    __shared__ cfx_bf16 oscr[8][16][32];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    // batched unconditional clamped loads + deferred zero-select (see
    // the plain kernel's stage_plane comment: a per-load select costs a
    // full vmcnt(0) drain per 8-byte load)
    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4 - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();
    // STG=1: one-iteration-ahead staging — plane z+1 is loaded during
    // iteration z-1, so the store below never waits on its own loads
    // (the phase timing showed 24% of the z period in store+barrier)
    bf16x4 va[LV];
    bool ka[LV];
    if (STG) plane_load(1, va, ka);

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    // EPI==3: each z's epilogue is DEFERRED into the next iteration's
    // mainloop region (fences off there) so the compiler can interleave
    // its ~60 VALU/LDS/store instructions into the MFMA issue shadows —
    // the phase timing showed the per-z barrier phase-locks the waves
    // and the pipe idles through the epilogue otherwise.
    f32x16 prevA = {};
    f32x16 prevB = {};
    for (int z = 0; z < D; ++z) {
        // TWO alternating accumulators: anything issued between two MFMAs
        // on the SAME accumulator costs a +43-cycle cliff (microarch
        // guide, per-instruction constants) and this loop has 2 LDS reads
        // + loop VALU per MFMA; alternating pairs makes consecutive MFMAs
        // hit different accumulators (~6 cyc/state instead). Accumulation
        // order becomes evens+odds (an f32 reorder vs _pl, inside the
        // bf16 engine tolerance).
        f32x16 accA = {};
        f32x16 accB = {};
        // fragment addressing for the flat pair pipeline (pair p: tap
        // p>>1, kk p&1); same PD=4 prefetch-distance scheme as the plain
        // kernel, split at pair 36 around the plane barrier
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
        };
        auto epilogue = [&](int zz, const f32x16& acc) {
            const int gy = y0 + wave;
            const int j = lane & 31;
            const float bj = (bias && j < K) ? bias[j] : 0.f;
#pragma unroll
            for (int h = 0; h < 2; ++h) {
#pragma unroll
                for (int r = 8 * h; r < 8 * h + 8; ++r) {
                    const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                    4 * (lane >> 5);
                    oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
                }
                const int xr = lane >> 2;
                const int ch = lane & 3;
                const int gx = x0 + 16 * h + xr;
                const int rem = K - ch * 8;
                if (gy < H && gx < W && rem >= 4) {
                    const int j0c = ch * 8;
                    long long o = ((((long long)n * D + zz) * H + gy) *
                                   W + gx) * K + j0c;
                    if (rem >= 8) {
                        bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                            &oscr[wave][xr][j0c]);
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x8w*>(out + o) = v;
                    } else {
                        bf16x4 v = *reinterpret_cast<const bf16x4*>(
                            &oscr[wave][xr][j0c]);
#pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x4*>(out + o) = v;
                    }
                }
            }
        };
        constexpr int PD = PDX;

        bf16x4 vals[LV];
        bool keep[LV];
        if (STG)
            plane_load(z + 2, vals, keep);  // lands during NEXT iteration
        else
            plane_load(z + 1, vals, keep);
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
            if (EPI == 3 && z > 0) epilogue(z - 1, prevA + prevB);
#pragma unroll
            for (int p = 0; p < 36; ++p) {  // dzi 0,1: planes z-1, z
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                if (SB) __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 36) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                if (SB) __builtin_amdgcn_sched_barrier(0);
            }
            if (STG)
                plane_store(z + 1, va, ka);
            else
                plane_store(z + 1, vals, keep);
            __syncthreads();
#pragma unroll
            for (int p = 36; p < 36 + PD; ++p) {
                abuf[p % PD] = *addrA(p);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = 36; p < 54; ++p) {  // dzi 2: plane z + 1
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                if (SB) __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                if (SB) __builtin_amdgcn_sched_barrier(0);
            }
        }

        if (STG) {
#pragma unroll
            for (int li = 0; li < LV; ++li) {
                va[li] = vals[li];
                ka[li] = keep[li];
            }
        }
        if (EPI == 3) {
            prevA = accA;
            prevB = accB;
            continue;
        }
        const f32x16 acc = accA + accB;
        if (!EPI) {  // timing ablation: keep acc alive, skip the real
                     // epilogue (WRONG results; CFX_ZRING_PL=10 only)
            out[tid] = (cfx_bf16)(acc[0] + acc[15]);
            continue;
        }
        if (EPI == 2) {
            // transposed epilogue: the j-per-lane layout stores 16
            // scattered 2-byte elements per lane per z; those 128 store
            // instructions per CU per z measured as 40% of the launch
            // (PL=10 ablation). Bounce the 32x32 tile
            // through wave-private LDS in two 16-row halves (the full
            // tile would need 16 KB of scratch and bust the 160 KB LDS
            // budget) and store x-major: one coalesced b128/b64 store
            // per lane per half. Residual reads ride the same wide
            // path; (acc+bias) rounds through bf16 before the res add
            // (bf16-engine tolerance, not bit-identical to _pl).
            const int gy = y0 + wave;
            const int j = lane & 31;
            const float bj = bias ? bias[j] : 0.f;
#pragma unroll
            for (int h = 0; h < 2; ++h) {
#pragma unroll
                for (int r = 8 * h; r < 8 * h + 8; ++r) {
                    const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                    4 * (lane >> 5);
                    oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
                }
                const int xr = lane >> 2;       // x row 0..15 in half
                const int ch = lane & 3;        // 16-byte chunk 0..3
                const int gx = x0 + 16 * h + xr;
                if (gy < H && gx < W) {
                    const int j0 = ch * 8;
                    long long o = ((((long long)n * D + z) * H + gy) *
                                   W + gx) * K + j0;
                    if (j0 + 8 <= K) {
                        bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                            &oscr[wave][xr][j0]);
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x8w*>(out + o) = v;
                    } else if (j0 < K) {        // tail chunk (K=28)
                        bf16x4 v = *reinterpret_cast<const bf16x4*>(
                            &oscr[wave][xr][j0]);
#pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x4*>(out + o) = v;
                    }
                }
            }
            continue;
        }
        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {  // batched residual reads (clamped addresses)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
    }
    if (EPI == 3) {
        // last z's epilogue (the in-loop lambda is out of scope here;
        // inline copy with zz = D-1)
        const f32x16 acc = prevA + prevB;
        const int gy = y0 + wave;
        const int j = lane & 31;
        const float bj = (bias && j < K) ? bias[j] : 0.f;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
#pragma unroll
            for (int r = 8 * h; r < 8 * h + 8; ++r) {
                const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                4 * (lane >> 5);
                oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
            }
            const int xr = lane >> 2;
            const int ch = lane & 3;
            const int gx = x0 + 16 * h + xr;
            const int rem = K - ch * 8;
            if (gy < H && gx < W && rem >= 4) {
                const int j0c = ch * 8;
                long long o = ((((long long)n * D + (D - 1)) * H + gy) *
                               W + gx) * K + j0c;
                if (rem >= 8) {
                    bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x8w*>(out + o) = v;
                } else {
                    bf16x4 v = *reinterpret_cast<const bf16x4*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 4; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x4*>(out + o) = v;
                }
            }
        }
    }
}


// 4-slot swizzled ring ("q2", CFX_ZRING_PL=14): slots hold planes
// z-1..z+2, so the plane stored during iteration z ((z+2) % 4) is
// DISJOINT from every slot read in iteration z -- no mid-iteration
// barrier; ONE barrier at iteration end bounds wave skew to <1
// iteration (the r2 phase timing put 24% of the z period in the old
// mid-iteration store+barrier). LDS fits because both ring and wall
// drop their +8 padding for the stride-32 XOR swizzle proven in the
// "w" kernel (chunk16 ^= (row>>2) & 3; conflict-free for every b128
// lane group): ring 4*10*34*32*2 = 87.0 KB + wall 55.3 + 8 KB
// transpose scratch = 150.3 KB. Epilogue = the transposed wide-store
// path. Accumulators alternate like "_a".
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_q2(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int RS = 32;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int LV = (STOT + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    typedef cfx_bf16 bf16x8w __attribute__((ext_vector_type(8)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[4 * SY * SX * RS];
    __shared__ cfx_bf16 wall[27 * 32 * RS];
    __shared__ cfx_bf16 oscr[8][16][32];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    auto sw = [](int row, int el) {
        return row * RS + ((((el >> 3) ^ (row >> 2)) & 3) << 3) + (el & 7);
    };

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[sw(tap * 32 + j, c)] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 4 + 4) % 4;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int row = (slot * SY + v / SX) * SX + v % SX;
            *reinterpret_cast<bf16x4*>(&ring[sw(row, c4 * 4)]) =
                keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV], v2[LV];
        bool k0[LV], k1[LV], k2[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_load(1, v2, k2);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
        plane_store(1, v2, k2);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 accA = {};
        f32x16 accB = {};
        int slots[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            slots[dzi] = ((z + dzi) % 4 + 4) % 4;
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            const int row = (slots[dzi] * SY + 1 + wave + dy) * SX +
                            (1 + dx) + ax;
            return reinterpret_cast<const bf16x8*>(
                &ring[sw(row, khalf + kk * 16)]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[sw(tap * 32 + ax, khalf + kk * 16)]);
        };
        constexpr int PD = 4;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 2, vals, keep);  // stored at the END of this iter
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 54; ++p) {  // all 27 taps, no mid barrier
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }
        plane_store(z + 2, vals, keep);  // slot (z+3)%4: read from z+1 on

        const f32x16 acc = accA + accB;
        const int gy = y0 + wave;
        const int j = lane & 31;
        const float bj = (bias && j < K) ? bias[j] : 0.f;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
#pragma unroll
            for (int r = 8 * h; r < 8 * h + 8; ++r) {
                const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                4 * (lane >> 5);
                oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
            }
            const int xr = lane >> 2;
            const int ch = lane & 3;
            const int gx = x0 + 16 * h + xr;
            const int rem = K - ch * 8;
            if (gy < H && gx < W && rem >= 4) {
                const int j0c = ch * 8;
                long long o = ((((long long)n * D + z) * H + gy) * W +
                               gx) * K + j0c;
                if (rem >= 8) {
                    bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x8w*>(out + o) = v;
                } else {
                    bf16x4 v = *reinterpret_cast<const bf16x4*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 4; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x4*>(out + o) = v;
                }
            }
        }
        __syncthreads();  // the ONLY barrier: bounds wave skew to < 1 iter
    }
}



// s_memtime phase-timing clone of the PL=13 default (CFX_ZRING_PL=12;
// WRONG results)
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_tm(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;
    constexpr int C4 = CP / 4;
    constexpr int LV = (SY * SX * C4 + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    typedef cfx_bf16 bf16x8w __attribute__((ext_vector_type(8)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];
    // wave-private transpose scratch for the EPI==2 epilogue:
    // [wave][x-row 16][j 32] bf16 (1 KB per wave; no cross-wave sync)
    __shared__ cfx_bf16 oscr[8][16][32];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    unsigned long long t_load = 0, t_mfma1 = 0, t_store = 0,
                       t_mfma2 = 0, t_epi = 0, t_mark;
#define TM_MARK() t_mark = __builtin_amdgcn_s_memtime()
#define TM_ACC(v) v += __builtin_amdgcn_s_memtime() - t_mark

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    // batched unconditional clamped loads + deferred zero-select (see
    // the plain kernel's stage_plane comment: a per-load select costs a
    // full vmcnt(0) drain per 8-byte load)
    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4 - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();
    // STG=1: one-iteration-ahead staging — plane z+1 is loaded during
    // iteration z-1, so the store below never waits on its own loads
    // (the phase timing showed 24% of the z period in store+barrier)
    bf16x4 va[LV];
    bool ka[LV];
    plane_load(1, va, ka);

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        // TWO alternating accumulators: anything issued between two MFMAs
        // on the SAME accumulator costs a +43-cycle cliff (microarch
        // guide, per-instruction constants) and this loop has 2 LDS reads
        // + loop VALU per MFMA; alternating pairs makes consecutive MFMAs
        // hit different accumulators (~6 cyc/state instead). Accumulation
        // order becomes evens+odds (an f32 reorder vs _pl, inside the
        // bf16 engine tolerance).
        f32x16 accA = {};
        f32x16 accB = {};
        // fragment addressing for the flat pair pipeline (pair p: tap
        // p>>1, kk p&1); same PD=4 prefetch-distance scheme as the plain
        // kernel, split at pair 36 around the plane barrier
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
        };
        constexpr int PD = 4;

        bf16x4 vals[LV];
        bool keep[LV];
        TM_MARK();
        plane_load(z + 2, vals, keep);
        TM_ACC(t_load);
        TM_MARK();
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 36; ++p) {  // dzi 0,1: planes z-1, z
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 36) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            TM_ACC(t_mfma1);
            TM_MARK();
            plane_store(z + 1, va, ka);
            __syncthreads();
            TM_ACC(t_store);
            TM_MARK();
#pragma unroll
            for (int p = 36; p < 36 + PD; ++p) {
                abuf[p % PD] = *addrA(p);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = 36; p < 54; ++p) {  // dzi 2: plane z + 1
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        TM_ACC(t_mfma2);
        TM_MARK();
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            va[li] = vals[li];
            ka[li] = keep[li];
        }
        const f32x16 acc = accA + accB;
                if (1) {
            // transposed epilogue: the j-per-lane layout stores 16
            // scattered 2-byte elements per lane per z; those 128 store
            // instructions per CU per z measured as 40% of the launch
            // (PL=10 ablation). Bounce the 32x32 tile
            // through wave-private LDS in two 16-row halves (the full
            // tile would need 16 KB of scratch and bust the 160 KB LDS
            // budget) and store x-major: one coalesced b128/b64 store
            // per lane per half. Residual reads ride the same wide
            // path; (acc+bias) rounds through bf16 before the res add
            // (bf16-engine tolerance, not bit-identical to _pl).
            const int gy = y0 + wave;
            const int j = lane & 31;
            const float bj = bias ? bias[j] : 0.f;
#pragma unroll
            for (int h = 0; h < 2; ++h) {
#pragma unroll
                for (int r = 8 * h; r < 8 * h + 8; ++r) {
                    const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                    4 * (lane >> 5);
                    oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
                }
                const int xr = lane >> 2;       // x row 0..15 in half
                const int ch = lane & 3;        // 16-byte chunk 0..3
                const int gx = x0 + 16 * h + xr;
                if (gy < H && gx < W) {
                    const int j0 = ch * 8;
                    long long o = ((((long long)n * D + z) * H + gy) *
                                   W + gx) * K + j0;
                    if (j0 + 8 <= K) {
                        bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                            &oscr[wave][xr][j0]);
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x8w*>(out + o) = v;
                    } else if (j0 < K) {        // tail chunk (K=28)
                        bf16x4 v = *reinterpret_cast<const bf16x4*>(
                            &oscr[wave][xr][j0]);
#pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x4*>(out + o) = v;
                    }
                }
            }
            continue;
        }
        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {  // batched residual reads (clamped addresses)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
        TM_ACC(t_epi);
    }
    if (tid == 0) {
        unsigned long long* dbg = reinterpret_cast<unsigned long long*>(out);
        const int wg = (blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x +
                       blockIdx.x;
        dbg[wg * 6 + 0] = t_load;
        dbg[wg * 6 + 1] = t_mfma1;
        dbg[wg * 6 + 2] = t_store;
        dbg[wg * 6 + 3] = t_mfma2;
        dbg[wg * 6 + 4] = t_epi;
        dbg[wg * 6 + 5] = t_load + t_mfma1 + t_store + t_mfma2 + t_epi;
    }
#undef TM_MARK
#undef TM_ACC
}


// one-iteration-ahead variant: double register buffers so the
// plane store never waits on its own loads (CFX_ZRING_PL=3)
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_pl2(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;
    constexpr int C4 = CP / 4;
    constexpr int LV = (SY * SX * C4 + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    // batched unconditional clamped loads + deferred zero-select (see
    // the plain kernel's stage_plane comment: a per-load select costs a
    // full vmcnt(0) drain per 8-byte load)
    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4 - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    bf16x4 va[LV];
    bool ka[LV];
    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    plane_load(1, va, ka);  // lands during iteration 0's mainloop
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 acc = {};
        // fragment addressing for the flat pair pipeline (pair p: tap
        // p>>1, kk p&1); same PD=4 prefetch-distance scheme as the plain
        // kernel, split at pair 36 around the plane barrier
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
        };
        constexpr int PD = 4;

        // plane z+2 issues NOW; plane z+1 (va) was issued one whole
        // iteration ago, so its store below never drains vmcnt — the
        // 61%-parked bottleneck of the single-buffer _pl (pmc_bf16 r02)
        bf16x4 vb[LV];
        bool kb[LV];
        plane_load(z + 2, vb, kb);
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 36; ++p) {  // dzi 0,1: planes z-1, z
                const int si = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 36) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            plane_store(z + 1, va, ka);
            __syncthreads();
#pragma unroll
            for (int p = 36; p < 36 + PD; ++p) {
                abuf[p % PD] = *addrA(p);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = 36; p < 54; ++p) {  // dzi 2: plane z + 1
                const int si = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

#pragma unroll
        for (int li = 0; li < LV; ++li) {
            va[li] = vb[li];
            ka[li] = kb[li];
        }

        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {  // batched residual reads (clamped addresses)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
    }
}


// Wide bf16 ring ("w"): 256 threads / 4 waves, each wave computing TWO
// y-rows per z so the B (weight) fragment read and the per-z loop/barrier
// overhead amortize over 108 MFMAs instead of 54. LDS drops the +8
// padding: rows are stride-32 (64 B) with an XOR swizzle on the 16-byte
// chunk index (chunk ^= (row>>2) & 3), which spreads the 4-lane
// same-bank classes of each b128 lane group across 4 chunks — measured
// conflict-free reasoning in DESIGN.md §10 r2. Ring 65.3 KB + wall
// 55.3 KB = 120.6 KB (the padded layout at this shape needs 182 KB).
// Accumulation order per output element is IDENTICAL to _pl.
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(256, 1) void k_conv3_zring_bf16_w(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int RS = 32;               // row stride, elements (64 B)
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    static_assert(C <= CP && K <= 32 && TY == 8 && TX == 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * RS];
    __shared__ cfx_bf16 wall[27 * 32 * RS];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;           // 0..3 -> y-row pair
    const int lane = tid & 63;

    // swizzled element offset within a row: chunk16 ^= (row >> 2) & 3
    auto sw = [](int row, int el) {
        return row * RS + ((((el >> 3) ^ (row >> 2)) & 3) << 3) + (el & 7);
    };

    for (int idx = tid; idx < 27 * 32 * CP; idx += 256) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[sw(tap * 32 + j, c)] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int LV = (STOT + 255) / 256;

    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 256, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 256;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[sw((slot * SY + v / SX) * SX + v % SX, c4 * 4)]) =
                keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 acc0 = {};
        f32x16 acc1 = {};
        int slots[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            slots[dzi] = ((z + dzi) % 3 + 3) % 3;
        auto addrA = [&](int p, int row2) {  // row2: 0/1 within the pair
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            const int r = (slots[dzi] * SY + 1 + 2 * wave + row2 + dy) *
                              SX + (1 + dx) + ax;
            return reinterpret_cast<const bf16x8*>(
                &ring[sw(r, khalf + kk * 16)]);
        };
        auto addrB = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            return reinterpret_cast<const bf16x8*>(
                &wall[sw(tap * 32 + ax, khalf + kk * 16)]);
        };
        constexpr int PD = 3;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 1, vals, keep);
        {
            bf16x8 a0buf[PD], a1buf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                a0buf[p] = *addrA(p, 0);
                a1buf[p] = *addrA(p, 1);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 36; ++p) {  // dzi 0,1: planes z-1, z
                const int si = p % PD;
                acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a0buf[si], bbuf[si], acc0, 0, 0, 0);
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a1buf[si], bbuf[si], acc1, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 36) {
                    a0buf[si] = *addrA(p + PD, 0);
                    a1buf[si] = *addrA(p + PD, 1);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            plane_store(z + 1, vals, keep);
            __syncthreads();
#pragma unroll
            for (int p = 36; p < 36 + PD; ++p) {
                a0buf[p % PD] = *addrA(p, 0);
                a1buf[p % PD] = *addrA(p, 1);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = 36; p < 54; ++p) {  // dzi 2: plane z + 1
                const int si = p % PD;
                acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a0buf[si], bbuf[si], acc0, 0, 0, 0);
                acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    a1buf[si], bbuf[si], acc1, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    a0buf[si] = *addrA(p + PD, 0);
                    a1buf[si] = *addrA(p + PD, 1);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        const int j = lane & 31;
#pragma unroll
        for (int row2 = 0; row2 < 2; ++row2) {
            const int gy = y0 + 2 * wave + row2;
            if (gy >= H || j >= K) continue;
            const f32x16& acc = row2 ? acc1 : acc0;
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
    }
}


// 16x16x32 ring ("t"): same staging/ring/wall as _pl but the mainloop
// uses v_mfma_f32_16x16x32_bf16, whose K=32 covers ALL channels in one
// op — ONE b128 LDS read per MFMA instead of two. The r2 phase ablation
//  showed the 32x32x16 loop's 2 reads/MFMA saturate the
// LDS array at exactly the MFMA-pipe time (3456 cyc each per z per CU),
// so the 32-wide ring is LDS-read co-bound; 16x16x32 trades a 20% lower
// MFMA peak rate (~5 vs ~8 cyc/CU per-flop-adjusted) for halving the LDS
// pressure. Per wave: 2 M-tiles (x) x 2 N-tiles (j) of f32x4, which also
// alternates accumulators between consecutive MFMAs for free.
// Fragment layouts probe-verified (tools/mfma_layout_probe.py, 16x16x32):
// A/B: row/col = lane&15, k = (lane>>4)*8 + e; C/D: col = lane&15,
// row = (lane>>4)*4 + reg.
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_t(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    typedef float f32x4 __attribute__((ext_vector_type(4)));
    static_assert(C <= CP && K <= 32 && TX == 32, "");

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int LV = (STOT + 511) / 512;

    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();

    const int li16 = lane & 15;          // fragment row/col
    const int kc = (lane >> 4) * 8;      // fragment k-chunk (channels)

    for (int z = 0; z < D; ++z) {
        f32x4 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int tap, int m) {
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + m * 16 +
                              li16) * PCB + kc]);
        };
        auto addrB = [&](int tap, int nt) {
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + nt * 16 + li16) * PCB + kc]);
        };
        constexpr int PD = 3;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 1, vals, keep);
        {
            bf16x8 a0[PD], a1[PD], b0[PD], b1[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                a0[p] = *addrA(p, 0);
                a1[p] = *addrA(p, 1);
                b0[p] = *addrB(p, 0);
                b1[p] = *addrB(p, 1);
            }
#pragma unroll
            for (int tap = 0; tap < 18; ++tap) {  // dzi 0,1
                const int si = tap % PD;
                acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0[si], b0[si], acc00, 0, 0, 0);
                acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0[si], b1[si], acc01, 0, 0, 0);
                acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a1[si], b0[si], acc10, 0, 0, 0);
                acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a1[si], b1[si], acc11, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (tap + PD < 18) {
                    a0[si] = *addrA(tap + PD, 0);
                    a1[si] = *addrA(tap + PD, 1);
                    b0[si] = *addrB(tap + PD, 0);
                    b1[si] = *addrB(tap + PD, 1);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            plane_store(z + 1, vals, keep);
            __syncthreads();
#pragma unroll
            for (int tap = 18; tap < 18 + PD; ++tap) {
                a0[tap % PD] = *addrA(tap, 0);
                a1[tap % PD] = *addrA(tap, 1);
                b0[tap % PD] = *addrB(tap, 0);
                b1[tap % PD] = *addrB(tap, 1);
            }
#pragma unroll
            for (int tap = 18; tap < 27; ++tap) {  // dzi 2
                const int si = tap % PD;
                acc00 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0[si], b0[si], acc00, 0, 0, 0);
                acc01 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a0[si], b1[si], acc01, 0, 0, 0);
                acc10 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a1[si], b0[si], acc10, 0, 0, 0);
                acc11 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a1[si], b1[si], acc11, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (tap + PD < 27) {
                    a0[si] = *addrA(tap + PD, 0);
                    a1[si] = *addrA(tap + PD, 1);
                    b0[si] = *addrB(tap + PD, 0);
                    b1[si] = *addrB(tap + PD, 1);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        const int gy = y0 + wave;
        if (gy < H) {
            const long long rowbase =
                (((long long)n * D + z) * H + gy) * (long long)W * K;
#pragma unroll
            for (int m = 0; m < 2; ++m) {
#pragma unroll
                for (int nt = 0; nt < 2; ++nt) {
                    const int j = nt * 16 + li16;
                    if (j >= K) continue;
                    const f32x4& acc = m == 0 ? (nt == 0 ? acc00 : acc01)
                                             : (nt == 0 ? acc10 : acc11);
                    const float bj = bias ? bias[j] : 0.f;
                    cfx_bf16 rv[4];
                    if (res) {
#pragma unroll
                        for (int r = 0; r < 4; ++r) {
                            const int row =
                                m * 16 + (lane >> 4) * 4 + r;
                            const int gx = min(x0 + row, W - 1);
                            rv[r] = res[rowbase + (long long)gx * K + j];
                        }
                    }
#pragma unroll
                    for (int r = 0; r < 4; ++r) {
                        const int row = m * 16 + (lane >> 4) * 4 + r;
                        const int gx = x0 + row;
                        if (gx >= W) continue;
                        float v = acc[r] + bj;
                        if (res) v += (float)rv[r];
                        if (do_elu) v = v > 0.f ? v : expm1f(v);
                        out[rowbase + (long long)gx * K + j] =
                            (cfx_bf16)v;
                    }
                }
            }
        }
    }
}



// q3 (CFX_ZRING_PL=15): 4-slot PADDED ring (PCB=40, constant-offset
// addressing preserved — the swizzled q2 lost more to per-lane address
// VALU than its barrier saved) + the weight wall halved to the kk0
// slices (PCB=24) with kk1 B-fragments streamed from global/L2 (the
// 55 KB pack is L2-hot on every XCD). The slot stored in iteration z,
// (z+3) % 4, is disjoint from every slot read in iterations z and z-1
// after the single end-of-iteration barrier — which bounds wave skew
// to < 1 iteration (the r2 phase timing put 40% of the non-epilogue z
// period in the old mid-iteration store+barrier). Epilogue = the
// transposed wide-store path; accumulators alternate.
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_q3(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB = CP + 8;          // ring x-stride (conflict-free)
    constexpr int WS = 24;               // wall row stride (16 used + 8)
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int LV = (STOT + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    typedef cfx_bf16 bf16x8w __attribute__((ext_vector_type(8)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[4 * SY * SX * PCB];   // 108.8 KB
    __shared__ cfx_bf16 wall[27 * 32 * WS];        // 41.5 KB (kk0 half)
    __shared__ cfx_bf16 oscr[8][16][32];           // 8 KB

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * 16; idx += 512) {
        const int c = idx % 16;
        const int j = (idx / 16) % 32;
        const int tap = idx / (16 * 32);
        wall[(tap * 32 + j) * WS + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 4 + 4) % 4;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV], v2[LV];
        bool k0[LV], k1[LV], k2[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_load(1, v2, k2);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
        plane_store(1, v2, k2);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 accA = {};
        f32x16 accB = {};
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] =
                &ring[(((z + dzi) % 4 + 4) % 4) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p >> 1, kk = p & 1;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {  // kk0: LDS wall; kk1: global (L2-hot)
            const int tap = p >> 1;
            if (p & 1)
                return reinterpret_cast<const bf16x8*>(
                    &wgt[(tap * 32 + ax) * 32 + khalf + 16]);
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * WS + khalf]);
        };
        constexpr int PD = 4;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 2, vals, keep);
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < 54; ++p) {
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    accA = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accA, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < 54) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }
        plane_store(z + 2, vals, keep);  // slot (z+3)%4: disjoint

        const f32x16 acc = accA + accB;
        const int gy = y0 + wave;
        const int j = lane & 31;
        const float bj = (bias && j < K) ? bias[j] : 0.f;
#pragma unroll
        for (int h = 0; h < 2; ++h) {
#pragma unroll
            for (int r = 8 * h; r < 8 * h + 8; ++r) {
                const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                4 * (lane >> 5);
                oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
            }
            const int xr = lane >> 2;
            const int ch = lane & 3;
            const int gx = x0 + 16 * h + xr;
            const int rem = K - ch * 8;
            if (gy < H && gx < W && rem >= 4) {
                const int j0c = ch * 8;
                long long o = ((((long long)n * D + z) * H + gy) * W +
                               gx) * K + j0c;
                if (rem >= 8) {
                    bf16x8w v = *reinterpret_cast<const bf16x8w*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 8; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x8w*>(out + o) = v;
                } else {
                    bf16x4 v = *reinterpret_cast<const bf16x4*>(
                        &oscr[wave][xr][j0c]);
#pragma unroll
                    for (int e = 0; e < 4; ++e) {
                        float t = (float)v[e];
                        if (res) t += (float)res[o + e];
                        if (do_elu) t = t > 0.f ? t : expm1f(t);
                        v[e] = (cfx_bf16)t;
                    }
                    *reinterpret_cast<bf16x4*>(out + o) = v;
                }
            }
        }
        __syncthreads();  // the only barrier per z
    }
}


// Sliced bf16 ring: the _pl kernel generalized to a c-slice [c0, c0+CL)
// of a wider channel dimension (runtime stride CS) and a j-tile
// [j0, j0+32) of a wider K (runtime stride KS). Widths 36 and 48 run as
// four launches — (c 0..31, CPV=32) + (c 32.., CPV=16), each for j-tiles
// j0=0 and j0=32 — reusing the PCB=CP+8 LDS layout whose strides (40/24
// elements) are the measured conflict-free ones. The second c-half launch
// chains through `res` = its own output (partial sums round through bf16
// between halves; bias/ELU apply only on the final half). Weight pack:
// [27][64][48] zero-padded, [tap][j][c].
template <int CPV, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_s(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int CS,
    int CL, int c0, int KS, int Ktot, int j0, int do_elu) {
    constexpr int CP = CPV;
    constexpr int PCB = CP + 8;
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int KK = CP / 16;
    constexpr int C4 = CP / 4;
    constexpr int LV = (SY * SX * C4 + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    constexpr int WPK = 64;  // weight pack j rows
    constexpr int WPC = 48;  // weight pack c width

    __shared__ cfx_bf16 ring[3 * SY * SX * PCB];
    __shared__ cfx_bf16 wall[27 * 32 * PCB];
    __shared__ cfx_bf16 oscr[8][16][32];  // transposed-epilogue scratch

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB + c] =
            wgt[(tap * WPK + j0 + j) * WPC + c0 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;

    auto plane_load = [&](int P, bf16x4 (&vals)[LV], bool (&keep)[LV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = min(tid + li * 512, SY * SX * C4 - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < CL;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + (((long long)n * D + (zin ? P : 0)) * H +
                      (ok ? gy : 0)) * (long long)W * CS +
                (ok ? gx : 0) * (long long)CS + c0 + (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[LV],
                           const bool (&keep)[LV]) {
        const int slot = ((P + 1) % 3 + 3) % 3;
#pragma unroll
        for (int li = 0; li < LV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= SY * SX * C4) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[LV], v1[LV];
        bool k0[LV], k1[LV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        f32x16 acc = {};   // even pairs; accB odd (same-acc-cliff dodge,
        f32x16 accB = {};  // see k_conv3_zring_bf16_a)
        const cfx_bf16* planes[3];
#pragma unroll
        for (int dzi = 0; dzi < 3; ++dzi)
            planes[dzi] = &ring[(((z + dzi) % 3 + 3) % 3) * SY * SX * PCB];
        auto addrA = [&](int p) {
            const int tap = p / KK, kk = p % KK;
            const int dzi = tap / 9, tl = tap % 9;
            const int dy = tl / 3 - 1, dx = tl % 3 - 1;
            return reinterpret_cast<const bf16x8*>(
                &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) * PCB +
                             khalf + kk * 16]);
        };
        auto addrB = [&](int p) {
            const int tap = p / KK, kk = p % KK;
            return reinterpret_cast<const bf16x8*>(
                &wall[(tap * 32 + ax) * PCB + khalf + kk * 16]);
        };
        constexpr int PD = 4;
        constexpr int NP1 = 18 * KK;  // dzi 0,1: planes z-1, z
        constexpr int NP2 = 27 * KK;

        bf16x4 vals[LV];
        bool keep[LV];
        plane_load(z + 1, vals, keep);
        {
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < NP1; ++p) {
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < NP1) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
            plane_store(z + 1, vals, keep);
            __syncthreads();
#pragma unroll
            for (int p = NP1; p < NP1 + PD; ++p) {
                abuf[p % PD] = *addrA(p);
                bbuf[p % PD] = *addrB(p);
            }
#pragma unroll
            for (int p = NP1; p < NP2; ++p) {
                const int si = p % PD;
                if (p & 1)
                    accB = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], accB, 0, 0, 0);
                else
                    acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                        abuf[si], bbuf[si], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < NP2) {
                    abuf[si] = *addrA(p + PD);
                    bbuf[si] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        acc = acc + accB;
        // transposed wide-store epilogue (see k_conv3_zring_bf16_a
        // EPI==2): two 16-row halves through wave-private LDS, one
        // coalesced b128/b64 store per lane per half. rem per 8-j chunk
        // is always 8, 4 or <=0 for K in {28, 36, 48}.
        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H) {
            const float bj =
                (bias && j0 + j < Ktot) ? bias[j0 + j] : 0.f;
#pragma unroll
            for (int h = 0; h < 2; ++h) {
#pragma unroll
                for (int r = 8 * h; r < 8 * h + 8; ++r) {
                    const int row = (r & 3) + 8 * ((r >> 2) & 1) +
                                    4 * (lane >> 5);
                    oscr[wave][row][j] = (cfx_bf16)(acc[r] + bj);
                }
                const int xr = lane >> 2;
                const int ch = lane & 3;
                const int gx = x0 + 16 * h + xr;
                const int rem = Ktot - j0 - ch * 8;
                if (gx < W && rem >= 4) {
                    const int j0c = ch * 8;
                    long long o = (((long long)n * D + z) * H + gy) *
                                      (long long)W * KS +
                                  (long long)gx * KS + j0 + j0c;
                    typedef cfx_bf16 bf16x8s
                        __attribute__((ext_vector_type(8)));
                    if (rem >= 8) {
                        bf16x8s v = *reinterpret_cast<const bf16x8s*>(
                            &oscr[wave][xr][j0c]);
#pragma unroll
                        for (int e = 0; e < 8; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x8s*>(out + o) = v;
                    } else {
                        bf16x4 v = *reinterpret_cast<const bf16x4*>(
                            &oscr[wave][xr][j0c]);
#pragma unroll
                        for (int e = 0; e < 4; ++e) {
                            float t = (float)v[e];
                            if (res) t += (float)res[o + e];
                            if (do_elu) t = t > 0.f ? t : expm1f(t);
                            v[e] = (cfx_bf16)t;
                        }
                        *reinterpret_cast<bf16x4*>(out + o) = v;
                    }
                }
            }
        }
    }
}


// 4-slot ring ("quad"): planes z-1, z, z+1 all RESIDENT when iteration z
// starts, plane z+2 loads during the full 54-MFMA mainloop and lands in
// the slot plane z-2 vacated — ONE barrier per z and a whole iteration
// (~3.5k cyc) of latency cover for the staging loads, vs the 3-slot
// ring where plane z+1 must land mid-iteration. The x-stride shrinks to
// PCB2=36 so 4 slots + the wall fit 160 KB LDS exactly (160,128 B);
// stride 72 B leaves a 2-way conflict on 2 of 32 banks per 8-lane b128
// phase — measured below the MFMA bound. Accumulation order matches the
// other bf16 rings bit-for-bit.
template <int C, int K, int TY, int TX>
__global__ __launch_bounds__(512, 1) void k_conv3_zring_bf16_q(
    const cfx_bf16* __restrict__ in, const cfx_bf16* __restrict__ wgt,
    const float* __restrict__ bias, const cfx_bf16* __restrict__ res,
    cfx_bf16* __restrict__ out, int N, int D, int H, int W, int do_elu) {
    constexpr int CP = 32;
    constexpr int PCB2 = CP + 4;         // 36: 4-slot ring must fit LDS
    constexpr int SX = TX + 2;
    constexpr int SY = TY + 2;
    constexpr int C4 = CP / 4;
    constexpr int STOT = SY * SX * C4;
    constexpr int SLV = (STOT + 511) / 512;
    typedef cfx_bf16 bf16x4 __attribute__((ext_vector_type(4)));
    static_assert(C <= CP && K <= 32, "");

    __shared__ cfx_bf16 ring[4 * SY * SX * PCB2];
    __shared__ cfx_bf16 wall[27 * 32 * PCB2];

    const int n = blockIdx.z;
    const int y0 = blockIdx.y * TY;
    const int x0 = blockIdx.x * TX;
    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;

    for (int idx = tid; idx < 27 * 32 * CP; idx += 512) {
        const int c = idx % CP;
        const int j = (idx / CP) % 32;
        const int tap = idx / (CP * 32);
        wall[(tap * 32 + j) * PCB2 + c] = wgt[(tap * 32 + j) * 32 + c];
    }

    const bool xy_interior = y0 >= 1 && y0 + TY + 1 <= H && x0 >= 1 &&
                             x0 + TX + 1 <= W;
    auto plane_load = [&](int P, bf16x4 (&vals)[SLV], bool (&keep)[SLV]) {
        const bool zin = P >= 0 && P < D;
        const bool interior = zin && xy_interior;
#pragma unroll
        for (int li = 0; li < SLV; ++li) {
            const int idx = min(tid + li * 512, STOT - 1);
            const int c4 = idx % C4;
            const int v = idx / C4;
            const int gy = y0 + v / SX - 1;
            const int gx = x0 + v % SX - 1;
            const bool cok = c4 * 4 < C;
            const bool ok = cok && zin &&
                            (interior || (gy >= 0 && gy < H && gx >= 0 &&
                                          gx < W));
            keep[li] = ok;
            vals[li] = *reinterpret_cast<const bf16x4*>(
                in + ((((long long)n * D + (zin ? P : 0)) * H +
                       (ok ? gy : 0)) * W + (ok ? gx : 0)) * C +
                (cok ? c4 * 4 : 0));
        }
    };
    auto plane_store = [&](int P, const bf16x4 (&vals)[SLV],
                           const bool (&keep)[SLV]) {
        const int slot = (P + 2) & 3;
#pragma unroll
        for (int li = 0; li < SLV; ++li) {
            const int idx = tid + li * 512;
            if (idx >= STOT) break;
            const int c4 = idx % C4;
            const int v = idx / C4;
            *reinterpret_cast<bf16x4*>(
                &ring[((slot * SY + v / SX) * SX + v % SX) * PCB2 +
                      c4 * 4]) = keep[li] ? vals[li] : bf16x4{};
        }
    };

    {
        bf16x4 v0[SLV], v1[SLV], v2[SLV];
        bool k0[SLV], k1[SLV], k2[SLV];
        plane_load(-1, v0, k0);
        plane_load(0, v1, k1);
        plane_load(1, v2, k2);
        plane_store(-1, v0, k0);
        plane_store(0, v1, k1);
        plane_store(1, v2, k2);
    }
    __syncthreads();

    const int ax = lane & 31;
    const int khalf = (lane >> 5) * 8;

    for (int z = 0; z < D; ++z) {
        bf16x4 vals[SLV];
        bool keep[SLV];
        plane_load(z + 2, vals, keep);  // a full iteration to land

        f32x16 acc = {};
        {
            const cfx_bf16* planes[3];
#pragma unroll
            for (int dzi = 0; dzi < 3; ++dzi)
                planes[dzi] =
                    &ring[((z + dzi + 1) & 3) * SY * SX * PCB2];
            auto addrA = [&](int p) {
                const int tap = p >> 1, kk = p & 1;
                const int dzi = tap / 9, tl = tap % 9;
                const int dy = tl / 3 - 1, dx = tl % 3 - 1;
                return reinterpret_cast<const bf16x8*>(
                    &planes[dzi][((1 + wave + dy) * SX + (1 + dx) + ax) *
                                     PCB2 + khalf + kk * 16]);
            };
            auto addrB = [&](int p) {
                const int tap = p >> 1, kk = p & 1;
                return reinterpret_cast<const bf16x8*>(
                    &wall[(tap * 32 + ax) * PCB2 + khalf + kk * 16]);
            };
            constexpr int PD = 4;
            constexpr int NP = 27 * 2;
            bf16x8 abuf[PD], bbuf[PD];
#pragma unroll
            for (int p = 0; p < PD; ++p) {
                abuf[p] = *addrA(p);
                bbuf[p] = *addrB(p);
            }
#pragma unroll
            for (int p = 0; p < NP; ++p) {
                const int sidx = p % PD;
                acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                    abuf[sidx], bbuf[sidx], acc, 0, 0, 0);
                __builtin_amdgcn_sched_barrier(0);
                if (p + PD < NP) {
                    abuf[sidx] = *addrA(p + PD);
                    bbuf[sidx] = *addrB(p + PD);
                }
                __builtin_amdgcn_sched_barrier(0);
            }
        }

        plane_store(z + 2, vals, keep);

        const int gy = y0 + wave;
        const int j = lane & 31;
        if (gy < H && j < K) {
            const float bj = bias ? bias[j] : 0.f;
            cfx_bf16 rv[16];
            if (res) {  // batched residual reads (clamped addresses)
#pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const int row =
                        (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                    const int gx = min(x0 + row, W - 1);
                    rv[r] = res[
                        ((((long long)n * D + z) * H + gy) * W + gx) * K +
                        j];
                }
            }
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int row = (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5);
                const int gx = x0 + row;
                if (gx >= W) continue;
                long long o =
                    ((((long long)n * D + z) * H + gy) * W + gx) * K + j;
                float v = acc[r] + bj;
                if (res) v += (float)rv[r];
                if (do_elu) v = v > 0.f ? v : expm1f(v);
                out[o] = (cfx_bf16)v;
            }
        }
        __syncthreads();
    }
}

}  // namespace

extern "C" int cfx_conv3_ndhwc_bf16(cfx_ctx* ctx, const void* in,
                                    const void* wgt, const float* bias,
                                    const void* residual, void* out, int N,
                                    int D, int H, int W, int C, int K,
                                    int do_elu) {
    if ((C == 36 && K == 36) || (C == 48 && K == 48)) {
        // sliced schedule: (c-half, j-tile) x 4 launches; wgt pack is
        // [27][64][48] zero-padded. Bias/ELU only on the final c-half,
        // which chains through res = its own output.
        dim3 grid((W + 31) / 32, (H + 7) / 8, (unsigned)N);
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
        const int CL2 = C - 32;
        for (int j0 = 0; j0 < K; j0 += 32) {
            hipLaunchKernelGGL((k_conv3_zring_bf16_s<32, 8, 32>), grid,
                               dim3(512), 0, ctx->stream,
                               (const cfx_bf16*)in, (const cfx_bf16*)wgt,
                               nullptr, (const cfx_bf16*)residual,
                               (cfx_bf16*)out, N, D, H, W, C, 32, 0, K, K,
                               j0, 0);
            hipLaunchKernelGGL((k_conv3_zring_bf16_s<16, 8, 32>), grid,
                               dim3(512), 0, ctx->stream,
                               (const cfx_bf16*)in, (const cfx_bf16*)wgt,
                               bias, (const cfx_bf16*)out, (cfx_bf16*)out,
                               N, D, H, W, C, CL2, 32, K, K, j0, do_elu);
        }
        CFX_CHECK(hipGetLastError());
        double flops = 2.0 * 27.0 * C * K * (double)N * D * H * W;
        if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
        return 0;
    }
    if (C != 28 || K != 28) {
        g_err = "cfx_conv3_ndhwc_bf16: only C == K == 28/36/48 instantiated";
        return -1;
    }
    dim3 grid((W + 31) / 32, (H + 7) / 8, (unsigned)N);
    // default 6 = pipelined + alternating accumulators (539.9 TF vs
    // 528.6 for _pl; the r2 ladder tried 4 structural variants — pl2
    // prefetch-double-buffer neutral, wide-swizzled 280 (1 wave/SIMD),
    // alternation +2% — the ring is pinned at ~530-540 TF by something
    // other than staging, LDS conflicts, or the same-acc issue cliff;
    // DESIGN.md §10-r2 item 6)
    static const int use_pl = [] {
        const char* e = getenv("CFX_ZRING_PL");
        return e ? atoi(e) : 13;
    }();
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    static const int mode = [] {
        const char* e = getenv("CFX_BF16_MODE");  // phase ablation (timing)
        return e ? atoi(e) : 0;
    }();
    if (use_pl == 8)       // ablation: deeper LDS prefetch
        hipLaunchKernelGGL((k_conv3_zring_bf16_a<28, 28, 8, 32, 8>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 9)  // ablation: no sched_barriers
        hipLaunchKernelGGL((k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 0>),
                           grid, dim3(512), 0, ctx->stream,
                           (const cfx_bf16*)in, (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 14)  // 4-slot swizzled ring, one barrier per z
        hipLaunchKernelGGL((k_conv3_zring_bf16_q2<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 16)  // deferred epilogue, fences off (interleave)
        hipLaunchKernelGGL(
            (k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 0, 3, 1>), grid,
            dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
            (const cfx_bf16*)wgt, bias, (const cfx_bf16*)residual,
            (cfx_bf16*)out, N, D, H, W, do_elu);
    else if (use_pl == 17)  // deferred epilogue, fences ON (control)
        hipLaunchKernelGGL(
            (k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 1, 3, 1>), grid,
            dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
            (const cfx_bf16*)wgt, bias, (const cfx_bf16*)residual,
            (cfx_bf16*)out, N, D, H, W, do_elu);
    else if (use_pl == 15)  // q3: 4-slot padded ring, half wall + L2 B
        hipLaunchKernelGGL((k_conv3_zring_bf16_q3<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 13)  // transposed epilogue + iteration-ahead staging
        hipLaunchKernelGGL(
            (k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 1, 2, 1>), grid,
            dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
            (const cfx_bf16*)wgt, bias, (const cfx_bf16*)residual,
            (cfx_bf16*)out, N, D, H, W, do_elu);
    else if (use_pl == 12)  // s_memtime phase timing (WRONG results)
        hipLaunchKernelGGL((k_conv3_zring_bf16_tm<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 11)  // transposed wide-store epilogue
        hipLaunchKernelGGL((k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 1, 2>),
                           grid, dim3(512), 0, ctx->stream,
                           (const cfx_bf16*)in, (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 10)  // ablation: cheap epilogue (WRONG results)
        hipLaunchKernelGGL((k_conv3_zring_bf16_a<28, 28, 8, 32, 4, 1, 0>),
                           grid, dim3(512), 0, ctx->stream,
                           (const cfx_bf16*)in, (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 7)
        hipLaunchKernelGGL((k_conv3_zring_bf16_t<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 6)
        hipLaunchKernelGGL((k_conv3_zring_bf16_a<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 4)
        hipLaunchKernelGGL((k_conv3_zring_bf16_w<28, 28, 8, 32>), grid,
                           dim3(256), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 3)
        hipLaunchKernelGGL((k_conv3_zring_bf16_pl2<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl == 2)
        hipLaunchKernelGGL((k_conv3_zring_bf16_q<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (use_pl)
        hipLaunchKernelGGL((k_conv3_zring_bf16_pl<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (mode == 1)
        hipLaunchKernelGGL((k_conv3_zring_bf16<28, 28, 8, 32, 1>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (mode == 2)
        hipLaunchKernelGGL((k_conv3_zring_bf16<28, 28, 8, 32, 2>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else if (mode == 3)
        hipLaunchKernelGGL((k_conv3_zring_bf16<28, 28, 8, 32, 3>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    else
        hipLaunchKernelGGL((k_conv3_zring_bf16<28, 28, 8, 32>), grid,
                           dim3(512), 0, ctx->stream, (const cfx_bf16*)in,
                           (const cfx_bf16*)wgt, bias,
                           (const cfx_bf16*)residual, (cfx_bf16*)out, N, D,
                           H, W, do_elu);
    CFX_CHECK(hipGetLastError());
    double flops = 2.0 * 27.0 * 28 * 28 * (double)N * D * H * W;
    if (prof_end(ctx, e0, CFX_K_CONV, flops)) return -1;
    return 0;
}
