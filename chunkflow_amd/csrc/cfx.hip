// chunkflow_amd hot-path kernels for MI355X (gfx950, CDNA4).
//
// Every kernel here is HBM-bandwidth-bound streaming work over f32 volumes
// laid out C-order z-y-x (x fastest): coalesced float4 lines along x,
// 256-thread workgroups (4 waves of 64), grid-stride over (channel, z, y)
// "lines" so a 512^3 chunk launches tens of thousands of workgroups (>>256
// CUs, fills all 8 XCDs). No LDS is needed: there is zero reuse beyond the
// overlap halo, which the line decomposition already exploits via L2.
// Compiled with -ffp-contract=off so the blend's multiply-then-add matches
// the reference's numpy arithmetic (two roundings, not one fused fma).
//
// Semantics follow seung-lab/chunkflow v1.1.7 (citations in
// include/chunkflow_amd.h and per kernel below).
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "cfx_internal.h"

#define CFX_VERSION 1

// ---------------------------------------------------------------------------
// error handling
// ---------------------------------------------------------------------------
thread_local std::string g_err;

extern "C" const char* cfx_last_error(void) { return g_err.c_str(); }
extern "C" int cfx_version(void) { return CFX_VERSION; }

// ---------------------------------------------------------------------------
// context (struct + CFX_CHECK in cfx_internal.h, shared with cc.hip)
// ---------------------------------------------------------------------------
extern "C" cfx_ctx* cfx_init(int device) {
    hipError_t e = hipSetDevice(device);
    if (e != hipSuccess) {
        g_err = std::string("hipSetDevice: ") + hipGetErrorString(e);
        return nullptr;
    }
    cfx_ctx* ctx = new cfx_ctx();
    ctx->device = device;
    if (hipMalloc(&ctx->dev_max, sizeof(unsigned int)) != hipSuccess) {
        g_err = "hipMalloc scratch failed";
        delete ctx;
        return nullptr;
    }
    return ctx;
}

extern "C" void cfx_destroy(cfx_ctx* ctx) {
    if (!ctx) return;
    for (auto& p : ctx->pending) {
        hipEventDestroy(p.e0);
        hipEventDestroy(p.e1);
    }
    if (ctx->dev_max) hipFree(ctx->dev_max);
    if (ctx->cc_counts) hipFree(ctx->cc_counts);
    delete ctx;
}

extern "C" int cfx_set_stream(cfx_ctx* ctx, void* hip_stream) {
    ctx->stream = reinterpret_cast<hipStream_t>(hip_stream);
    return 0;
}

extern "C" int cfx_sync(cfx_ctx* ctx) {
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    return 0;
}

// profiling helpers ---------------------------------------------------------
int prof_begin(cfx_ctx* ctx, hipEvent_t* e0) {
    if (!ctx->profile) return 0;
    CFX_CHECK(hipEventCreate(e0));
    CFX_CHECK(hipEventRecord(*e0, ctx->stream));
    return 0;
}

int prof_end(cfx_ctx* ctx, hipEvent_t e0, int kid, double bytes) {
    if (!ctx->profile) return 0;
    hipEvent_t e1;
    CFX_CHECK(hipEventCreate(&e1));
    CFX_CHECK(hipEventRecord(e1, ctx->stream));
    ctx->pending.push_back({e0, e1, kid, bytes});
    return 0;
}

extern "C" int cfx_profile_enable(cfx_ctx* ctx, int enable) {
    ctx->profile = enable != 0;
    return 0;
}

extern "C" int cfx_profile_reset(cfx_ctx* ctx) {
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    for (auto& p : ctx->pending) {
        hipEventDestroy(p.e0);
        hipEventDestroy(p.e1);
    }
    ctx->pending.clear();
    memset(ctx->prof_count, 0, sizeof(ctx->prof_count));
    memset(ctx->prof_ms, 0, sizeof(ctx->prof_ms));
    memset(ctx->prof_bytes, 0, sizeof(ctx->prof_bytes));
    return 0;
}

extern "C" int cfx_profile_get(cfx_ctx* ctx, int kernel_id,
                               unsigned long long* count, double* total_ms,
                               double* bytes) {
    if (kernel_id < 0 || kernel_id >= CFX_K_COUNT) {
        g_err = "bad kernel id";
        return -1;
    }
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    for (auto& p : ctx->pending) {
        float ms = 0.f;
        CFX_CHECK(hipEventElapsedTime(&ms, p.e0, p.e1));
        ctx->prof_count[p.kid] += 1;
        ctx->prof_ms[p.kid] += ms;
        ctx->prof_bytes[p.kid] += p.bytes;
        hipEventDestroy(p.e0);
        hipEventDestroy(p.e1);
    }
    ctx->pending.clear();
    *count = ctx->prof_count[kernel_id];
    *total_ms = ctx->prof_ms[kernel_id];
    *bytes = ctx->prof_bytes[kernel_id];
    return 0;
}

// ---------------------------------------------------------------------------
// elementwise u8 -> f32 kernels (normalize-intensity, int->unit cast)
// ---------------------------------------------------------------------------
// out = (float)in * a + b; vectorized uchar4 -> float4. a/b are chosen by the
// host so the arithmetic matches the reference exactly:
//   normalize-intensity: x/127.5 - 1   (flow.py:1664-1666)
//   unit cast:           x/255        (inferencer.py:395-399)
// f32 division by a constant and multiplication by its reciprocal differ in
// the last ulp, so we divide, like numpy does.
__global__ void k_u8_to_f32(const unsigned char* __restrict__ in,
                            float* __restrict__ out, long long n4,
                            float divisor, float bias) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    const uchar4* in4 = reinterpret_cast<const uchar4*>(in);
    float4* out4 = reinterpret_cast<float4*>(out);
    for (; i < n4; i += stride) {
        uchar4 v = in4[i];
        float4 r;
        r.x = (float)v.x / divisor + bias;
        r.y = (float)v.y / divisor + bias;
        r.z = (float)v.z / divisor + bias;
        r.w = (float)v.w / divisor + bias;
        out4[i] = r;
    }
}

__global__ void k_u8_to_f32_tail(const unsigned char* __restrict__ in,
                                 float* __restrict__ out, long long start,
                                 long long n, float divisor, float bias) {
    long long i = start + blockIdx.x * (long long)blockDim.x + threadIdx.x;
    if (i < n) out[i] = (float)in[i] / divisor + bias;
}

static int u8_to_f32(cfx_ctx* ctx, const unsigned char* in, float* out,
                     long long n, float divisor, float bias, int kid) {
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    long long n4 = n / 4;
    if (n4 > 0) {
        int threads = 256;
        long long want = (n4 + threads - 1) / threads;
        int blocks = (int)std::min<long long>(want, 8192);
        hipLaunchKernelGGL(k_u8_to_f32, dim3(blocks), dim3(threads), 0,
                           ctx->stream, in, out, n4, divisor, bias);
    }
    if (n % 4) {
        hipLaunchKernelGGL(k_u8_to_f32_tail, dim3(1), dim3(256), 0,
                           ctx->stream, in, out, n4 * 4, n, divisor, bias);
    }
    CFX_CHECK(hipGetLastError());
    if (prof_end(ctx, e0, kid, (double)n * 5.0)) return -1;  // 1B read + 4B write
    return 0;
}

extern "C" int cfx_normalize_intensity(cfx_ctx* ctx, const unsigned char* in,
                                       float* out, long long n) {
    return u8_to_f32(ctx, in, out, n, 127.5f, -1.0f, CFX_K_NORMALIZE);
}

extern "C" int cfx_cast_u8_f32_div(cfx_ctx* ctx, const unsigned char* in,
                                   float* out, long long n, float divisor) {
    return u8_to_f32(ctx, in, out, n, divisor, 0.0f, CFX_K_CAST);
}

// ---------------------------------------------------------------------------
// patch extraction: gather windows of the f32 chunk into the batch buffer
// ---------------------------------------------------------------------------
// One launch per batch. Block b of the grid walks "lines" (patch, channel,
// z, y); threads stride x. Patch starts ride in kernarg (<=64 per launch).
struct ExtractArgs {
    int n;                 // number of patches this launch
    int starts[64 * 3];    // chunk-local (z, y, x) starts
};

// one wave per x-line, G=4 lines batched per wave iteration (k_blend note)
template <bool VEC>
__global__ void k_extract(const float* __restrict__ chunk, int C, int D,
                          int H, int W, ExtractArgs args, int pz, int py,
                          int px, float* __restrict__ out) {
    constexpr int G = 2;  // same sweet spot as the blend (G=2 beat G=4)
    long long n_lines = (long long)args.n * C * pz * py;
    long long n_groups = (n_lines + G - 1) / G;
    long long stride = (long long)gridDim.x * blockDim.y;
    const int px4 = px / 4;
    for (long long g = (long long)blockIdx.x * blockDim.y + threadIdx.y;
         g < n_groups; g += stride) {
        long long line0 = g * G;
        int nl = (int)(n_lines - line0 < G ? n_lines - line0 : G);
        const float* src[G];
        float* dst[G];
#pragma unroll
        for (int j = 0; j < G; ++j) {
            long long line = line0 + (j < nl ? j : 0);
            int y = (int)(line % py);
            long long t = line / py;
            int z = (int)(t % pz);
            t /= pz;
            int c = (int)(t % C);
            int p = (int)(t / C);
            int z0 = args.starts[p * 3 + 0];
            int y0 = args.starts[p * 3 + 1];
            int x0 = args.starts[p * 3 + 2];
            src[j] = chunk +
                (((long long)c * D + z0 + z) * H + y0 + y) * W + x0;
            dst[j] = out +
                ((((long long)p * C + c) * pz + z) * py + y) * px;
        }
        if (VEC) {
            for (int x = threadIdx.x; x < px4; x += 64) {
                float4 v[G];
#pragma unroll
                for (int j = 0; j < G; ++j)
                    if (j < nl)
                        v[j] = reinterpret_cast<const float4*>(src[j])[x];
#pragma unroll
                for (int j = 0; j < G; ++j)
                    if (j < nl)
                        reinterpret_cast<float4*>(dst[j])[x] = v[j];
            }
        } else {
            for (int x = threadIdx.x; x < px; x += 64) {
#pragma unroll
                for (int j = 0; j < G; ++j)
                    if (j < nl) dst[j][x] = src[j][x];
            }
        }
    }
}

extern "C" int cfx_extract_patches(cfx_ctx* ctx, const float* chunk,
                                   int channels, const int chunk_dims[3],
                                   const int* starts_zyx, int n_patches,
                                   const int patch_size[3], float* out) {
    int D = chunk_dims[0], H = chunk_dims[1], W = chunk_dims[2];
    int pz = patch_size[0], py = patch_size[1], px = patch_size[2];
    long long pvox = (long long)pz * py * px;
    for (int base = 0; base < n_patches; base += 64) {
        int n = std::min(64, n_patches - base);
        ExtractArgs args;
        args.n = n;
        bool vec = (W % 4 == 0) && (px % 4 == 0);
        for (int i = 0; i < n; ++i) {
            args.starts[i * 3 + 0] = starts_zyx[(base + i) * 3 + 0];
            args.starts[i * 3 + 1] = starts_zyx[(base + i) * 3 + 1];
            args.starts[i * 3 + 2] = starts_zyx[(base + i) * 3 + 2];
            vec = vec && (args.starts[i * 3 + 2] % 4 == 0);
        }
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
        long long n_lines = (long long)n * channels * pz * py;
        long long n_groups = (n_lines + 1) / 2;  // G=2
        int blocks = (int)std::min<long long>((n_groups + 3) / 4, 8192);
        float* dst = out + (long long)base * channels * pvox;
        if (vec)
            hipLaunchKernelGGL(k_extract<true>, dim3(blocks), dim3(64, 4),
                               0, ctx->stream, chunk, channels, D, H, W,
                               args, pz, py, px, dst);
        else
            hipLaunchKernelGGL(k_extract<false>, dim3(blocks), dim3(64, 4),
                               0, ctx->stream, chunk, channels, D, H, W,
                               args, pz, py, px, dst);
        CFX_CHECK(hipGetLastError());
        double bytes = (double)n * channels * pvox * 8.0;  // read + write f32
        if (prof_end(ctx, e0, CFX_K_EXTRACT, bytes)) return -1;
    }
    return 0;
}

// ---------------------------------------------------------------------------
// blend-accumulate: out[c, region] += patch[c, region'] * mask[region']
// ---------------------------------------------------------------------------
// The highest-traffic kernel of the whole path (~13.6 GB per
// 512^3 chunk). One launch per patch (keeps the accumulate deterministic and
// atomics-free: within a launch each output voxel is written exactly once;
// patches overlap only ACROSS launches, which the stream serializes in the
// reference's own blend order). Lines along x, float4 when aligned.
// One WAVE (64 lanes) per x-line, 4 waves per {64,4} block, and each wave
// iteration batches G=4 lines: the lane issues all 4 lines' loads (up to 12
// independent float4 loads in flight) before any accumulate, which is what
// hides HBM latency here — a single float4 per lane per line measured only
// ~44% of peak (latency-bound), the flat 256-thread-per-line form ~35%
// (3/4 idle lanes).
template <bool VEC, bool MASKED>
__global__ void k_blend(float* __restrict__ out, int OD, int OH, int OW,
                        const float* __restrict__ patch, int PD, int PH,
                        int PW, const float* __restrict__ mask, int C,
                        int dz0, int dy0, int dx0,   // region start in out
                        int pz0, int py0, int px0,   // region start in patch
                        int rz, int ry, int rx) {
    constexpr int G = 4;  // lines batched per wave iteration
    long long n_lines = (long long)C * rz * ry;
    long long n_groups = (n_lines + G - 1) / G;
    long long stride = (long long)gridDim.x * blockDim.y;
    const int rx4 = rx / 4;
    for (long long g = (long long)blockIdx.x * blockDim.y + threadIdx.y;
         g < n_groups; g += stride) {
        long long line0 = g * G;
        int nl = (int)(n_lines - line0 < G ? n_lines - line0 : G);
        float* o[G];
        const float* p[G];
        const float* m[G];
#pragma unroll
        for (int j = 0; j < G; ++j) {
            long long line = line0 + (j < nl ? j : 0);
            int y = (int)(line % ry);
            long long t = line / ry;
            int z = (int)(t % rz);
            int c = (int)(t / rz);
            o[j] = out +
                (((long long)c * OD + dz0 + z) * OH + dy0 + y) * OW + dx0;
            p[j] = patch +
                (((long long)c * PD + pz0 + z) * PH + py0 + y) * PW + px0;
            if (MASKED)
                m[j] = mask +
                    (((long long)(pz0 + z)) * PH + py0 + y) * PW + px0;
        }
        if (VEC) {
            for (int x = threadIdx.x; x < rx4; x += 64) {
                float4 ov[G], pv[G], mv[G];
#pragma unroll
                for (int j = 0; j < G; ++j) {
                    if (j < nl) {
                        ov[j] = reinterpret_cast<float4*>(o[j])[x];
                        pv[j] = reinterpret_cast<const float4*>(p[j])[x];
                        if (MASKED)
                            mv[j] = reinterpret_cast<const float4*>(m[j])[x];
                    }
                }
#pragma unroll
                for (int j = 0; j < G; ++j) {
                    if (j < nl) {
                        if (MASKED) {
                            ov[j].x += pv[j].x * mv[j].x;
                            ov[j].y += pv[j].y * mv[j].y;
                            ov[j].z += pv[j].z * mv[j].z;
                            ov[j].w += pv[j].w * mv[j].w;
                        } else {
                            ov[j].x += pv[j].x;
                            ov[j].y += pv[j].y;
                            ov[j].z += pv[j].z;
                            ov[j].w += pv[j].w;
                        }
                        reinterpret_cast<float4*>(o[j])[x] = ov[j];
                    }
                }
            }
        } else {
            for (int x = threadIdx.x; x < rx; x += 64) {
#pragma unroll
                for (int j = 0; j < G; ++j)
                    if (j < nl)
                        o[j][x] += MASKED ? p[j][x] * m[j][x] : p[j][x];
            }
        }
    }
}

// ---------------------------------------------------------------------------
// batched blend: many DISJOINT patches in one launch
// ---------------------------------------------------------------------------
// The per-patch launch (~34-52 MB) pays a ~1.5-2 us dependent-kernel
// boundary per patch and leaves each wave a single line-group of work.
// Overlapping patches must stay ordered (f32 accumulation order = the
// reference's sequential blend up to f32 commutativity), and patches whose
// CLIPPED output regions are disjoint share one launch: the host groups
// patches with a first-fit coloring (chunkflow_amd/grouping.py), so each
// launch writes every output voxel at most once — no atomics; numerics note
// in grouping.py (reordering bounded by f32 associativity ulps).
struct BlendBatchArgs {
    int n;                  // patches in this launch (<= 32)
    int d0[32 * 3];         // clipped region start in out
    int p0[32 * 3];         // clipped region start in patch
    int r[32 * 3];          // clipped region size
    int pidx[32];           // batch index into the patch buffer
    int line_start[33];     // prefix sum of C*rz*ry per patch
};

template <int G, bool VEC, bool MASKED, bool NT>
__global__ void k_blend_batch(float* __restrict__ out, int OD, int OH,
                              int OW, const float* __restrict__ patch,
                              int PD, int PH, int PW,
                              const float* __restrict__ mask, int C,
                              BlendBatchArgs a, int rxmax) {
    long long n_lines = a.line_start[a.n];
    long long n_groups = (n_lines + G - 1) / G;
    long long stride = (long long)gridDim.x * blockDim.y;
    for (long long g = (long long)blockIdx.x * blockDim.y + threadIdx.y;
         g < n_groups; g += stride) {
        long long line0 = g * G;
        int nl = (int)(n_lines - line0 < G ? n_lines - line0 : G);
        float* o[G];
        const float* p[G];
        const float* m[G];
        int rx[G];
#pragma unroll
        for (int j = 0; j < G; ++j) {
            int line = (int)(line0 + (j < nl ? j : 0));
            int k = 0;
            while (line >= a.line_start[k + 1]) ++k;  // n <= 32, wave-uniform
            int local = line - a.line_start[k];
            int ry = a.r[k * 3 + 1];
            int rz = a.r[k * 3 + 0];
            int y = local % ry;
            int t = local / ry;
            int z = t % rz;
            int c = t / rz;
            rx[j] = a.r[k * 3 + 2];
            o[j] = out + (((long long)c * OD + a.d0[k * 3] + z) * OH +
                          a.d0[k * 3 + 1] + y) * OW + a.d0[k * 3 + 2];
            p[j] = patch + ((((long long)a.pidx[k] * C + c) * PD +
                             a.p0[k * 3] + z) * PH +
                            a.p0[k * 3 + 1] + y) * PW + a.p0[k * 3 + 2];
            if (MASKED)
                m[j] = mask + (((long long)(a.p0[k * 3] + z)) * PH +
                               a.p0[k * 3 + 1] + y) * PW + a.p0[k * 3 + 2];
        }
        if (VEC) {
            for (int x = threadIdx.x; x < rxmax / 4; x += 64) {
                float4 ov[G], pv[G], mv[G];
#pragma unroll
                for (int j = 0; j < G; ++j) {
                    if (j < nl && x < rx[j] / 4) {
                        ov[j] = reinterpret_cast<float4*>(o[j])[x];
                        if (NT) {
                            typedef float f4v
                                __attribute__((ext_vector_type(4)));
                            f4v t = __builtin_nontemporal_load(
                                reinterpret_cast<const f4v*>(p[j]) + x);
                            pv[j] = *reinterpret_cast<float4*>(&t);
                        } else {
                            pv[j] =
                                reinterpret_cast<const float4*>(p[j])[x];
                        }
                        if (MASKED)
                            mv[j] = reinterpret_cast<const float4*>(m[j])[x];
                    }
                }
#pragma unroll
                for (int j = 0; j < G; ++j) {
                    if (j < nl && x < rx[j] / 4) {
                        if (MASKED) {
                            ov[j].x += pv[j].x * mv[j].x;
                            ov[j].y += pv[j].y * mv[j].y;
                            ov[j].z += pv[j].z * mv[j].z;
                            ov[j].w += pv[j].w * mv[j].w;
                        } else {
                            ov[j].x += pv[j].x;
                            ov[j].y += pv[j].y;
                            ov[j].z += pv[j].z;
                            ov[j].w += pv[j].w;
                        }
                        reinterpret_cast<float4*>(o[j])[x] = ov[j];
                    }
                }
            }
        } else {
            for (int x = threadIdx.x; x < rxmax; x += 64) {
#pragma unroll
                for (int j = 0; j < G; ++j)
                    if (j < nl && x < rx[j])
                        o[j][x] += MASKED ? p[j][x] * m[j][x] : p[j][x];
            }
        }
    }
}

// clip the patch span against the output bounds (Chunk.blend semantics,
// chunk/base.py:796-807); returns false when the intersection is empty
static bool clip_region(const int out_dims[3], const int patch_dims[3],
                        const int offset[3], int* d0, int* p0, int* r) {
    for (int a = 0; a < 3; ++a) {
        int lo = offset[a] > 0 ? offset[a] : 0;
        int hi = offset[a] + patch_dims[a] < out_dims[a]
                     ? offset[a] + patch_dims[a]
                     : out_dims[a];
        if (hi <= lo) return false;
        d0[a] = lo;
        p0[a] = lo - offset[a];
        r[a] = hi - lo;
    }
    return true;
}

static int blend_one(cfx_ctx* ctx, float* out, int C, const int out_dims[3],
                     const float* patch, const int patch_dims[3],
                     const int offset[3], const float* mask) {
    int d0[3], p0[3], r[3];
    if (!clip_region(out_dims, patch_dims, offset, d0, p0, r)) return 0;
    bool vec = (out_dims[2] % 4 == 0) && (patch_dims[2] % 4 == 0) &&
               (d0[2] % 4 == 0) && (p0[2] % 4 == 0) && (r[2] % 4 == 0);
    long long n_lines = (long long)C * r[0] * r[1];
    long long n_groups = (n_lines + 3) / 4;          // G=4 lines per group
    int blocks = (int)std::min<long long>((n_groups + 3) / 4, 8192);
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
#define CFX_LAUNCH_BLEND(V, M)                                              \
    hipLaunchKernelGGL((k_blend<V, M>), dim3(blocks), dim3(64, 4), 0,       \
                       ctx->stream, out, out_dims[0], out_dims[1],          \
                       out_dims[2], patch, patch_dims[0], patch_dims[1],    \
                       patch_dims[2], mask, C, d0[0], d0[1], d0[2], p0[0],  \
                       p0[1], p0[2], r[0], r[1], r[2])
    if (vec && mask) CFX_LAUNCH_BLEND(true, true);
    else if (vec) CFX_LAUNCH_BLEND(true, false);
    else if (mask) CFX_LAUNCH_BLEND(false, true);
    else CFX_LAUNCH_BLEND(false, false);
#undef CFX_LAUNCH_BLEND
    CFX_CHECK(hipGetLastError());
    double rv = (double)r[0] * r[1] * r[2];
    // algorithmic bytes: out read+write + patch read (per channel) + mask read
    double bytes = rv * C * 12.0 + (mask ? rv * 4.0 : 0.0);
    if (prof_end(ctx, e0, CFX_K_BLEND, bytes)) return -1;
    return 0;
}

extern "C" int cfx_blend_accumulate(cfx_ctx* ctx, float* out, int channels,
                                    const int out_dims[3], const float* patch,
                                    const int patch_dims[3],
                                    const int offset_zyx[3],
                                    const float* mask) {
    return blend_one(ctx, out, channels, out_dims, patch, patch_dims,
                     offset_zyx, mask);
}

// items: n * 4 ints (batch_index, oz, oy, ox). The CALLER guarantees the
// clipped output regions are pairwise disjoint (first-fit grouping in the
// host, chunkflow_amd/grouping.py); patches clipped to nothing are skipped.
extern "C" int cfx_blend_batch(cfx_ctx* ctx, float* out, int channels,
                               const int out_dims[3], const float* patch,
                               const int patch_dims[3], const int* items,
                               int n, const float* mask) {
    for (int base = 0; base < n; base += 32) {
        int nb = std::min(32, n - base);
        BlendBatchArgs a;
        a.n = 0;
        a.line_start[0] = 0;
        bool vec = (out_dims[2] % 4 == 0) && (patch_dims[2] % 4 == 0);
        int rxmax = 0;
        double bytes = 0.0;
        for (int i = 0; i < nb; ++i) {
            const int* it = items + (base + i) * 4;
            int off[3] = {it[1], it[2], it[3]};
            int d0[3], p0[3], r[3];
            if (!clip_region(out_dims, patch_dims, off, d0, p0, r)) continue;
            int k = a.n;
            for (int ax = 0; ax < 3; ++ax) {
                a.d0[k * 3 + ax] = d0[ax];
                a.p0[k * 3 + ax] = p0[ax];
                a.r[k * 3 + ax] = r[ax];
            }
            a.pidx[k] = it[0];
            a.line_start[k + 1] =
                a.line_start[k] + channels * r[0] * r[1];
            vec = vec && (d0[2] % 4 == 0) && (p0[2] % 4 == 0) &&
                  (r[2] % 4 == 0);
            rxmax = std::max(rxmax, r[2]);
            double rv = (double)r[0] * r[1] * r[2];
            bytes += rv * channels * 12.0 + (mask ? rv * 4.0 : 0.0);
            a.n = k + 1;
        }
        if (a.n == 0) continue;
        // tuning knobs (measured via tools/blend_tune.py):
        //   CFX_BLEND_G  = lines batched per wave iteration (2/4/8)
        //   CFX_BLEND_NT = nontemporal loads for the read-once patch stream
        static int env_g = [] {
            const char* s = getenv("CFX_BLEND_G");
            int v = s ? atoi(s) : 2;  // measured: G=2 5.8 TB/s, G=4 4.0, G=8 1.1
            return (v == 2 || v == 4 || v == 8) ? v : 2;
        }();
        static bool env_nt = [] {
            const char* s = getenv("CFX_BLEND_NT");
            return s && atoi(s) != 0;
        }();
        long long n_lines = a.line_start[a.n];
        long long n_groups = (n_lines + env_g - 1) / env_g;
        int blocks = (int)std::min<long long>((n_groups + 3) / 4, 8192);
        hipEvent_t e0;
        if (prof_begin(ctx, &e0)) return -1;
#define CFX_LAUNCH_BB(G, V, M, NT)                                           \
    hipLaunchKernelGGL((k_blend_batch<G, V, M, NT>), dim3(blocks),           \
                       dim3(64, 4), 0, ctx->stream, out, out_dims[0],        \
                       out_dims[1], out_dims[2], patch, patch_dims[0],       \
                       patch_dims[1], patch_dims[2], mask, channels, a,      \
                       rxmax)
#define CFX_DISPATCH_G(G)                                                    \
    do {                                                                     \
        if (vec && mask && env_nt) CFX_LAUNCH_BB(G, true, true, true);       \
        else if (vec && mask) CFX_LAUNCH_BB(G, true, true, false);           \
        else if (vec && env_nt) CFX_LAUNCH_BB(G, true, false, true);         \
        else if (vec) CFX_LAUNCH_BB(G, true, false, false);                  \
        else if (mask) CFX_LAUNCH_BB(G, false, true, false);                 \
        else CFX_LAUNCH_BB(G, false, false, false);                          \
    } while (0)
        if (env_g == 2) CFX_DISPATCH_G(2);
        else if (env_g == 8) CFX_DISPATCH_G(8);
        else CFX_DISPATCH_G(4);
#undef CFX_DISPATCH_G
#undef CFX_LAUNCH_BB
        CFX_CHECK(hipGetLastError());
        if (prof_end(ctx, e0, CFX_K_BLEND, bytes)) return -1;
    }
    return 0;
}

// ---------------------------------------------------------------------------
// chunk-mask build: zero, blend the patch mask at every offset, reciprocal
// ---------------------------------------------------------------------------
__global__ void k_reciprocal(float* __restrict__ buf, long long n) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride) buf[i] = 1.0f / buf[i];
}

extern "C" int cfx_reciprocal(cfx_ctx* ctx, float* buf, long long n) {
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    int threads = 256;
    long long want = (n + threads - 1) / threads;
    int blocks = (int)std::min<long long>(want, 8192);
    hipLaunchKernelGGL(k_reciprocal, dim3(blocks), dim3(threads), 0,
                       ctx->stream, buf, n);
    CFX_CHECK(hipGetLastError());
    if (prof_end(ctx, e0, CFX_K_RECIPROCAL, (double)n * 8.0)) return -1;
    return 0;
}

extern "C" int cfx_build_chunk_mask(cfx_ctx* ctx, float* mask_out,
                                    const int out_dims[3],
                                    const float* patch_mask,
                                    const int patch_dims[3],
                                    const int* offsets_zyx, int n) {
    long long nvox =
        (long long)out_dims[0] * out_dims[1] * out_dims[2];
    CFX_CHECK(hipMemsetAsync(mask_out, 0, nvox * sizeof(float), ctx->stream));
    for (int i = 0; i < n; ++i) {
        if (blend_one(ctx, mask_out, 1, out_dims, patch_mask, patch_dims,
                      offsets_zyx + i * 3, nullptr))
            return -1;
    }
    return cfx_reciprocal(ctx, mask_out, nvox);
}

// ---------------------------------------------------------------------------
// mask-normalize multiply: out[c, i] *= mask[i]
// ---------------------------------------------------------------------------
// order-preserving bit transform lives below (f32_ord); forward-declare
__host__ __device__ inline unsigned int f32_ord(float f);

// mask-normalize: out[c, i] *= mask[i] for every channel, mask read ONCE
// (channels in the inner loop — a per-channel grid re-reads the 0.54 GB
// chunk mask from HBM, it does not fit the 256 MB L3). U=2 float4s per
// thread per iteration -> 2 mask + 2*C out loads in flight. Optionally
// fuses the reference's "< 1.0001" sanity scan (inferencer.py:463-466)
// via a wave-reduced atomic max, saving a full read pass of the output.
template <bool WITHMAX>
__global__ void k_maskmul(float* __restrict__ out,
                          const float* __restrict__ mask, long long n4,
                          int C, long long cstride4,
                          unsigned int* __restrict__ result) {
    constexpr int U = 2;
    long long i0 = ((long long)blockIdx.x * blockDim.x + threadIdx.x) * U;
    long long stride = (long long)gridDim.x * blockDim.x * U;
    const float4* m4 = reinterpret_cast<const float4*>(mask);
    float4* o4 = reinterpret_cast<float4*>(out);
    float vmax = -INFINITY;
    for (long long i = i0; i < n4; i += stride) {
        float4 mv[U];
        int nu = (int)(n4 - i < U ? n4 - i : U);
#pragma unroll
        for (int u = 0; u < U; ++u)
            if (u < nu) mv[u] = m4[i + u];
        for (int c = 0; c < C; ++c) {
            float4 ov[U];
#pragma unroll
            for (int u = 0; u < U; ++u)
                if (u < nu) ov[u] = o4[c * cstride4 + i + u];
#pragma unroll
            for (int u = 0; u < U; ++u) {
                if (u < nu) {
                    ov[u].x *= mv[u].x;
                    ov[u].y *= mv[u].y;
                    ov[u].z *= mv[u].z;
                    ov[u].w *= mv[u].w;
                    o4[c * cstride4 + i + u] = ov[u];
                    if (WITHMAX) {
                        vmax = fmaxf(vmax, fmaxf(fmaxf(ov[u].x, ov[u].y),
                                                 fmaxf(ov[u].z, ov[u].w)));
                    }
                }
            }
        }
    }
    if (WITHMAX) {
        for (int off = 32; off > 0; off >>= 1)
            vmax = fmaxf(vmax, __shfl_down(vmax, off, 64));
        if ((threadIdx.x & 63) == 0) atomicMax(result, f32_ord(vmax));
    }
}

__global__ void k_maskmul_scalar(float* __restrict__ out,
                                 const float* __restrict__ mask,
                                 long long n, int C) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride)
        for (int c = 0; c < C; ++c) out[c * n + i] *= mask[i];
}

static int multiply_mask_impl(cfx_ctx* ctx, float* out, const float* mask,
                              int channels, long long n_voxels,
                              bool with_max) {
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    int threads = 256;
    if (n_voxels % 4 == 0) {
        long long n4 = n_voxels / 4;
        long long want = (n4 + threads * 2 - 1) / (threads * 2);
        int blocks = (int)std::min<long long>(want, 8192);
        if (with_max)
            hipLaunchKernelGGL(k_maskmul<true>, dim3(blocks), dim3(threads),
                               0, ctx->stream, out, mask, n4, channels, n4,
                               ctx->dev_max);
        else
            hipLaunchKernelGGL(k_maskmul<false>, dim3(blocks), dim3(threads),
                               0, ctx->stream, out, mask, n4, channels, n4,
                               nullptr);
    } else {
        long long want = (n_voxels + threads - 1) / threads;
        int blocks = (int)std::min<long long>(want, 8192);
        hipLaunchKernelGGL(k_maskmul_scalar, dim3(blocks), dim3(threads), 0,
                           ctx->stream, out, mask, n_voxels, channels);
        with_max = false;  // caller falls back to cfx_max
    }
    CFX_CHECK(hipGetLastError());
    double bytes = (double)n_voxels * (channels * 8.0 + 4.0);
    if (prof_end(ctx, e0, CFX_K_MASKMUL, bytes)) return -1;
    return 0;
}

extern "C" int cfx_multiply_mask(cfx_ctx* ctx, float* out, const float* mask,
                                 int channels, long long n_voxels) {
    return multiply_mask_impl(ctx, out, mask, channels, n_voxels, false);
}

/* fused mask-normalize + max scan; returns -2 when the fused max is
 * unavailable (unaligned size) and the caller must cfx_max separately */
extern "C" int cfx_multiply_mask_max(cfx_ctx* ctx, float* out,
                                     const float* mask, int channels,
                                     long long n_voxels, float* host_max) {
    if (n_voxels % 4 != 0) {
        int rc = multiply_mask_impl(ctx, out, mask, channels, n_voxels,
                                    false);
        return rc != 0 ? rc : -2;
    }
    unsigned int init = f32_ord(-INFINITY);
    CFX_CHECK(hipMemcpyAsync(ctx->dev_max, &init, sizeof(init),
                             hipMemcpyHostToDevice, ctx->stream));
    int rc = multiply_mask_impl(ctx, out, mask, channels, n_voxels, true);
    if (rc) return rc;
    unsigned int outv;
    CFX_CHECK(hipMemcpyAsync(&outv, ctx->dev_max, sizeof(outv),
                             hipMemcpyDeviceToHost, ctx->stream));
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    unsigned int u = (outv & 0x80000000u) ? (outv & 0x7fffffffu) : ~outv;
    float f;
    __builtin_memcpy(&f, &u, sizeof(f));
    *host_max = f;
    return 0;
}

// ---------------------------------------------------------------------------
// max reduce (sanity assert: all(out < 1.0001), inferencer.py:463-466)
// ---------------------------------------------------------------------------
// order-preserving bit transform so unsigned atomicMax works for any float
__host__ __device__ inline unsigned int f32_ord(float f) {
    unsigned int u;
    __builtin_memcpy(&u, &f, sizeof(u));
    return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__global__ void k_max(const float* __restrict__ buf, long long n,
                      unsigned int* __restrict__ result) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    float m = -INFINITY;
    for (; i < n; i += stride) m = fmaxf(m, buf[i]);
    // wave reduce (64 lanes), then one atomic per wave
    for (int off = 32; off > 0; off >>= 1)
        m = fmaxf(m, __shfl_down(m, off, 64));
    if ((threadIdx.x & 63) == 0) atomicMax(result, f32_ord(m));
}

extern "C" int cfx_max(cfx_ctx* ctx, const float* buf, long long n,
                       float* host_max) {
    unsigned int init = f32_ord(-INFINITY);
    CFX_CHECK(hipMemcpyAsync(ctx->dev_max, &init, sizeof(init),
                             hipMemcpyHostToDevice, ctx->stream));
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    int threads = 256;
    long long want = (n + threads - 1) / threads;
    int blocks = (int)std::min<long long>(want, 4096);
    hipLaunchKernelGGL(k_max, dim3(blocks), dim3(threads), 0, ctx->stream,
                       buf, n, ctx->dev_max);
    CFX_CHECK(hipGetLastError());
    if (prof_end(ctx, e0, CFX_K_MAX, (double)n * 4.0)) return -1;
    unsigned int out;
    CFX_CHECK(hipMemcpyAsync(&out, ctx->dev_max, sizeof(out),
                             hipMemcpyDeviceToHost, ctx->stream));
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    unsigned int u = (out & 0x80000000u) ? (out & 0x7fffffffu) : ~out;
    *host_max = __builtin_bit_cast(float, u);
    return 0;
}

// ---------------------------------------------------------------------------
// crop-margin: contiguous copy dropping margins (chunk/base.py:691-726)
// ---------------------------------------------------------------------------
// one wave per x-line, 4 lines per block (see k_blend note)
template <bool VEC>
__global__ void k_crop(const float* __restrict__ in, int D, int H, int W,
                       float* __restrict__ out, int C, int oD, int oH, int oW,
                       int m0, int m1, int m2) {
    long long n_lines = (long long)C * oD * oH;
    long long stride = (long long)gridDim.x * blockDim.y;
    for (long long line = (long long)blockIdx.x * blockDim.y + threadIdx.y;
         line < n_lines; line += stride) {
        int y = (int)(line % oH);
        long long t = line / oH;
        int z = (int)(t % oD);
        int c = (int)(t / oD);
        const float* src =
            in + (((long long)c * D + m0 + z) * H + m1 + y) * W + m2;
        float* dst = out + (((long long)c * oD + z) * oH + y) * oW;
        if (VEC) {
            const float4* s4 = reinterpret_cast<const float4*>(src);
            float4* d4 = reinterpret_cast<float4*>(dst);
            for (int x = threadIdx.x; x < oW / 4; x += 64)
                d4[x] = s4[x];
        } else {
            for (int x = threadIdx.x; x < oW; x += 64)
                dst[x] = src[x];
        }
    }
}

extern "C" int cfx_crop_margin(cfx_ctx* ctx, const float* in, float* out,
                               int channels, const int in_dims[3],
                               const int margins[6]) {
    int D = in_dims[0], H = in_dims[1], W = in_dims[2];
    int oD = D - margins[0] - margins[3];
    int oH = H - margins[1] - margins[4];
    int oW = W - margins[2] - margins[5];
    if (oD <= 0 || oH <= 0 || oW <= 0) {
        g_err = "crop_margin: empty output";
        return -1;
    }
    bool vec = (W % 4 == 0) && (oW % 4 == 0) && (margins[2] % 4 == 0);
    long long n_lines = (long long)channels * oD * oH;
    int blocks = (int)std::min<long long>((n_lines + 3) / 4, 8192);
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    if (vec)
        hipLaunchKernelGGL(k_crop<true>, dim3(blocks), dim3(64, 4), 0,
                           ctx->stream, in, D, H, W, out, channels, oD, oH,
                           oW, margins[0], margins[1], margins[2]);
    else
        hipLaunchKernelGGL(k_crop<false>, dim3(blocks), dim3(64, 4), 0,
                           ctx->stream, in, D, H, W, out, channels, oD, oH,
                           oW, margins[0], margins[1], margins[2]);
    CFX_CHECK(hipGetLastError());
    double bytes = (double)channels * oD * oH * oW * 8.0;
    if (prof_end(ctx, e0, CFX_K_CROP, bytes)) return -1;
    return 0;
}

// ---------------------------------------------------------------------------
// myelin mask: out[c] = in[c] * (in[C-1] < threshold)  (chunk/base.py:685-689)
// ---------------------------------------------------------------------------
__global__ void k_myelin(const float* __restrict__ in,
                         float* __restrict__ out, long long n, int cout,
                         float threshold) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    const float* last = in + (long long)cout * n;
    for (; i < n; i += stride) {
        float keep = last[i] < threshold ? 1.0f : 0.0f;
        for (int c = 0; c < cout; ++c)
            out[c * n + i] = in[c * n + i] * keep;
    }
}

extern "C" int cfx_mask_using_last_channel(cfx_ctx* ctx, const float* in,
                                           float* out, int channels,
                                           const int dims[3],
                                           float threshold) {
    long long n = (long long)dims[0] * dims[1] * dims[2];
    int cout = channels - 1;
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    int threads = 256;
    long long want = (n + threads - 1) / threads;
    int blocks = (int)std::min<long long>(want, 8192);
    hipLaunchKernelGGL(k_myelin, dim3(blocks), dim3(threads), 0, ctx->stream,
                       in, out, n, cout, threshold);
    CFX_CHECK(hipGetLastError());
    double bytes = (double)n * (channels + cout) * 4.0;
    if (prof_end(ctx, e0, CFX_K_MYELIN, bytes)) return -1;
    return 0;
}

// ---------------------------------------------------------------------------
// host-side patch mask (f64 pipeline; patch/patch_mask.py:15-68)
// ---------------------------------------------------------------------------
// The product's Python host path computes this with numpy (bit-identical to
// the reference); this C implementation completes the C-ABI for non-Python
// hosts and is tested against the same golden CRCs to <=1e-12 relative.
extern "C" int cfx_make_patch_mask(const int patch_size[3],
                                   const int overlap[3], float* out) {
    const int pz = patch_size[0], py = patch_size[1], px = patch_size[2];
    const long long n = (long long)pz * py * px;
    std::vector<double> bump((size_t)n);
    // bump map on centered grids
    std::vector<double> gx(px), gy(py), gz(pz);
    for (int i = 0; i < px; ++i)
        gx[i] = (i + 1.0) / (px + 1.0) * 2.0 - 1.0;
    for (int i = 0; i < py; ++i)
        gy[i] = (i + 1.0) / (py + 1.0) * 2.0 - 1.0;
    for (int i = 0; i < pz; ++i)
        gz[i] = (i + 1.0) / (pz + 1.0) * 2.0 - 1.0;
    double bmin = 1e300, bmax = -1e300;
    for (int z = 0; z < pz; ++z)
        for (int y = 0; y < py; ++y)
            for (int x = 0; x < px; ++x) {
                double v = std::exp(-1.0 / (1.0 - gx[x] * gx[x]) +
                                    -1.0 / (1.0 - gy[y] * gy[y]) +
                                    -1.0 / (1.0 - gz[z] * gz[z]));
                bump[((long long)z * py + y) * px + x] = v;
                if (v < bmin) bmin = v;
                if (v > bmax) bmax = v;
            }
    // np.interp remap of (bmin, bmax) -> (1, 1e6)
    const double slope = (1e6 - 1.0) / (bmax - bmin);
    for (long long i = 0; i < n; ++i) {
        double v = bump[i];
        if (v <= bmin) v = 1.0;
        else if (v >= bmax) v = 1e6;
        else v = slope * (v - bmin) + 1.0;
        bump[i] = v;
    }
    // 3x3x3 shifted self-accumulation, normalize by the center crop
    const int sz = pz - overlap[0], sy = py - overlap[1], sx = px - overlap[2];
    const int BD = pz + 2 * sz, BH = py + 2 * sy, BW = px + 2 * sx;
    std::vector<double> base((size_t)BD * BH * BW, 0.0);
    for (int nz = 0; nz < 3; ++nz)
        for (int ny = 0; ny < 3; ++ny)
            for (int nx = 0; nx < 3; ++nx)
                for (int z = 0; z < pz; ++z)
                    for (int y = 0; y < py; ++y) {
                        double* b = &base[((long long)(nz * sz + z) * BH +
                                           ny * sy + y) * BW + nx * sx];
                        const double* s = &bump[((long long)z * py + y) * px];
                        for (int x = 0; x < px; ++x) b[x] += s[x];
                    }
    for (int z = 0; z < pz; ++z)
        for (int y = 0; y < py; ++y)
            for (int x = 0; x < px; ++x) {
                long long i = ((long long)z * py + y) * px + x;
                bump[i] /= base[((long long)(sz + z) * BH + sy + y) * BW +
                                sx + x];
            }
    for (long long i = 0; i < n; ++i) out[i] = (float)bump[i];
    return 0;
}
