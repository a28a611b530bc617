// Connected-components labeling on gfx950 (threshold + 6/18/26-connectivity
// union-find) — the cc3d replacement for the config-4 operator chain
// (reference flow/flow.py:1803-1829 + chunk/base.py:128-137; cc3d's labels
// are unpinned by any reference test, SURVEY.md §8c, so parity is defined
// against scipy.ndimage.label's partition AND numbering: labels are
// assigned 1..N in raster-scan first-encounter order).
//
// Algorithm: one merge pass of atomic union-find over the backward neighbor
// set (min-index root wins, so each component's root IS its first raster
// voxel), a path-compression pass, then a three-step rank assignment
// (per-chunk root counts -> host exclusive scan of the small count array ->
// wave-ballot local ranks), and a final gather. All buffers are
// caller-owned except the small per-chunk count array (context scratch).
#include <hip/hip_runtime.h>

#include <cstdio>
#include <string>
#include <vector>

// shared context/error plumbing lives in cfx.hip; this file is compiled
// into the same shared object (see csrc/Makefile and build.py)
#include "cfx_internal.h"

namespace {

constexpr int SCAN_CHUNK = 16384;  // elements ranked per workgroup

// backward (already-scanned) neighbor offsets per connectivity
// 6-conn: 3 face neighbors; 18: +6 edge; 26: +4 corner (13 total)
__device__ __constant__ int BWD[13][3] = {
    {0, 0, -1}, {0, -1, 0}, {-1, 0, 0},                    // 6
    {0, -1, -1}, {0, -1, 1}, {-1, 0, -1}, {-1, 0, 1},      // 18 (edges)
    {-1, -1, 0}, {-1, 1, 0},
    {-1, -1, -1}, {-1, -1, 1}, {-1, 1, -1}, {-1, 1, 1},    // 26 (corners)
};

// Every parent access is an agent-scope relaxed atomic (lowered to an
// sc1 load/store that bypasses the per-CU L1): a mid-chain node j can only
// ever READ as "p[j]==j" from a stale init-era L1 line — plain loads
// produced exactly that (~0.7% of voxels compressed to a non-root on real
// volumes; MI355X_MICROARCH.md §inter-workgroup visibility).
__device__ inline unsigned int cc_ld(const unsigned int* p, unsigned int x) {
    return __hip_atomic_load(&p[x], __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_AGENT);
}

__device__ inline void cc_st(unsigned int* p, unsigned int x,
                             unsigned int v) {
    __hip_atomic_store(&p[x], v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// read-only find: used where a halving store could overwrite another
// thread's already-compressed root (k_cc_compress: thread T2 shortcutting
// THROUGH node i with a stale-read mid-chain ancestor would clobber T1's
// final parent[i]=root store — observed as ~1e-4 of voxels labeled from
// unwritten scratch)
__device__ inline unsigned int cc_find_ro(const unsigned int* p,
                                          unsigned int x) {
    unsigned int px = cc_ld(p, x);
    while (px != x) {
        x = px;
        px = cc_ld(p, x);
    }
    return x;
}

__device__ inline unsigned int cc_find(unsigned int* __restrict__ p,
                                       unsigned int x) {
    unsigned int px = cc_ld(p, x);
    while (px != x) {
        unsigned int ppx = cc_ld(p, px);
        cc_st(p, x, ppx);  // path halving (ancestor store, benign race)
        x = ppx;
        px = cc_ld(p, x);
    }
    return x;
}

__device__ inline void cc_union(unsigned int* __restrict__ p,
                                unsigned int a, unsigned int b) {
    while (true) {
        a = cc_find(p, a);
        b = cc_find(p, b);
        if (a == b) return;
        unsigned int lo = a < b ? a : b;
        unsigned int hi = a ^ b ^ lo;
        unsigned int old = atomicCAS(&p[hi], hi, lo);
        if (old == hi) return;
        a = old;
        b = lo;
    }
}

__global__ void k_cc_init(const unsigned char* __restrict__ fg,
                          unsigned int* __restrict__ parent, long long n) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride)
        cc_st(parent, (unsigned int)i,
              fg[i] ? (unsigned int)i : 0xFFFFFFFFu);
}

__global__ void k_cc_merge(const unsigned char* __restrict__ fg,
                           unsigned int* __restrict__ parent, int D, int H,
                           int W, int ndirs) {
    long long n = (long long)D * H * W;
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride) {
        if (!fg[i]) continue;
        int x = (int)(i % W);
        long long t = i / W;
        int y = (int)(t % H);
        int z = (int)(t / H);
        for (int d = 0; d < ndirs; ++d) {
            int nz = z + BWD[d][0];
            int ny = y + BWD[d][1];
            int nx = x + BWD[d][2];
            if (nz < 0 || ny < 0 || ny >= H || nx < 0 || nx >= W) continue;
            long long j = ((long long)nz * H + ny) * W + nx;
            if (fg[j]) cc_union(parent, (unsigned int)i, (unsigned int)j);
        }
    }
}

__global__ void k_cc_compress(const unsigned char* __restrict__ fg,
                              unsigned int* __restrict__ parent,
                              long long n) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride)
        if (fg[i])
            cc_st(parent, (unsigned int)i,
                  cc_find_ro(parent, (unsigned int)i));
}

// one 64-lane wave per SCAN_CHUNK: count roots (parent[i] == i)
__global__ void k_cc_count(const unsigned int* __restrict__ parent,
                           long long n,
                           unsigned int* __restrict__ counts) {
    long long c0 = (long long)blockIdx.x * SCAN_CHUNK;
    if (c0 >= n) return;
    long long c1 = c0 + SCAN_CHUNK < n ? c0 + SCAN_CHUNK : n;
    unsigned int cnt = 0;
    for (long long i = c0 + threadIdx.x; i < c1; i += 64)
        if (cc_ld(parent, (unsigned int)i) == (unsigned int)i) ++cnt;
    for (int off = 32; off > 0; off >>= 1)
        cnt += __shfl_down(cnt, off, 64);
    if (threadIdx.x == 0)
        __hip_atomic_store(&counts[blockIdx.x], cnt, __ATOMIC_RELAXED,
                           __HIP_MEMORY_SCOPE_AGENT);
}

// assign newlabel[root] = chunk_offset + local_rank + 1 via wave ballots
__global__ void k_cc_rank(const unsigned int* __restrict__ parent,
                          long long n,
                          const unsigned int* __restrict__ offsets,
                          unsigned int* __restrict__ newlabel) {
    long long c0 = (long long)blockIdx.x * SCAN_CHUNK;
    if (c0 >= n) return;
    long long c1 = c0 + SCAN_CHUNK < n ? c0 + SCAN_CHUNK : n;
    unsigned int base = __hip_atomic_load(
        &offsets[blockIdx.x], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
    for (long long i0 = c0; i0 < c1; i0 += 64) {
        long long i = i0 + threadIdx.x;
        bool is_root =
            i < c1 &&
            cc_ld(parent, (unsigned int)i) == (unsigned int)i;
        unsigned long long ballot = __ballot(is_root);
        if (is_root) {
            unsigned int before = (unsigned int)__popcll(
                ballot & ((1ull << threadIdx.x) - 1ull));
            __hip_atomic_store(&newlabel[i], base + before + 1,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        base += (unsigned int)__popcll(ballot);
    }
}

__global__ void k_cc_final(const unsigned char* __restrict__ fg,
                           const unsigned int* __restrict__ parent,
                           const unsigned int* __restrict__ newlabel,
                           unsigned int* __restrict__ labels, long long n) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride)
        labels[i] =
            fg[i] ? cc_ld(newlabel, cc_ld(parent, (unsigned int)i)) : 0u;
}

__global__ void k_threshold(const float* __restrict__ in,
                            unsigned char* __restrict__ fg, long long n,
                            float threshold) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride) fg[i] = in[i] > threshold ? 1 : 0;
}

__global__ void k_nonzero_u8(const unsigned char* __restrict__ in,
                             unsigned char* __restrict__ fg, long long n) {
    long long i = blockIdx.x * (long long)blockDim.x + threadIdx.x;
    long long stride = gridDim.x * (long long)blockDim.x;
    for (; i < n; i += stride) fg[i] = in[i] != 0 ? 1 : 0;
}

inline int grid_for(long long work, int threads) {
    long long want = (work + threads - 1) / threads;
    return (int)(want < 8192 ? want : 8192);
}

}  // namespace

extern "C" int cfx_threshold(cfx_ctx* ctx, const float* in,
                             unsigned char* fg, long long n,
                             float threshold) {
    hipLaunchKernelGGL(k_threshold, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, in, fg, n, threshold);
    CFX_CHECK(hipGetLastError());
    return 0;
}

extern "C" int cfx_nonzero_u8(cfx_ctx* ctx, const unsigned char* in,
                              unsigned char* fg, long long n) {
    hipLaunchKernelGGL(k_nonzero_u8, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, in, fg, n);
    CFX_CHECK(hipGetLastError());
    return 0;
}

/* fg: device u8 foreground mask; labels: device u32 out (also used as the
 * union-find parent array); scratch: device u32, same length (root->label
 * table); n_components written on the host after a sync. */
extern "C" int cfx_connected_components(cfx_ctx* ctx,
                                        const unsigned char* fg,
                                        const int dims[3], int connectivity,
                                        unsigned int* labels,
                                        unsigned int* scratch,
                                        long long* n_components) {
    int D = dims[0], H = dims[1], W = dims[2];
    long long n = (long long)D * H * W;
    if (n >= 0xFFFFFFFFll) {
        g_err = "connected_components: volume exceeds u32 indexing";
        return -1;
    }
    int ndirs = connectivity == 6 ? 3 : connectivity == 18 ? 9
                : connectivity == 26 ? 13 : -1;
    if (ndirs < 0) {
        g_err = "connectivity must be 6, 18 or 26";
        return -1;
    }
    hipEvent_t e0;
    if (prof_begin(ctx, &e0)) return -1;
    hipLaunchKernelGGL(k_cc_init, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, fg, labels, n);
    hipLaunchKernelGGL(k_cc_merge, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, fg, labels, D, H, W, ndirs);
    hipLaunchKernelGGL(k_cc_compress, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, fg, labels, n);

    int nchunks = (int)((n + SCAN_CHUNK - 1) / SCAN_CHUNK);
    if (ctx->cc_counts_cap < nchunks) {
        if (ctx->cc_counts) hipFree(ctx->cc_counts);
        CFX_CHECK(hipMalloc(&ctx->cc_counts,
                            (size_t)nchunks * sizeof(unsigned int)));
        ctx->cc_counts_cap = nchunks;
    }
    hipLaunchKernelGGL(k_cc_count, dim3(nchunks), dim3(64), 0, ctx->stream,
                       labels, n, ctx->cc_counts);
    // exclusive scan of the (small) per-chunk counts on the host
    std::vector<unsigned int> counts(nchunks);
    CFX_CHECK(hipMemcpyAsync(counts.data(), ctx->cc_counts,
                             (size_t)nchunks * sizeof(unsigned int),
                             hipMemcpyDeviceToHost, ctx->stream));
    CFX_CHECK(hipStreamSynchronize(ctx->stream));
    unsigned long long total = 0;
    for (int i = 0; i < nchunks; ++i) {
        unsigned int c = counts[i];
        counts[i] = (unsigned int)total;
        total += c;
    }
    *n_components = (long long)total;
    CFX_CHECK(hipMemcpyAsync(ctx->cc_counts, counts.data(),
                             (size_t)nchunks * sizeof(unsigned int),
                             hipMemcpyHostToDevice, ctx->stream));
    hipLaunchKernelGGL(k_cc_rank, dim3(nchunks), dim3(64), 0, ctx->stream,
                       labels, n, ctx->cc_counts, scratch);
    hipLaunchKernelGGL(k_cc_final, dim3(grid_for(n, 256)), dim3(256), 0,
                       ctx->stream, fg, labels, scratch, labels, n);
    CFX_CHECK(hipGetLastError());
    // algorithmic bytes: fg reads x3 + parent RMW passes + final gather
    double bytes = (double)n * (3.0 * 1.0 + 4.0 * 6.0);
    if (prof_end(ctx, e0, CFX_K_CC, bytes)) return -1;
    return 0;
}
