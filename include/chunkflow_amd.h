/* chunkflow_amd C-ABI — the MI355X (gfx950) hot-path extension beneath the
 * `chunkflow inference` operator surface.
 *
 * Each entry point replaces a reference hot-path function (SURVEY.md §8a/§8b;
 * reference = seung-lab/chunkflow v1.1.7):
 *   cfx_make_patch_mask      <- patch/patch_mask.py:15-68  (host, f64 internally)
 *   cfx_cast_u8_f32_div      <- inferencer.py:395-399      (int chunk -> f32 / dtype_max)
 *   cfx_normalize_intensity  <- flow/flow.py:1650-1669     (u8 -> f32, x/127.5 - 1)
 *   cfx_extract_patches      <- inferencer.py:408-411 + chunk/base.py:761-781
 *   cfx_blend_accumulate     <- inferencer.py:436-455 + chunk/base.py:792-807
 *                               (fuses the engine's patch-mask multiply,
 *                                pytorch.py:113, when mask != NULL)
 *   cfx_build_chunk_mask     <- inferencer.py:294-333
 *   cfx_reciprocal           <- inferencer.py:333
 *   cfx_multiply_mask        <- inferencer.py:460-461 (ufunc chunk/base.py:418-453)
 *   cfx_max                  <- inferencer.py:463-466 (assert < 1.0001)
 *   cfx_crop_margin          <- chunk/base.py:691-726 (flow.py:2053-2084)
 *   cfx_mask_using_last_channel <- chunk/base.py:685-689 (inferencer.py:468-477)
 *
 * Conventions: the caller owns every buffer (device pointers come from
 * torch-ROCm tensors' data_ptr()); layout is contiguous C-order z-y-x with x
 * fastest; all calls are stream-ordered on the context stream (adopt torch's
 * stream via cfx_set_stream); int return = 0 on success, nonzero with
 * cfx_last_error() set; one context per GPU rank, single-threaded per
 * context.
 */
#ifndef CHUNKFLOW_AMD_H
#define CHUNKFLOW_AMD_H

#ifdef __cplusplus
extern "C" {
#endif

typedef struct cfx_ctx cfx_ctx;

/* ---- lifecycle -------------------------------------------------------- */
cfx_ctx* cfx_init(int device);
void     cfx_destroy(cfx_ctx* ctx);
const char* cfx_last_error(void);
int      cfx_version(void);
/* adopt an existing HIP stream (e.g. torch.cuda.current_stream().cuda_stream);
 * stream may be NULL for the legacy default stream */
int cfx_set_stream(cfx_ctx* ctx, void* hip_stream);
int cfx_sync(cfx_ctx* ctx);

/* ---- host precompute --------------------------------------------------- */
/* Wu bump-weight patch mask, float64 pipeline, f32 result (pz*py*px floats).
 * Runs on the HOST; upload the result once per geometry. */
int cfx_make_patch_mask(const int patch_size[3], const int overlap[3],
                        float* out);

/* ---- device kernels (all pointers are DEVICE pointers) ----------------- */
/* out[i] = (float)in[i] / 127.5f - 1.0f */
int cfx_normalize_intensity(cfx_ctx* ctx, const unsigned char* in, float* out,
                            long long n);
/* out[i] = (float)in[i] / divisor  (divisor = integer dtype max) */
int cfx_cast_u8_f32_div(cfx_ctx* ctx, const unsigned char* in, float* out,
                        long long n, float divisor);
/* gather n_patches windows of an f32 chunk (channels, D, H, W) into a batch
 * buffer (n_patches, channels, pz, py, px); starts_zyx is a HOST array of
 * n_patches*3 ints, chunk-local coordinates */
int cfx_extract_patches(cfx_ctx* ctx, const float* chunk, int channels,
                        const int chunk_dims[3], const int* starts_zyx,
                        int n_patches, const int patch_size[3], float* out);
/* out[c, region] += patch[c, region'] * mask[region'], clipped to the output
 * bounds; offset_zyx is the patch start relative to the output buffer origin;
 * mask may be NULL (plain accumulate, e.g. a pre-masked universal plugin) */
int cfx_blend_accumulate(cfx_ctx* ctx, float* out, int channels,
                         const int out_dims[3], const float* patch,
                         const int patch_dims[3], const int offset_zyx[3],
                         const float* mask);
/* many-patch blend in ONE launch: items is a HOST array of n*4 ints
 * (batch_index, oz, oy, ox). The caller must guarantee the clipped output
 * regions are pairwise DISJOINT (see the first-fit grouping in
 * chunkflow_amd/grouping.py) so the accumulate stays atomics-free and
 * bit-ordered; >32 items are split across launches. */
int cfx_blend_batch(cfx_ctx* ctx, float* out, int channels,
                    const int out_dims[3], const float* patch,
                    const int patch_dims[3], const int* items, int n,
                    const float* mask);
/* zero mask_out (out_dims), blend patch_mask at each of n offsets (HOST
 * array, n*3 ints, relative to the output origin), then reciprocal */
int cfx_build_chunk_mask(cfx_ctx* ctx, float* mask_out, const int out_dims[3],
                         const float* patch_mask, const int patch_dims[3],
                         const int* offsets_zyx, int n);
int cfx_reciprocal(cfx_ctx* ctx, float* buf, long long n);
/* out[c*n + i] *= mask[i] for every channel c (mask-normalize) */
int cfx_multiply_mask(cfx_ctx* ctx, float* out, const float* mask,
                      int channels, long long n_voxels);
/* synchronous max over n floats (the <1.0001 sanity assert) */
int cfx_max(cfx_ctx* ctx, const float* buf, long long n, float* host_max);
/* fused mask-normalize + max scan (saves a full output read pass);
 * returns -2 when n_voxels is not float4-aligned — multiply done, call
 * cfx_max yourself */
int cfx_multiply_mask_max(cfx_ctx* ctx, float* out, const float* mask,
                          int channels, long long n_voxels,
                          float* host_max);
/* contiguous copy dropping margins[6] = -z,-y,-x,+z,+y,+x */
int cfx_crop_margin(cfx_ctx* ctx, const float* in, float* out, int channels,
                    const int in_dims[3], const int margins[6]);
/* out[(channels-1), dims] = in[:channels-1] * (in[channels-1] < threshold) */
int cfx_mask_using_last_channel(cfx_ctx* ctx, const float* in, float* out,
                                int channels, const int dims[3],
                                float threshold);

/* ---- connected components (cc3d replacement; config-4 chain) ----------- */
/* fg[i] = in[i] > threshold */
int cfx_threshold(cfx_ctx* ctx, const float* in, unsigned char* fg,
                  long long n, float threshold);
/* fg[i] = in[i] != 0 */
int cfx_nonzero_u8(cfx_ctx* ctx, const unsigned char* in, unsigned char* fg,
                   long long n);
/* 6/18/26-connectivity union-find labeling of a u8 foreground mask.
 * labels (u32, caller-owned) doubles as the union-find parent array;
 * scratch is a second u32 buffer of the same length. Labels are 1..N in
 * raster-scan first-encounter order (scipy.ndimage.label-compatible);
 * background is 0. */
int cfx_connected_components(cfx_ctx* ctx, const unsigned char* fg,
                             const int dims[3], int connectivity,
                             unsigned int* labels, unsigned int* scratch,
                             long long* n_components);

/* ---- image normalization (normalize-contrast operator) ----------------- */
/* zero + accumulate nsec 256-bin u32 histograms, one per contiguous
 * n_per_sec-voxel section of a u8 volume */
int cfx_hist_u8(cfx_ctx* ctx, const unsigned char* in, long long n_per_sec,
                int nsec, unsigned int* hist);
/* in-place per-section LUT gather: buf[i] = lut[sec][buf[i]] */
int cfx_lut_apply_u8(cfx_ctx* ctx, unsigned char* buf, long long n_per_sec,
                     int nsec, const unsigned char* lut);

/* ---- hand-written MFMA convolution (RSUNet ResBlock 3x3x3) -------------- */
/* NDHWC f32, stride 1, pad 1, C == K in {28, 36, 48, 64}; wgt layout
 * (27, C, K) with tap = ((dz+1)*3 + (dy+1))*3 + (dx+1); bias may be NULL;
 * residual (same layout as out) may be NULL; do_elu applies ELU(alpha=1)
 * after bias/residual. profile bytes field carries FLOPs for this id. */
int cfx_conv3_ndhwc(cfx_ctx* ctx, const float* in, const float* wgt,
                    const float* bias, const float* residual, float* out,
                    int N, int D, int H, int W, int C, int K, int do_elu);
/* the persistent-z ring variant (weights LDS-resident, one input plane
 * staged per z; C == K == 28 instantiated) */
int cfx_conv3_ndhwc_zring(cfx_ctx* ctx, const float* in, const float* wgt,
                          const float* bias, const float* residual,
                          float* out, int N, int D, int H, int W, int C,
                          int K, int do_elu);
/* bf16 persistent-z ring (v_mfma_f32_32x32x16_bf16; C == K in
 * {28, 36, 48}): in/out/residual are bf16 NDHWC, wgt is a bf16 zero-padded
 * [tap][j][c] pack — (27, 32, 32) for C == 28, (27, 64, 48) for 36/48
 * (those run as four c-half x j-tile launches; partial sums round through
 * bf16 between the halves) — bias stays f32; epilogue math in f32 */
int cfx_conv3_ndhwc_bf16(cfx_ctx* ctx, const void* in, const void* wgt,
                         const float* bias, const void* residual,
                         void* out, int N, int D, int H, int W, int C,
                         int K, int do_elu);

/* ---- up/down-sampling convs (RSUNet (1,2,2)-kernel, (1,2,2)-stride) ----- */
/* ConvTranspose3d: out (N,D,2H,2W,K) from in (N,D,H,W,C), NDHWC; wgt
 * packed [parity q=(py<<1)|px][C][K] in the compute dtype; bias f32 or
 * NULL; is_bf16 selects bf16 vs f32 tensors. HBM-streaming VALU kernel. */
int cfx_upconv_2x2(cfx_ctx* ctx, const void* in, const void* wgt,
                   const float* bias, void* out, int N, int D, int H,
                   int W, int C, int K, int is_bf16);
/* strided Conv3d (downsample): out (N,D,H/2,W/2,K) from in (N,D,H,W,C);
 * same wgt pack/convention as cfx_upconv_2x2 */
int cfx_downconv_2x2(cfx_ctx* ctx, const void* in, const void* wgt,
                     const float* bias, void* out, int N, int D, int H,
                     int W, int C, int K, int is_bf16);
/* single-channel (1,5,5) conv, pad (0,2,2) — RSUNet conv_in; in
 * (N,D,H,W,1), out (N,D,H,W,K) NDHWC; wgt [K][25] row-major (dy*5+dx) in
 * the compute dtype; K <= 32 */
int cfx_conv155_c1(cfx_ctx* ctx, const void* in, const void* wgt,
                   const float* bias, void* out, int N, int D, int H,
                   int W, int K, int is_bf16);
/* few-output-channel (1,5,5) conv, pad (0,2,2) — RSUNet conv_out; in
 * (N,D,H,W,C), out (N,D,H,W,K) NDHWC; wgt [K][25][C] (k, dy*5+dx, c) in
 * the compute dtype; C == 28, K == 3 instantiated */
int cfx_conv155_out(cfx_ctx* ctx, const void* in, const void* wgt,
                    const float* bias, void* out, int N, int D, int H,
                    int W, int C, int K, int is_bf16);
/* the 32x32x2-MFMA variant (C == K == 28 instantiated) */
int cfx_conv3_ndhwc_w32(cfx_ctx* ctx, const float* in, const float* wgt,
                        const float* bias, const float* residual,
                        float* out, int N, int D, int H, int W, int C,
                        int K, int do_elu);

/* ---- kernel timing (HIP events on the context stream) ------------------ */
enum cfx_kernel_id {
    CFX_K_BLEND = 0,
    CFX_K_EXTRACT = 1,
    CFX_K_NORMALIZE = 2,
    CFX_K_CAST = 3,
    CFX_K_RECIPROCAL = 4,
    CFX_K_MASKMUL = 5,
    CFX_K_CROP = 6,
    CFX_K_MAX = 7,
    CFX_K_MYELIN = 8,
    CFX_K_CC = 9,
    CFX_K_CONV = 10,
    CFX_K_CONV_STREAM = 11,  /* up/down-sample + (1,5,5) stream convs */
    CFX_K_COUNT = 12
};
int cfx_profile_enable(cfx_ctx* ctx, int enable);
int cfx_profile_reset(cfx_ctx* ctx);
/* drains pending events (syncs the stream); bytes = accumulated ALGORITHMIC
 * bytes of the profiled launches (computed from region sizes, not counters) */
int cfx_profile_get(cfx_ctx* ctx, int kernel_id, unsigned long long* count,
                    double* total_ms, double* bytes);

#ifdef __cplusplus
}
#endif
#endif /* CHUNKFLOW_AMD_H */
