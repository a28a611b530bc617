"""ctypes binding to the in-tree HIP C-ABI extension (libchunkflow_amd.so).

The product GPU path calls exclusively through this module; if the extension
is missing on a machine with a GPU, every entry raises — there is no silent
eager/CPU fallback (DESIGN.md 'no fallback' rule).
"""
import ctypes
import os

import numpy as np

from .build import SO_PATH, build

# every extern "C" symbol the header declares (kept in sync with
# include/chunkflow_amd.h; tests/test_abi.py checks the list against the
# header text)
SYMBOLS = [
    'cfx_init', 'cfx_destroy', 'cfx_last_error', 'cfx_version',
    'cfx_set_stream', 'cfx_sync', 'cfx_make_patch_mask',
    'cfx_normalize_intensity', 'cfx_cast_u8_f32_div', 'cfx_extract_patches',
    'cfx_blend_accumulate', 'cfx_blend_batch', 'cfx_build_chunk_mask',
    'cfx_reciprocal',
    'cfx_multiply_mask', 'cfx_multiply_mask_max', 'cfx_max',
    'cfx_crop_margin',
    'cfx_mask_using_last_channel', 'cfx_threshold', 'cfx_nonzero_u8',
    'cfx_connected_components', 'cfx_hist_u8', 'cfx_lut_apply_u8',
    'cfx_conv3_ndhwc', 'cfx_conv3_ndhwc_w32',
    'cfx_conv3_ndhwc_zring', 'cfx_conv3_ndhwc_bf16',
    'cfx_upconv_2x2', 'cfx_downconv_2x2', 'cfx_conv155_c1',
    'cfx_conv155_out',
    'cfx_profile_enable', 'cfx_profile_reset',
    'cfx_profile_get',
]

KERNEL_IDS = {
    'blend': 0, 'extract': 1, 'normalize': 2, 'cast': 3, 'reciprocal': 4,
    'maskmul': 5, 'crop': 6, 'max': 7, 'myelin': 8, 'cc': 9, 'conv': 10,
    'conv_stream': 11,
}

_lib = None


class CfxError(RuntimeError):
    pass


def _i3(v):
    return (ctypes.c_int * 3)(*[int(x) for x in v])


def _i6(v):
    return (ctypes.c_int * 6)(*[int(x) for x in v])


def load_library(build_if_missing: bool = True):
    """Load (building if needed) the extension; raises on failure."""
    global _lib
    if _lib is not None:
        return _lib
    path = SO_PATH
    if not os.path.exists(path):
        if not build_if_missing:
            raise CfxError(f'HIP extension not built: {path}')
        path = build()
    lib = ctypes.CDLL(path)
    for sym in SYMBOLS:
        if not hasattr(lib, sym):
            raise CfxError(f'{path} is missing symbol {sym}')
    lib.cfx_init.restype = ctypes.c_void_p
    lib.cfx_init.argtypes = [ctypes.c_int]
    lib.cfx_last_error.restype = ctypes.c_char_p
    lib.cfx_destroy.argtypes = [ctypes.c_void_p]
    _lib = lib
    return lib


def extension_available() -> bool:
    try:
        load_library(build_if_missing=False)
        return True
    except Exception:
        return False


class CfxContext:
    """One per GPU rank. Wraps the cfx_ctx and adopts torch's stream so
    kernels interleave correctly with the conv forward."""

    def __init__(self, device: int = 0):
        self.lib = load_library()
        self.ctx = self.lib.cfx_init(ctypes.c_int(device))
        if not self.ctx:
            raise CfxError('cfx_init failed: '
                           + self.lib.cfx_last_error().decode())
        self.device = device

    def _chk(self, rc: int, what: str):
        if rc != 0:
            raise CfxError(f'{what}: {self.lib.cfx_last_error().decode()}')

    def close(self):
        if getattr(self, 'ctx', None):
            self.lib.cfx_destroy(ctypes.c_void_p(self.ctx))
            self.ctx = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    # --- stream ------------------------------------------------------------
    def adopt_torch_stream(self):
        import torch
        s = torch.cuda.current_stream(self.device).cuda_stream
        self.set_stream(s)

    def set_stream(self, stream_ptr: int):
        self._chk(self.lib.cfx_set_stream(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(stream_ptr)),
            'cfx_set_stream')

    def sync(self):
        self._chk(self.lib.cfx_sync(ctypes.c_void_p(self.ctx)), 'cfx_sync')

    # --- kernels (device pointers are ints from tensor.data_ptr()) ---------
    def normalize_intensity(self, in_ptr, out_ptr, n):
        self._chk(self.lib.cfx_normalize_intensity(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(out_ptr), ctypes.c_longlong(n)),
            'cfx_normalize_intensity')

    def cast_u8_f32_div(self, in_ptr, out_ptr, n, divisor):
        self._chk(self.lib.cfx_cast_u8_f32_div(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(out_ptr), ctypes.c_longlong(n),
            ctypes.c_float(divisor)), 'cfx_cast_u8_f32_div')

    def extract_patches(self, chunk_ptr, channels, chunk_dims, starts,
                        patch_size, out_ptr):
        starts = np.ascontiguousarray(starts, dtype=np.int32)
        n = starts.shape[0]
        self._chk(self.lib.cfx_extract_patches(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(chunk_ptr),
            ctypes.c_int(channels), _i3(chunk_dims),
            starts.ctypes.data_as(ctypes.POINTER(ctypes.c_int)),
            ctypes.c_int(n), _i3(patch_size), ctypes.c_void_p(out_ptr)),
            'cfx_extract_patches')

    def blend_accumulate(self, out_ptr, channels, out_dims, patch_ptr,
                         patch_dims, offset, mask_ptr=None):
        self._chk(self.lib.cfx_blend_accumulate(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(out_ptr),
            ctypes.c_int(channels), _i3(out_dims), ctypes.c_void_p(patch_ptr),
            _i3(patch_dims), _i3(offset),
            ctypes.c_void_p(mask_ptr) if mask_ptr else None),
            'cfx_blend_accumulate')

    def blend_batch(self, out_ptr, channels, out_dims, patch_ptr,
                    patch_dims, items, mask_ptr=None):
        """items: (n, 4) int32 array of (batch_index, oz, oy, ox); clipped
        regions must be pairwise disjoint."""
        items = np.ascontiguousarray(items, dtype=np.int32)
        n = items.shape[0]
        self._chk(self.lib.cfx_blend_batch(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(out_ptr),
            ctypes.c_int(channels), _i3(out_dims), ctypes.c_void_p(patch_ptr),
            _i3(patch_dims),
            items.ctypes.data_as(ctypes.POINTER(ctypes.c_int)),
            ctypes.c_int(n),
            ctypes.c_void_p(mask_ptr) if mask_ptr else None),
            'cfx_blend_batch')

    def build_chunk_mask(self, mask_out_ptr, out_dims, patch_mask_ptr,
                         patch_dims, offsets):
        offsets = np.ascontiguousarray(offsets, dtype=np.int32)
        n = offsets.shape[0]
        self._chk(self.lib.cfx_build_chunk_mask(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(mask_out_ptr),
            _i3(out_dims), ctypes.c_void_p(patch_mask_ptr), _i3(patch_dims),
            offsets.ctypes.data_as(ctypes.POINTER(ctypes.c_int)),
            ctypes.c_int(n)), 'cfx_build_chunk_mask')

    def reciprocal(self, ptr, n):
        self._chk(self.lib.cfx_reciprocal(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(ptr),
            ctypes.c_longlong(n)), 'cfx_reciprocal')

    def multiply_mask(self, out_ptr, mask_ptr, channels, n_voxels):
        self._chk(self.lib.cfx_multiply_mask(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(out_ptr),
            ctypes.c_void_p(mask_ptr), ctypes.c_int(channels),
            ctypes.c_longlong(n_voxels)), 'cfx_multiply_mask')

    def multiply_mask_max(self, out_ptr, mask_ptr, channels, n_voxels):
        """Returns the post-multiply max, or None if the fused path was
        unavailable (caller should use max())."""
        out = ctypes.c_float(0)
        rc = self.lib.cfx_multiply_mask_max(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(out_ptr),
            ctypes.c_void_p(mask_ptr), ctypes.c_int(channels),
            ctypes.c_longlong(n_voxels), ctypes.byref(out))
        if rc == -2:
            return None
        self._chk(rc, 'cfx_multiply_mask_max')
        return out.value

    def max(self, ptr, n) -> float:
        out = ctypes.c_float(0)
        self._chk(self.lib.cfx_max(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(ptr),
            ctypes.c_longlong(n), ctypes.byref(out)), 'cfx_max')
        return out.value

    def crop_margin(self, in_ptr, out_ptr, channels, in_dims, margins):
        self._chk(self.lib.cfx_crop_margin(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(out_ptr), ctypes.c_int(channels), _i3(in_dims),
            _i6(margins)), 'cfx_crop_margin')

    def mask_using_last_channel(self, in_ptr, out_ptr, channels, dims,
                                threshold):
        self._chk(self.lib.cfx_mask_using_last_channel(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(out_ptr), ctypes.c_int(channels), _i3(dims),
            ctypes.c_float(threshold)), 'cfx_mask_using_last_channel')

    # --- connected components ----------------------------------------------
    def threshold(self, in_ptr, fg_ptr, n, thresh):
        self._chk(self.lib.cfx_threshold(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(fg_ptr), ctypes.c_longlong(n),
            ctypes.c_float(thresh)), 'cfx_threshold')

    def nonzero_u8(self, in_ptr, fg_ptr, n):
        self._chk(self.lib.cfx_nonzero_u8(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(fg_ptr), ctypes.c_longlong(n)),
            'cfx_nonzero_u8')

    def connected_components(self, fg_ptr, dims, connectivity, labels_ptr,
                             scratch_ptr) -> int:
        ncomp = ctypes.c_longlong(0)
        self._chk(self.lib.cfx_connected_components(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(fg_ptr), _i3(dims),
            ctypes.c_int(connectivity), ctypes.c_void_p(labels_ptr),
            ctypes.c_void_p(scratch_ptr), ctypes.byref(ncomp)),
            'cfx_connected_components')
        return ncomp.value

    def conv3_ndhwc_bf16(self, in_ptr, wgt_ptr, bias_ptr, residual_ptr,
                         out_ptr, n, d, h, w, c, k, do_elu=False):
        self._chk(self.lib.cfx_conv3_ndhwc_bf16(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(residual_ptr) if residual_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(c),
            ctypes.c_int(k), ctypes.c_int(1 if do_elu else 0)),
            'cfx_conv3_ndhwc_bf16')

    def upconv_2x2(self, in_ptr, wgt_ptr, bias_ptr, out_ptr, n, d, h, w,
                   c, k, bf16=False):
        self._chk(self.lib.cfx_upconv_2x2(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(c),
            ctypes.c_int(k), ctypes.c_int(1 if bf16 else 0)),
            'cfx_upconv_2x2')

    def downconv_2x2(self, in_ptr, wgt_ptr, bias_ptr, out_ptr, n, d, h, w,
                     c, k, bf16=False):
        self._chk(self.lib.cfx_downconv_2x2(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(c),
            ctypes.c_int(k), ctypes.c_int(1 if bf16 else 0)),
            'cfx_downconv_2x2')

    def conv155_c1(self, in_ptr, wgt_ptr, bias_ptr, out_ptr, n, d, h, w,
                   k, bf16=False):
        self._chk(self.lib.cfx_conv155_c1(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(k),
            ctypes.c_int(1 if bf16 else 0)), 'cfx_conv155_c1')

    def conv155_out(self, in_ptr, wgt_ptr, bias_ptr, out_ptr, n, d, h, w,
                    c, k, bf16=False):
        self._chk(self.lib.cfx_conv155_out(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(c),
            ctypes.c_int(k), ctypes.c_int(1 if bf16 else 0)),
            'cfx_conv155_out')

    # --- image normalization -------------------------------------------------
    def hist_u8(self, in_ptr, n_per_sec, nsec, hist_ptr):
        self._chk(self.lib.cfx_hist_u8(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_longlong(n_per_sec), ctypes.c_int(nsec),
            ctypes.c_void_p(hist_ptr)), 'cfx_hist_u8')

    def lut_apply_u8(self, buf_ptr, n_per_sec, nsec, lut_ptr):
        self._chk(self.lib.cfx_lut_apply_u8(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(buf_ptr),
            ctypes.c_longlong(n_per_sec), ctypes.c_int(nsec),
            ctypes.c_void_p(lut_ptr)), 'cfx_lut_apply_u8')

    def conv3_ndhwc(self, in_ptr, wgt_ptr, bias_ptr, residual_ptr, out_ptr,
                    n, d, h, w, c, k, do_elu=False, w32=False,
                    zring=False):
        fn = (self.lib.cfx_conv3_ndhwc_zring if zring
              else self.lib.cfx_conv3_ndhwc_w32 if w32
              else self.lib.cfx_conv3_ndhwc)
        self._chk(fn(
            ctypes.c_void_p(self.ctx), ctypes.c_void_p(in_ptr),
            ctypes.c_void_p(wgt_ptr),
            ctypes.c_void_p(bias_ptr) if bias_ptr else None,
            ctypes.c_void_p(residual_ptr) if residual_ptr else None,
            ctypes.c_void_p(out_ptr), ctypes.c_int(n), ctypes.c_int(d),
            ctypes.c_int(h), ctypes.c_int(w), ctypes.c_int(c),
            ctypes.c_int(k), ctypes.c_int(1 if do_elu else 0)),
            'cfx_conv3_ndhwc')

    # --- profiling ----------------------------------------------------------
    def profile_enable(self, enable=True):
        self._chk(self.lib.cfx_profile_enable(
            ctypes.c_void_p(self.ctx), ctypes.c_int(1 if enable else 0)),
            'cfx_profile_enable')

    def profile_reset(self):
        self._chk(self.lib.cfx_profile_reset(ctypes.c_void_p(self.ctx)),
                  'cfx_profile_reset')

    def profile_get(self, kernel: str):
        kid = KERNEL_IDS[kernel]
        count = ctypes.c_ulonglong(0)
        ms = ctypes.c_double(0)
        bytes_ = ctypes.c_double(0)
        self._chk(self.lib.cfx_profile_get(
            ctypes.c_void_p(self.ctx), ctypes.c_int(kid),
            ctypes.byref(count), ctypes.byref(ms), ctypes.byref(bytes_)),
            'cfx_profile_get')
        return {'count': count.value, 'total_ms': ms.value,
                'bytes': bytes_.value}


def make_patch_mask_c(patch_size, overlap) -> np.ndarray:
    """Host-side C implementation of the bump patch mask (C-ABI
    completeness; the Python product path uses chunkflow_amd.patch_mask)."""
    lib = load_library()
    out = np.empty(tuple(patch_size), dtype=np.float32)
    rc = lib.cfx_make_patch_mask(
        _i3(patch_size), _i3(overlap),
        out.ctypes.data_as(ctypes.POINTER(ctypes.c_float)))
    if rc != 0:
        raise CfxError('cfx_make_patch_mask failed')
    return out
