"""normalize-contrast: per-section histogram -> LUT apply (PRODUCT code).

Replicates the reference Image.normalize_contrast OUTPUT
(chunk/image/base.py:30-132): per-section uint8 histogram
(np.bincount(..., minlength=255) — length 255 unless value 255 occurs),
clamping values from the CDF at the clip fractions (pure black removed),
affine LUT rounded+clipped to [minval, maxval].

Reference control-flow quirks replicated, not fixed:
  1. the per-section for loop's `else` clause always runs, so after the
     per-section pass the WHOLE chunk is normalized again with a LUT from
     the already-normalized data;
  2. per_section=False (--whole) is a no-op (the entire block is skipped).

Device chunks: gfx950 histogram/LUT-apply kernels (csrc/image.hip) with the
LUT math on the host from the device histograms (a few KB per chunk). Host
chunks: the same numpy math.
"""
import numpy as np
import torch

from .chunk import Chunk


def find_clamping_values(hist, lower_clip_fraction, upper_clip_fraction):
    # chunk/image/base.py:30-62
    filtered = hist
    filtered[0] = 0
    cdf = np.cumsum(filtered.astype(np.uint64))
    total = cdf[-1]
    if total == 0:
        return 0, 0
    lower = 0
    for i, val in enumerate(cdf):
        if float(val) / float(total) > lower_clip_fraction:
            break
        lower = i
    upper = 0
    for i, val in enumerate(cdf):
        if float(val) / float(total) > 1 - upper_clip_fraction:
            break
        upper = i
    return lower, upper


def hist_to_lut(hist, lower_clip_fraction, upper_clip_fraction, minval=1,
                maxval=255):
    # chunk/image/base.py:64-91; returns None when no transform is needed
    lower, upper = find_clamping_values(hist, lower_clip_fraction,
                                        upper_clip_fraction)
    if lower == upper:
        return None
    lut = np.arange(0, 256, dtype=np.float32)
    lut = (lut - float(lower)) * (maxval / (float(upper) - float(lower)))
    np.clip(lut, minval, maxval, out=lut)
    return np.round(lut).astype(np.uint8)


def _bincount_like(hist256: np.ndarray) -> np.ndarray:
    """A 256-bin device histogram, trimmed to what np.bincount(...,
    minlength=255) would return (length 255 when value 255 never occurs —
    the trailing bin affects the ucf=0 edge case)."""
    if hist256[255] == 0:
        return hist256[:255].copy()
    return hist256.copy()


def _normalize_host(arr, lcf, ucf, minval, maxval):
    hist = np.bincount(arr.flatten(), minlength=255)
    lut = hist_to_lut(hist, lcf, ucf, minval=minval, maxval=maxval)
    if lut is not None:
        arr = lut[arr]
    return arr


def normalize_contrast(chunk: Chunk, lower_clip_fraction=0.01,
                       upper_clip_fraction=0.01, minval=1, maxval=255,
                       per_section=True) -> Chunk:
    """In-place on the (cloned) chunk's array; returns the chunk."""
    if not per_section:
        return chunk  # reference --whole mode is a no-op (quirk 2)

    if chunk.is_device:
        from .ops import HipOps
        t = chunk.array
        assert t.dtype == torch.uint8 and t.ndim == 3
        ops = HipOps(t.device.index or 0)
        D = t.shape[0]
        n_per_sec = t.shape[1] * t.shape[2]

        # int32 device bins bound each count by voxels-per-section; the
        # whole-chunk pass's histogram is derived on host in int64 by
        # pushing the per-section histograms through their LUTs (exact:
        # the apply is pointwise v -> lut[v]), so no 2^31 whole-chunk limit
        assert n_per_sec < 2**31, \
            'per-section voxel count overflows the int32 device histogram'
        identity = np.arange(256, dtype=np.uint8)

        def luts_from_hist(hist, nsec):
            luts = np.empty((nsec, 256), dtype=np.uint8)
            for s in range(nsec):
                lut = hist_to_lut(_bincount_like(hist[s]),
                                  lower_clip_fraction, upper_clip_fraction,
                                  minval=minval, maxval=maxval)
                luts[s] = identity if lut is None else lut
            return luts

        hist_dev = torch.empty((D, 256), dtype=torch.int32, device=t.device)
        ops.cfx.hist_u8(t.data_ptr(), n_per_sec, D, hist_dev.data_ptr())
        hist = hist_dev.cpu().numpy().astype(np.int64)
        luts = luts_from_hist(hist, D)                  # per-section pass
        lut_dev = torch.from_numpy(luts).to(t.device)
        ops.cfx.lut_apply_u8(t.data_ptr(), n_per_sec, D, lut_dev.data_ptr())

        whole = np.zeros((1, 256), dtype=np.int64)      # post-apply hist
        for s in range(D):
            np.add.at(whole[0], luts[s], hist[s])
        luts_w = luts_from_hist(whole, 1)               # for-else whole pass
        lut_w_dev = torch.from_numpy(luts_w).to(t.device)
        ops.cfx.lut_apply_u8(t.data_ptr(), D * n_per_sec, 1,
                             lut_w_dev.data_ptr())
        return chunk

    arr = chunk.numpy().array
    assert arr.dtype == np.uint8 and arr.ndim == 3
    for z in range(arr.shape[0]):
        arr[z] = _normalize_host(arr[z], lower_clip_fraction,
                                 upper_clip_fraction, minval, maxval)
    arr = _normalize_host(arr, lower_clip_fraction, upper_clip_fraction,
                          minval, maxval)
    chunk.array = arr
    return chunk
