"""normalize-contrast CPU oracle restatement (test infrastructure only).

Restates Image.normalize_contrast (reference chunk/image/base.py:30-132):
per-section uint8 histogram (np.bincount minlength=255), clamping values
from the CDF at the clip fractions (pure black removed), affine LUT
rounded+clipped to [minval, maxval].

Reference quirks replicated: (1) the per-section loop's
`else` clause ALWAYS runs afterwards, so per_section=True normalizes each
section and then re-normalizes the WHOLE chunk with a LUT computed on the
already-normalized data; (2) per_section=False (--whole) is a NO-OP in the
reference — the whole `if per_section:` block is skipped. We replicate the
OUTPUT of that control flow.
"""
import numpy as np


def find_clamping_values(hist, lower_clip_fraction, upper_clip_fraction):
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


def hist_to_lut(hist, lower_clip_fraction, upper_clip_fraction,
                minval=1, maxval=255):
    lower, upper = find_clamping_values(hist, lower_clip_fraction,
                                        upper_clip_fraction)
    if lower == upper:
        return None
    lut = np.arange(0, 256, dtype=np.float32)
    lut = (lut - float(lower)) * (maxval / (float(upper) - float(lower)))
    np.clip(lut, minval, maxval, out=lut)
    return np.round(lut).astype(np.uint8)


def _normalize_array(arr, lcf, ucf, minval, maxval):
    hist = np.bincount(arr.flatten(), minlength=255)
    lut = hist_to_lut(hist, lcf, ucf, minval=minval, maxval=maxval)
    if lut is not None:
        arr = lut[arr]
    return arr


def oracle_normalize_contrast(arr, lower_clip_fraction=0.01,
                              upper_clip_fraction=0.01, minval=1,
                              maxval=255, per_section=True):
    arr = np.array(arr)
    assert arr.dtype == np.uint8 and arr.ndim == 3
    if not per_section:
        return arr  # reference --whole mode is a no-op (quirk 2)
    for z in range(arr.shape[0]):
        arr[z] = _normalize_array(arr[z], lower_clip_fraction,
                                  upper_clip_fraction, minval, maxval)
    # the reference's for-else: the whole-chunk pass always follows
    arr = _normalize_array(arr, lower_clip_fraction, upper_clip_fraction,
                           minval, maxval)
    return arr
