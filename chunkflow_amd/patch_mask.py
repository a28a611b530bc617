"""Bump-weight patch mask — PRODUCT host-side precompute.

Host float64 pipeline replicating the reference bit-for-bit
(chunkflow/flow/divid_conquer/patch/patch_mask.py:15-68): Wu bump function
on centered grids, np.interp remap to [1, 1e6], 3x3x3 shifted
self-accumulation normalize, f32 cast. Computed once per geometry on the
host and uploaded to HBM; the blend kernel fuses the per-voxel multiply.

This file is product code (NOT the oracle): the oracle keeps its own
restatement under oracle/ and only tests may import that one. Both are pinned
to the same golden CRCs (tests/golden).
"""
from functools import lru_cache

import numpy as np


def make_bump_map(patch_size) -> np.ndarray:
    pz, py, px = patch_size
    zv, yv, xv = np.meshgrid(np.arange(pz), np.arange(py), np.arange(px),
                             indexing='ij')
    xv = (xv + 1.0) / (px + 1.0) * 2.0 - 1.0
    yv = (yv + 1.0) / (py + 1.0) * 2.0 - 1.0
    zv = (zv + 1.0) / (pz + 1.0) * 2.0 - 1.0
    bump = np.exp(-1.0 / (1.0 - xv * xv)
                  - 1.0 / (1.0 - yv * yv)
                  - 1.0 / (1.0 - zv * zv))
    bump = np.interp(bump, (bump.min(), bump.max()), (1, 1e6))
    return np.asarray(bump, dtype=np.float64)


@lru_cache(maxsize=8)
def make_patch_mask(patch_size: tuple, overlap: tuple,
                    dtype: str = 'float32') -> np.ndarray:
    """f32 (pz,py,px) weights; interior [ov, ps-ov) is exactly 1 and the
    27-neighbor weights sum to 1 per voxel (asserted, like the reference)."""
    bump = make_bump_map(patch_size)
    stride = tuple(p - o for p, o in zip(patch_size, overlap))
    base = np.zeros(tuple(p + 2 * s for p, s in zip(patch_size, stride)),
                    dtype='float64')
    for nz in range(3):
        for ny in range(3):
            for nx in range(3):
                base[nz * stride[0]:nz * stride[0] + patch_size[0],
                     ny * stride[1]:ny * stride[1] + patch_size[1],
                     nx * stride[2]:nx * stride[2] + patch_size[2]] += bump
    bump /= base[stride[0]:stride[0] + patch_size[0],
                 stride[1]:stride[1] + patch_size[1],
                 stride[2]:stride[2] + patch_size[2]]
    np.testing.assert_array_equal(
        bump[overlap[0]:-overlap[0], overlap[1]:-overlap[1],
             overlap[2]:-overlap[2]], 1)
    out = bump.astype(dtype)
    out.setflags(write=False)
    return out
