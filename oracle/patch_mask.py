"""Bump-weight patch mask — CPU oracle restatement (test infrastructure only).

Restates chunkflow/flow/divid_conquer/patch/patch_mask.py:15-68 (reference
v1.1.7): a Wu bump function on centered grids, affinely remapped to [1, 1e6],
then normalized by a 3x3x3 shifted self-accumulation (a simulation of the
blending of neighboring patches) so that interior voxels sum to exactly 1
across overlapping patches.

The whole pipeline is float64 and only the final result is cast to the
requested dtype, exactly as the reference does. Golden CRC32s of the f32
result at the BASELINE geometries are committed under tests/golden/.
"""
import numpy as np


def make_bump_map(patch_size):
    # chunkflow patch_mask.py:51-68: grids are (index+1)/(n+1)*2-1 in (-1,1),
    # bump = exp(-1/(1-x^2) - 1/(1-y^2) - 1/(1-z^2)), then np.interp-remapped
    # so min->1 and max->1e6.
    pz, py, px = patch_size
    zv, yv, xv = np.meshgrid(
        np.arange(pz), np.arange(py), np.arange(px), indexing='ij')
    xv = (xv + 1.0) / (px + 1.0) * 2.0 - 1.0
    yv = (yv + 1.0) / (py + 1.0) * 2.0 - 1.0
    zv = (zv + 1.0) / (pz + 1.0) * 2.0 - 1.0
    bump = np.exp(-1.0 / (1.0 - xv * xv)
                  - 1.0 / (1.0 - yv * yv)
                  - 1.0 / (1.0 - zv * zv))
    bump = np.interp(bump, (bump.min(), bump.max()), (1, 1e6))
    return np.asarray(bump, dtype=np.float64)


def make_patch_mask(patch_size, overlap, dtype='float32'):
    # chunkflow patch_mask.py:15-48: 3x3x3 shifted accumulation of the bump
    # map at stride = patch_size - overlap simulates the blend of the 27
    # neighboring patches; dividing by the center crop normalizes the weights
    # so that each interior voxel's weights sum to 1 across patches.
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
    # reference asserts the interior [overlap, size-overlap) is exactly 1
    # (patch_mask.py:43-46)
    np.testing.assert_array_equal(
        bump[overlap[0]:-overlap[0],
             overlap[1]:-overlap[1],
             overlap[2]:-overlap[2]], 1)
    return bump.astype(dtype)
