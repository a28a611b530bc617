"""Connected components (threshold + 6/18/26-connectivity labeling).

The config-4 operator chain's segmentation step (reference
flow/flow.py:1803-1829 + chunk/base.py:128-137, which delegate to the cc3d
C++ wheel). Parity pin: the reference pins no golden labels for cc3d
(SURVEY.md §8c), so the pin is scipy.ndimage.label with the matching
structuring element — SAME partition AND SAME numbering (labels 1..N in
raster-scan first-encounter order), which the GPU union-find reproduces
exactly (min-index roots ranked in flat order; csrc/cc.hip).

Device chunks run the gfx950 union-find; host chunks use scipy. Both
implement the binary semantics (threshold > t, or nonzero foreground);
cc3d's multi-label equal-value semantics is out of the pinned scope.
"""
import numpy as np
import torch
from scipy import ndimage

from .chunk import Chunk

_STRUCTS = {
    6: ndimage.generate_binary_structure(3, 1),
    18: ndimage.generate_binary_structure(3, 2),
    26: ndimage.generate_binary_structure(3, 3),
}


def _first_channel(arr):
    if arr.ndim == 4:
        return arr[0]
    return arr


def connected_component_gpu(chunk: Chunk, threshold: float = None,
                            connectivity: int = 6) -> Chunk:
    from .ops import HipOps
    ops = HipOps(int(str(chunk.array.device).split(':')[-1])
                 if ':' in str(chunk.array.device) else 0)
    t = chunk.array
    t3 = _first_channel(t).contiguous()
    dims = tuple(t3.shape)
    n = t3.numel()
    fg = torch.empty(dims, dtype=torch.uint8, device=t3.device)
    if threshold is not None:
        if t3.dtype != torch.float32:
            t3 = t3.to(torch.float32)
        ops.cfx.threshold(t3.data_ptr(), fg.data_ptr(), n, float(threshold))
    else:
        if t3.dtype == torch.uint8:
            ops.cfx.nonzero_u8(t3.data_ptr(), fg.data_ptr(), n)
        else:
            fg = (t3 != 0).to(torch.uint8).contiguous()
    labels = torch.empty(dims, dtype=torch.int32, device=t3.device)
    scratch = torch.empty(dims, dtype=torch.int32, device=t3.device)
    ops.cfx.connected_components(fg.data_ptr(), dims, connectivity,
                                 labels.data_ptr(), scratch.data_ptr())
    return Chunk(labels, voxel_offset=chunk.voxel_offset,
                 voxel_size=chunk.voxel_size)


def connected_component(chunk: Chunk, threshold: float = None,
                        connectivity: int = 6) -> Chunk:
    assert connectivity in _STRUCTS
    if chunk.is_device:
        return connected_component_gpu(chunk, threshold=threshold,
                                       connectivity=connectivity)
    c = chunk.numpy()
    arr = _first_channel(c.array)
    seg = arr > threshold if threshold is not None else arr
    labels, _ = ndimage.label(np.ascontiguousarray(seg),
                              structure=_STRUCTS[connectivity])
    return Chunk(labels.astype(np.uint32), voxel_offset=c.voxel_offset,
                 voxel_size=c.voxel_size)
