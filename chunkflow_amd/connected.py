"""Connected components (threshold + 6/18/26-connectivity labeling).

The config-4 operator chain's segmentation step (reference
flow/flow.py:1803-1829 + chunk/base.py:128-137, which delegate to the cc3d
C++ wheel). Parity pin: the reference pins no golden labels for cc3d
(SURVEY.md §8c), so the pin is scipy.ndimage.label with the matching
structuring element — SAME partition AND SAME numbering (labels 1..N in
raster-scan first-encounter order), which the GPU union-find reproduces
exactly (min-index roots ranked in flat order; csrc/cc.hip).

Device chunks run the gfx950 union-find; host chunks use scipy. With a
threshold both implement the binary semantics (threshold > t). Without a
threshold the reference's cc3d keeps EQUAL-VALUE semantics (each connected
region of one value is one component; 0 is background) — implemented here
as `equal_value_label` (per-value scipy labeling renumbered in raster-scan
first-encounter order; cc3d's own numbering/dtype is implementation-defined
and unpinned). A multi-valued device chunk without a
threshold falls back to the host path (correctness over speed; warned).
"""
import warnings
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


def equal_value_label(arr: np.ndarray, connectivity: int = 6) -> np.ndarray:
    """cc3d-style multi-label CC: one component per connected region of
    equal nonzero value (reference chunk/base.py:128-137 semantics).
    Numbering: labels 1..N in raster-scan first-encounter order (our CC
    convention — cc3d's numbering is implementation-defined, unpinned)."""
    assert connectivity in _STRUCTS
    out = np.zeros(arr.shape, dtype=np.int64)
    offset = 0
    for v in np.unique(arr):
        if v == 0:
            continue
        lab, n = ndimage.label(arr == v, structure=_STRUCTS[connectivity])
        sel = lab > 0
        out[sel] = lab[sel].astype(np.int64) + offset
        offset += n
    # renumber by global raster-scan first encounter
    flat = out.ravel()
    uniq, first = np.unique(flat, return_index=True)
    order = np.argsort(first, kind='stable')
    remap = np.zeros(offset + 1, dtype=np.uint32)
    next_id = 1
    for u in uniq[order]:
        if u != 0:
            remap[u] = next_id
            next_id += 1
    return remap[flat].reshape(arr.shape)


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


def _is_multivalued(arr3) -> bool:
    """>1 distinct nonzero value (numpy or torch array)."""
    if isinstance(arr3, torch.Tensor):
        u = torch.unique(arr3)
        return int((u != 0).sum()) > 1
    u = np.unique(arr3)
    return int(np.count_nonzero(u)) > 1


def connected_component(chunk: Chunk, threshold: float = None,
                        connectivity: int = 6) -> Chunk:
    assert connectivity in _STRUCTS
    if chunk.is_device:
        if threshold is None and _is_multivalued(_first_channel(chunk.array)):
            # cc3d equal-value semantics; no GPU kernel for it yet — the
            # binary union-find would merge touching different-label regions
            warnings.warn('connected-components without --threshold on a '
                          'multi-valued device chunk: falling back to the '
                          'host equal-value path (cc3d semantics)')
            c = chunk.numpy()
            labels = equal_value_label(
                np.ascontiguousarray(_first_channel(c.array)), connectivity)
            return Chunk(torch.from_numpy(labels).to(chunk.array.device),
                         voxel_offset=c.voxel_offset, voxel_size=c.voxel_size)
        return connected_component_gpu(chunk, threshold=threshold,
                                       connectivity=connectivity)
    c = chunk.numpy()
    arr = _first_channel(c.array)
    if threshold is None and _is_multivalued(arr):
        labels = equal_value_label(np.ascontiguousarray(arr), connectivity)
        return Chunk(labels, voxel_offset=c.voxel_offset,
                     voxel_size=c.voxel_size)
    seg = arr > threshold if threshold is not None else arr
    labels, _ = ndimage.label(np.ascontiguousarray(seg),
                              structure=_STRUCTS[connectivity])
    return Chunk(labels.astype(np.uint32), voxel_offset=c.voxel_offset,
                 voxel_size=c.voxel_size)
