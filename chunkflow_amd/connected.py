"""Connected components (threshold + 6/18/26-connectivity labeling).

Host implementation for the config-4 operator chain (reference
flow/flow.py:1803-1829 + chunk/base.py:128-137, which delegate to the cc3d
C++ wheel). The reference pins no golden labels for cc3d (SURVEY.md §8c:
parity unpinned at that boundary); we pin against scipy.ndimage.label with
the matching structuring element, which defines the same partition up to
label renumbering. A gfx950 union-find kernel is the first 'next' row
 for a later round.
"""
import numpy as np
from scipy import ndimage

from .chunk import Chunk

_STRUCTS = {
    6: ndimage.generate_binary_structure(3, 1),
    18: ndimage.generate_binary_structure(3, 2),
    26: ndimage.generate_binary_structure(3, 3),
}


def connected_component(chunk: Chunk, threshold: float = None,
                        connectivity: int = 6) -> Chunk:
    assert connectivity in _STRUCTS
    c = chunk.numpy()
    arr = c.array
    if threshold is not None:
        if arr.ndim == 4:
            assert arr.shape[0] >= 1
            arr = arr[0]
        seg = arr > threshold
    else:
        seg = arr
    labels, _ = ndimage.label(np.ascontiguousarray(seg),
                              structure=_STRUCTS[connectivity])
    return Chunk(labels.astype(np.uint32), voxel_offset=c.voxel_offset,
                 voxel_size=c.voxel_size)
