"""Debug the GPU connected-components against scipy on small cases."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
from scipy import ndimage

from chunkflow_amd.ops import HipOps

ops = HipOps(0)


def run_case(name, fg_np, connectivity=6):
    D, H, W = fg_np.shape
    n = fg_np.size
    fg = torch.from_numpy(fg_np.astype(np.uint8)).cuda().contiguous()
    labels = torch.empty((D, H, W), dtype=torch.int32, device='cuda')
    scratch = torch.full((D, H, W), -7, dtype=torch.int32, device='cuda')
    ncomp = ops.cfx.connected_components(fg.data_ptr(), (D, H, W),
                                         connectivity, labels.data_ptr(),
                                         scratch.data_ptr())
    got = labels.cpu().numpy()
    ref, nref = ndimage.label(
        fg_np, structure=ndimage.generate_binary_structure(
            3, {6: 1, 18: 2, 26: 3}[connectivity]))
    ok = np.array_equal(got.astype(np.int64), ref)
    print(f'{name}: n={n} ncomp={ncomp} nref={nref} exact={ok}')
    if not ok:
        bad = got.astype(np.int64) != ref
        print('  mismatched voxels:', bad.sum(), 'of', n)
        idx = np.argwhere(bad)[:5]
        for z, y, x in idx:
            print(f'   vox ({z},{y},{x}) got={got[z,y,x]} ref={ref[z,y,x]} '
                  f'fg={fg_np[z,y,x]}')
        # partition check ignoring numbering
        m = {}
        consistent = True
        for g, r in zip(got[fg_np > 0].ravel(), ref[fg_np > 0].ravel()):
            if g in m and m[g] != r:
                consistent = False
                break
            m[g] = r
        print('  partition consistent (renumber only):', consistent,
              'unique got:', len(np.unique(got)) - 1)
    return ok


rng = np.random.RandomState(30)
run_case('rand40', (rng.rand(40, 50, 60) > 0.7).astype(np.uint8))
run_case('rand-small', (rng.rand(8, 8, 8) > 0.5).astype(np.uint8))
run_case('ones', np.ones((8, 8, 8), dtype=np.uint8))
run_case('zeros', np.zeros((8, 8, 8), dtype=np.uint8))
run_case('two-slabs', np.pad(np.ones((4, 8, 8), np.uint8),
                             ((0, 4), (0, 0), (0, 0))))
vol = (rng.rand(30, 40, 50) > 0.4).astype(np.uint8)
run_case('dense-26', vol, 26)
run_case('dense-18', vol, 18)
# the config-4 failing shape: dense sin blobs
from chunkflow_amd.chunk import Chunk
sinv = Chunk.create(size=(20, 72, 88), dtype='uint8', pattern='sin')
run_case('sin-blobs', (sinv.array.astype(np.float32) / 255.0 > 0.3)
         .astype(np.uint8))
