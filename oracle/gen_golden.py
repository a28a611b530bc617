"""Generate golden parity fixtures by running the REFERENCE in this container.

TEST INFRASTRUCTURE ONLY. Run from the repo root:

    python -m oracle.gen_golden

Writes tests/golden/golden.npz + tests/golden/golden.json (+ the reference
conv model file/weights used by both the reference run and the product GPU
parity tests). Requires /root/reference (the build container); the committed
fixtures are what travels to the GPU box.

Golden cases (all geometries small enough to run in seconds):
  * patch masks at the BASELINE geometries: CRC32 + stats (+ full array for
    the small test geometry)
  * patch slices list for the config-2 512^3 geometry (288 entries)
  * Chunk.create 'sin' pattern arrays/CRCs
  * end-to-end masked identity inference (non-aligned chunk)
  * end-to-end masked identity inference with 4 channels + myelin mask
  * end-to-end masked inference through a seeded 2-layer torch-CPU conv net
"""
import json
import os
import zlib

import numpy as np

from .ref_harness import import_reference, REFERENCE_PATH

HERE = os.path.dirname(os.path.abspath(__file__))
GOLDEN_DIR = os.path.join(HERE, '..', 'tests', 'golden')

MODEL_FILE_SRC = '''\
"""Seeded 2-layer 3D conv net, reference `pytorch` framework model-file
contract (chunkflow/flow/divid_conquer/patch/pytorch.py:48-60): exposes
`InstantiatedModel`; weights are loaded from the --convnet-weight-path file.
Used as a golden conv-parity pin (torch-CPU reference vs MI355X path)."""
import torch
import torch.nn as nn


class GoldenNet(nn.Module):
    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv3d(1, 8, 3, padding=1)
        self.conv2 = nn.Conv3d(8, 3, 3, padding=1)

    def forward(self, x):
        return torch.sigmoid(self.conv2(torch.relu(self.conv1(x))))


InstantiatedModel = GoldenNet()
'''


def crc(arr) -> int:
    return zlib.crc32(np.ascontiguousarray(arr).tobytes())


def main():
    assert os.path.isdir(REFERENCE_PATH), 'golden generation needs the reference'
    os.makedirs(GOLDEN_DIR, exist_ok=True)
    Inferencer, Chunk, make_patch_mask = import_reference()

    arrays = {}
    meta = {'reference': 'seung-lab/chunkflow v1.1.7', 'cases': {}}

    # --- patch masks -------------------------------------------------------
    geometries = {
        'mask_20x256x256_ov4x64x64': ((20, 256, 256), (4, 64, 64)),
        'mask_20x128x128_ov4x32x32': ((20, 128, 128), (4, 32, 32)),
        'mask_32x256x256_ov4x64x64': ((32, 256, 256), (4, 64, 64)),
        'mask_10x32x32_ov2x8x8': ((10, 32, 32), (2, 8, 8)),
    }
    for name, (ps, ov) in geometries.items():
        m = make_patch_mask(ps, ov, dtype='float32')
        meta['cases'][name] = {
            'patch_size': ps, 'overlap': ov, 'crc32': crc(m),
            'min': float(m.min()), 'sum': float(m.sum()),
            'corner': float(m[0, 0, 0]),
        }
        if name == 'mask_10x32x32_ov2x8x8':
            arrays[name] = m

    # --- patch slices list, config-2 geometry ------------------------------
    inf = Inferencer(None, None, (20, 256, 256),
                     output_patch_overlap=(4, 64, 64), framework='identity',
                     num_output_channels=3, batch_size=12,
                     mask_output_chunk=True, dry_run=True)
    inf.input_size = (512, 512, 512)
    inf._construct_patch_slices_list((0, 0, 0))
    starts = np.array(
        [[s.start for s in inp] + [s.start for s in outp]
         for inp, outp in inf.patch_slices_list], dtype=np.int32)
    arrays['slices_512_p20x256x256_ov4x64x64'] = starts
    meta['cases']['slices_512'] = {'n': int(starts.shape[0])}

    # --- Chunk.create 'sin' pattern ----------------------------------------
    small_sin = Chunk.create(size=(20, 68, 72), dtype='uint8',
                             pattern='sin').array
    arrays['sin_20x68x72_u8'] = small_sin
    meta['cases']['sin_20x68x72_u8'] = {'crc32': crc(small_sin)}
    big_sin = Chunk.create(size=(512, 512, 512), dtype='uint8',
                           pattern='sin').array
    meta['cases']['sin_512_u8'] = {'crc32': crc(big_sin),
                                   'sum': int(big_sin.astype(np.int64).sum())}
    del big_sin

    # --- end-to-end identity inference (masked, non-aligned chunk) ---------
    rng = np.random.RandomState(0)
    chunk_arr = rng.randint(0, 256, size=(20, 68, 72), dtype=np.uint8)
    arrays['e2e_input_u8'] = chunk_arr

    with Inferencer(None, None, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='identity',
                    num_output_channels=3, batch_size=3,
                    mask_output_chunk=True) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(0, 0, 0)))
    arrays['e2e_identity_out'] = np.asarray(out.array)
    meta['cases']['e2e_identity'] = {
        'patch_size': (10, 32, 32), 'overlap': (2, 8, 8), 'batch_size': 3}

    # the same with a nonzero voxel offset (offset arithmetic pin)
    with Inferencer(None, None, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='identity',
                    num_output_channels=3, batch_size=4,
                    mask_output_chunk=True) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(7, 11, 13)))
    arrays['e2e_identity_offset_out'] = np.asarray(out.array)

    # --- identity + myelin mask (4 channels, threshold) --------------------
    with Inferencer(None, None, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='identity',
                    num_output_channels=4, batch_size=3,
                    mask_output_chunk=True,
                    mask_myelin_threshold=0.3) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(0, 0, 0)))
    arrays['e2e_identity_myelin_out'] = np.asarray(out.array)

    # --- torch-CPU conv engine (framework='pytorch', batch 1) --------------
    import torch
    torch.manual_seed(0)
    model_path = os.path.join(GOLDEN_DIR, 'ref_model.py')
    with open(model_path, 'w') as f:
        f.write(MODEL_FILE_SRC)
    import importlib.util
    spec = importlib.util.spec_from_file_location('golden_model', model_path)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    weight_path = os.path.join(GOLDEN_DIR, 'ref_model_weights.pt')
    torch.save(mod.InstantiatedModel.state_dict(), weight_path)

    with Inferencer(model_path, weight_path, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='pytorch',
                    num_output_channels=3, batch_size=1,
                    mask_output_chunk=True) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(0, 0, 0)))
    arrays['e2e_pytorch_out'] = np.asarray(out.array)
    meta['cases']['e2e_pytorch'] = {
        'model': 'tests/golden/ref_model.py',
        'weights': 'tests/golden/ref_model_weights.pt',
        'patch_size': (10, 32, 32), 'overlap': (2, 8, 8), 'batch_size': 1}

    # --- TTA (augment=True) goldens: identity + conv engines ---------------
    # (reference transform.py:114-156 driven by inferencer.py:420-431)
    with Inferencer(None, None, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='identity',
                    num_output_channels=3, batch_size=3,
                    mask_output_chunk=True, augment=True) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(0, 0, 0)))
    arrays['e2e_identity_augment_out'] = np.asarray(out.array)

    with Inferencer(model_path, weight_path, (10, 32, 32),
                    output_patch_overlap=(2, 8, 8), framework='pytorch',
                    num_output_channels=3, batch_size=1,
                    mask_output_chunk=True, augment=True) as inferencer:
        out = inferencer(Chunk(chunk_arr.copy(), voxel_offset=(0, 0, 0)))
    arrays['e2e_pytorch_augment_out'] = np.asarray(out.array)
    meta['cases']['e2e_augment'] = {
        'patch_size': (10, 32, 32), 'overlap': (2, 8, 8),
        'note': 'augment=True goldens for identity (batch 3) and the '
                '2-layer conv net (batch 1), same input/geometry as e2e'}

    # --- full-RSUNet conv parity pin at a config-2-like geometry -----------
    # (the real benchmark net through the reference
    # Inferencer on torch-CPU; patch/pytorch.py:98-119 semantics)
    rs_model = os.path.join(HERE, '..', 'examples', 'nets', 'rsunet.py')
    rs_weights = os.path.join(GOLDEN_DIR, 'rsunet_weights.pt')
    spec2 = importlib.util.spec_from_file_location('golden_rsunet', rs_model)
    mod2 = importlib.util.module_from_spec(spec2)
    spec2.loader.exec_module(mod2)
    torch.save(mod2.InstantiatedModel.state_dict(), rs_weights)
    wsum = {k: float(v.to(torch.float64).sum())
            for k, v in mod2.InstantiatedModel.state_dict().items()}

    rs_in = Chunk.create(size=(64, 256, 256), dtype='uint8',
                         pattern='sin').array
    with Inferencer(rs_model, rs_weights, (20, 128, 128),
                    output_patch_overlap=(4, 32, 32), framework='pytorch',
                    num_output_channels=3, batch_size=1,
                    mask_output_chunk=True) as inferencer:
        out = inferencer(Chunk(rs_in.copy(), voxel_offset=(0, 0, 0)))
    rs_out = np.asarray(out.array)
    assert rs_out.shape == (3, 64, 256, 256)
    # full array is 50 MB — commit a strided subsample + a seeded random
    # voxel sample + f64 global stats instead
    arrays['rsunet_64x256x256_sub'] = rs_out[:, ::4, ::8, ::8].copy()
    rng2 = np.random.RandomState(123)
    idx = rng2.choice(rs_out.size, size=8192, replace=False)
    idx.sort()
    arrays['rsunet_64x256x256_sample_idx'] = idx.astype(np.int64)
    arrays['rsunet_64x256x256_sample_val'] = rs_out.ravel()[idx].copy()
    meta['cases']['rsunet_64x256x256'] = {
        'model': 'examples/nets/rsunet.py',
        'weights': 'tests/golden/rsunet_weights.pt',
        'weight_f64_sums': wsum,
        'input': "Chunk.create((64,256,256), uint8, pattern='sin')",
        'input_crc32': crc(rs_in),
        'patch_size': (20, 128, 128), 'overlap': (4, 32, 32),
        'batch_size': 1,
        'sum_f64': float(rs_out.astype(np.float64).sum()),
        'min': float(rs_out.min()), 'max': float(rs_out.max()),
    }

    np.savez_compressed(os.path.join(GOLDEN_DIR, 'golden.npz'), **arrays)
    with open(os.path.join(GOLDEN_DIR, 'golden.json'), 'w') as f:
        json.dump(meta, f, indent=1, default=str)
    total = sum(a.nbytes for a in arrays.values())
    print(f'wrote {len(arrays)} arrays ({total/1e6:.1f} MB raw) '
          f'+ {len(meta["cases"])} meta cases to {GOLDEN_DIR}')


if __name__ == '__main__':
    main()
