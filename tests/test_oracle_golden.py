"""Pin the CPU oracle against golden vectors generated from the reference.

The fixtures in tests/golden/ were produced by running the UNMODIFIED
reference (seung-lab/chunkflow v1.1.7) via oracle/gen_golden.py in the build
container; the CRC32s of the three BASELINE patch-mask geometries match
SURVEY.md Appendix A.2. These tests run everywhere (no GPU, no reference
checkout needed).
"""
import zlib

import numpy as np
import pytest

from oracle import make_patch_mask, oracle_inference, patch_slices_list


def crc(arr):
    return zlib.crc32(np.ascontiguousarray(arr).tobytes())


@pytest.mark.parametrize('name', [
    'mask_20x256x256_ov4x64x64',
    'mask_20x128x128_ov4x32x32',
    'mask_10x32x32_ov2x8x8',
])
def test_patch_mask_crc(golden, name):
    meta, arrays = golden
    case = meta['cases'][name]
    m = make_patch_mask(tuple(case['patch_size']), tuple(case['overlap']))
    assert crc(m) == case['crc32']
    assert m.min() == np.float32(case['min'])
    if name in arrays:
        np.testing.assert_array_equal(m, arrays[name])


def test_patch_mask_invariants():
    # interior exactly 1; sum == stride-cell volume (neighbor weights sum to
    # 1 per voxel by construction) — SURVEY.md A.2 invariants
    ps, ov = (20, 128, 128), (4, 32, 32)
    m = make_patch_mask(ps, ov).astype(np.float64)
    np.testing.assert_array_equal(m[4:-4, 32:-32, 32:-32], 1.0)
    stride_vol = (ps[0] - ov[0]) * (ps[1] - ov[1]) * (ps[2] - ov[2])
    assert abs(m.sum() - stride_vol) < 1e-2


def test_slices_list_config2(golden):
    meta, arrays = golden
    ref = arrays['slices_512_p20x256x256_ov4x64x64']
    got = patch_slices_list((512, 512, 512), (20, 256, 256), (4, 64, 64))
    assert len(got) == 288 == ref.shape[0]
    got_arr = np.array([list(i) + list(o) for i, o in got], dtype=np.int32)
    np.testing.assert_array_equal(got_arr, ref)


def test_slices_tail_clamp_duplicates():
    # a size whose tail clamps onto the previous position must DUPLICATE the
    # clamped entry, exactly like the reference loop (inferencer.py:268-285)
    got = patch_slices_list((20, 33, 32), (10, 32, 32), (2, 8, 8))
    starts = [i for i, _ in got]
    assert len(starts) != len(set(starts)) or all(
        s[1] in (0, 1) for s in starts)


def test_e2e_identity(golden):
    meta, arrays = golden
    chunk = arrays['e2e_input_u8']
    out = oracle_inference(chunk, (10, 32, 32), (2, 8, 8),
                           num_output_channels=3, batch_size=3)
    np.testing.assert_allclose(out, arrays['e2e_identity_out'],
                               rtol=1e-6, atol=1e-7)
    # identity through the full blend+normalize is the input/255 (1e-5 pin,
    # the reference's own strongest test: test_inferencer.py:141-169)
    np.testing.assert_allclose(out[0], chunk.astype(np.float32) / 255.0,
                               rtol=1e-5, atol=1e-5)


def test_e2e_identity_offset(golden):
    meta, arrays = golden
    chunk = arrays['e2e_input_u8']
    out = oracle_inference(chunk, (10, 32, 32), (2, 8, 8),
                           num_output_channels=3, batch_size=4,
                           offset=(7, 11, 13))
    np.testing.assert_allclose(out, arrays['e2e_identity_offset_out'],
                               rtol=1e-6, atol=1e-7)


def test_e2e_identity_myelin(golden):
    meta, arrays = golden
    chunk = arrays['e2e_input_u8']
    out = oracle_inference(chunk, (10, 32, 32), (2, 8, 8),
                           num_output_channels=4, batch_size=3,
                           mask_myelin_threshold=0.3)
    np.testing.assert_allclose(out, arrays['e2e_identity_myelin_out'],
                               rtol=1e-6, atol=1e-7)


def test_e2e_pytorch_cpu(golden, golden_dir):
    """Oracle with the seeded 2-layer torch-CPU conv engine vs the reference
    `pytorch` framework output (the conv-parity golden pin)."""
    torch = pytest.importorskip('torch')
    import importlib.util
    import os
    meta, arrays = golden
    spec = importlib.util.spec_from_file_location(
        'golden_model_t', os.path.join(golden_dir, 'ref_model.py'))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    model = mod.InstantiatedModel
    model.load_state_dict(torch.load(
        os.path.join(golden_dir, 'ref_model_weights.pt')))
    model.eval()
    mask = make_patch_mask((10, 32, 32), (2, 8, 8))

    def engine(batch):
        with torch.no_grad():
            out = model(torch.from_numpy(batch)).numpy()
        return out * mask

    chunk = arrays['e2e_input_u8']
    out = oracle_inference(chunk, (10, 32, 32), (2, 8, 8),
                           num_output_channels=3, batch_size=1,
                           engine=engine)
    np.testing.assert_allclose(out, arrays['e2e_pytorch_out'],
                               rtol=1e-5, atol=1e-6)


@pytest.mark.skipif(not __import__('os').path.isdir('/root/reference'),
                    reason='reference checkout not present (GPU box)')
def test_oracle_vs_live_reference():
    """When the reference is mounted (build container), also check the oracle
    against a LIVE reference run on a fresh random case."""
    from oracle.ref_harness import import_reference
    Inferencer, Chunk, _ = import_reference()
    rng = np.random.RandomState(42)
    arr = rng.randint(0, 256, size=(24, 70, 66), dtype=np.uint8)
    with Inferencer(None, None, (12, 32, 32),
                    output_patch_overlap=(4, 8, 8), framework='identity',
                    num_output_channels=2, batch_size=5,
                    mask_output_chunk=True) as inf:
        ref_out = inf(Chunk(arr.copy(), voxel_offset=(3, 4, 5)))
    out = oracle_inference(arr, (12, 32, 32), (4, 8, 8),
                           num_output_channels=2, batch_size=5,
                           offset=(3, 4, 5))
    np.testing.assert_allclose(out, np.asarray(ref_out.array),
                               rtol=1e-6, atol=1e-7)


@pytest.mark.skipif(not __import__('os').path.isdir('/root/reference'),
                    reason='reference checkout not present (GPU box)')
@pytest.mark.parametrize('seed', [101, 202, 303])
def test_oracle_vs_live_reference_fuzz(seed):
    """Randomized geometries against the LIVE reference (build container
    only): chunk/patch/overlap drawn per seed, identity engine, masked
    mode, random offset."""
    from oracle.ref_harness import import_reference
    Inferencer, Chunk, _ = import_reference()
    rng = np.random.RandomState(seed)
    patch = tuple(int(rng.randint(6, 13) * 2) for _ in range(3))
    ov = tuple(max(2, (p // 4) // 2 * 2) for p in patch)
    chunk = tuple(p + int(rng.randint(0, 3)) * (p - o)
                  + int(rng.randint(0, 4))
                  for p, o in zip(patch, ov))
    off = tuple(int(v) for v in rng.randint(0, 9, size=3))
    nch = int(rng.randint(1, 4))
    bs = int(rng.randint(1, 5))
    arr = rng.randint(0, 256, size=chunk).astype(np.uint8)
    with Inferencer(None, None, patch, output_patch_overlap=ov,
                    framework='identity', num_output_channels=nch,
                    batch_size=bs, mask_output_chunk=True) as inf:
        ref_out = inf(Chunk(arr.copy(), voxel_offset=off))
    got = oracle_inference(arr, patch, ov, num_output_channels=nch,
                           batch_size=bs, offset=off)
    np.testing.assert_allclose(got, np.asarray(ref_out.array),
                               rtol=1e-6, atol=1e-7)
