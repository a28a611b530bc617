"""GPU parity tests: the HIP hot path vs the oracle / reference goldens.

Every test here asserts the HIP extension IS the compute path (no eager
fallback). Tolerances: identity (pure tiler/blend/normalize) at 1e-5 — the
reference's own strongest pin (test_inferencer.py:141-169); conv paths at
1e-4 fp32 vs the torch-CPU reference (the north-star tolerance,
BASELINE.json)."""
import os

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope='module')
def cfx():
    from chunkflow_amd.hip import CfxContext
    ctx = CfxContext(0)
    ctx.adopt_torch_stream()
    return ctx


@pytest.fixture(scope='module')
def hip_ops():
    from chunkflow_amd.ops import HipOps
    return HipOps(0)


# --------------------------------------------------------------------------
# kernel-level parity
# --------------------------------------------------------------------------
class TestKernels:
    def test_normalize_intensity(self, hip_ops):
        rng = np.random.RandomState(0)
        arr = rng.randint(0, 256, size=(7, 33, 41), dtype=np.uint8)
        t = torch.from_numpy(arr).cuda()
        out = hip_ops.normalize_intensity(t).cpu().numpy()
        ref = arr.astype(np.float32) / 127.5 - 1.0
        np.testing.assert_array_equal(out, ref)

    def test_cast_div(self, hip_ops):
        rng = np.random.RandomState(1)
        arr = rng.randint(0, 256, size=(5, 17, 23), dtype=np.uint8)
        t = torch.from_numpy(arr).cuda()
        out = hip_ops.cast_div(t, 255.0).cpu().numpy()
        ref = arr.astype(np.float32) / 255.0
        np.testing.assert_array_equal(out, ref)

    def test_extract_blend_roundtrip(self, hip_ops):
        """extract + unmasked blend at disjoint offsets reproduces a copy."""
        rng = np.random.RandomState(2)
        chunk = torch.from_numpy(
            rng.rand(16, 32, 32).astype(np.float32)).cuda()
        starts = np.array([[0, 0, 0], [8, 0, 0]], dtype=np.int32)
        batch = torch.empty((2, 1, 8, 32, 32), dtype=torch.float32,
                            device='cuda')
        hip_ops.extract(chunk, starts, (8, 32, 32), batch)
        np.testing.assert_array_equal(batch[0, 0].cpu().numpy(),
                                      chunk[:8].cpu().numpy())
        out = torch.zeros((1, 16, 32, 32), dtype=torch.float32,
                          device='cuda')
        hip_ops.blend(out, batch, 0, (0, 0, 0))
        hip_ops.blend(out, batch, 1, (8, 0, 0))
        np.testing.assert_array_equal(out[0].cpu().numpy(),
                                      chunk.cpu().numpy())

    def test_blend_clipping_and_mask(self, hip_ops):
        rng = np.random.RandomState(3)
        patch = rng.rand(2, 2, 6, 8, 10).astype(np.float32)
        mask = rng.rand(6, 8, 10).astype(np.float32)
        out_np = np.zeros((2, 10, 12, 14), dtype=np.float32)
        from oracle.inference import blend_into
        blend_into(out_np, (0, 0, 0), patch[0] * mask, (-2, 9, 5))
        blend_into(out_np, (0, 0, 0), patch[1] * mask, (3, -1, -3))

        out = torch.zeros((2, 10, 12, 14), dtype=torch.float32,
                          device='cuda')
        pb = torch.from_numpy(patch).cuda()
        m = torch.from_numpy(mask).cuda()
        hip_ops.blend(out, pb, 0, (-2, 9, 5), mask=m)
        hip_ops.blend(out, pb, 1, (3, -1, -3), mask=m)
        np.testing.assert_allclose(out.cpu().numpy(), out_np,
                                   rtol=1e-6, atol=1e-7)

    def test_build_chunk_mask_vs_oracle(self, hip_ops):
        from oracle.inference import build_chunk_mask, patch_slices_list
        from oracle.patch_mask import make_patch_mask
        ps, ov = (10, 32, 32), (2, 8, 8)
        size = (20, 68, 72)
        pm = make_patch_mask(ps, ov)
        slices = patch_slices_list(size, ps, ov)
        ref = build_chunk_mask(size, (0, 0, 0), slices, pm)
        offsets = np.array([o for _, o in slices], dtype=np.int32)
        pm_t = torch.from_numpy(pm.copy()).cuda()
        got = hip_ops.build_chunk_mask(size, pm_t, offsets).cpu().numpy()
        np.testing.assert_allclose(got, ref, rtol=1e-6, atol=0)

    def test_multiply_mask_and_max(self, hip_ops):
        rng = np.random.RandomState(4)
        out = rng.rand(3, 9, 11, 13).astype(np.float32)
        mask = rng.rand(9, 11, 13).astype(np.float32)
        t = torch.from_numpy(out.copy()).cuda()
        m = torch.from_numpy(mask).cuda()
        hip_ops.multiply_mask(t, m)
        np.testing.assert_array_equal(t.cpu().numpy(), out * mask)
        assert abs(hip_ops.max(t) - (out * mask).max()) < 1e-7
        # negative values handled by the ordered-bits atomic max
        neg = torch.full((1000,), -3.5, device='cuda')
        neg[123] = -0.25
        assert hip_ops.max(neg) == -0.25

    def test_crop_margin(self, hip_ops):
        rng = np.random.RandomState(5)
        arr = rng.rand(3, 12, 14, 16).astype(np.float32)
        t = torch.from_numpy(arr).cuda()
        got = hip_ops.crop_margin(t, [1, 2, 3, 2, 1, 0]).cpu().numpy()
        np.testing.assert_array_equal(got, arr[:, 1:-2, 2:-1, 3:])
        got4 = hip_ops.crop_margin(t, [1, 1, 4, 1, 1, 4]).cpu().numpy()
        np.testing.assert_array_equal(got4, arr[:, 1:-1, 1:-1, 4:-4])

    def test_mask_using_last_channel(self, hip_ops):
        rng = np.random.RandomState(6)
        arr = rng.rand(4, 7, 9, 11).astype(np.float32)
        t = torch.from_numpy(arr).cuda()
        got = hip_ops.mask_using_last_channel(t, 0.3).cpu().numpy()
        keep = arr[-1] < 0.3
        np.testing.assert_array_equal(got, arr[:3] * keep)

    def test_reciprocal(self, cfx):
        rng = np.random.RandomState(7)
        arr = (rng.rand(1000).astype(np.float32) + 1e-6) * 10
        t = torch.from_numpy(arr.copy()).cuda()
        cfx.reciprocal(t.data_ptr(), t.numel())
        cfx.sync()
        np.testing.assert_array_equal(t.cpu().numpy(),
                                      (1.0 / arr).astype(np.float32))


# --------------------------------------------------------------------------
# end-to-end inferencer parity (HIP path vs reference goldens)
# --------------------------------------------------------------------------
def _hip_inferencer(**kw):
    from chunkflow_amd.inferencer import Inferencer
    inf = Inferencer(kw.pop('model', None), kw.pop('weights', None),
                     kw.pop('patch_size', (10, 32, 32)),
                     output_patch_overlap=kw.pop('overlap', (2, 8, 8)),
                     compute_device='cuda:0', **kw)
    assert inf.ops.is_hip
    return inf


class TestInferencerGPU:
    def test_identity_golden(self, golden):
        from chunkflow_amd.chunk import Chunk
        _, arrays = golden
        inf = _hip_inferencer(framework='identity', num_output_channels=3,
                              batch_size=3, mask_output_chunk=True)
        out = inf(Chunk(arrays['e2e_input_u8'].copy()))
        assert out.is_device
        np.testing.assert_allclose(out.numpy().array,
                                   arrays['e2e_identity_out'],
                                   rtol=1e-5, atol=1e-6)

    def test_identity_offset_golden(self, golden):
        from chunkflow_amd.chunk import Chunk
        _, arrays = golden
        inf = _hip_inferencer(framework='identity', num_output_channels=3,
                              batch_size=4, mask_output_chunk=True)
        out = inf(Chunk(arrays['e2e_input_u8'].copy(),
                        voxel_offset=(7, 11, 13)))
        np.testing.assert_allclose(out.numpy().array,
                                   arrays['e2e_identity_offset_out'],
                                   rtol=1e-5, atol=1e-6)

    def test_identity_myelin_golden(self, golden):
        from chunkflow_amd.chunk import Chunk
        _, arrays = golden
        inf = _hip_inferencer(framework='identity', num_output_channels=4,
                              batch_size=3, mask_output_chunk=True,
                              mask_myelin_threshold=0.3)
        out = inf(Chunk(arrays['e2e_input_u8'].copy()))
        np.testing.assert_allclose(out.numpy().array,
                                   arrays['e2e_identity_myelin_out'],
                                   rtol=1e-5, atol=1e-6)

    def test_pytorch_conv_golden_1e4(self, golden, golden_dir):
        """GPU conv (MIOpen) + HIP blend vs the torch-CPU reference output:
        the north-star 1e-4 fp32 gate."""
        from chunkflow_amd.chunk import Chunk
        _, arrays = golden
        inf = _hip_inferencer(
            model=os.path.join(golden_dir, 'ref_model.py'),
            weights=os.path.join(golden_dir, 'ref_model_weights.pt'),
            framework='pytorch', num_output_channels=3, batch_size=1,
            mask_output_chunk=True)
        out = inf(Chunk(arrays['e2e_input_u8'].copy()))
        np.testing.assert_allclose(out.numpy().array,
                                   arrays['e2e_pytorch_out'],
                                   rtol=1e-4, atol=1e-4)

    def test_nonaligned_identity_1e5(self):
        """The reference's strongest parity case shape
        (test_non_aligned_input_chunk, odd sizes, rtol/atol 1e-5): identity
        through the full masked path == input/255."""
        from chunkflow_amd.chunk import Chunk
        rng = np.random.RandomState(8)
        arr = rng.randint(0, 256, size=(28, 145, 151), dtype=np.uint8)
        inf = _hip_inferencer(patch_size=(10, 64, 64), overlap=(2, 16, 16),
                              framework='identity', num_output_channels=3,
                              batch_size=5, mask_output_chunk=True)
        out = inf(Chunk(arr)).numpy().array
        expect = arr.astype(np.float32) / 255.0
        for c in range(3):
            np.testing.assert_allclose(out[c], expect, rtol=1e-5, atol=1e-5)

    def test_aligned_mode_ref_geometry(self):
        """The reference's aligned test geometry (test_aligned_input_size:
        18x224x224, patch 10x128x128, overlap 2x32x32, no chunk mask)."""
        from chunkflow_amd.chunk import Chunk
        chunk = Chunk.create(size=(18, 224, 224), dtype='uint8',
                             pattern='sin')
        inf = _hip_inferencer(patch_size=(10, 128, 128),
                              overlap=(2, 32, 32), framework='identity',
                              num_output_channels=3, batch_size=1,
                              mask_output_chunk=False,
                              input_size=(18, 224, 224))
        out = inf(chunk)
        assert out.shape == (3, 14, 160, 160)
        expect = chunk.array.astype(np.float32)[2:-2, 32:-32, 32:-32] / 255.0
        np.testing.assert_allclose(out.numpy().array[0], expect,
                                   atol=1.0 / 255.0)

    def test_f16_aligned(self):
        """f16 output dtype in aligned mode (the only f16 mode the reference
        survives — SURVEY.md A.1); rtol 1e-3 like the reference f16 test."""
        from chunkflow_amd.chunk import Chunk
        chunk = Chunk.create(size=(18, 56, 56), dtype='uint8',
                             pattern='sin')
        inf = _hip_inferencer(patch_size=(10, 32, 32), overlap=(2, 8, 8),
                              framework='identity', num_output_channels=1,
                              batch_size=5, dtype='float16',
                              mask_output_chunk=False,
                              input_size=(18, 56, 56))
        out = inf(chunk)
        assert out.numpy().array.dtype == np.float16
        expect = chunk.array.astype(np.float32)[2:-2, 8:-8, 8:-8] / 255.0
        np.testing.assert_allclose(
            out.numpy().array[0].astype(np.float32), expect,
            rtol=1e-3, atol=2.0 / 255.0)

    def test_universal_plugin_gpu(self, golden, tmp_path):
        from chunkflow_amd.chunk import Chunk
        _, arrays = golden
        plugin = tmp_path / 'universal_identity.py'
        plugin.write_text(
            'import numpy as np\n'
            'class PatchInferencer:\n'
            '    def __init__(self, weight_path, mask):\n'
            '        self.mask = mask\n'
            '    def __call__(self, patch):\n'
            '        out = np.repeat(patch.astype(np.float32), 3, axis=1)\n'
            '        return out * self.mask\n')
        inf = _hip_inferencer(model=str(plugin), framework='universal',
                              num_output_channels=3, batch_size=3,
                              mask_output_chunk=True)
        out = inf(Chunk(arrays['e2e_input_u8'].copy()))
        np.testing.assert_allclose(out.numpy().array,
                                   arrays['e2e_identity_out'],
                                   rtol=1e-5, atol=1e-6)

    def test_config2_identity_512(self):
        """Config-2 geometry (512^3, patch 20x256x256, ov 4x64x64, 288
        patches, batch 12) with the identity engine: masked output ==
        input/255 at 1e-5 (SURVEY.md A.1 measured the reference at max err
        5.4e-7 on this exact case)."""
        from chunkflow_amd.chunk import Chunk
        chunk = Chunk.create(size=(512, 512, 512), dtype='uint8',
                             pattern='sin')
        inf = _hip_inferencer(patch_size=(20, 256, 256),
                              overlap=(4, 64, 64), framework='identity',
                              num_output_channels=3, batch_size=12,
                              mask_output_chunk=True)
        out = inf(Chunk(chunk.array))
        got = out.numpy().array
        expect = chunk.array.astype(np.float32) / 255.0
        err = np.abs(got[0] - expect).max()
        assert err < 1e-5, f'max err {err}'


def test_cli_gpu_pipeline(tmp_path, golden):
    """create-chunk -> inference (identity, HIP) -> crop-margin (HIP) ->
    save-npy through the real CLI."""
    from click.testing import CliRunner
    from chunkflow_amd.flow import main
    out = tmp_path / 'out.npy'
    r = CliRunner().invoke(main, [
        'create-chunk', '--size', '20', '68', '72', '--dtype', 'uint8',
        '--pattern', 'sin',
        'inference', '-s', '10', '32', '32',
        '--output-patch-overlap', '2', '8', '8',
        '--framework', 'identity', '--batch-size', '3',
        '--num-output-channels', '3', '--mask-output-chunk',
        'crop-margin', '-m', '1', '2', '3', '1', '2', '3',
        'save-npy', '-f', str(out)], catch_exceptions=False)
    assert r.exit_code == 0, r.output
    got = np.load(out)
    assert got.shape == (3, 18, 64, 66)
    _, arrays = golden
    sin = arrays['sin_20x68x72_u8'].astype(np.float32) / 255.0
    np.testing.assert_allclose(got[0], sin[1:-1, 2:-2, 3:-3],
                               rtol=1e-5, atol=1e-5)


def test_patch_num_aligned_gpu():
    from chunkflow_amd.chunk import Chunk
    chunk = Chunk.create(size=(18, 56, 56), dtype='uint8', pattern='sin')
    inf = _hip_inferencer(patch_size=(10, 32, 32), overlap=(2, 8, 8),
                          framework='identity', num_output_channels=2,
                          batch_size=5, dtype='float16',
                          mask_output_chunk=False, patch_num=(2, 2, 2))
    out = inf(chunk)
    assert out.shape == (2, 14, 40, 40)
    expect = chunk.array.astype(np.float32)[2:-2, 8:-8, 8:-8] / 255.0
    np.testing.assert_allclose(out.numpy().array[0].astype(np.float32),
                               expect, rtol=1e-3, atol=2.0 / 255.0)


def test_nonaligned_vs_oracle_gpu():
    from chunkflow_amd.chunk import Chunk
    from oracle import oracle_inference
    rng = np.random.RandomState(11)
    arr = rng.randint(0, 256, size=(23, 71, 66), dtype=np.uint8)
    inf = _hip_inferencer(patch_size=(12, 32, 32), overlap=(4, 8, 8),
                          framework='identity', num_output_channels=2,
                          batch_size=7, mask_output_chunk=True)
    out = inf(Chunk(arr.copy(), voxel_offset=(1, 2, 3)))
    ref = oracle_inference(arr, (12, 32, 32), (4, 8, 8),
                           num_output_channels=2, batch_size=7,
                           offset=(1, 2, 3))
    np.testing.assert_allclose(out.numpy().array, ref, rtol=1e-5, atol=1e-6)


def test_cli_gpu_config4_chain(tmp_path):
    """BASELINE config 4 shape at reduced size: create-chunk ->
    normalize-intensity (HIP) -> inference (HIP) -> crop-margin (HIP) ->
    connected-components -> save-npy, all chained through the CLI.
    normalize-intensity feeds [-1,1] floats into inference (the reference
    chain does the same: float input skips the /255 normalize)."""
    from click.testing import CliRunner
    from chunkflow_amd.flow import main
    out = tmp_path / 'seg.npy'
    r = CliRunner().invoke(main, [
        'create-chunk', '--size', '24', '80', '96', '--dtype', 'uint8',
        '--pattern', 'sin',
        'normalize-intensity',
        'inference', '-s', '12', '32', '32',
        '--output-patch-overlap', '4', '8', '8',
        '--framework', 'identity', '--batch-size', '6',
        '--num-output-channels', '3', '--mask-output-chunk',
        'crop-margin', '-m', '2', '4', '4', '2', '4', '4',
        'connected-components', '-t', '0.0', '-c', '6',
        'save-npy', '-f', str(out)], catch_exceptions=False)
    assert r.exit_code == 0, r.output
    seg = np.load(out)
    assert seg.shape == (20, 72, 88)
    assert seg.dtype in (np.uint32, np.int32)
    assert seg.max() >= 1

    # the same chain against the oracle: normalize-intensity then identity
    # inference == normalized input, cropped
    from oracle import oracle_inference
    from chunkflow_amd.chunk import Chunk
    sin = Chunk.create(size=(24, 80, 96), dtype='uint8',
                       pattern='sin').array
    norm = sin.astype(np.float32) / 127.5 - 1.0
    ref = oracle_inference(norm, (12, 32, 32), (4, 8, 8),
                           num_output_channels=3, batch_size=6)
    ref_crop = ref[:, 2:-2, 4:-4, 4:-4]
    from scipy import ndimage
    labels, _ = ndimage.label(
        ref_crop[0] > 0.0, structure=ndimage.generate_binary_structure(3, 1))
    assert seg.max() == labels.max()


def test_bf16_engine_close_to_f32(golden, golden_dir):
    """--dtype bfloat16 conv path (config-5 wiring): bf16 MFMA forward +
    f32 accumulation stays within bf16 tolerance of the f32 golden."""
    from chunkflow_amd.chunk import Chunk
    _, arrays = golden
    inf = _hip_inferencer(
        model=os.path.join(golden_dir, 'ref_model.py'),
        weights=os.path.join(golden_dir, 'ref_model_weights.pt'),
        framework='pytorch', num_output_channels=3, batch_size=1,
        dtype='bfloat16', mask_output_chunk=True)
    out = inf(Chunk(arrays['e2e_input_u8'].copy()))
    np.testing.assert_allclose(out.numpy().array,
                               arrays['e2e_pytorch_out'],
                               rtol=0.05, atol=0.02)


def test_scalar_paths_non_x4_geometry():
    """x dims not divisible by 4 exercise every kernel's scalar fallback
    (blend_batch, extract, crop, maskmul via cfx_max path)."""
    from chunkflow_amd.chunk import Chunk
    from oracle import oracle_inference
    rng = np.random.RandomState(21)
    arr = rng.randint(0, 256, size=(17, 45, 51), dtype=np.uint8)
    inf = _hip_inferencer(patch_size=(9, 21, 27), overlap=(3, 7, 9),
                          framework='identity', num_output_channels=2,
                          batch_size=5, mask_output_chunk=True)
    out = inf(Chunk(arr.copy()))
    ref = oracle_inference(arr, (9, 21, 27), (3, 7, 9),
                           num_output_channels=2, batch_size=5)
    np.testing.assert_allclose(out.numpy().array, ref, rtol=1e-5, atol=1e-6)


def test_f16_masked_mode_no_crash():
    """--dtype float16 + --mask-output-chunk CRASHES the reference (f16
    reciprocal overflow -> NaN -> assert). This build
    accumulates in f32 and only casts the output — documented divergence."""
    from chunkflow_amd.chunk import Chunk
    chunk = Chunk.create(size=(28, 100, 100), dtype='uint8', pattern='sin')
    inf = _hip_inferencer(patch_size=(20, 64, 64), overlap=(4, 16, 16),
                          framework='identity', num_output_channels=1,
                          batch_size=4, dtype='float16',
                          mask_output_chunk=True)
    out = inf(chunk)
    got = out.numpy().array
    assert got.dtype == np.float16
    assert np.isfinite(got).all()
    expect = chunk.array.astype(np.float32) / 255.0
    np.testing.assert_allclose(got[0].astype(np.float32), expect,
                               rtol=2e-3, atol=2.0 / 255.0)


class TestConnectedComponentsGPU:
    """GPU union-find vs scipy.ndimage.label: identical labels (same
    partition AND same first-encounter numbering)."""

    def _check(self, vol, threshold, connectivity):
        from scipy import ndimage
        from chunkflow_amd.chunk import Chunk
        from chunkflow_amd.connected import connected_component, _STRUCTS
        dev = Chunk(torch.from_numpy(vol.copy()).cuda())
        got = connected_component(dev, threshold=threshold,
                                  connectivity=connectivity)
        got_np = got.numpy().array
        seg = vol > threshold if threshold is not None else vol != 0
        ref, nref = ndimage.label(seg, structure=_STRUCTS[connectivity])
        np.testing.assert_array_equal(got_np.astype(np.int64), ref)
        assert int(got_np.max()) == nref

    def test_random_6(self):
        rng = np.random.RandomState(30)
        vol = rng.rand(40, 50, 60).astype(np.float32)
        self._check(vol, 0.7, 6)

    def test_random_18_26(self):
        rng = np.random.RandomState(31)
        vol = rng.rand(30, 40, 50).astype(np.float32)
        self._check(vol, 0.6, 18)
        self._check(vol, 0.6, 26)

    def test_dense_blobs(self):
        # low threshold -> few huge components spanning the volume
        from chunkflow_amd.chunk import Chunk as C
        vol = C.create(size=(48, 64, 64), dtype='uint8',
                       pattern='sin').array.astype(np.float32) / 255.0
        self._check(vol, 0.5, 6)
        self._check(vol, 0.5, 26)

    def test_nonzero_u8_path(self):
        rng = np.random.RandomState(32)
        vol = (rng.rand(20, 30, 40) > 0.6).astype(np.uint8)
        self._check(vol, None, 6)

    def test_all_foreground_and_empty(self):
        self._check(np.ones((8, 8, 8), dtype=np.float32), 0.5, 6)
        self._check(np.zeros((8, 8, 8), dtype=np.float32), 0.5, 6)


def test_normalize_contrast_gpu_exact():
    """Device histogram+LUT path == host numpy path bit-for-bit (integer
    pipeline, deterministic histograms)."""
    from chunkflow_amd.chunk import Chunk
    from chunkflow_amd.contrast import normalize_contrast
    rng = np.random.RandomState(10)
    for arr in ((rng.rand(6, 64, 72) * 256).astype(np.uint8),
                (rng.rand(5, 33, 41) * 60 + 90).astype(np.uint8)):
        host = normalize_contrast(Chunk(arr.copy())).numpy().array
        dev = normalize_contrast(
            Chunk(torch.from_numpy(arr.copy()).cuda()))
        np.testing.assert_array_equal(dev.numpy().array, host)


def test_normalize_contrast_cli_gpu(tmp_path, golden):
    from click.testing import CliRunner
    from chunkflow_amd.flow import main
    out = tmp_path / 'nc.npy'
    r = CliRunner().invoke(main, [
        'create-chunk', '--size', '8', '40', '48', '--dtype', 'uint8',
        '--pattern', 'sin',
        'normalize-contrast',
        'save-npy', '-f', str(out)], catch_exceptions=False)
    assert r.exit_code == 0, r.output
    got = np.load(out)
    from oracle.contrast import oracle_normalize_contrast
    from chunkflow_amd.chunk import Chunk as C
    sin = C.create(size=(8, 40, 48), dtype='uint8', pattern='sin').array
    np.testing.assert_array_equal(got, oracle_normalize_contrast(sin))


def test_fuzz_geometries_vs_oracle():
    """Randomized geometry fuzz: odd sizes, tail clamps, duplicate
    positions, varying batch/channels/offsets — HIP path vs oracle at
    1e-5 on six cases."""
    from chunkflow_amd.chunk import Chunk
    from oracle import oracle_inference
    rng = np.random.RandomState(99)
    for case in range(6):
        pz = int(rng.randint(6, 14))
        py = int(rng.randint(16, 40))
        px = int(rng.randint(16, 40))
        ov = (max(1, pz // 4), max(2, py // 4), max(2, px // 4))
        size = (pz + int(rng.randint(0, 2 * pz)),
                py + int(rng.randint(0, 2 * py)),
                px + int(rng.randint(0, 2 * px)))
        nc = int(rng.randint(1, 4))
        bs = int(rng.randint(1, 7))
        off = tuple(int(v) for v in rng.randint(0, 20, 3))
        arr = rng.randint(0, 256, size=size).astype(np.uint8)
        inf = _hip_inferencer(patch_size=(pz, py, px), overlap=ov,
                              framework='identity',
                              num_output_channels=nc, batch_size=bs,
                              mask_output_chunk=True)
        got = inf(Chunk(arr.copy(), voxel_offset=off)).numpy().array
        ref = oracle_inference(arr, (pz, py, px), ov,
                               num_output_channels=nc, batch_size=bs,
                               offset=off)
        np.testing.assert_allclose(
            got, ref, rtol=1e-5, atol=1e-6,
            err_msg=f'case {case}: size={size} patch={(pz,py,px)} '
                    f'ov={ov} nc={nc} bs={bs} off={off}')


def test_aligned_mode_vs_oracle():
    """Aligned mode (no chunk mask) vs the aligned-mode oracle."""
    from chunkflow_amd.chunk import Chunk
    from oracle.inference import oracle_inference_aligned
    rng = np.random.RandomState(41)
    ps, ov = (8, 24, 24), (2, 8, 8)
    size = (8 + 2 * 6, 24 + 2 * 16, 24 + 16)  # aligned: (i-o)%s==0
    arr = rng.randint(0, 256, size=size).astype(np.uint8)
    inf = _hip_inferencer(patch_size=ps, overlap=ov, framework='identity',
                          num_output_channels=2, batch_size=3,
                          mask_output_chunk=False, input_size=size)
    got = inf(Chunk(arr.copy())).numpy().array
    ref = oracle_inference_aligned(arr, ps, ov, num_output_channels=2,
                                   batch_size=3)
    np.testing.assert_allclose(got, ref, rtol=1e-5, atol=1e-6)


class TestFastConv:
    """Hand-written MFMA 3x3x3 conv (csrc/conv.hip): parity vs torch conv3d
    (both paths and all instantiated widths). Perf is currently 92% of
    MIOpen's ck kernels, so CFX_FASTCONV stays opt-in."""

    def _check(self, C, shape=(2, 8, 40, 48), w32=False, elu=False,
               residual=False):
        import torch.nn.functional as F
        from chunkflow_amd.fastconv import get_cfx
        cl = torch.channels_last_3d
        torch.manual_seed(C)
        n, d, h, w = shape
        x = torch.randn(n, C, d, h, w, device='cuda').contiguous(
            memory_format=cl)
        wt = torch.randn(C, C, 3, 3, 3, device='cuda') * 0.05
        bias = torch.randn(C, device='cuda') * 0.1
        wtap = wt.permute(2, 3, 4, 1, 0).reshape(27, C, C).contiguous()
        res = torch.randn_like(x).contiguous(memory_format=cl) \
            if residual else None
        out = torch.empty_like(x)
        get_cfx(0).conv3_ndhwc(
            x.data_ptr(), wtap.data_ptr(), bias.data_ptr(),
            res.data_ptr() if res is not None else None, out.data_ptr(),
            n, d, h, w, C, C, do_elu=elu, w32=w32)
        ref = F.conv3d(x, wt, bias, padding=1)
        if res is not None:
            ref = ref + res
        if elu:
            ref = torch.nn.functional.elu(ref)
        torch.cuda.synchronize()
        np.testing.assert_allclose(out.cpu().numpy(), ref.cpu().numpy(),
                                   rtol=1e-4, atol=5e-5)

    def test_widths(self):
        for C in (28, 36, 48, 64):
            self._check(C)

    def test_w32_path(self):
        self._check(28, w32=True)

    def test_fused_elu_residual(self):
        self._check(28, elu=True, residual=True)
        self._check(48, elu=True)

    def test_odd_extents(self):
        self._check(28, shape=(1, 5, 23, 37))
        self._check(64, shape=(1, 3, 17, 19))

    def test_engine_fastconv_end_to_end(self, monkeypatch):
        """RSUNet through the engine with fastconv ON vs OFF at 1e-4."""
        from chunkflow_amd.chunk import Chunk
        chunk = Chunk.create(size=(20, 128, 128), dtype='uint8',
                             pattern='sin')
        model = os.path.join(REPO, 'examples', 'nets', 'rsunet.py')

        def run():
            inf = _hip_inferencer(model=model, framework='pytorch',
                                  patch_size=(20, 64, 64),
                                  overlap=(4, 16, 16),
                                  num_output_channels=3, batch_size=2,
                                  mask_output_chunk=True)
            return inf(chunk).numpy().array

        monkeypatch.setenv('CFX_FASTCONV', '0')
        base = run()
        monkeypatch.setenv('CFX_FASTCONV', '1')
        fast = run()
        np.testing.assert_allclose(fast, base, rtol=1e-4, atol=1e-4)


def test_augment_identity_roundtrip_gpu():
    """--augment on the HIP path (TTA forward/backward on device)."""
    from chunkflow_amd.chunk import Chunk
    rng = np.random.RandomState(77)
    arr = rng.randint(0, 256, size=(16, 40, 40), dtype=np.uint8)
    kw = dict(overlap=(2, 8, 8), framework='identity',
              num_output_channels=2, batch_size=3, mask_output_chunk=True)
    plain = _hip_inferencer(patch_size=(8, 24, 24), **kw)(Chunk(arr.copy()))
    aug = _hip_inferencer(patch_size=(8, 24, 24), augment=True,
                          **kw)(Chunk(arr.copy()))
    np.testing.assert_allclose(aug.numpy().array, plain.numpy().array,
                               rtol=1e-5, atol=1e-6)


def test_zring_parity_and_fusion():
    """Persistent-z ring conv (the variant that beats MIOpen at C=28):
    parity incl. fused ELU+residual epilogue."""
    import torch.nn.functional as F
    from chunkflow_amd.fastconv import get_cfx
    cl = torch.channels_last_3d
    torch.manual_seed(5)
    n, C, d, h, w = 2, 28, 9, 37, 41
    x = torch.randn(n, C, d, h, w, device='cuda').contiguous(
        memory_format=cl)
    wt = torch.randn(C, C, 3, 3, 3, device='cuda') * 0.05
    bias = torch.randn(C, device='cuda') * 0.1
    wtap = wt.permute(2, 3, 4, 1, 0).reshape(27, C, C).contiguous()
    res = torch.randn_like(x).contiguous(memory_format=cl)
    out = torch.empty_like(x)
    get_cfx(0).conv3_ndhwc(x.data_ptr(), wtap.data_ptr(), bias.data_ptr(),
                           res.data_ptr(), out.data_ptr(), n, d, h, w, C,
                           C, do_elu=True, zring=True)
    ref = torch.nn.functional.elu(F.conv3d(x, wt, bias, padding=1) + res)
    torch.cuda.synchronize()
    np.testing.assert_allclose(out.cpu().numpy(), ref.cpu().numpy(),
                               rtol=1e-4, atol=5e-5)


def test_bf16_zring_parity():
    """bf16 z-ring conv vs torch bf16 conv (both accumulate f32): agree
    within bf16 rounding of the activations."""
    import torch.nn.functional as F
    from chunkflow_amd.fastconv import CfxConv3dBF16
    cl = torch.channels_last_3d
    torch.manual_seed(7)
    conv = torch.nn.Conv3d(28, 28, 3, padding=1).cuda()
    m = CfxConv3dBF16(conv).cuda()
    x = (torch.randn(2, 28, 7, 33, 41, device='cuda') * 0.5) \
        .to(torch.bfloat16).contiguous(memory_format=cl)
    with torch.no_grad():
        got = m(x).float()
        ref = F.conv3d(x.float(), conv.weight.float()
                       .to(torch.bfloat16).float(),
                       conv.bias.float(), padding=1)
    np.testing.assert_allclose(got.cpu().numpy(), ref.cpu().numpy(),
                               rtol=0.05, atol=0.03)


def test_bf16_engine_fastconv_close_to_f32(golden_dir):
    """config-5 wiring: the bf16 engine with bf16 fastconv stays within
    bf16 tolerance of the f32 MIOpen path on a small RSUNet run."""
    from chunkflow_amd.chunk import Chunk
    chunk = Chunk.create(size=(20, 128, 128), dtype='uint8', pattern='sin')
    model = os.path.join(REPO, 'examples', 'nets', 'rsunet.py')

    def run(dtype, fast):
        os.environ['CFX_FASTCONV'] = fast
        try:
            inf = _hip_inferencer(model=model, framework='pytorch',
                                  patch_size=(20, 64, 64),
                                  overlap=(4, 16, 16), dtype=dtype,
                                  num_output_channels=3, batch_size=2,
                                  mask_output_chunk=True)
            return inf(chunk).numpy().array
        finally:
            os.environ.pop('CFX_FASTCONV', None)

    base = run('float32', '0')
    fast_bf16 = run('bfloat16', '1')
    np.testing.assert_allclose(fast_bf16, base, rtol=0.05, atol=0.03)


# --------------------------------------------------------------------------
# round-2 reference-golden pins: TTA and the full benchmark RSUNet
# --------------------------------------------------------------------------
def test_augment_identity_reference_golden(golden):
    """TTA (augment=True) on GPU vs the live-reference golden
    (transform.py:114-156 semantics), identity engine."""
    from chunkflow_amd.chunk import Chunk
    _, arrays = golden
    inf = _hip_inferencer(framework='identity', num_output_channels=3,
                          batch_size=3, mask_output_chunk=True,
                          augment=True)
    out = inf(Chunk(arrays['e2e_input_u8'].copy()))
    np.testing.assert_allclose(out.numpy().array,
                               arrays['e2e_identity_augment_out'],
                               rtol=1e-5, atol=1e-6)


def test_augment_pytorch_reference_golden(golden, golden_dir):
    """TTA through the conv engine on GPU (the flips/transposes actually
    change conv outputs) vs the live-reference golden, 1e-4 fp32."""
    from chunkflow_amd.chunk import Chunk
    _, arrays = golden
    inf = _hip_inferencer(
        model=os.path.join(golden_dir, 'ref_model.py'),
        weights=os.path.join(golden_dir, 'ref_model_weights.pt'),
        framework='pytorch', num_output_channels=3, batch_size=1,
        mask_output_chunk=True, augment=True)
    out = inf(Chunk(arrays['e2e_input_u8'].copy()))
    np.testing.assert_allclose(out.numpy().array,
                               arrays['e2e_pytorch_augment_out'],
                               rtol=1e-4, atol=1e-4)


def test_rsunet_config2_conv_parity_1e4(golden, golden_dir):
    """the REAL benchmark RSUNet through the full GPU
    path (fastconv MFMA rings + MIOpen + HIP tiler/blend) vs the live
    reference Inferencer on torch-CPU at a config-2-like geometry
    (64x256x256, patch 20x128x128, overlap 4x32x32), <=1e-4 fp32."""
    from chunkflow_amd.chunk import Chunk
    meta, arrays = golden
    case = meta['cases']['rsunet_64x256x256']
    rs_in = Chunk.create(size=(64, 256, 256), dtype='uint8', pattern='sin')
    inf = _hip_inferencer(
        model=os.path.join(REPO, 'examples', 'nets', 'rsunet.py'),
        weights=os.path.join(golden_dir, 'rsunet_weights.pt'),
        patch_size=(20, 128, 128), overlap=(4, 32, 32),
        framework='pytorch', num_output_channels=3, batch_size=4,
        mask_output_chunk=True)
    out = np.asarray(inf(rs_in).numpy().array)
    assert out.shape == (3, 64, 256, 256)
    np.testing.assert_allclose(out[:, ::4, ::8, ::8],
                               arrays['rsunet_64x256x256_sub'],
                               rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(
        out.ravel()[arrays['rsunet_64x256x256_sample_idx']],
        arrays['rsunet_64x256x256_sample_val'], rtol=1e-4, atol=1e-4)
    assert abs(out.astype(np.float64).sum() - case['sum_f64']) < 200.0
    assert case['min'] - 1e-3 < out.min() and out.max() < case['max'] + 1e-3


# --------------------------------------------------------------------------
# round-2 kernels: up/down-sampling convs + sliced bf16 ring (C=36/48)
# --------------------------------------------------------------------------
@pytest.mark.parametrize('C,K,bias', [(36, 28, True), (48, 36, True),
                                      (64, 48, True), (28, 28, False)])
def test_upconv_2x2_f32_vs_torch(C, K, bias):
    """csrc/updown.hip transposed conv vs torch ConvTranspose3d, f32."""
    from chunkflow_amd.fastconv import CfxUpConv3d
    torch.manual_seed(3)
    conv = torch.nn.ConvTranspose3d(C, K, (1, 2, 2), stride=(1, 2, 2),
                                    bias=bias).cuda()
    x = torch.randn(2, C, 3, 13, 21, device='cuda') \
        .contiguous(memory_format=torch.channels_last_3d)
    want = conv(x)
    got = CfxUpConv3d(conv, 0, bf16=False).cuda()(x)
    assert got.shape == want.shape
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize('C,K', [(36, 28), (64, 48)])
def test_upconv_2x2_bf16_vs_torch(C, K):
    from chunkflow_amd.fastconv import CfxUpConv3d
    torch.manual_seed(4)
    conv = torch.nn.ConvTranspose3d(C, K, (1, 2, 2),
                                    stride=(1, 2, 2)).cuda()
    x = torch.randn(2, C, 3, 9, 33, device='cuda')
    want = conv.to(torch.bfloat16)(
        x.to(torch.bfloat16)
        .contiguous(memory_format=torch.channels_last_3d)).float()
    got = CfxUpConv3d(conv, 0, bf16=True).cuda()(
        x.to(torch.bfloat16)
        .contiguous(memory_format=torch.channels_last_3d)).float()
    torch.testing.assert_close(got, want, rtol=0.05, atol=0.05)


@pytest.mark.parametrize('C,K,bias', [(28, 36, True), (36, 48, True),
                                      (48, 64, False)])
def test_downconv_2x2_f32_vs_torch(C, K, bias):
    from chunkflow_amd.fastconv import CfxDownConv3d
    torch.manual_seed(5)
    conv = torch.nn.Conv3d(C, K, (1, 2, 2), stride=(1, 2, 2),
                           bias=bias).cuda()
    x = torch.randn(2, C, 3, 14, 22, device='cuda') \
        .contiguous(memory_format=torch.channels_last_3d)
    want = conv(x)
    got = CfxDownConv3d(conv, 0, bf16=False).cuda()(x)
    assert got.shape == want.shape
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


def test_downconv_2x2_bf16_vs_torch():
    from chunkflow_amd.fastconv import CfxDownConv3d
    torch.manual_seed(6)
    conv = torch.nn.Conv3d(36, 48, (1, 2, 2), stride=(1, 2, 2)).cuda()
    x = torch.randn(1, 36, 4, 16, 40, device='cuda').to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last_3d)
    want = conv.to(torch.bfloat16)(x).float()
    got = CfxDownConv3d(conv, 0, bf16=True).cuda()(x).float()
    torch.testing.assert_close(got, want, rtol=0.05, atol=0.05)


@pytest.mark.parametrize('C', [36, 48])
def test_bf16_sliced_ring_vs_f32_conv(C):
    """The 4-launch sliced bf16 ring (c-halves x j-tiles) vs the same conv
    in f32: the bf16 engine contract (partial sums round through bf16)."""
    from chunkflow_amd.fastconv import CfxConv3dBF16
    torch.manual_seed(7)
    conv = torch.nn.Conv3d(C, C, 3, padding=1).cuda()
    x = torch.randn(2, C, 5, 37, 41, device='cuda') \
        .contiguous(memory_format=torch.channels_last_3d)
    want = conv(x).float()
    got = CfxConv3dBF16(conv, 0).cuda()._run(
        x.to(torch.bfloat16)
        .contiguous(memory_format=torch.channels_last_3d)).float()
    # bf16 fragments + bf16 inter-half rounding: bf16-level tolerance,
    # tight in the mean
    torch.testing.assert_close(got, want, rtol=0.08, atol=0.08)
    assert float((got - want).abs().mean()) < 0.02


@pytest.mark.parametrize('C', [36, 48])
def test_bf16_sliced_ring_residual_elu(C):
    """Fused bias+residual+ELU epilogue of the sliced schedule vs torch."""
    from chunkflow_amd.fastconv import CfxConv3dBF16
    torch.manual_seed(8)
    conv = torch.nn.Conv3d(C, C, 3, padding=1).cuda()
    x = torch.randn(1, C, 4, 18, 35, device='cuda')
    r = torch.randn(1, C, 4, 18, 35, device='cuda')
    want = torch.nn.functional.elu(conv(x) + r).float()
    xb = x.to(torch.bfloat16).cuda() \
        .contiguous(memory_format=torch.channels_last_3d)
    rb = r.to(torch.bfloat16).cuda() \
        .contiguous(memory_format=torch.channels_last_3d)
    got = CfxConv3dBF16(conv, 0).cuda()._run(xb, residual=rb,
                                             elu=True).float()
    torch.testing.assert_close(got, want, rtol=0.08, atol=0.08)
    assert float((got - want).abs().mean()) < 0.02


def test_rsunet_bf16_full_surgery_runs():
    """Full RSUNet bf16 surgery (rings 28/36/48 + up/down kernels) stays
    close to the torch-bf16 model output on a real patch shape."""
    import importlib.util
    from chunkflow_amd.fastconv import (maybe_accelerate_bf16,
                                        accelerate_updown)
    spec = importlib.util.spec_from_file_location(
        'rsunet_t', os.path.join(REPO, 'examples', 'nets', 'rsunet.py'))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    model = mod.InstantiatedModel.cuda().eval().to(torch.bfloat16) \
        .to(memory_format=torch.channels_last_3d)
    x = torch.randn(1, 1, 4, 64, 64, device='cuda').to(torch.bfloat16) \
        .contiguous(memory_format=torch.channels_last_3d)
    with torch.no_grad():
        want = model(x).float()
    import copy
    m2 = copy.deepcopy(model)
    n_rb = maybe_accelerate_bf16(m2, 0)
    n_ud = accelerate_updown(m2, 0, bf16=True)
    from chunkflow_amd.fastconv import BF16_WIDTHS, UP_SURGERY_WIDTHS
    want_rb = 2 * len([w for w in (28, 36, 48) if w in BF16_WIDTHS])
    assert n_rb == want_rb, (n_rb, want_rb)  # enc+dec blocks per width
    assert n_ud == len(UP_SURGERY_WIDTHS), n_ud  # measured-winning ups
    with torch.no_grad():
        got = m2(x).float()
    # sigmoid output in [0,1]: absolute tolerance
    torch.testing.assert_close(got, want, rtol=0.1, atol=0.03)


@pytest.mark.parametrize('bf16', [False, True])
def test_conv155_c1_vs_torch(bf16):
    """Single-channel (1,5,5) input-conv stencil vs torch/MIOpen."""
    from chunkflow_amd.fastconv import CfxConvIn155
    torch.manual_seed(9)
    conv = torch.nn.Conv3d(1, 28, (1, 5, 5), padding=(0, 2, 2)).cuda()
    x = torch.randn(2, 1, 3, 37, 83, device='cuda')
    dt = torch.bfloat16 if bf16 else torch.float32
    xc = x.to(dt).contiguous(memory_format=torch.channels_last_3d)
    want = conv.to(dt)(xc).float()
    got = CfxConvIn155(conv, 0, bf16=bf16).cuda()(xc).float()
    tol = 0.05 if bf16 else 1e-4
    torch.testing.assert_close(got, want, rtol=tol, atol=tol)


@pytest.mark.parametrize('bf16', [False, True])
def test_conv155_out_vs_torch(bf16):
    """Few-output-channel (1,5,5) output-conv kernel vs torch/MIOpen."""
    from chunkflow_amd.fastconv import CfxConvOut155
    torch.manual_seed(10)
    conv = torch.nn.Conv3d(28, 3, (1, 5, 5), padding=(0, 2, 2)).cuda()
    x = torch.randn(2, 28, 3, 41, 77, device='cuda')
    dt = torch.bfloat16 if bf16 else torch.float32
    xc = x.to(dt).contiguous(memory_format=torch.channels_last_3d)
    want = conv.to(dt)(xc).float()
    got = CfxConvOut155(conv, 0, bf16=bf16).cuda()(xc).float()
    tol = 0.05 if bf16 else 1e-4
    torch.testing.assert_close(got, want, rtol=tol, atol=tol)


def test_blend_reference_order_gpu(golden, monkeypatch):
    """The per-patch reference-order blend mode on GPU matches the grouped
    default at 1e-6 and the reference golden at 1e-5."""
    from chunkflow_amd.chunk import Chunk
    _, arrays = golden

    def run():
        inf = _hip_inferencer(framework='identity', num_output_channels=3,
                              batch_size=3, mask_output_chunk=True)
        return np.asarray(
            inf(Chunk(arrays['e2e_input_u8'].copy())).numpy().array)

    base = run()
    monkeypatch.setenv('CFX_BLEND_REFORDER', '1')
    ref_order = run()
    np.testing.assert_allclose(ref_order, base, rtol=1e-6, atol=1e-7)
    np.testing.assert_allclose(ref_order, arrays['e2e_identity_out'],
                               rtol=1e-5, atol=1e-6)


def test_cli_worker_stitch_gpu_world1(tmp_path, monkeypatch):
    """The config-3 CLI worker pipeline (generate-tasks sharding + stitch
    operator) on GPU at world 1: inference runs the HIP path per task and
    the stitch places each task's sub-volume correctly."""
    from click.testing import CliRunner
    from chunkflow_amd.flow import main
    for k in ('RANK', 'WORLD_SIZE', 'MASTER_ADDR', 'MASTER_PORT'):
        monkeypatch.delenv(k, raising=False)
    out = tmp_path / 'vol.npy'
    r = CliRunner().invoke(main, [
        'generate-tasks', '-c', '16', '40', '48',
        '--roi-size', '16', '40', '96',
        'create-chunk',
        'inference', '-s', '8', '16', '16',
        '--output-patch-overlap', '2', '4', '4',
        '--framework', 'identity', '--batch-size', '4',
        '--num-output-channels', '3', '--mask-output-chunk',
        'stitch', '-f', str(out)], catch_exceptions=False)
    assert r.exit_code == 0, r.output
    vol = np.load(out)
    assert vol.shape == (3, 16, 40, 96)
    # identity inference of the sin chunk == chunk/255 at every position;
    # create-chunk from bbox uses the same sin for both tasks
    from chunkflow_amd.chunk import Chunk
    sin = np.asarray(Chunk.create(size=(16, 40, 48), dtype='uint8',
                                  pattern='sin').array)
    ref = sin.astype(np.float32) / 255.0
    for t in range(2):
        np.testing.assert_allclose(vol[0, :, :, t * 48:(t + 1) * 48], ref,
                                   rtol=1e-5, atol=1e-6)


def test_upconv_fuzz_shapes():
    """Random-shape fuzz of the up-conv kernel vs torch, f32 (odd sizes,
    varied channels)."""
    from chunkflow_amd.fastconv import CfxUpConv3d
    rng = np.random.RandomState(11)
    for _ in range(6):
        C = int(rng.choice([3, 17, 36, 48, 64]))
        K = int(rng.choice([1, 5, 28, 33, 64]))
        n = int(rng.randint(1, 3))
        d, h, w = (int(rng.randint(1, 5)), int(rng.randint(1, 20)),
                   int(rng.randint(1, 70)))
        conv = torch.nn.ConvTranspose3d(C, K, (1, 2, 2),
                                        stride=(1, 2, 2)).cuda()
        x = torch.randn(n, C, d, h, w, device='cuda') \
            .contiguous(memory_format=torch.channels_last_3d)
        want = conv(x)
        got = CfxUpConv3d(conv, 0, bf16=False).cuda()(x)
        torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-4)


def test_conv155_fuzz_shapes():
    """Random-shape fuzz of the (1,5,5) conv_in/conv_out kernels, f32."""
    from chunkflow_amd.fastconv import CfxConvIn155, CfxConvOut155
    rng = np.random.RandomState(12)
    for _ in range(5):
        K = int(rng.choice([1, 7, 28, 32]))
        n, d = int(rng.randint(1, 3)), int(rng.randint(1, 4))
        h, w = int(rng.randint(1, 30)), int(rng.randint(1, 300))
        conv = torch.nn.Conv3d(1, K, (1, 5, 5), padding=(0, 2, 2)).cuda()
        x = torch.randn(n, 1, d, h, w, device='cuda') \
            .contiguous(memory_format=torch.channels_last_3d)
        torch.testing.assert_close(
            CfxConvIn155(conv, 0).cuda()(x), conv(x),
            rtol=1e-4, atol=1e-4)
    for _ in range(3):
        n, d = int(rng.randint(1, 3)), int(rng.randint(1, 4))
        h, w = int(rng.randint(1, 30)), int(rng.randint(1, 300))
        conv = torch.nn.Conv3d(28, 3, (1, 5, 5), padding=(0, 2, 2)).cuda()
        x = torch.randn(n, 28, d, h, w, device='cuda') \
            .contiguous(memory_format=torch.channels_last_3d)
        torch.testing.assert_close(
            CfxConvOut155(conv, 0).cuda()(x), conv(x),
            rtol=1e-4, atol=1e-4)


def test_sliced_ring_fuzz_shapes():
    """Random-shape fuzz of the sliced bf16 ring (C=36/48) vs torch f32
    at bf16 tolerance."""
    from chunkflow_amd.fastconv import CfxConv3dBF16
    rng = np.random.RandomState(13)
    for _ in range(4):
        C = int(rng.choice([36, 48]))
        n, d = int(rng.randint(1, 3)), int(rng.randint(1, 6))
        h, w = int(rng.randint(1, 40)), int(rng.randint(1, 40))
        conv = torch.nn.Conv3d(C, C, 3, padding=1).cuda()
        x = (torch.randn(n, C, d, h, w, device='cuda') * 0.3)
        want = conv(x).float()
        got = CfxConv3dBF16(conv, 0).cuda()._run(
            x.to(torch.bfloat16)
            .contiguous(memory_format=torch.channels_last_3d)).float()
        torch.testing.assert_close(got, want, rtol=0.1, atol=0.05)


@pytest.mark.timeout(600)
def test_zring_bf16_variant_sweep():
    """Every bf16 ring variant behind CFX_ZRING_PL computes the same conv
    (vs torch bf16). The env is read once per process, so each variant
    runs in a subprocess. PL=10 is a timing-only ablation and excluded."""
    import subprocess
    import sys
    code = (
        'import sys, torch; sys.path.insert(0, ".")\n'
        'from chunkflow_amd.fastconv import CfxConv3dBF16\n'
        'torch.manual_seed(0)\n'
        'conv = torch.nn.Conv3d(28, 28, 3, padding=1).cuda()\n'
        'm = CfxConv3dBF16(conv).cuda()\n'
        'x = (torch.randn(2, 28, 5, 37, 41, device="cuda") * 0.3)'
        '.to(torch.bfloat16)'
        '.contiguous(memory_format=torch.channels_last_3d)\n'
        'r = (torch.randn(2, 28, 5, 37, 41, device="cuda") * 0.3)'
        '.to(torch.bfloat16)'
        '.contiguous(memory_format=torch.channels_last_3d)\n'
        'want = torch.nn.functional.elu(\n'
        '    torch.nn.functional.conv3d(x, conv.weight.to(torch.bfloat16)'
        '.to(memory_format=torch.channels_last_3d),'
        ' conv.bias.to(torch.bfloat16), padding=1) + r).float()\n'
        'got = m._run(x, residual=r, elu=True).float()\n'
        'torch.testing.assert_close(got, want, rtol=0.06, atol=0.06)\n'
        'print("OK")\n')
    for pl in ('0', '1', '2', '3', '6', '8', '9', '11', '13'):
        env = dict(os.environ, CFX_ZRING_PL=pl)
        p = subprocess.run([sys.executable, '-c', code], env=env,
                           capture_output=True, text=True, timeout=240,
                           cwd=REPO)
        assert p.returncode == 0 and 'OK' in p.stdout, \
            (pl, p.stdout[-500:], p.stderr[-500:])
