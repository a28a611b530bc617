"""Property-based coverage (hypothesis): host plumbing vs the oracle, and
the IO codecs under random shapes/dtypes.

Derandomized with small example counts so the CPU suite stays fast;
failures print the minimal counterexample geometry.
"""
import numpy as np
from hypothesis import given, settings, HealthCheck, strategies as st

from oracle import (build_chunk_mask, make_patch_mask, oracle_inference,
                    patch_slices_list)

SET = settings(max_examples=12, deadline=None, derandomize=True,
               suppress_health_check=[HealthCheck.too_slow])


@st.composite
def geom(draw):
    chunk, patch, ov = [], [], []
    for _ in range(3):
        p = draw(st.integers(6, 16))
        o = draw(st.integers(1, max(1, p // 3)))
        e = draw(st.integers(0, 14))
        chunk.append(p + e)
        patch.append(p)
        ov.append(o)
    return tuple(chunk), tuple(patch), tuple(ov)


@SET
@given(geom())
def test_tiler_covers_chunk(g):
    """Every output voxel is covered by >= 1 patch, and every patch fits
    the chunk (inferencer.py:255-292 semantics)."""
    chunk, patch, ov = g
    slices = patch_slices_list(chunk, patch, ov)  # global START coords
    cover = np.zeros(chunk, dtype=np.int32)
    for _, ostart in slices:
        for a, p, n in zip(ostart, patch, chunk):
            assert 0 <= a and a + p <= n
        cover[tuple(slice(a, a + p) for a, p in zip(ostart, patch))] += 1
    assert cover.min() >= 1


@SET
@given(geom())
def test_chunk_mask_reciprocal_normalizes(g):
    """sum(shifted masks) * chunk-mask == 1 everywhere (the <1.0001
    invariant the reference asserts, inferencer.py:463-466)."""
    chunk, patch, ov = g
    slices = patch_slices_list(chunk, patch, ov)
    pm = make_patch_mask(patch, ov)
    acc = np.zeros(chunk, dtype=np.float32)
    for istart, _ in slices:
        sl = tuple(slice(a, a + p) for a, p in zip(istart, patch))
        acc[sl] += pm
    mask = build_chunk_mask(chunk, (0, 0, 0), slices, pm)
    np.testing.assert_allclose(acc * mask, 1.0, atol=1e-4)


@SET
@given(geom(), st.integers(0, 2**31 - 1))
def test_identity_inference_matches_input(g, seed):
    """Identity engine through the full blend+normalize returns
    input/255 (the reference's strongest own test, generalized)."""
    chunk, patch, ov = g
    rng = np.random.RandomState(seed)
    arr = rng.randint(0, 256, size=chunk, dtype=np.uint8)
    out = oracle_inference(arr, patch, ov, num_output_channels=2,
                           batch_size=3)
    np.testing.assert_allclose(out[0], arr.astype(np.float32) / 255.0,
                               rtol=1e-4, atol=1e-4)


@SET
@given(st.integers(1, 4), st.integers(1, 40), st.integers(1, 40),
       st.sampled_from(['uint8', 'uint16', 'int32', 'float32', 'float64']),
       st.sampled_from(['', 'zlib']))
def test_tiff_roundtrip_random(npage, h, w, dtype, comp):
    import tempfile
    from chunkflow_amd import tiffio
    rng = np.random.RandomState(npage * 1000 + h * 40 + w)
    if dtype.startswith('float'):
        arr = rng.randn(npage, h, w).astype(dtype)
    else:
        arr = rng.randint(0, np.iinfo(dtype).max,
                          size=(npage, h, w)).astype(dtype)
    with tempfile.TemporaryDirectory() as td:
        p = f'{td}/r.tif'
        tiffio.imwrite(p, arr, compression=comp)
        got = tiffio.imread(p)
    np.testing.assert_array_equal(np.atleast_3d(got).reshape(arr.shape),
                                  arr)


@SET
@given(st.integers(1, 3), st.integers(1, 12), st.integers(1, 12),
       st.integers(1, 12),
       st.sampled_from(['uint8', 'uint16', 'uint32', 'uint64', 'int8',
                        'int16', 'int64', 'float32', 'float64']))
def test_h5_roundtrip_random(d0, d1, d2, d3, dtype):
    import tempfile
    from chunkflow_amd import h5io
    rng = np.random.RandomState(d0 + d1 * 13 + d2 * 169 + d3)
    shape = (d1, d2, d3)[:max(1, d0)]
    if dtype.startswith('float'):
        arr = rng.randn(*shape).astype(dtype)
    else:
        arr = rng.randint(0, 100, size=shape).astype(dtype)
    with tempfile.TemporaryDirectory() as td:
        p = f'{td}/r.h5'
        h5io.write_h5(p, {'main': arr})
        np.testing.assert_array_equal(h5io.read_h5(p)['main'], arr)
