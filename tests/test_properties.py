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


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.integers(1, 6), st.integers(1, 4), st.integers(0, 2**31 - 1))
def test_stitch_placement_property(ntasks, chunk_scale, seed):
    """Property: for ANY 1-D task decomposition, stitch_to_rank0 (world 1)
    places every task's block exactly at its bbox."""
    import torch
    from chunkflow_amd.cartesian import BoundingBoxes
    from chunkflow_amd.dispatch import stitch_to_rank0
    rng = np.random.RandomState(seed)
    cz, cy, cx = 2, 3, 2 * chunk_scale
    bboxes = BoundingBoxes.from_manual_setup(
        (cz, cy, cx), roi_size=(cz, cy, cx * ntasks))
    blocks = {i: torch.from_numpy(
        rng.rand(2, cz, cy, cx).astype(np.float32)) for i in range(ntasks)}
    vol = stitch_to_rank0(bboxes, dict(blocks), 2, 0, 1, 'cpu')
    assert vol.shape == (2, cz, cy, cx * ntasks)
    for i in range(ntasks):
        np.testing.assert_array_equal(
            vol[:, :, :, i * cx:(i + 1) * cx].numpy(), blocks[i].numpy())


@settings(max_examples=25, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.integers(0, 2**31 - 1), st.integers(2, 5))
def test_equal_value_label_properties(seed, nvals):
    """Properties of the cc3d-semantics labeler: (1) the label partition
    refines the value partition (no component spans two values); (2)
    labels are dense 1..N; (3) background stays 0; (4) per-value
    partition matches scipy run on that value alone."""
    from scipy import ndimage
    from chunkflow_amd.connected import equal_value_label
    rng = np.random.RandomState(seed)
    arr = rng.randint(0, nvals, size=(5, 9, 8)).astype(np.uint32)
    lab = equal_value_label(arr, 6)
    assert (lab[arr == 0] == 0).all()
    u = np.unique(lab)
    u = u[u != 0]
    assert np.array_equal(u, np.arange(1, len(u) + 1))
    struct = ndimage.generate_binary_structure(3, 1)
    for lb in u:
        vals = np.unique(arr[lab == lb])
        assert len(vals) == 1 and vals[0] != 0
    for v in np.unique(arr):
        if v == 0:
            continue
        ref, n = ndimage.label(arr == v, structure=struct)
        got_ids = lab[arr == v]
        # same number of components and a bijection between labelings
        assert len(np.unique(got_ids)) == n
        pairs = {(a, b) for a, b in zip(ref[arr == v], got_ids)}
        assert len(pairs) == n


@settings(max_examples=20, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(st.tuples(st.integers(3, 10), st.integers(6, 24),
                 st.integers(6, 24)),
       st.integers(1, 2), st.integers(1, 5), st.integers(1, 5))
def test_patch_mask_partition_of_unity(ps, oz, oy, ox):
    """SURVEY A.2 invariants for ANY geometry: the bump patch mask's
    interior [ov, ps-ov) is exactly 1.0 and its SUM equals the
    stride-cell volume prod(ps - ov) — i.e. neighboring patch weights
    form a partition of unity."""
    from chunkflow_amd.patch_mask import make_patch_mask
    ov = (min(oz, ps[0] // 2 - 1) or 1, min(oy, ps[1] // 2 - 1) or 1,
          min(ox, ps[2] // 2 - 1) or 1)
    if any(o < 1 or 2 * o >= p for o, p in zip(ov, ps)):
        return
    m = make_patch_mask(tuple(ps), ov)
    inner = m[ov[0]:ps[0] - ov[0], ov[1]:ps[1] - ov[1],
              ov[2]:ps[2] - ov[2]]
    np.testing.assert_array_equal(inner, 1.0)
    want = float(np.prod([p - o for p, o in zip(ps, ov)]))
    assert abs(float(m.sum()) - want) / want < 1e-5
    assert float(m.min()) > 0
