"""CPU oracle of the chunkflow Inferencer hot path (test infrastructure only).

Plain-numpy restatement of the reference algorithm; each function cites the
reference (seung-lab/chunkflow v1.1.7) file:line it follows. No classes, no
Chunk wrapper — arrays plus explicit (z, y, x) offsets, which is all the hot
path actually uses.

Geometry convention (reference inferencer.py:85-143): with equal input/output
patch sizes (the only mode the BASELINE configs use),
    output_patch_crop_margin = 0
    input_patch_overlap      = output_patch_overlap
    stride                   = patch_size - overlap
"""
import numpy as np

from .patch_mask import make_patch_mask


def patch_slices_list(input_size, patch_size, overlap, offset=(0, 0, 0),
                      crop_margin=(0, 0, 0), output_patch_size=None):
    """The (input, output) patch start list, tail-clamped.

    Restates inferencer.py:255-292: a z/y/x triple loop with step = stride;
    any patch that would run past the chunk end is clamped back to
    (size - patch_size) — tail clamping can DUPLICATE positions, and the
    reference double-blends those (the chunk mask double-counts the same
    slices so normalization cancels it); we replicate that exactly.

    Returns a list of ((iz,iy,ix), (oz,oy,ox)) global start coordinates.
    """
    if output_patch_size is None:
        output_patch_size = patch_size
    stride = tuple(p - o for p, o in zip(patch_size, overlap))
    out = []
    for iz in range(0, input_size[0] - overlap[0], stride[0]):
        if iz + patch_size[0] > input_size[0]:
            iz = input_size[0] - patch_size[0]
            assert iz >= 0
        iz += offset[0]
        oz = iz + crop_margin[0]
        for iy in range(0, input_size[1] - overlap[1], stride[1]):
            if iy + patch_size[1] > input_size[1]:
                iy = input_size[1] - patch_size[1]
                assert iy >= 0
            iy += offset[1]
            oy = iy + crop_margin[1]
            for ix in range(0, input_size[2] - overlap[2], stride[2]):
                if ix + patch_size[2] > input_size[2]:
                    ix = input_size[2] - patch_size[2]
                    assert ix >= 0
                ix += offset[2]
                ox = ix + crop_margin[2]
                out.append(((iz, iy, ix), (oz, oy, ox)))
    return out


def blend_into(out, out_offset, patch, patch_offset):
    """out[region] += patch, clipped to the intersection.

    Restates Chunk.blend (chunk/base.py:792-807): the patch region is clipped
    against the output buffer bounds, then accumulated with +=. `out` is
    (C, D, H, W) or (D, H, W); `patch` has matching leading dims; offsets are
    global (z, y, x) starts of the trailing 3 dims.
    """
    shape3 = out.shape[-3:]
    pshape3 = patch.shape[-3:]
    dst = []
    src = []
    for s, o, h, p in zip(patch_offset, out_offset, shape3, pshape3):
        d0 = max(s - o, 0)
        d1 = min(s - o + p, h)
        dst.append(slice(d0, d1))
        src.append(slice(d0 - (s - o), d1 - (s - o)))
    out[..., dst[0], dst[1], dst[2]] += patch[..., src[0], src[1], src[2]]


def build_chunk_mask(output_size3, out_offset, slices, patch_mask,
                     dtype='float32'):
    """Blend every output patch's mask into a zero buffer, then reciprocal.

    Restates inferencer.py:294-333. Returns the RECIPROCAL mask (the
    reference stores 1.0/mask and later multiplies the output by it).
    """
    mask = np.zeros(output_size3, dtype=dtype)
    for _, ostart in slices:
        blend_into(mask, out_offset, patch_mask, ostart)
    return 1.0 / mask


def identity_engine(patch5d, patch_mask, num_output_channels):
    """The reference test-oracle engine (patch/identity.py:29-51):
    f32 cast, (no crop at equal patch sizes), multiply by the patch mask,
    then channel-repeat."""
    out = patch5d.astype(np.float32)
    out = out * patch_mask
    if num_output_channels > 1:
        out = np.repeat(out, num_output_channels, axis=1)
    return out


def oracle_inference(chunk, patch_size, overlap, num_output_channels=3,
                     batch_size=1, engine=None, dtype='float32',
                     mask_output_chunk=True, offset=(0, 0, 0),
                     patch_mask=None, mask_myelin_threshold=None):
    """Full masked-mode inference of one chunk — the parity pin.

    Restates Inferencer.__call__ (inferencer.py:360-479) in masked mode
    (mask_output_chunk=True, the BASELINE default: output size == input size):

      1. integer chunk -> dtype, /= dtype_max            (:395-399)
      2. per batch: cutout patches -> engine -> blend    (:404-455)
      3. output *= 1/chunk_mask                          (:460-461)
      4. assert output < 1.0001                          (:463-466)
      5. optional myelin mask via last channel           (:468-477)

    `engine(batch5d) -> (B, C, pz, py, px) f32, already masked` mirrors the
    PatchInferencer contract (patch/base.py:47-58). engine=None uses the
    identity engine with the mask applied inside (patch/identity.py).
    """
    assert mask_output_chunk, 'oracle covers the masked (non-aligned) mode'
    chunk = np.asarray(chunk)
    input_size = chunk.shape[-3:]
    if patch_mask is None:
        patch_mask = make_patch_mask(patch_size, overlap, dtype=dtype)

    slices = patch_slices_list(input_size, patch_size, overlap, offset=offset)
    out_size = (num_output_channels,) + tuple(input_size)
    out = np.zeros(out_size, dtype=dtype)
    recip_mask = build_chunk_mask(input_size, offset, slices, patch_mask,
                                  dtype=dtype)

    if np.issubdtype(chunk.dtype, np.integer):
        dtype_max = np.iinfo(chunk.dtype).max
        chunk = chunk.astype(dtype)
        chunk /= dtype_max

    if engine is None:
        def engine(batch):
            return identity_engine(batch, patch_mask, num_output_channels)

    buf = np.zeros((batch_size, 1) + tuple(patch_size), dtype=dtype)
    for i in range(0, len(slices), batch_size):
        batch = slices[i:i + batch_size]
        for bi, ((iz, iy, ix), _) in enumerate(batch):
            z0, y0, x0 = iz - offset[0], iy - offset[1], ix - offset[2]
            buf[bi, 0] = chunk[z0:z0 + patch_size[0],
                               y0:y0 + patch_size[1],
                               x0:x0 + patch_size[2]]
        out_patch = engine(buf)
        for bi, (_, ostart) in enumerate(batch):
            blend_into(out, offset, out_patch[bi], ostart)

    out *= recip_mask
    np.testing.assert_array_less(out, 1.0001)

    if mask_myelin_threshold is not None:
        # chunk/base.py:685-689 via inferencer.py:468-477
        myelin_mask = out[-1] < mask_myelin_threshold
        out = out[:-1] * myelin_mask
    return out


def oracle_inference_aligned(chunk, patch_size, overlap,
                             num_output_channels=3, batch_size=1,
                             engine=None, dtype='float32',
                             output_crop_margin=None, patch_mask=None):
    """Aligned-mode inference (mask_output_chunk=False): the output buffer
    is input_size - 2*crop_margin (crop_margin defaults to the overlap,
    inferencer.py:98-107), patches blend bump-masked WITHOUT the chunk-mask
    normalize, and the buffer clipping drops the margins
    (inferencer.py:124-139, 369-370; blend clipping chunk/base.py:796-807).
    """
    chunk = np.asarray(chunk)
    input_size = chunk.shape[-3:]
    if output_crop_margin is None:
        output_crop_margin = overlap
    if patch_mask is None:
        patch_mask = make_patch_mask(patch_size, overlap, dtype=dtype)
    stride = tuple(p - o for p, o in zip(patch_size, overlap))
    for i, s, o in zip(input_size, stride, overlap):
        assert (i - o) % s == 0, 'aligned mode needs aligned input'

    slices = patch_slices_list(input_size, patch_size, overlap)
    out_size3 = tuple(i - 2 * m for i, m in
                      zip(input_size, output_crop_margin))
    out = np.zeros((num_output_channels,) + out_size3, dtype=dtype)

    if np.issubdtype(chunk.dtype, np.integer):
        dtype_max = np.iinfo(chunk.dtype).max
        chunk = chunk.astype(dtype)
        chunk /= dtype_max

    if engine is None:
        def engine(batch):
            return identity_engine(batch, patch_mask, num_output_channels)

    buf = np.zeros((batch_size, 1) + tuple(patch_size), dtype=dtype)
    for i in range(0, len(slices), batch_size):
        batch = slices[i:i + batch_size]
        for bi, ((iz, iy, ix), _) in enumerate(batch):
            buf[bi, 0] = chunk[iz:iz + patch_size[0],
                               iy:iy + patch_size[1],
                               ix:ix + patch_size[2]]
        out_patch = engine(buf)
        for bi, (_, (oz, oy, ox)) in enumerate(batch):
            blend_into(out, output_crop_margin, out_patch[bi],
                       (oz, oy, ox))
    np.testing.assert_array_less(out, 1.0001)
    return out
