"""In-repo TIFF codec: round trips, PIL cross-validation, CLI ops.

PIL (independent TIFF implementation, present in this image) acts as the
codec's oracle: files we write must read back identically through PIL, and
PIL-written files must read identically through our reader. Reference
semantics under test: chunk/base.py:209-263, flow.py:918-974.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

from chunkflow_amd import tiffio
from chunkflow_amd.chunk import Chunk

PIL = pytest.importorskip('PIL.Image')

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def pil_read_stack(path):
    img = PIL.open(path)
    pages = []
    for i in range(getattr(img, 'n_frames', 1)):
        img.seek(i)
        pages.append(np.array(img))
    return pages[0] if len(pages) == 1 else np.stack(pages)


@pytest.mark.parametrize('dtype', ['uint8', 'uint16', 'int32', 'float32'])
@pytest.mark.parametrize('compression', ['', 'zlib'])
def test_roundtrip_and_pil_read(tmp_path, dtype, compression):
    rng = np.random.RandomState(7)
    if dtype == 'float32':
        arr = rng.randn(5, 13, 17).astype(dtype)
    else:
        arr = rng.randint(0, np.iinfo(dtype).max, size=(5, 13, 17)) \
            .astype(dtype)
    p = str(tmp_path / 'a.tif')
    tiffio.imwrite(p, arr, compression=compression)
    np.testing.assert_array_equal(tiffio.imread(p), arr)
    # PIL must agree page-for-page (int32 full-range: PIL mode 'I' handles
    # it; float32 reads as mode 'F')
    np.testing.assert_array_equal(pil_read_stack(p).astype(dtype), arr)


def test_read_pil_written(tmp_path):
    rng = np.random.RandomState(8)
    arr = rng.randint(0, 255, size=(4, 10, 12)).astype(np.uint8)
    p = str(tmp_path / 'pil.tif')
    frames = [PIL.fromarray(a) for a in arr]
    frames[0].save(p, save_all=True, append_images=frames[1:])
    np.testing.assert_array_equal(tiffio.imread(p), arr)


def test_read_single_page_2d(tmp_path):
    arr = np.arange(42, dtype=np.uint8).reshape(6, 7)
    p = str(tmp_path / 's.tif')
    tiffio.imwrite(p, arr)
    got = tiffio.imread(p)
    assert got.ndim == 2
    np.testing.assert_array_equal(got, arr)


def test_chunk_to_tif_float32_quirk(tmp_path):
    """float32 chunks save as *255 uint8 — the reference's ImageJ
    visual-scaling quirk (chunk/base.py:241-247)."""
    arr = np.linspace(0, 1, 3 * 4 * 5, dtype=np.float32).reshape(3, 4, 5)
    c = Chunk(arr.copy(), voxel_offset=(1, 2, 3))
    p = str(tmp_path / 'f.tif')
    c.to_tif(p)
    got = tiffio.imread(p)
    np.testing.assert_array_equal(got, (arr * 255).astype(np.uint8))


def test_chunk_tif_4d_pages(tmp_path):
    rng = np.random.RandomState(9)
    arr = rng.randint(0, 255, size=(2, 3, 6, 7)).astype(np.uint8)
    c = Chunk(arr.copy())
    p = str(tmp_path / 'c.tif')
    c.to_tif(p)
    got = tiffio.imread(p)
    np.testing.assert_array_equal(got, arr.reshape(-1, 6, 7))


def test_from_tif_directory(tmp_path):
    """Directory of per-section files -> sorted stack
    (chunk/base.py:218-233)."""
    rng = np.random.RandomState(10)
    arr = rng.randint(0, 255, size=(5, 8, 9)).astype(np.uint8)
    d = tmp_path / 'secs'
    d.mkdir()
    for z in range(5):
        tiffio.imwrite(str(d / f'sec_{z:04d}.tif'), arr[z])
    c = Chunk.from_tif(str(d), voxel_offset=(10, 0, 0))
    np.testing.assert_array_equal(c.array, arr)
    assert tuple(c.voxel_offset) == (10, 0, 0)


def test_cli_save_load_tif(tmp_path):
    """save-tif then load-tif through the CLI, uint8 end to end."""
    rng = np.random.RandomState(11)
    arr = rng.randint(0, 255, size=(6, 11, 13)).astype(np.uint8)
    src = str(tmp_path / 'in.npy')
    np.save(src, arr)
    tif = str(tmp_path / 'mid.tif')
    out = str(tmp_path / 'out.npy')
    env = dict(os.environ, PYTHONPATH=REPO)
    # two invocations: load-tif's exists=True path check runs at parse
    # time, before save-tif could create the file (reference matches)
    subprocess.run(
        [sys.executable, '-m', 'chunkflow_amd.flow',
         'load-npy', '-f', src, 'save-tif', '-f', tif],
        check=True, env=env, cwd=REPO)
    subprocess.run(
        [sys.executable, '-m', 'chunkflow_amd.flow',
         'load-tif', '-f', tif, 'save-npy', '-f', out],
        check=True, env=env, cwd=REPO)
    np.testing.assert_array_equal(np.load(out), arr)


def test_big_endian_read(tmp_path):
    """MM (big-endian) files read correctly — hand-built header."""
    arr = np.arange(12, dtype=np.uint16).reshape(3, 4)
    import struct
    data = arr.astype('>u2').tobytes()
    tags = [(256, 3, 1, 4), (257, 3, 1, 3), (258, 3, 1, 16), (259, 3, 1, 1),
            (262, 3, 1, 1), (273, 4, 1, 8), (277, 3, 1, 1), (278, 4, 1, 3),
            (279, 4, 1, len(data)), (339, 3, 1, 1)]
    ifd = struct.pack('>H', len(tags))
    for tag, ty, cnt, val in tags:
        if ty == 3:
            ifd += struct.pack('>HHIHH', tag, ty, cnt, val, 0)
        else:
            ifd += struct.pack('>HHII', tag, ty, cnt, val)
    ifd += struct.pack('>I', 0)
    buf = b'MM' + struct.pack('>HI', 42, 8 + len(data)) + data + ifd
    p = str(tmp_path / 'be.tif')
    with open(p, 'wb') as f:
        f.write(buf)
    np.testing.assert_array_equal(tiffio.imread(p), arr)
