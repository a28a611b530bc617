"""Minimal TIFF 6.0 codec for the pipeline edge (load-tif / save-tif).

The reference uses tifffile (chunk/base.py:209-263); that library is not in
this image, so this module implements the subset the hot-path pipeline
needs from scratch: single-sample (grayscale) images, strip-organized,
uncompressed or zlib/deflate, little- or big-endian on read, multi-page
stacks for 3-D/4-D chunks. It is pure numpy + the stdlib zlib — no GPU
involvement; chunks cross this boundary as host arrays.

Layout on write: header, then per page [strip data][IFD], IFDs chained.
One strip per page (RowsPerStrip = height) keeps the writer simple; the
reader accepts any strip split.
"""
import glob
import os
import struct
import zlib

import numpy as np

# tag ids (TIFF 6.0)
T_WIDTH, T_LENGTH, T_BPS, T_COMP, T_PHOTO = 256, 257, 258, 259, 262
T_STRIP_OFF, T_SPP, T_ROWS_PER_STRIP, T_STRIP_CNT = 273, 277, 278, 279
T_PLANAR, T_SAMPLE_FMT = 284, 339
TY_SHORT, TY_LONG = 3, 4

_FMT_UINT, _FMT_INT, _FMT_FLOAT = 1, 2, 3

_DTYPE_TO_TIFF = {
    np.dtype('uint8'): (8, _FMT_UINT), np.dtype('uint16'): (16, _FMT_UINT),
    np.dtype('uint32'): (32, _FMT_UINT), np.dtype('uint64'): (64, _FMT_UINT),
    np.dtype('int8'): (8, _FMT_INT), np.dtype('int16'): (16, _FMT_INT),
    np.dtype('int32'): (32, _FMT_INT), np.dtype('int64'): (64, _FMT_INT),
    np.dtype('float16'): (16, _FMT_FLOAT),
    np.dtype('float32'): (32, _FMT_FLOAT),
    np.dtype('float64'): (64, _FMT_FLOAT),
}


def _tiff_to_dtype(bits, fmt, bo):
    kind = {_FMT_UINT: 'u', _FMT_INT: 'i', _FMT_FLOAT: 'f'}[fmt]
    return np.dtype(f'{bo}{kind}{bits // 8}')


def imwrite(file_name: str, arr: np.ndarray, compression: str = 'zlib'):
    """Write a 2-D page or a 3-D stack of pages (page-major)."""
    arr = np.asarray(arr)
    if arr.ndim == 2:
        arr = arr[None]
    assert arr.ndim == 3, 'imwrite takes a 2-D page or 3-D page stack'
    if arr.dtype not in _DTYPE_TO_TIFF:
        raise NotImplementedError(f'tiff write for dtype {arr.dtype}')
    bits, fmt = _DTYPE_TO_TIFF[arr.dtype]
    if compression in ('', None, 'none', 'raw'):
        comp_id, pack = 1, lambda b: b
    elif compression == 'zlib':
        comp_id, pack = 8, lambda b: zlib.compress(b, 6)
    else:
        raise NotImplementedError(
            f'compression {compression!r} (supported: "", "zlib")')
    npage, h, w = arr.shape
    le = '<'
    out = bytearray()
    out += b'II*\x00' + struct.pack('<I', 0)  # first-IFD offset patched below
    ifd_pos = []
    for p in range(npage):
        data = pack(np.ascontiguousarray(arr[p]).astype(
            arr.dtype.newbyteorder(le), copy=False).tobytes())
        strip_off = len(out)
        out += data
        if len(out) & 1:
            out += b'\x00'  # word-align the IFD
        ifd_pos.append(len(out))
        tags = [
            (T_WIDTH, TY_LONG, 1, w),
            (T_LENGTH, TY_LONG, 1, h),
            (T_BPS, TY_SHORT, 1, bits),
            (T_COMP, TY_SHORT, 1, comp_id),
            (T_PHOTO, TY_SHORT, 1, 1),          # BlackIsZero
            (T_STRIP_OFF, TY_LONG, 1, strip_off),
            (T_SPP, TY_SHORT, 1, 1),
            (T_ROWS_PER_STRIP, TY_LONG, 1, h),
            (T_STRIP_CNT, TY_LONG, 1, len(data)),
            (T_SAMPLE_FMT, TY_SHORT, 1, fmt),
        ]
        out += struct.pack('<H', len(tags))
        for tag, ty, cnt, val in tags:
            if ty == TY_SHORT:  # inline, left-justified in the value field
                out += struct.pack('<HHIHH', tag, ty, cnt, val, 0)
            else:
                out += struct.pack('<HHII', tag, ty, cnt, val)
        out += struct.pack('<I', 0)  # next-IFD, patched for all but last
    # chain the IFDs
    struct.pack_into('<I', out, 4, ifd_pos[0])
    for i in range(npage - 1):
        # next-IFD field sits after the tag table of IFD i
        ntag_off = ifd_pos[i]
        (ntags,) = struct.unpack_from('<H', out, ntag_off)
        struct.pack_into('<I', out, ntag_off + 2 + 12 * ntags, ifd_pos[i + 1])
    with open(file_name, 'wb') as f:
        f.write(bytes(out))


def _read_tag_values(buf, bo, ty, cnt, raw):
    size = {1: 1, TY_SHORT: 2, TY_LONG: 4}.get(ty)
    if size is None:
        raise NotImplementedError(f'tiff tag type {ty}')
    fmt = {1: 'B', TY_SHORT: 'H', TY_LONG: 'I'}[ty]
    if size * cnt <= 4:
        vals = struct.unpack_from(f'{bo}{cnt}{fmt}', raw, 0)
    else:
        (off,) = struct.unpack_from(f'{bo}I', raw, 0)
        vals = struct.unpack_from(f'{bo}{cnt}{fmt}', buf, off)
    return list(vals)


def imread(file_name: str) -> np.ndarray:
    """Read a grayscale strip-organized TIFF; multi-page stacks come back
    as (npage, h, w)."""
    with open(file_name, 'rb') as f:
        buf = f.read()
    if buf[:2] == b'II':
        bo = '<'
    elif buf[:2] == b'MM':
        bo = '>'
    else:
        raise ValueError(f'{file_name}: not a TIFF (no II/MM byte order)')
    (magic, ifd_off) = struct.unpack_from(f'{bo}HI', buf, 2)
    if magic != 42:
        raise ValueError(f'{file_name}: bad TIFF magic {magic}')
    pages = []
    while ifd_off:
        (ntags,) = struct.unpack_from(f'{bo}H', buf, ifd_off)
        tags = {}
        for i in range(ntags):
            tag, ty, cnt = struct.unpack_from(
                f'{bo}HHI', buf, ifd_off + 2 + 12 * i)
            raw = buf[ifd_off + 2 + 12 * i + 8: ifd_off + 2 + 12 * i + 12]
            if tag in (T_WIDTH, T_LENGTH, T_BPS, T_COMP, T_PHOTO,
                       T_STRIP_OFF, T_SPP, T_ROWS_PER_STRIP, T_STRIP_CNT,
                       T_PLANAR, T_SAMPLE_FMT):
                tags[tag] = _read_tag_values(buf, bo, ty, cnt, raw)
        (ifd_off,) = struct.unpack_from(
            f'{bo}I', buf, ifd_off + 2 + 12 * ntags)
        w, h = tags[T_WIDTH][0], tags[T_LENGTH][0]
        spp = tags.get(T_SPP, [1])[0]
        if spp != 1:
            raise NotImplementedError(
                f'{file_name}: {spp} samples/pixel (grayscale only)')
        bits_l = tags.get(T_BPS, [8])
        bits = bits_l[0]
        fmt = tags.get(T_SAMPLE_FMT, [_FMT_UINT])[0]
        comp = tags.get(T_COMP, [1])[0]
        if 273 not in tags:
            raise NotImplementedError(f'{file_name}: tiled TIFF unsupported')
        offs, cnts = tags[T_STRIP_OFF], tags[T_STRIP_CNT]
        raw = b''.join(
            zlib.decompress(buf[o:o + c]) if comp in (8, 32946)
            else buf[o:o + c]
            for o, c in zip(offs, cnts))
        if comp not in (1, 8, 32946):
            raise NotImplementedError(
                f'{file_name}: compression {comp} (raw/deflate only)')
        dt = _tiff_to_dtype(bits, fmt, bo)
        page = np.frombuffer(raw, dtype=dt, count=h * w).reshape(h, w)
        pages.append(page.astype(dt.newbyteorder('='), copy=False))
    if not pages:
        raise ValueError(f'{file_name}: no pages')
    if len(pages) == 1:
        return pages[0]
    return np.stack(pages, axis=0)


def read_volume(file_name: str, dtype=None) -> np.ndarray:
    """Reference from_tif source logic (chunk/base.py:213-233): a file is
    read directly; a directory is a sorted stack of per-section files."""
    assert os.path.exists(file_name)
    if os.path.isfile(file_name):
        arr = imread(file_name)
        if dtype:
            arr = arr.astype(dtype)
        return arr
    fnames = sorted(glob.glob(f'{file_name}/*.tif*'))
    if not fnames:
        raise FileNotFoundError(f'no *.tif* sections under {file_name}')
    section = imread(fnames[0])
    if dtype is None:
        dtype = section.dtype
    arr = np.empty((len(fnames), *section.shape[-2:]), dtype=dtype)
    arr[0] = section
    for idx, fname in enumerate(fnames[1:]):
        arr[idx + 1] = imread(fname)
    return arr


def write_volume(file_name: str, arr: np.ndarray,
                 compression: str = 'zlib'):
    """Reference to_tif source logic (chunk/base.py:238-263): float32 is
    visual-scaled to uint8 (the reference's ImageJ quirk), 3-D saves as a
    ZYX page stack, 4-D as CZYX with the leading axes flattened to pages."""
    if arr.dtype == np.float32:
        arr = (arr * 255).astype(np.uint8)
    if arr.ndim == 4:
        arr = arr.reshape(-1, *arr.shape[-2:])
    imwrite(file_name, arr, compression=compression)
