"""Minimal HDF5 codec for the pipeline edge (load-h5 / save-h5).

The reference stores chunks via h5py (chunk/base.py:267-410); h5py is not
in this image's python, so this module implements the HDF5 subset those
ops need from scratch (pure numpy + stdlib zlib), following the public
HDF5 file-format spec:

- write: superblock v0, old-style root group (B-tree v1 + local heap +
  SNOD), one object header v1 per dataset, CONTIGUOUS little-endian
  storage — the plainest conforming layout; validated against libhdf5's
  h5dump and h5py 3.3 (both present under /opt/conda in the build
  container — tests/test_h5io.py cross-checks both directions);
- read: everything we write PLUS what stock h5py emits: chunked layout
  (v3 class 2) with B-tree v1 chunk indexes and deflate/shuffle filters,
  object-header continuation blocks, nested old-style groups.

Out of subset (clear errors): superblock v2/v3 (libver='latest'),
object header v2, compact/virtual layouts, szip/lzf, attributes.
"""
import os
import struct
import zlib

import numpy as np

UNDEF = 0xFFFFFFFFFFFFFFFF

# ---------------------------------------------------------------------------
# datatype message <-> numpy dtype
# ---------------------------------------------------------------------------


def _dtype_message(dt: np.dtype) -> bytes:
    """Datatype message body (version 1) for LE fixed-point/float."""
    if dt.kind in 'ui':
        bitfield = 0x08 if dt.kind == 'i' else 0x00  # bit3: signed 2's comp
        return struct.pack('<B3BI2H', 0x10, bitfield, 0, 0, dt.itemsize,
                           0, dt.itemsize * 8)
    if dt.kind == 'f' and dt.itemsize in (4, 8):
        if dt.itemsize == 4:
            sign, exp_loc, exp_sz, man_sz, bias = 31, 23, 8, 23, 127
        else:
            sign, exp_loc, exp_sz, man_sz, bias = 63, 52, 11, 52, 1023
        return struct.pack('<B3BI2H4BI', 0x11, 0x20, sign, 0, dt.itemsize,
                           0, dt.itemsize * 8, exp_loc, exp_sz, 0, man_sz,
                           bias)
    raise NotImplementedError(f'h5 write for dtype {dt}')


def _parse_dtype(body: bytes) -> np.dtype:
    ver_cls = body[0]
    cls = ver_cls & 0x0F
    bitfield = body[1]
    (size,) = struct.unpack_from('<I', body, 4)
    bo = '>' if (bitfield & 1) else '<'
    if cls == 0:  # fixed-point
        kind = 'i' if (bitfield & 0x08) else 'u'
        return np.dtype(f'{bo}{kind}{size}')
    if cls == 1:  # float
        return np.dtype(f'{bo}f{size}')
    raise NotImplementedError(f'h5 datatype class {cls}')


# ---------------------------------------------------------------------------
# writer
# ---------------------------------------------------------------------------


def _msg(mtype: int, body: bytes, flags: int = 0) -> bytes:
    if len(body) % 8:
        body += b'\x00' * (8 - len(body) % 8)
    return struct.pack('<2HB3x', mtype, len(body), flags) + body


def _object_header(messages) -> bytes:
    total = sum(len(m) for m in messages)
    hdr = struct.pack('<BxHII4x', 1, len(messages), 1, total)
    return hdr + b''.join(messages)


def write_h5(file_name: str, datasets: dict):
    """Write {name: ndarray} as root-level contiguous datasets."""
    names = sorted(datasets)
    assert len(names) <= 8, 'one SNOD (8 entries) supported'
    arrays = {n: np.ascontiguousarray(datasets[n]) for n in names}

    # local heap data: offset 0 = empty string, then NUL-padded names
    heap_data = bytearray(8)
    name_off = {}
    for n in names:
        name_off[n] = len(heap_data)
        nb = n.encode() + b'\x00'
        heap_data += nb + b'\x00' * (-len(nb) % 8)
    free_off = len(heap_data)
    heap_data += struct.pack('<QQ', 1, 16)  # free block: no next, 16 bytes

    # layout: superblock | root OH | btree | heap hdr | heap data | SNOD
    #         | per-dataset OH | data...
    K = 4  # group leaf/internal k (both 4: one SNOD child)
    sb_size = 96
    root_oh_addr = sb_size
    root_oh = None  # built once btree/heap addrs known (fixed sizes first)
    btree_size = 24 + (2 * K + 1) * 8 + 2 * K * 8
    heap_hdr_size = 32
    snod_size = 8 + 2 * K * 40

    root_oh_size = 16 + len(_msg(0x0011, b'\x00' * 16))
    btree_addr = root_oh_addr + root_oh_size
    heap_hdr_addr = btree_addr + btree_size
    heap_data_addr = heap_hdr_addr + heap_hdr_size
    snod_addr = heap_data_addr + len(heap_data)
    cursor = snod_addr + snod_size

    oh_addr, data_addr = {}, {}
    oh_blobs = {}
    for n in names:
        a = arrays[n]
        space = struct.pack('<BBB5x', 1, a.ndim, 1)
        space += b''.join(struct.pack('<Q', d) for d in a.shape) * 2
        dtype_b = _dtype_message(a.dtype)
        fill = struct.pack('<4BI', 2, 2, 2, 1, 0)
        oh_addr[n] = cursor
        layout_placeholder = struct.pack('<BBQQ', 3, 1, 0, a.nbytes)
        msgs = [_msg(0x0001, space), _msg(0x0003, dtype_b, 1),
                _msg(0x0005, fill, 1), _msg(0x0008, layout_placeholder)]
        blob = _object_header(msgs)
        oh_blobs[n] = blob
        cursor += len(blob)
    for n in names:
        a = arrays[n]
        data_addr[n] = cursor
        cursor += a.nbytes
    eof = cursor

    out = bytearray()
    out += b'\x89HDF\r\n\x1a\n'
    out += struct.pack('<8B2HI', 0, 0, 0, 0, 0, 8, 8, 0, K, K, 0)
    out += struct.pack('<4Q', 0, UNDEF, eof, UNDEF)
    # root symbol table entry: name offset 0, OH addr, cached stab(1)
    out += struct.pack('<QQII', 0, root_oh_addr, 1, 0)
    out += struct.pack('<QQ', btree_addr, heap_hdr_addr)
    assert len(out) == sb_size

    root_oh = _object_header(
        [_msg(0x0011, struct.pack('<QQ', btree_addr, heap_hdr_addr))])
    out += root_oh

    bt = bytearray(b'TREE' + struct.pack('<BBH2Q', 0, 0, 1, UNDEF, UNDEF))
    max_name = max(name_off.values()) if names else 0
    bt += struct.pack('<QQQ', 0, snod_addr, max_name)
    bt += b'\x00' * (btree_size - len(bt))
    out += bt

    out += b'HEAP' + struct.pack('<B3xQQQ', 0, len(heap_data), free_off,
                                 heap_data_addr)
    out += heap_data

    snod = bytearray(b'SNOD' + struct.pack('<BxH', 1, len(names)))
    for n in names:
        snod += struct.pack('<QQII16x', name_off[n], oh_addr[n], 0, 0)
    snod += b'\x00' * (snod_size - len(snod))
    out += snod

    for n in names:
        blob = oh_blobs[n]
        layout = _msg(0x0008, struct.pack('<BBQQ', 3, 1, data_addr[n],
                                          arrays[n].nbytes))
        # patch the placeholder layout message (it is the last message)
        out += blob[:-len(layout)] + layout
    for n in names:
        a = arrays[n]
        le = a.astype(a.dtype.newbyteorder('<'), copy=False)
        out += le.tobytes()
    assert len(out) == eof
    with open(file_name, 'wb') as f:
        f.write(bytes(out))


# ---------------------------------------------------------------------------
# reader
# ---------------------------------------------------------------------------


class _Reader:
    def __init__(self, buf: bytes, file_name: str):
        self.buf = buf
        self.fn = file_name
        if buf[:8] != b'\x89HDF\r\n\x1a\n':
            raise ValueError(f'{file_name}: not an HDF5 file')
        ver = buf[8]
        if ver != 0:
            raise NotImplementedError(
                f'{file_name}: superblock v{ver} (only v0 / default h5py '
                'files supported)')
        self.offsz, self.lensz = buf[13], buf[14]
        if (self.offsz, self.lensz) != (8, 8):
            raise NotImplementedError('non-8-byte offsets/lengths')
        # superblock v0: sig(8) + versions/sizes(8) + k's(4) + flags(4) +
        # base/freespace/eof/driver(32) = 56, then the root symbol table
        # entry: link name offset (8), object header address (8)
        (self.root_oh,) = struct.unpack_from('<Q', buf, 64)

    # -- object headers (v1, with continuations) --
    def messages(self, addr):
        buf = self.buf
        ver = buf[addr]
        if ver != 1:
            raise NotImplementedError(
                f'{self.fn}: object header v{ver} (only v1 supported)')
        (nmsg,) = struct.unpack_from('<H', buf, addr + 2)
        blocks = [(addr + 16, struct.unpack_from('<I', buf, addr + 8)[0])]
        out = []
        while blocks and len(out) < nmsg:
            pos, remaining = blocks.pop(0)
            while remaining >= 8 and len(out) < nmsg:
                mtype, msize, mflags = struct.unpack_from('<2HB', buf, pos)
                body = buf[pos + 8: pos + 8 + msize]
                pos += 8 + msize
                remaining -= 8 + msize
                if mtype == 0x0010:  # continuation
                    coff, clen = struct.unpack_from('<QQ', body, 0)
                    blocks.append((coff, clen))
                else:
                    out.append((mtype, body))
        return out

    # -- group traversal (old-style symbol tables) --
    def group_entries(self, oh_addr):
        msgs = dict_multi(self.messages(oh_addr))
        stab = msgs.get(0x0011)
        if not stab:
            return None
        btree, heap = struct.unpack_from('<QQ', stab[0], 0)
        (heap_data,) = struct.unpack_from('<Q', self.buf, heap + 24)
        entries = []
        self._walk_gnode(btree, heap_data, entries)
        return entries

    def _heap_str(self, heap_data, off):
        end = self.buf.index(b'\x00', heap_data + off)
        return self.buf[heap_data + off:end].decode()

    def _walk_gnode(self, addr, heap_data, out):
        buf = self.buf
        assert buf[addr:addr + 4] == b'TREE', 'bad group b-tree node'
        level, nent = struct.unpack_from('<BH', buf, addr + 5)
        pos = addr + 8 + 16  # skip siblings
        for i in range(nent):
            (child,) = struct.unpack_from('<Q', buf, pos + 8)
            pos += 16
            if level > 0:
                self._walk_gnode(child, heap_data, out)
            else:
                assert buf[child:child + 4] == b'SNOD'
                (nsym,) = struct.unpack_from('<H', buf, child + 6)
                for s in range(nsym):
                    noff, oh = struct.unpack_from('<QQ', buf,
                                                  child + 8 + s * 40)
                    out.append((self._heap_str(heap_data, noff), oh))

    # -- datasets --
    def read_dataset(self, oh_addr):
        msgs = dict_multi(self.messages(oh_addr))
        if 0x0011 in msgs:
            return None  # a group, not a dataset
        space = msgs[0x0001][0]
        sver, rank = space[0], space[1]
        if sver == 1:
            dims = struct.unpack_from(f'<{rank}Q', space, 8)
        elif sver == 2:
            dims = struct.unpack_from(f'<{rank}Q', space, 4)
        else:
            raise NotImplementedError(f'dataspace v{sver}')
        dt = _parse_dtype(msgs[0x0003][0])
        filters = self._parse_filters(msgs.get(0x000B))
        layout = msgs[0x0008][0]
        lver, lclass = layout[0], layout[1]
        if lver != 3:
            raise NotImplementedError(f'data layout v{lver}')
        n = int(np.prod(dims)) if dims else 1
        if lclass == 1:  # contiguous
            addr, size = struct.unpack_from('<QQ', layout, 2)
            if addr == UNDEF:
                arr = np.zeros(dims, dtype=dt)
            else:
                arr = np.frombuffer(self.buf, dtype=dt, count=n,
                                    offset=addr).reshape(dims)
        elif lclass == 2:  # chunked, b-tree v1 index
            ndim_p1 = layout[2]
            (btree,) = struct.unpack_from('<Q', layout, 3)
            cdims = struct.unpack_from(f'<{ndim_p1}I', layout, 11)[:-1]
            arr = np.zeros(dims, dtype=dt)
            if btree != UNDEF:
                self._walk_chunks(btree, arr, cdims, dt, filters, ndim_p1)
        else:
            raise NotImplementedError(f'data layout class {lclass}')
        return np.ascontiguousarray(arr.astype(dt.newbyteorder('='),
                                               copy=False))

    def _parse_filters(self, bodies):
        if not bodies:
            return []
        body = bodies[0]
        ver, nf = body[0], body[1]
        pos = 8 if ver == 1 else 2
        out = []
        for _ in range(nf):
            fid, namelen = struct.unpack_from('<HH', body, pos)
            _, ncv = struct.unpack_from('<HH', body, pos + 4)
            pos += 8
            if ver == 1 or fid >= 256:
                pos += namelen + (-namelen % 8 if ver == 1 else 0)
            elif namelen:
                pos += namelen
            pos += 4 * ncv
            if ver == 1 and ncv % 2:
                pos += 4
            out.append(fid)
        return out

    def _walk_chunks(self, addr, arr, cdims, dt, filters, ndim_p1):
        buf = self.buf
        assert buf[addr:addr + 4] == b'TREE', 'bad chunk b-tree node'
        level, nent = struct.unpack_from('<BH', buf, addr + 5)
        keysz = 8 + 8 * ndim_p1
        pos = addr + 24
        for i in range(nent):
            nbytes, fmask = struct.unpack_from('<II', buf, pos)
            offs = struct.unpack_from(f'<{ndim_p1}Q', buf, pos + 8)[:-1]
            (child,) = struct.unpack_from('<Q', buf, pos + keysz)
            pos += keysz + 8
            if level > 0:
                self._walk_chunks(child, arr, cdims, dt, filters, ndim_p1)
                continue
            raw = buf[child:child + nbytes]
            for fi, fid in enumerate(reversed(filters)):
                if fmask & (1 << (len(filters) - 1 - fi)):
                    continue
                if fid == 1:  # deflate
                    raw = zlib.decompress(raw)
                elif fid == 2:  # shuffle
                    a = np.frombuffer(raw, np.uint8)
                    raw = a.reshape(dt.itemsize, -1).T.tobytes()
                elif fid == 3:  # fletcher32 checksum: strip trailing 4 B
                    raw = raw[:-4]
                else:
                    raise NotImplementedError(f'h5 filter id {fid}')
            chunk = np.frombuffer(raw, dtype=dt,
                                  count=int(np.prod(cdims))).reshape(cdims)
            sl = tuple(slice(o, min(o + c, s))
                       for o, c, s in zip(offs, cdims, arr.shape))
            csl = tuple(slice(0, s.stop - s.start) for s in sl)
            arr[sl] = chunk[csl]


def dict_multi(pairs):
    d = {}
    for k, v in pairs:
        d.setdefault(k, []).append(v)
    return d


def read_h5(file_name: str) -> dict:
    """Read all datasets as {path: ndarray} (root + nested groups)."""
    with open(file_name, 'rb') as f:
        buf = f.read()
    r = _Reader(buf, file_name)
    out = {}

    def walk(prefix, oh):
        entries = r.group_entries(oh)
        if entries is None:
            return
        for name, child_oh in entries:
            path = f'{prefix}{name}'
            arr = r.read_dataset(child_oh)
            if arr is None:
                walk(f'{path}/', child_oh)
            else:
                out[path] = arr
    walk('', r.root_oh)
    return out


def is_hdf5(file_name: str) -> bool:
    try:
        with open(file_name, 'rb') as f:
            return f.read(8) == b'\x89HDF\r\n\x1a\n'
    except OSError:
        return False
