"""h5lite — minimal in-repo HDF5 reader/writer (no libhdf5 dependency).

The reference stores pretraining shards as HDF5 files of int32/int8
datasets (utils/encode_data.py:204-210; read at src/dataset.py:217-222).
This ROCm image ships neither h5py nor libhdf5, so the framework carries
its own small HDF5 implementation:

* ``read(path)`` parses real HDF5: superblock v0/v2/v3, object headers
  v1/v2 (with continuation blocks), old-style symbol-table groups and
  new-style compact link messages, contiguous and chunked datasets,
  deflate (gzip) and shuffle filters — enough to ingest shards written
  by h5py/libhdf5 as well as our own.
* ``write(path, {...})`` emits spec-conformant HDF5 (superblock v2,
  v2 object headers with Jenkins lookup3 checksums, compact links,
  contiguous storage) that round-trips through ``read`` and is laid out
  for mmap-friendly sequential access.

Only the features the framework needs are implemented; unknown header
messages are skipped, unknown filters raise.
"""

from __future__ import annotations

import struct
import zlib
from typing import Dict

import numpy as np

_UNDEF = 0xFFFFFFFFFFFFFFFF
_M32 = 0xFFFFFFFF
_SIG = b"\x89HDF\r\n\x1a\n"


# ---------------------------------------------------------------------------
# Jenkins lookup3 ("hashlittle") — the checksum HDF5 v2 structures use.
# ---------------------------------------------------------------------------
def _rot(x: int, k: int) -> int:
    return ((x << k) | (x >> (32 - k))) & _M32


def lookup3(data: bytes, initval: int = 0) -> int:
    length = len(data)
    a = b = c = (0xDEADBEEF + length + initval) & _M32
    i = 0
    while length > 12:
        a = (a + int.from_bytes(data[i : i + 4], "little")) & _M32
        b = (b + int.from_bytes(data[i + 4 : i + 8], "little")) & _M32
        c = (c + int.from_bytes(data[i + 8 : i + 12], "little")) & _M32
        a = (a - c) & _M32; a ^= _rot(c, 4); c = (c + b) & _M32
        b = (b - a) & _M32; b ^= _rot(a, 6); a = (a + c) & _M32
        c = (c - b) & _M32; c ^= _rot(b, 8); b = (b + a) & _M32
        a = (a - c) & _M32; a ^= _rot(c, 16); c = (c + b) & _M32
        b = (b - a) & _M32; b ^= _rot(a, 19); a = (a + c) & _M32
        c = (c - b) & _M32; c ^= _rot(b, 4); b = (b + a) & _M32
        i += 12
        length -= 12
    tail = data[i:]
    if not tail:
        return c
    k = tail + b"\x00" * (12 - len(tail))
    a = (a + int.from_bytes(k[0:4], "little")) & _M32
    b = (b + int.from_bytes(k[4:8], "little")) & _M32
    c = (c + int.from_bytes(k[8:12], "little")) & _M32
    # final mix
    c ^= b; c = (c - _rot(b, 14)) & _M32
    a ^= c; a = (a - _rot(c, 11)) & _M32
    b ^= a; b = (b - _rot(a, 25)) & _M32
    c ^= b; c = (c - _rot(b, 16)) & _M32
    a ^= c; a = (a - _rot(c, 4)) & _M32
    b ^= a; b = (b - _rot(a, 14)) & _M32
    c ^= b; c = (c - _rot(b, 24)) & _M32
    return c


# ---------------------------------------------------------------------------
# Writer
# ---------------------------------------------------------------------------
def _dtype_message(dt: np.dtype) -> bytes:
    dt = np.dtype(dt)
    if dt.kind in ("i", "u"):
        signed = dt.kind == "i"
        bits = 0x08 if signed else 0x00  # LE, no pad, sign bit3
        body = bytes([0x10, bits, 0x00, 0x00])  # version1|class0, bitfield
        body += struct.pack("<I", dt.itemsize)
        body += struct.pack("<HH", 0, dt.itemsize * 8)  # bit offset, precision
        return body
    if dt == np.float32 or dt == np.float64:
        # class 1 floating point, IEEE LE
        body = bytes([0x11, 0x20, 0x3F, 0x00])
        body += struct.pack("<I", dt.itemsize)
        if dt == np.float32:
            body += struct.pack("<HHBBBBI", 0, 32, 23, 8, 0, 23, 127)
        else:
            body += struct.pack("<HHBBBBI", 0, 64, 52, 11, 0, 52, 1023)
        return body
    raise TypeError(f"h5lite writer: unsupported dtype {dt}")


def _v2_message(mtype: int, body: bytes) -> bytes:
    return struct.pack("<BHB", mtype, len(body), 0) + body


def _v2_object_header(messages: list[bytes]) -> bytes:
    payload = b"".join(messages)
    # flags 0x02: size-of-chunk0 field is 4 bytes; no times/attrs stored
    head = b"OHDR" + bytes([2, 0x02]) + struct.pack("<I", len(payload))
    block = head + payload
    return block + struct.pack("<I", lookup3(block))


def _dataset_ohdr(arr: np.ndarray, data_addr: int) -> bytes:
    msgs = []
    # dataspace v2: version, rank, flags, type(simple=1), dims
    ds = struct.pack("<BBBB", 2, arr.ndim, 0, 1)
    for d in arr.shape:
        ds += struct.pack("<Q", d)
    msgs.append(_v2_message(0x01, ds))
    msgs.append(_v2_message(0x03, _dtype_message(arr.dtype)))
    # fill value v2: version, alloc time(2=late), write time(0), defined(0)
    msgs.append(_v2_message(0x05, struct.pack("<BBBB", 2, 2, 0, 0)))
    # layout v3 contiguous: version, class, addr, size
    msgs.append(
        _v2_message(0x08, struct.pack("<BBQQ", 3, 1, data_addr, arr.nbytes))
    )
    return _v2_object_header(msgs)


def _root_ohdr(links: Dict[str, int]) -> bytes:
    msgs = []
    # link info v0: version, flags, fractal heap addr, name-index btree addr
    msgs.append(_v2_message(0x02, struct.pack("<BBQQ", 0, 0, _UNDEF, _UNDEF)))
    # group info v0
    msgs.append(_v2_message(0x0A, struct.pack("<BB", 0, 0)))
    for name, addr in links.items():
        nm = name.encode()
        if len(nm) > 255:
            raise ValueError("h5lite: link name too long")
        body = struct.pack("<BBB", 1, 0, len(nm)) + nm + struct.pack("<Q", addr)
        msgs.append(_v2_message(0x06, body))
    return _v2_object_header(msgs)


def write(path: str, datasets: Dict[str, np.ndarray]) -> None:
    """Write a flat HDF5 file: one root group of named datasets."""
    arrays = {k: np.ascontiguousarray(v) for k, v in datasets.items()}
    offset = 48  # superblock v2 size
    data_addrs = {}
    for name, arr in arrays.items():
        # 8-byte align data blocks
        offset = (offset + 7) & ~7
        data_addrs[name] = offset
        offset += arr.nbytes
    ohdr_addrs = {}
    ohdr_blobs = {}
    for name, arr in arrays.items():
        offset = (offset + 7) & ~7
        blob = _dataset_ohdr(arr, data_addrs[name])
        ohdr_addrs[name] = offset
        ohdr_blobs[name] = blob
        offset += len(blob)
    offset = (offset + 7) & ~7
    root_addr = offset
    root = _root_ohdr(ohdr_addrs)
    eof = root_addr + len(root)

    sb = _SIG + bytes([2, 8, 8, 0]) + struct.pack("<QQQQ", 0, _UNDEF, eof, root_addr)
    sb += struct.pack("<I", lookup3(sb))

    with open(path, "wb") as f:
        f.write(sb)
        for name, arr in arrays.items():
            _pad_to(f, data_addrs[name])
            f.write(arr.tobytes())
        for name in arrays:
            _pad_to(f, ohdr_addrs[name])
            f.write(ohdr_blobs[name])
        _pad_to(f, root_addr)
        f.write(root)


def _pad_to(f, addr: int) -> None:
    cur = f.tell()
    if cur < addr:
        f.write(b"\x00" * (addr - cur))


# ---------------------------------------------------------------------------
# Reader
# ---------------------------------------------------------------------------
class _Message:
    __slots__ = ("mtype", "body")

    def __init__(self, mtype: int, body: bytes):
        self.mtype = mtype
        self.body = body


def _parse_ohdr(buf: memoryview, addr: int) -> list[_Message]:
    if bytes(buf[addr : addr + 4]) == b"OHDR":
        return _parse_ohdr_v2(buf, addr)
    return _parse_ohdr_v1(buf, addr)


def _parse_ohdr_v2(buf: memoryview, addr: int) -> list[_Message]:
    pos = addr + 4
    version, flags = buf[pos], buf[pos + 1]
    if version != 2:
        raise ValueError("h5lite: bad OHDR version")
    pos += 2
    if flags & 0x20:  # times stored
        pos += 16
    if flags & 0x10:  # storage phase change
        pos += 4
    size_bytes = 1 << (flags & 0x3)
    chunk_size = int.from_bytes(buf[pos : pos + size_bytes], "little")
    pos += size_bytes
    track_order = bool(flags & 0x04)
    msgs: list[_Message] = []
    _parse_v2_messages(buf, pos, chunk_size, track_order, msgs, first=True)
    return msgs


def _parse_v2_messages(buf, pos, length, track_order, msgs, first):
    # first block: `length` counts message bytes only (checksum follows);
    # continuation block: `length` includes OCHK signature and checksum.
    end = pos + length
    if not first:
        pos += 4  # OCHK signature
        end -= 4  # trailing checksum
    while pos + 4 <= end:
        mtype = buf[pos]
        msize = int.from_bytes(buf[pos + 1 : pos + 3], "little")
        pos += 4
        if track_order:
            pos += 2
        body = bytes(buf[pos : pos + msize])
        pos += msize
        if mtype == 0x10:  # continuation
            caddr, clen = struct.unpack_from("<QQ", body)
            _parse_v2_messages(buf, caddr, clen, track_order, msgs, first=False)
        else:
            msgs.append(_Message(mtype, body))


def _parse_ohdr_v1(buf: memoryview, addr: int) -> list[_Message]:
    version = buf[addr]
    if version != 1:
        raise ValueError(f"h5lite: unsupported object header version {version}")
    nmsgs = int.from_bytes(buf[addr + 2 : addr + 4], "little")
    hsize = int.from_bytes(buf[addr + 8 : addr + 12], "little")
    msgs: list[_Message] = []
    blocks = [(addr + 16, hsize)]  # 12B prefix + 4B alignment pad
    count = 0
    while blocks and count < nmsgs:
        pos, length = blocks.pop(0)
        end = pos + length
        while pos + 8 <= end and count < nmsgs:
            mtype = int.from_bytes(buf[pos : pos + 2], "little")
            msize = int.from_bytes(buf[pos + 2 : pos + 4], "little")
            body = bytes(buf[pos + 8 : pos + 8 + msize])
            pos += 8 + msize
            pos = (pos + 7) & ~7
            count += 1
            if mtype == 0x10:
                caddr, clen = struct.unpack_from("<QQ", body)
                blocks.append((caddr, clen))
            else:
                msgs.append(_Message(mtype, body))
    return msgs


def _parse_dtype(body: bytes) -> np.dtype:
    cls = body[0] & 0x0F
    bits0 = body[1]
    size = struct.unpack_from("<I", body, 4)[0]
    if cls == 0:  # fixed point
        signed = bool(bits0 & 0x08)
        if bits0 & 0x01:
            raise ValueError("h5lite: big-endian data unsupported")
        return np.dtype(f"{'i' if signed else 'u'}{size}")
    if cls == 1:  # float
        return np.dtype(f"f{size}")
    raise ValueError(f"h5lite: unsupported datatype class {cls}")


def _read_chunked(buf, btree_addr, shape, dtype, chunk_dims, filters):
    arr = np.zeros(shape, dtype=dtype)
    rank = len(shape)

    def walk(addr):
        if addr == _UNDEF:
            return
        if bytes(buf[addr : addr + 4]) != b"TREE":
            raise ValueError("h5lite: bad chunk B-tree node")
        level = buf[addr + 5]
        nentries = int.from_bytes(buf[addr + 6 : addr + 8], "little")
        pos = addr + 24
        key_size = 8 + (rank + 1) * 8
        for _ in range(nentries):
            nbytes_, fmask = struct.unpack_from("<II", buf, pos)
            offsets = struct.unpack_from(f"<{rank + 1}Q", buf, pos + 8)
            child = struct.unpack_from("<Q", buf, pos + key_size)[0]
            pos += key_size + 8
            if level > 0:
                walk(child)
                continue
            raw = bytes(buf[child : child + nbytes_])
            for fid, cdata in reversed(filters):
                if fmask:  # filter skipped for this chunk per mask bit
                    pass
                if fid == 1:  # deflate
                    raw = zlib.decompress(raw)
                elif fid == 2:  # shuffle
                    esize = cdata[0] if cdata else dtype.itemsize
                    n = len(raw) // esize
                    raw = (
                        np.frombuffer(raw, np.uint8)
                        .reshape(esize, n)
                        .T.tobytes()
                    )
                else:
                    raise ValueError(f"h5lite: unsupported filter id {fid}")
            chunk = np.frombuffer(raw, dtype=dtype)
            chunk = chunk[: int(np.prod(chunk_dims))].reshape(chunk_dims)
            sel_dst, sel_src = [], []
            for d in range(rank):
                start = offsets[d]
                stop = min(start + chunk_dims[d], shape[d])
                sel_dst.append(slice(start, stop))
                sel_src.append(slice(0, stop - start))
            arr[tuple(sel_dst)] = chunk[tuple(sel_src)]

    walk(btree_addr)
    return arr


def _load_dataset(buf: memoryview, msgs: list[_Message]) -> np.ndarray:
    shape = dtype = None
    layout = None
    filters: list[tuple[int, list[int]]] = []
    for m in msgs:
        if m.mtype == 0x01:  # dataspace
            v = m.body[0]
            if v == 1:
                rank, flags = m.body[1], m.body[2]
                off = 8
            else:
                rank, flags = m.body[1], m.body[2]
                off = 4
            shape = struct.unpack_from(f"<{rank}Q", m.body, off)
        elif m.mtype == 0x03:
            dtype = _parse_dtype(m.body)
        elif m.mtype == 0x08:
            layout = m.body
        elif m.mtype == 0x0B:
            filters = _parse_filters(m.body)
    if shape is None or dtype is None or layout is None:
        raise ValueError("h5lite: dataset missing required messages")

    version = layout[0]
    if version == 3:
        cls = layout[1]
        if cls == 1:  # contiguous
            addr, size = struct.unpack_from("<QQ", layout, 2)
            if addr == _UNDEF:
                return np.zeros(shape, dtype=dtype)
            return (
                np.frombuffer(buf, dtype=dtype, count=int(np.prod(shape)), offset=addr)
                .reshape(shape)
                .copy()
            )
        if cls == 2:  # chunked
            dimensionality = layout[2]
            btree_addr = struct.unpack_from("<Q", layout, 3)[0]
            cdims = struct.unpack_from(f"<{dimensionality - 1}I", layout, 11)
            return _read_chunked(buf, btree_addr, shape, dtype, cdims, filters)
        if cls == 0:  # compact
            size = struct.unpack_from("<H", layout, 2)[0]
            return (
                np.frombuffer(layout, dtype=dtype, count=int(np.prod(shape)), offset=4)
                .reshape(shape)
                .copy()
            )
    raise ValueError(f"h5lite: unsupported layout version {version}")


def _parse_filters(body: bytes) -> list[tuple[int, list[int]]]:
    version = body[0]
    nfilters = body[1]
    filters = []
    pos = 8 if version == 1 else 2
    for _ in range(nfilters):
        fid = struct.unpack_from("<H", body, pos)[0]
        if version == 1 or fid >= 256:
            namelen = struct.unpack_from("<H", body, pos + 2)[0]
            ncvals = struct.unpack_from("<H", body, pos + 6)[0]
            pos += 8 + namelen
        else:
            ncvals = struct.unpack_from("<H", body, pos + 6)[0]
            pos += 8
        cvals = list(struct.unpack_from(f"<{ncvals}I", body, pos))
        pos += 4 * ncvals
        if version == 1 and ncvals % 2 == 1:
            pos += 4
        filters.append((fid, cvals))
    return filters


def _group_children(buf: memoryview, msgs: list[_Message]) -> Dict[str, int]:
    children: Dict[str, int] = {}
    for m in msgs:
        if m.mtype == 0x06:  # link message
            version, flags = m.body[0], m.body[1]
            pos = 2
            if flags & 0x08:
                ltype = m.body[pos]
                pos += 1
            else:
                ltype = 0
            if flags & 0x04:
                pos += 8  # creation order
            if flags & 0x10:
                pos += 1  # charset
            lsize = 1 << (flags & 0x3)
            namelen = int.from_bytes(m.body[pos : pos + lsize], "little")
            pos += lsize
            name = m.body[pos : pos + namelen].decode()
            pos += namelen
            if ltype == 0:
                children[name] = struct.unpack_from("<Q", m.body, pos)[0]
        elif m.mtype == 0x11:  # symbol table (old-style group)
            btree, heap = struct.unpack_from("<QQ", m.body)
            children.update(_symbol_table_children(buf, btree, heap))
    return children


def _symbol_table_children(buf, btree_addr, heap_addr) -> Dict[str, int]:
    if bytes(buf[heap_addr : heap_addr + 4]) != b"HEAP":
        raise ValueError("h5lite: bad local heap")
    heap_data = struct.unpack_from("<Q", buf, heap_addr + 24)[0]
    out: Dict[str, int] = {}

    def name_at(off):
        end = off
        while buf[heap_data + end] != 0:
            end += 1
        return bytes(buf[heap_data + off : heap_data + end]).decode()

    def walk(addr):
        sig = bytes(buf[addr : addr + 4])
        if sig == b"TREE":
            level = buf[addr + 5]
            nentries = int.from_bytes(buf[addr + 6 : addr + 8], "little")
            pos = addr + 24 + 8  # skip first key
            for _ in range(nentries):
                child = struct.unpack_from("<Q", buf, pos)[0]
                pos += 16  # child ptr + next key
                walk(child)
        elif sig == b"SNOD":
            nsyms = int.from_bytes(buf[addr + 6 : addr + 8], "little")
            pos = addr + 8
            for _ in range(nsyms):
                name_off, ohdr = struct.unpack_from("<QQ", buf, pos)
                out[name_at(name_off)] = ohdr
                pos += 40
        else:
            raise ValueError("h5lite: bad group btree node")

    walk(btree_addr)
    return out


class H5LiteFile:
    """Read-only flat HDF5 file: ``f['name']`` / ``f.keys()`` like h5py."""

    def __init__(self, path: str):
        with open(path, "rb") as f:
            self._raw = f.read()
        buf = memoryview(self._raw)
        if bytes(buf[0:8]) != _SIG:
            raise ValueError(f"{path}: not an HDF5 file")
        sbver = buf[8]
        if sbver in (2, 3):
            root_addr = struct.unpack_from("<Q", buf, 36)[0]
        elif sbver == 0:
            # v0: versions block (24B incl. sizes/k) then base/free/eof/driver
            # then root group symbol table entry: link name offset(8) + ohdr(8)
            root_addr = struct.unpack_from("<Q", buf, 24 + 32 + 8)[0]
        else:
            raise ValueError(f"h5lite: unsupported superblock version {sbver}")
        self._buf = buf
        self._children = _group_children(buf, _parse_ohdr(buf, root_addr))
        self._cache: Dict[str, np.ndarray] = {}

    def keys(self):
        return self._children.keys()

    def __contains__(self, name: str) -> bool:
        return name in self._children

    def __getitem__(self, name: str) -> np.ndarray:
        if name not in self._cache:
            addr = self._children[name]
            self._cache[name] = _load_dataset(self._buf, _parse_ohdr(self._buf, addr))
        return self._cache[name]

    def close(self):
        self._cache.clear()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


def read(path: str) -> Dict[str, np.ndarray]:
    with H5LiteFile(path) as f:
        return {k: f[k] for k in f.keys()}
