"""In-repo TensorFlow checkpoint-bundle reader/writer (no TensorFlow).

The reference imports Google's published BERT weights with
``tf.train.list_variables`` / ``load_variable``
(``src/modeling.py:58-116``); TensorFlow is not available in this
environment, so — like ``h5lite`` replaces h5py — this module parses
the TF "tensor bundle" format directly:

* ``<prefix>.index`` — a LevelDB/SSTable file: prefix-compressed
  key/value blocks (optionally snappy-compressed), an index block
  addressing them, and a 48-byte footer ending in the table magic
  ``0xdb4775248b80fb57``. Keys are tensor names ("" is the bundle
  header); values are protobuf ``BundleHeaderProto`` /
  ``BundleEntryProto`` messages (dtype, shape, shard, offset, size).
* ``<prefix>.data-NNNNN-of-MMMMM`` — raw little-endian tensor bytes.

The protobuf wire decoding is hand-rolled (the messages use only
varint / length-delimited / fixed32 fields), as is the snappy
decompressor. ``TFBundleWriter`` emits the same format (uncompressed
blocks, correct crc32c trailers) and exists primarily as the test
fixture for the reader — reading checkpoints written by real
TensorFlow is the target; writing ones TF accepts is best-effort
(untestable here: no TF in the image).

Checksums: block trailers and per-entry crc32c values are written
correctly but NOT verified on read (Google's published checkpoints
are assumed intact; verification would double the read cost).
"""

from __future__ import annotations

import os
import struct
from typing import Dict, List, Tuple

import numpy as np

_TABLE_MAGIC = 0xDB4775248B80FB57

# TF DataType enum values -> numpy dtypes (the ones BERT checkpoints use)
_DTYPES = {
    1: np.dtype("<f4"),   # DT_FLOAT
    2: np.dtype("<f8"),   # DT_DOUBLE
    3: np.dtype("<i4"),   # DT_INT32
    9: np.dtype("<i8"),   # DT_INT64
    14: np.dtype("<u2"),  # DT_BFLOAT16 (raw 16-bit payloads)
    19: np.dtype("<f2"),  # DT_HALF
}
_DTYPE_IDS = {v: k for k, v in _DTYPES.items()}


# ---------------------------------------------------------------------------
# varint / protobuf wire helpers
# ---------------------------------------------------------------------------
def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7
        if shift > 63:
            raise ValueError("varint too long")


def _write_varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _pb_fields(buf: bytes) -> List[Tuple[int, int, object]]:
    """Decode a protobuf message into (field_no, wire_type, value)."""
    fields = []
    pos = 0
    while pos < len(buf):
        tag, pos = _read_varint(buf, pos)
        field, wt = tag >> 3, tag & 7
        if wt == 0:  # varint
            val, pos = _read_varint(buf, pos)
        elif wt == 2:  # length-delimited
            ln, pos = _read_varint(buf, pos)
            val = buf[pos:pos + ln]
            pos += ln
        elif wt == 5:  # fixed32
            val = struct.unpack_from("<I", buf, pos)[0]
            pos += 4
        elif wt == 1:  # fixed64
            val = struct.unpack_from("<Q", buf, pos)[0]
            pos += 8
        else:
            raise ValueError(f"unsupported wire type {wt}")
        fields.append((field, wt, val))
    return fields


def _pb_key(field: int, wt: int) -> bytes:
    return _write_varint((field << 3) | wt)


# ---------------------------------------------------------------------------
# snappy (decode only; the writer emits uncompressed blocks)
# ---------------------------------------------------------------------------
def snappy_decompress(data: bytes) -> bytes:
    total, pos = _read_varint(data, 0)
    out = bytearray()
    while pos < len(data):
        tag = data[pos]
        pos += 1
        kind = tag & 3
        if kind == 0:  # literal
            ln = tag >> 2
            if ln >= 60:
                nbytes = ln - 59
                ln = int.from_bytes(data[pos:pos + nbytes], "little")
                pos += nbytes
            ln += 1
            out += data[pos:pos + ln]
            pos += ln
        else:
            if kind == 1:
                ln = ((tag >> 2) & 7) + 4
                off = ((tag >> 5) << 8) | data[pos]
                pos += 1
            elif kind == 2:
                ln = (tag >> 2) + 1
                off = int.from_bytes(data[pos:pos + 2], "little")
                pos += 2
            else:
                ln = (tag >> 2) + 1
                off = int.from_bytes(data[pos:pos + 4], "little")
                pos += 4
            if off == 0 or off > len(out):
                raise ValueError("bad snappy copy offset")
            start = len(out) - off
            for i in range(ln):  # may overlap: byte-by-byte
                out.append(out[start + i])
    if len(out) != total:
        raise ValueError("snappy length mismatch")
    return bytes(out)


# ---------------------------------------------------------------------------
# crc32c (software, Castagnoli) — for writing valid block trailers
# ---------------------------------------------------------------------------
_CRC_TABLE = []


def _crc_table():
    global _CRC_TABLE
    if not _CRC_TABLE:
        poly = 0x82F63B78
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            _CRC_TABLE.append(c)
    return _CRC_TABLE


def crc32c(data: bytes) -> int:
    tab = _crc_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = tab[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# SSTable block parsing
# ---------------------------------------------------------------------------
def _parse_block(contents: bytes) -> List[Tuple[bytes, bytes]]:
    """Decode a LevelDB table block into (key, value) pairs."""
    if len(contents) < 4:
        return []
    n_restarts = struct.unpack_from("<I", contents, len(contents) - 4)[0]
    data_end = len(contents) - 4 - 4 * n_restarts
    pairs = []
    pos = 0
    key = b""
    while pos < data_end:
        shared, pos = _read_varint(contents, pos)
        non_shared, pos = _read_varint(contents, pos)
        vlen, pos = _read_varint(contents, pos)
        key = key[:shared] + contents[pos:pos + non_shared]
        pos += non_shared
        val = contents[pos:pos + vlen]
        pos += vlen
        pairs.append((key, val))
    return pairs


class TFBundleReader:
    """Read a TensorFlow tensor-bundle checkpoint (``bert_model.ckpt``)."""

    def __init__(self, prefix: str):
        self.prefix = prefix
        index_path = prefix + ".index"
        with open(index_path, "rb") as f:
            raw = f.read()
        if len(raw) < 48:
            raise ValueError(f"{index_path}: not an SSTable (too short)")
        footer = raw[-48:]
        magic = struct.unpack_from("<Q", footer, 40)[0]
        if magic != _TABLE_MAGIC:
            raise ValueError(f"{index_path}: bad table magic")
        # footer: metaindex handle, then index handle (varint pairs)
        pos = 0
        _, pos = _read_varint(footer, pos)  # metaindex offset
        _, pos = _read_varint(footer, pos)  # metaindex size
        idx_off, pos = _read_varint(footer, pos)
        idx_size, pos = _read_varint(footer, pos)

        self._entries: Dict[str, dict] = {}
        self.num_shards = 1
        for _, handle_val in _parse_block(self._block(raw, idx_off, idx_size)):
            hpos = 0
            doff, hpos = _read_varint(handle_val, hpos)
            dsize, hpos = _read_varint(handle_val, hpos)
            for key, val in _parse_block(self._block(raw, doff, dsize)):
                name = key.decode("utf-8", "replace")
                if name == "":
                    for field, _, v in _pb_fields(val):
                        if field == 1:  # num_shards
                            self.num_shards = int(v)
                    continue
                self._entries[name] = self._parse_entry(val)

    @staticmethod
    def _block(raw: bytes, off: int, size: int) -> bytes:
        if off + size + 5 > len(raw):  # contents + 5-byte trailer
            raise ValueError("truncated SSTable block")
        contents = raw[off:off + size]
        ctype = raw[off + size]  # 1-byte type in the 5-byte trailer
        if ctype == 1:
            contents = snappy_decompress(contents)
        elif ctype != 0:
            raise ValueError(f"unsupported block compression {ctype}")
        return contents

    @staticmethod
    def _parse_entry(val: bytes) -> dict:
        entry = {"dtype": 1, "shape": [], "shard": 0, "offset": 0, "size": 0}
        for field, _, v in _pb_fields(val):
            if field == 1:
                entry["dtype"] = int(v)
            elif field == 2:  # TensorShapeProto
                dims = []
                for f2, _, v2 in _pb_fields(v):
                    if f2 == 2:  # repeated Dim
                        for f3, _, v3 in _pb_fields(v2):
                            if f3 == 1:
                                dims.append(int(v3))
                entry["shape"] = dims
            elif field == 3:
                entry["shard"] = int(v)
            elif field == 4:
                entry["offset"] = int(v)
            elif field == 5:
                entry["size"] = int(v)
        return entry

    def list_variables(self) -> List[Tuple[str, List[int]]]:
        return sorted(
            (name, e["shape"]) for name, e in self._entries.items()
        )

    def load_variable(self, name: str) -> np.ndarray:
        if name not in self._entries:
            raise KeyError(f"tensor {name!r} not in bundle {self.prefix}")
        e = self._entries[name]
        dt = _DTYPES.get(e["dtype"])
        if dt is None:
            raise ValueError(f"{name}: unsupported TF dtype {e['dtype']}")
        shard_path = (
            f"{self.prefix}.data-{e['shard']:05d}-of-{self.num_shards:05d}"
        )
        with open(shard_path, "rb") as f:
            f.seek(e["offset"])
            buf = f.read(e["size"])
        if len(buf) != e["size"]:
            raise ValueError(f"{shard_path}: truncated at {name!r}")
        arr = np.frombuffer(buf, dtype=dt)
        return arr.reshape(e["shape"]) if e["shape"] else arr.reshape(())


class TFBundleWriter:
    """Write a tensor bundle this module's reader (and, best-effort,
    TensorFlow's BundleReader) can load. Uncompressed blocks, restart
    point at every key (shared-prefix length 0 — valid, just not
    maximally compact)."""

    def __init__(self, prefix: str):
        self.prefix = prefix
        self._tensors: List[Tuple[str, np.ndarray]] = []

    def add(self, name: str, array: np.ndarray) -> None:
        a = np.asarray(array)
        # ascontiguousarray promotes 0-d to 1-d; keep scalar shape
        self._tensors.append((name, np.ascontiguousarray(a).reshape(a.shape)))

    @staticmethod
    def _entry_proto(dtype_id: int, shape, shard: int, offset: int,
                     size: int, crc: int) -> bytes:
        dims = b"".join(
            _pb_key(2, 2)
            + _write_varint(len(_pb_key(1, 0) + _write_varint(d)))
            + _pb_key(1, 0) + _write_varint(d)
            for d in shape
        )
        out = _pb_key(1, 0) + _write_varint(dtype_id)
        out += _pb_key(2, 2) + _write_varint(len(dims)) + dims
        if shard:
            out += _pb_key(3, 0) + _write_varint(shard)
        if offset:
            out += _pb_key(4, 0) + _write_varint(offset)
        out += _pb_key(5, 0) + _write_varint(size)
        out += _pb_key(6, 5) + struct.pack("<I", crc)
        return out

    @staticmethod
    def _block(pairs: List[Tuple[bytes, bytes]]) -> bytes:
        body = bytearray()
        restarts = []
        for key, val in pairs:
            restarts.append(len(body))
            body += _write_varint(0)  # shared
            body += _write_varint(len(key))
            body += _write_varint(len(val))
            body += key + val
        for r in restarts:
            body += struct.pack("<I", r)
        body += struct.pack("<I", len(restarts))
        return bytes(body)

    @staticmethod
    def _emit_block(out: bytearray, contents: bytes) -> Tuple[int, int]:
        off = len(out)
        out += contents
        trailer_type = b"\x00"  # no compression
        crc = _masked_crc(contents + trailer_type)
        out += trailer_type + struct.pack("<I", crc)
        return off, len(contents)

    def save(self) -> None:
        # data shard
        data = bytearray()
        entries: List[Tuple[bytes, bytes]] = []
        for name, arr in sorted(self._tensors):
            if arr.dtype == np.float64:
                arr = arr.astype(np.float32)
            dtype_id = _DTYPE_IDS.get(np.dtype(arr.dtype.newbyteorder("<")))
            if dtype_id is None:
                raise ValueError(f"{name}: unsupported dtype {arr.dtype}")
            raw = arr.astype(arr.dtype.newbyteorder("<")).tobytes()
            off = len(data)
            data += raw
            entries.append((
                name.encode(),
                self._entry_proto(dtype_id, arr.shape, 0, off, len(raw),
                                  _masked_crc(raw)),
            ))
        with open(f"{self.prefix}.data-00000-of-00001", "wb") as f:
            f.write(bytes(data))

        # header entry (key "") must sort first
        header = _pb_key(1, 0) + _write_varint(1)  # num_shards = 1
        pairs = [(b"", header)] + entries

        out = bytearray()
        doff, dsize = self._emit_block(out, self._block(pairs))
        moff, msize = self._emit_block(out, self._block([]))  # metaindex
        # index block: one separator key >= last data key
        last_key = pairs[-1][0] + b"\xff"
        handle = _write_varint(doff) + _write_varint(dsize)
        ioff, isize = self._emit_block(out, self._block([(last_key, handle)]))
        footer = bytearray()
        footer += _write_varint(moff) + _write_varint(msize)
        footer += _write_varint(ioff) + _write_varint(isize)
        footer += b"\x00" * (40 - len(footer))
        footer += struct.pack("<Q", _TABLE_MAGIC)
        out += footer
        with open(f"{self.prefix}.index", "wb") as f:
            f.write(bytes(out))
