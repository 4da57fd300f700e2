"""TensorFlow TensorBundle (checkpoint V2) reader/writer — no TensorFlow.

The released code2vec models (README.md:86, 1.4 GB java14m) are saved with
`tf.compat.v1.train.Saver` (reference tensorflow_model.py:370-377), which
writes the V2 "tensor bundle" format:

- `<prefix>.index` — an SSTable (TensorFlow's fork of the LevelDB table
  format): prefix-compressed key/value blocks, each with a 5-byte trailer
  (compression type byte + crc32c), an index block mapping last-keys to
  block handles, and a 48-byte footer ending in the magic
  0xdb4775248b80fb57. The empty key holds a BundleHeaderProto; every other
  key is a tensor name with a BundleEntryProto value (dtype, shape,
  shard/offset/size, masked crc32c of the payload).
- `<prefix>.data-00000-of-00001` — raw little-endian tensor payloads.

This module implements just enough of the stack to load those five model
tensors (and Adam slots) into the MI355X engine and to write round-trip
fixtures: varint/protobuf mini-codecs for the two messages, a snappy
decompressor for compressed index blocks, crc32c for payload verification,
and a writer producing uncompressed single-data-block tables that the reader
(and TF itself) accepts.
"""

import os
import struct
from typing import Dict, List, Tuple

import numpy as np

TABLE_MAGIC = 0xdb4775248b80fb57

# tensorflow DataType enum values we support
DT_FLOAT, DT_DOUBLE, DT_INT32, DT_INT64, DT_HALF, DT_BFLOAT16 = 1, 2, 3, 9, 19, 14
_DTYPES = {
    DT_FLOAT: np.dtype('<f4'),
    DT_DOUBLE: np.dtype('<f8'),
    DT_INT32: np.dtype('<i4'),
    DT_INT64: np.dtype('<i8'),
    DT_HALF: np.dtype('<f2'),
}
_NP_TO_DT = {np.dtype('float32'): DT_FLOAT, np.dtype('float64'): DT_DOUBLE,
             np.dtype('int32'): DT_INT32, np.dtype('int64'): DT_INT64,
             np.dtype('float16'): DT_HALF}


# ---------------------------------------------------------------------------
# crc32c (Castagnoli), with the masking scheme LevelDB/TF applies to stored
# checksums so that checksumming a value that contains checksums is safe.
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


def crc32c(data: bytes, crc: int = 0) -> int:
    tbl = _crc_table()
    c = crc ^ 0xFFFFFFFF
    for b in data:
        c = tbl[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def crc_mask(crc: int) -> int:
    return (((crc >> 15) | (crc << 17)) + 0xa282ead8) & 0xFFFFFFFF


def crc_unmask(masked: int) -> int:
    rot = (masked - 0xa282ead8) & 0xFFFFFFFF
    return ((rot >> 17) | (rot << 15)) & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# varints
# ---------------------------------------------------------------------------

def read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def write_varint(value: int) -> bytes:
    out = bytearray()
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


# ---------------------------------------------------------------------------
# snappy decompression (index blocks may be block-compressed)
# ---------------------------------------------------------------------------

def snappy_decompress(data: bytes) -> bytes:
    n, pos = read_varint(data, 0)
    out = bytearray()
    while pos < len(data):
        tag = data[pos]
        pos += 1
        kind = tag & 3
        if kind == 0:                       # literal
            length = (tag >> 2) + 1
            if length > 60:
                nbytes = length - 60
                length = int.from_bytes(data[pos:pos + nbytes], 'little') + 1
                pos += nbytes
            out += data[pos:pos + length]
            pos += length
            continue
        if kind == 1:                       # copy with 1-byte offset
            length = ((tag >> 2) & 0x7) + 4
            offset = ((tag >> 5) << 8) | data[pos]
            pos += 1
        elif kind == 2:                     # copy with 2-byte offset
            length = (tag >> 2) + 1
            offset = int.from_bytes(data[pos:pos + 2], 'little')
            pos += 2
        else:                               # copy with 4-byte offset
            length = (tag >> 2) + 1
            offset = int.from_bytes(data[pos:pos + 4], 'little')
            pos += 4
        start = len(out) - offset
        for i in range(length):             # overlapping copies are legal
            out.append(out[start + i])
    assert len(out) == n, 'snappy length mismatch'
    return bytes(out)


# ---------------------------------------------------------------------------
# protobuf mini-codec for BundleHeaderProto / BundleEntryProto
# ---------------------------------------------------------------------------

def _iter_fields(buf: bytes):
    pos = 0
    while pos < len(buf):
        key, pos = read_varint(buf, pos)
        field, wire = key >> 3, key & 7
        if wire == 0:                       # varint
            val, pos = read_varint(buf, pos)
        elif wire == 1:                     # fixed64
            val = struct.unpack_from('<Q', buf, pos)[0]
            pos += 8
        elif wire == 2:                     # length-delimited
            ln, pos = read_varint(buf, pos)
            val = buf[pos:pos + ln]
            pos += ln
        elif wire == 5:                     # fixed32
            val = struct.unpack_from('<I', buf, pos)[0]
            pos += 4
        else:
            raise ValueError('unsupported wire type %d' % wire)
        yield field, wire, val


class BundleEntry:
    def __init__(self):
        self.dtype = DT_FLOAT
        self.shape: List[int] = []
        self.shard_id = 0
        self.offset = 0
        self.size = 0
        self.crc32c = 0

    @classmethod
    def parse(cls, buf: bytes) -> 'BundleEntry':
        e = cls()
        for field, _wire, val in _iter_fields(buf):
            if field == 1:
                e.dtype = val
            elif field == 2:                # TensorShapeProto
                for f2, _w2, v2 in _iter_fields(val):
                    if f2 == 2:             # Dim
                        for f3, _w3, v3 in _iter_fields(v2):
                            if f3 == 1:
                                e.shape.append(v3)
            elif field == 3:
                e.shard_id = val
            elif field == 4:
                e.offset = val
            elif field == 5:
                e.size = val
            elif field == 6:
                e.crc32c = val
        return e

    def serialize(self) -> bytes:
        out = bytearray()
        out += write_varint(1 << 3) + write_varint(self.dtype)
        shape = bytearray()
        for d in self.shape:
            dim = write_varint(1 << 3) + write_varint(d)
            shape += write_varint((2 << 3) | 2) + write_varint(len(dim)) + dim
        out += write_varint((2 << 3) | 2) + write_varint(len(shape)) + shape
        if self.shard_id:
            out += write_varint(3 << 3) + write_varint(self.shard_id)
        out += write_varint(4 << 3) + write_varint(self.offset)
        out += write_varint(5 << 3) + write_varint(self.size)
        out += write_varint((6 << 3) | 5) + struct.pack('<I', self.crc32c)
        return bytes(out)


def _serialize_header(num_shards: int = 1) -> bytes:
    # BundleHeaderProto: num_shards=1, endianness little (0, default),
    # version { producer: 1 }
    version = write_varint(1 << 3) + write_varint(1)
    return (write_varint(1 << 3) + write_varint(num_shards)
            + write_varint((3 << 3) | 2) + write_varint(len(version)) + version)


# ---------------------------------------------------------------------------
# SSTable block / table parsing
# ---------------------------------------------------------------------------

def _parse_block(block: bytes) -> List[Tuple[bytes, bytes]]:
    """Parse one (already-decompressed) table block into key/value pairs."""
    if len(block) < 4:
        return []
    num_restarts = struct.unpack_from('<I', block, len(block) - 4)[0]
    data_end = len(block) - 4 - 4 * num_restarts
    entries = []
    pos = 0
    key = b''
    while pos < data_end:
        shared, pos = read_varint(block, pos)
        non_shared, pos = read_varint(block, pos)
        value_len, pos = read_varint(block, pos)
        key = key[:shared] + block[pos:pos + non_shared]
        pos += non_shared
        value = block[pos:pos + value_len]
        pos += value_len
        entries.append((key, value))
    return entries


def _read_block(buf: bytes, offset: int, size: int) -> bytes:
    """Read a block given its handle; the 5-byte trailer is
    [compression type][crc32c of data+type]."""
    data = buf[offset:offset + size]
    ctype = buf[offset + size]
    if ctype == 1:
        data = snappy_decompress(data)
    elif ctype != 0:
        raise ValueError('unsupported block compression %d' % ctype)
    return data


def read_index_file(path: str) -> Dict[str, BundleEntry]:
    with open(path, 'rb') as f:
        buf = f.read()
    magic = struct.unpack_from('<Q', buf, len(buf) - 8)[0]
    if magic != TABLE_MAGIC:
        raise ValueError('%s is not a TensorFlow checkpoint index '
                         '(bad table magic)' % path)
    footer = buf[len(buf) - 48:]
    pos = 0
    _meta_off, pos = read_varint(footer, pos)
    _meta_size, pos = read_varint(footer, pos)
    index_off, pos = read_varint(footer, pos)
    index_size, pos = read_varint(footer, pos)
    entries: Dict[str, BundleEntry] = {}
    header = None
    for _ikey, handle in _parse_block(_read_block(buf, index_off, index_size)):
        hpos = 0
        off, hpos = read_varint(handle, hpos)
        size, hpos = read_varint(handle, hpos)
        for key, value in _parse_block(_read_block(buf, off, size)):
            if key == b'':
                header = value  # BundleHeaderProto
                continue
            entries[key.decode('utf-8')] = BundleEntry.parse(value)
    if header is not None:
        for field, _w, val in _iter_fields(header):
            if field == 2 and val == 1:
                raise ValueError('big-endian checkpoints are not supported')
    return entries


class TFCheckpointReader:
    """Random access to the tensors of a TF V2 checkpoint prefix."""

    def __init__(self, prefix: str):
        self.prefix = prefix
        self.entries = read_index_file(prefix + '.index')
        self._shards: Dict[int, np.memmap] = {}

    @staticmethod
    def is_tf_checkpoint(prefix: str) -> bool:
        return os.path.isfile(prefix + '.index')

    def tensor_names(self) -> List[str]:
        return sorted(self.entries)

    def has_tensor(self, name: str) -> bool:
        return name in self.entries

    def _shard(self, shard_id: int, num_shards: int) -> np.memmap:
        if shard_id not in self._shards:
            path = '%s.data-%05d-of-%05d' % (self.prefix, shard_id, num_shards)
            self._shards[shard_id] = np.memmap(path, dtype=np.uint8, mode='r')
        return self._shards[shard_id]

    def get_tensor(self, name: str, verify: bool = False) -> np.ndarray:
        e = self.entries[name]
        num_shards = max(s.shard_id for s in self.entries.values()) + 1
        shard = self._shard(e.shard_id, num_shards)
        raw = bytes(shard[e.offset:e.offset + e.size])
        if verify and e.crc32c:
            if crc32c(raw) != crc_unmask(e.crc32c):
                raise ValueError('crc mismatch for tensor %s' % name)
        if e.dtype == DT_BFLOAT16:
            u16 = np.frombuffer(raw, dtype='<u2').astype(np.uint32) << 16
            arr = u16.view(np.float32)
        else:
            arr = np.frombuffer(raw, dtype=_DTYPES[e.dtype])
        return arr.reshape(e.shape).copy()


# ---------------------------------------------------------------------------
# writer (round-trip fixtures + TF-compatible export)
# ---------------------------------------------------------------------------

def _build_block(pairs: List[Tuple[bytes, bytes]]) -> bytes:
    """Uncompressed table block: full keys (no prefix sharing), one restart."""
    out = bytearray()
    for key, value in pairs:
        out += write_varint(0) + write_varint(len(key)) \
            + write_varint(len(value)) + key + value
    out += struct.pack('<I', 0)             # restart point at offset 0
    out += struct.pack('<I', 1)             # num_restarts
    return bytes(out)


def _append_block(file_buf: bytearray, block: bytes) -> Tuple[int, int]:
    offset = len(file_buf)
    file_buf += block
    ctype = b'\x00'
    file_buf += ctype
    file_buf += struct.pack('<I', crc_mask(crc32c(block + ctype)))
    return offset, len(block)


def write_checkpoint(prefix: str, tensors: Dict[str, np.ndarray]):
    """Write a V2 checkpoint (single shard, uncompressed index) readable by
    TFCheckpointReader and by TensorFlow's BundleReader."""
    data = bytearray()
    pairs: List[Tuple[bytes, bytes]] = [(b'', _serialize_header())]
    for name in sorted(tensors):
        arr = np.ascontiguousarray(tensors[name])
        e = BundleEntry()
        e.dtype = _NP_TO_DT[arr.dtype]
        e.shape = list(arr.shape)
        e.offset = len(data)
        raw = arr.tobytes()
        e.size = len(raw)
        e.crc32c = crc_mask(crc32c(raw))
        data += raw
        pairs.append((name.encode('utf-8'), e.serialize()))

    index = bytearray()
    data_off, data_size = _append_block(index, _build_block(pairs))
    meta_off, meta_size = _append_block(index, _build_block([]))
    last_key = pairs[-1][0]
    handle = write_varint(data_off) + write_varint(data_size)
    idx_off, idx_size = _append_block(index,
                                      _build_block([(last_key, handle)]))
    footer = write_varint(meta_off) + write_varint(meta_size) \
        + write_varint(idx_off) + write_varint(idx_size)
    footer = footer.ljust(40, b'\x00') + struct.pack('<Q', TABLE_MAGIC)
    index += footer

    with open(prefix + '.index', 'wb') as f:
        f.write(bytes(index))
    with open(prefix + '.data-00000-of-00001', 'wb') as f:
        f.write(bytes(data))
