"""TensorFlow tensor_bundle reader/writer (the `variables/` checkpoint
of a non-frozen SavedModel).

TF Serving loads SavedModels whose weights live in
`variables/variables.index` + `variables/variables.data-NNNNN-of-MMMMM`
rather than as Const nodes; the reference proxies such models untouched
(the bundle is TF Serving's problem). Here the engine loads them
directly: the index is a LevelDB-style SSTable (tensorflow/core/lib/io
table format — prefix-compressed blocks, restart arrays, BlockHandle
footer, magic 0xdb4775248b80fb57) whose values are BundleEntryProto
records (tensorflow/core/protobuf/tensor_bundle.proto) pointing at
byte ranges of the data shards.

Scope: single-shard, uncompressed-block, full (non-sliced) numeric
tensors — what standard `tf.saved_model.save` / Saver V2 exports
produce. Snappy-compressed index blocks and tensor slices raise.
The writer exists for fixtures/tests (this environment has no TF) and
emits the same format, valid masked CRC32Cs included.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, List, Tuple

import numpy as np

from ..wire.pb import Message
from ..wire import messages as m

MAGIC = 0xDB4775248B80FB57
INDEX_SUFFIX = ".index"
DATA_PATTERN = "{prefix}.data-{shard:05d}-of-{num:05d}"

_NP_OF_DT = {
    m.DT_FLOAT: np.float32, m.DT_DOUBLE: np.float64,
    m.DT_INT32: np.int32, m.DT_INT64: np.int64,
    m.DT_UINT8: np.uint8, m.DT_INT8: np.int8,
    m.DT_INT16: np.int16, m.DT_UINT16: np.uint16,
    m.DT_BOOL: np.bool_,
}
_DT_OF_NP = {np.dtype(v): k for k, v in _NP_OF_DT.items()}


class BundleError(Exception):
    pass


# -- protos (tensor_bundle.proto) -------------------------------------------
class TensorShapeDim(Message):
    FIELDS = [("size", 1, "int64"), ("name", 2, "string")]


class TensorShape(Message):
    FIELDS = [("dim", 2, "message", dict(msg_cls=TensorShapeDim,
                                         repeated=True)),
              ("unknown_rank", 3, "bool")]


class BundleHeader(Message):
    FIELDS = [("num_shards", 1, "int32"), ("endianness", 2, "int64")]


class BundleEntry(Message):
    FIELDS = [("dtype", 1, "int64"),
              ("shape", 2, "message", dict(msg_cls=TensorShape)),
              ("shard_id", 3, "int32"), ("offset", 4, "int64"),
              ("size", 5, "int64"), ("crc32c", 6, "fixed32")]


# -- crc32c (Castagnoli), masked like TF/LevelDB ----------------------------
def _make_crc_table():
    poly = 0x82F63B78
    table = []
    for i in range(256):
        c = i
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        table.append(c)
    return table


_CRC_TABLE = _make_crc_table()


def crc32c(data: bytes, crc: int = 0) -> int:
    c = crc ^ 0xFFFFFFFF
    for b in data:
        c = _CRC_TABLE[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def masked_crc32c(data: bytes) -> int:
    c = crc32c(data)
    return ((c >> 15) | (c << 17)) + 0xA282EAD8 & 0xFFFFFFFF


# -- varints ----------------------------------------------------------------
def _put_varint(out: bytearray, v: int) -> None:
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)


def _get_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    shift = v = 0
    while True:
        b = buf[pos]
        pos += 1
        v |= (b & 0x7F) << shift
        if not b & 0x80:
            return v, pos
        shift += 7


# -- SSTable reading --------------------------------------------------------
def _read_block(data: bytes, offset: int, size: int) -> Dict[bytes, bytes]:
    """Decode one table block (entries + restart array) into a dict."""
    block = data[offset:offset + size]
    ctype = data[offset + size]
    if ctype != 0:
        raise BundleError(
            f"compressed index block (type {ctype}) not supported")
    n_restarts = struct.unpack_from("<I", block, len(block) - 4)[0]
    limit = len(block) - 4 * (n_restarts + 1)
    entries: Dict[bytes, bytes] = {}
    pos = 0
    key = b""
    while pos < limit:
        shared, pos = _get_varint(block, pos)
        non_shared, pos = _get_varint(block, pos)
        vlen, pos = _get_varint(block, pos)
        key = key[:shared] + block[pos:pos + non_shared]
        pos += non_shared
        entries[key] = block[pos:pos + vlen]
        pos += vlen
    return entries


def _read_table(path: str) -> Dict[bytes, bytes]:
    with open(path, "rb") as f:
        data = f.read()
    if len(data) < 48:
        raise BundleError(f"{path}: too short for an SSTable")
    footer = data[-48:]
    magic = struct.unpack("<Q", footer[40:48])[0]
    if magic != MAGIC:
        raise BundleError(f"{path}: bad table magic {magic:#x}")
    pos = 0
    _mi_off, pos = _get_varint(footer, pos)
    _mi_size, pos = _get_varint(footer, pos)
    idx_off, pos = _get_varint(footer, pos)
    idx_size, pos = _get_varint(footer, pos)
    index = _read_block(data, idx_off, idx_size)
    entries: Dict[bytes, bytes] = {}
    for _sep, handle in sorted(index.items()):
        hpos = 0
        b_off, hpos = _get_varint(handle, hpos)
        b_size, hpos = _get_varint(handle, hpos)
        entries.update(_read_block(data, b_off, b_size))
    return entries


def read_bundle(prefix: str) -> Dict[str, np.ndarray]:
    """Load every full tensor of the bundle at `prefix`
    (e.g. <version_dir>/variables/variables) into name -> ndarray."""
    raw = _read_table(prefix + INDEX_SUFFIX)
    header_bytes = raw.pop(b"", None)
    num_shards = 1
    if header_bytes is not None:
        num_shards = BundleHeader.decode(header_bytes).num_shards or 1
    shards = {}
    for shard in range(num_shards):
        p = DATA_PATTERN.format(prefix=prefix, shard=shard,
                                num=num_shards)
        with open(p, "rb") as f:
            shards[shard] = f.read()
    out: Dict[str, np.ndarray] = {}
    for key, val in raw.items():
        entry = BundleEntry.decode(val)
        np_dt = _NP_OF_DT.get(entry.dtype)
        if np_dt is None:
            continue                    # strings/resources: not servable
        shape = tuple(int(d.size) for d in entry.shape.dim) \
            if entry.shape is not None else ()
        blob = shards[entry.shard_id][entry.offset:
                                      entry.offset + entry.size]
        arr = np.frombuffer(blob, dtype=np.dtype(np_dt).newbyteorder("<"))
        out[key.decode()] = arr.reshape(shape).astype(np_dt, copy=False)
    return out


# -- writing (fixtures/tests) -----------------------------------------------
def _build_block(items: List[Tuple[bytes, bytes]],
                 restart_interval: int = 16) -> bytes:
    out = bytearray()
    restarts = []
    prev = b""
    for i, (key, val) in enumerate(items):
        shared = 0
        if i % restart_interval == 0:
            restarts.append(len(out))
        else:
            while (shared < len(prev) and shared < len(key) and
                   prev[shared] == key[shared]):
                shared += 1
        _put_varint(out, shared)
        _put_varint(out, len(key) - shared)
        _put_varint(out, len(val))
        out += key[shared:]
        out += val
        prev = key
    if not restarts:
        restarts.append(0)
    for r in restarts:
        out += struct.pack("<I", r)
    out += struct.pack("<I", len(restarts))
    return bytes(out)


class _TableWriter:
    def __init__(self):
        self.buf = bytearray()

    def add_block(self, block: bytes) -> bytes:
        """Appends block + trailer; returns its encoded BlockHandle."""
        off = len(self.buf)
        self.buf += block
        self.buf.append(0)              # no compression
        self.buf += struct.pack("<I", masked_crc32c(block + b"\x00"))
        handle = bytearray()
        _put_varint(handle, off)
        _put_varint(handle, len(block))
        return bytes(handle)

    def finish(self, items: List[Tuple[bytes, bytes]]) -> bytes:
        data_handle = self.add_block(_build_block(items))
        meta_handle = self.add_block(_build_block([]))
        last_key = items[-1][0] if items else b""
        index_handle = self.add_block(
            _build_block([(last_key + b"\x00", data_handle)]))
        footer = bytearray()
        footer += meta_handle
        footer += index_handle
        footer += b"\x00" * (40 - len(footer))
        footer += struct.pack("<Q", MAGIC)
        self.buf += footer
        return bytes(self.buf)


def write_bundle(prefix: str, tensors: Dict[str, np.ndarray]) -> None:
    """Write a single-shard bundle: <prefix>.index +
    <prefix>.data-00000-of-00001 (sorted keys, valid CRCs)."""
    os.makedirs(os.path.dirname(prefix), exist_ok=True)
    data = bytearray()
    items: List[Tuple[bytes, bytes]] = [
        (b"", BundleHeader(num_shards=1, endianness=0).encode())]
    for name in sorted(tensors):
        arr = np.ascontiguousarray(tensors[name])
        dt = _DT_OF_NP.get(arr.dtype)
        if dt is None:
            raise BundleError(f"unsupported dtype {arr.dtype} for {name}")
        blob = arr.astype(arr.dtype.newbyteorder("<"), copy=False).tobytes()
        entry = BundleEntry(
            dtype=dt,
            shape=TensorShape(dim=[TensorShapeDim(size=int(s))
                                   for s in arr.shape]),
            shard_id=0, offset=len(data), size=len(blob),
            crc32c=masked_crc32c(blob))
        data += blob
        items.append((name.encode(), entry.encode()))
    with open(DATA_PATTERN.format(prefix=prefix, shard=0, num=1),
              "wb") as f:
        f.write(bytes(data))
    with open(prefix + INDEX_SUFFIX, "wb") as f:
        f.write(_TableWriter().finish(items))
