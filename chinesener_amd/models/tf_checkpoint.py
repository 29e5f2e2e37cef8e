"""TF v2 tensor-bundle checkpoint reader (no TensorFlow needed).

The reference loads pretrained BERT with ``init_from_checkpoint`` from a
TF checkpoint (reference tools/train_utils.py:91-102) — on disk that is
the TensorFlow "tensor bundle" V2 format every TF-1.x Saver writes:

* ``prefix.index`` — a LevelDB-table (SSTable) file mapping variable
  names to ``BundleEntryProto`` records (dtype, shape, shard, offset,
  size, crc32c); key ""  holds the ``BundleHeaderProto``.
* ``prefix.data-NNNNN-of-MMMMM`` — raw little-endian tensor bytes at
  the recorded offsets.

This module parses both from scratch (varint/proto-wire + SSTable block
layout per the public LevelDB ``table_format.md`` and TF
``tensor_bundle.proto``), so ``bert_loader.load_tf_bert`` can consume a
real ``bert_model.ckpt`` in this no-TF image:

    weights = read_tf_checkpoint("/path/bert_model.ckpt")
    load_tf_bert(model.bert, weights)

Uncompressed blocks only (TF's BundleWriter disables block compression);
a snappy-compressed block raises with a clear message.
"""
from __future__ import annotations

import os
import struct
from typing import Dict, List, Tuple

import numpy as np

_FOOTER_SIZE = 48
_TABLE_MAGIC = 0xDB4775248B80FB57

# TF DataType enum -> numpy (the types BERT checkpoints actually use)
_DTYPES = {
    1: np.float32,    # DT_FLOAT
    2: np.float64,    # DT_DOUBLE
    3: np.int32,      # DT_INT32
    9: np.int64,      # DT_INT64
    14: np.uint16,    # DT_BFLOAT16 (raw 16-bit payload)
    19: np.float16,   # DT_HALF
}
_BF16 = 14


def _varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _block_handle(buf: bytes, pos: int) -> Tuple[int, int, int]:
    off, pos = _varint(buf, pos)
    size, pos = _varint(buf, pos)
    return off, size, pos


def _read_block(data: bytes, off: int, size: int) -> bytes:
    """Block payload + 5-byte trailer (1 compression type + 4 crc32c)."""
    blk = data[off:off + size]
    ctype = data[off + size]
    if ctype == 1:
        raise ValueError(
            "snappy-compressed SSTable block: TF BundleWriter emits "
            "uncompressed blocks, so this file was not written by a "
            "standard TF Saver")
    if ctype != 0:
        raise ValueError(f"unknown SSTable block compression {ctype}")
    return blk


def _iter_block_entries(blk: bytes):
    """Yield (key, value) from a LevelDB table block (sequential scan
    honoring the prefix compression; restart array at the tail)."""
    if len(blk) < 4:
        return
    num_restarts = struct.unpack("<I", blk[-4:])[0]
    limit = len(blk) - 4 - 4 * num_restarts
    pos = 0
    key = b""
    while pos < limit:
        shared, pos = _varint(blk, pos)
        non_shared, pos = _varint(blk, pos)
        vlen, pos = _varint(blk, pos)
        key = key[:shared] + blk[pos:pos + non_shared]
        pos += non_shared
        value = blk[pos:pos + vlen]
        pos += vlen
        yield key, value


def _parse_entry_proto(buf: bytes) -> Dict:
    """Minimal proto-wire parse of BundleEntryProto."""
    out = {"dtype": 0, "shape": [], "shard_id": 0, "offset": 0, "size": 0}
    pos = 0
    while pos < len(buf):
        tag, pos = _varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            val, pos = _varint(buf, pos)
            if field == 1:
                out["dtype"] = val
            elif field == 3:
                out["shard_id"] = val
            elif field == 4:
                out["offset"] = val
            elif field == 5:
                out["size"] = val
        elif wire == 2:
            ln, pos = _varint(buf, pos)
            sub = buf[pos:pos + ln]
            pos += ln
            if field == 2:               # TensorShapeProto
                out["shape"] = _parse_shape_proto(sub)
        elif wire == 5:
            pos += 4                     # fixed32 (crc32c)
        elif wire == 1:
            pos += 8
        else:
            raise ValueError(f"unsupported proto wire type {wire}")
    return out


def _parse_shape_proto(buf: bytes) -> List[int]:
    dims: List[int] = []
    pos = 0
    while pos < len(buf):
        tag, pos = _varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if field == 2 and wire == 2:     # repeated Dim
            ln, pos = _varint(buf, pos)
            sub = buf[pos:pos + ln]
            pos += ln
            dpos = 0
            while dpos < len(sub):
                dtag, dpos = _varint(sub, dpos)
                dfield, dwire = dtag >> 3, dtag & 7
                if dfield == 1 and dwire == 0:   # Dim.size
                    v, dpos = _varint(sub, dpos)
                    dims.append(v)
                elif dwire == 2:                  # Dim.name
                    ln2, dpos = _varint(sub, dpos)
                    dpos += ln2
                elif dwire == 0:
                    _, dpos = _varint(sub, dpos)
        elif wire == 0:
            _, pos = _varint(buf, pos)
        elif wire == 2:
            ln, pos = _varint(buf, pos)
            pos += ln
    return dims


def read_index(index_path: str) -> Dict[str, Dict]:
    """Parse prefix.index -> {variable_name: entry dict}."""
    with open(index_path, "rb") as f:
        data = f.read()
    if len(data) < _FOOTER_SIZE:
        raise ValueError(f"{index_path}: too short for an SSTable")
    footer = data[-_FOOTER_SIZE:]
    magic = struct.unpack("<Q", footer[-8:])[0]
    if magic != _TABLE_MAGIC:
        raise ValueError(f"{index_path}: bad SSTable magic "
                         f"{magic:#x} (not a TF checkpoint index)")
    pos = 0
    _, _, pos = _block_handle(footer, pos)       # metaindex (unused)
    idx_off, idx_size, _ = _block_handle(footer, pos)
    index_blk = _read_block(data, idx_off, idx_size)
    entries: Dict[str, Dict] = {}
    for _, handle in _iter_block_entries(index_blk):
        doff, dsize, _ = _block_handle(handle, 0)
        for key, value in _iter_block_entries(_read_block(data, doff, dsize)):
            name = key.decode("utf-8", "replace")
            if name == "":
                continue                          # BundleHeaderProto
            # slice metadata keys look like "name/part_0" only for
            # partitioned vars; BERT checkpoints are unpartitioned
            entries[name] = _parse_entry_proto(value)
    return entries


def read_tf_checkpoint(prefix: str) -> Dict[str, np.ndarray]:
    """Load every tensor of a TF v2 checkpoint ``prefix`` (the path
    passed to ``Saver.restore``, e.g. ``.../bert_model.ckpt``)."""
    index_path = prefix + ".index"
    if not os.path.exists(index_path):
        raise FileNotFoundError(index_path)
    entries = read_index(index_path)
    shards: Dict[int, np.memmap] = {}
    nshards = max((e["shard_id"] for e in entries.values()), default=0) + 1
    out: Dict[str, np.ndarray] = {}
    for name, e in entries.items():
        np_dtype = _DTYPES.get(e["dtype"])
        if np_dtype is None:
            continue                              # e.g. DT_STRING metadata
        sid = e["shard_id"]
        if sid not in shards:
            path = f"{prefix}.data-{sid:05d}-of-{nshards:05d}"
            shards[sid] = np.memmap(path, dtype=np.uint8, mode="r")
        raw = bytes(shards[sid][e["offset"]:e["offset"] + e["size"]])
        arr = np.frombuffer(raw, dtype=np_dtype).reshape(e["shape"])
        if e["dtype"] == _BF16:                   # bf16 payload -> fp32
            arr = (arr.astype(np.uint32) << 16).view(np.float32)
        out[name] = arr
    return out


def load_tf_checkpoint_into_bert(bert_model, prefix: str):
    """read_tf_checkpoint + bert_loader name mapping in one call
    (reference init_from_checkpoint parity)."""
    from .bert_loader import load_tf_bert
    return load_tf_bert(bert_model, read_tf_checkpoint(prefix))


# ----------------------------------------------------------------- writer
# Spec-faithful writer for the same format: used to build test fixtures
# in this no-TF image and to EXPORT weights a TF-1.x Saver can restore.

def _crc32c(data: bytes) -> int:
    """CRC-32C (Castagnoli), as LevelDB uses."""
    poly = 0x82F63B78
    tab = getattr(_crc32c, "_tab", None)
    if tab is None:
        tab = []
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            tab.append(c)
        _crc32c._tab = tab
    crc = 0xFFFFFFFF
    for b in data:
        crc = tab[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


def _put_varint(out: bytearray, v: int):
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _encode_block(entries: List[Tuple[bytes, bytes]]) -> bytes:
    """LevelDB table block: no prefix sharing (every entry a restart)."""
    out = bytearray()
    restarts = []
    for key, value in entries:
        restarts.append(len(out))
        _put_varint(out, 0)                  # shared
        _put_varint(out, len(key))           # non_shared
        _put_varint(out, len(value))
        out += key
        out += value
    for r in restarts:
        out += struct.pack("<I", r)
    out += struct.pack("<I", len(restarts))
    return bytes(out)


def _entry_proto(dtype: int, shape: List[int], offset: int, size: int,
                 crc: int) -> bytes:
    out = bytearray()
    if dtype:
        out.append(1 << 3)
        _put_varint(out, dtype)
    shp = bytearray()
    for d in shape:
        dim = bytearray()
        dim.append(1 << 3)
        _put_varint(dim, d)
        shp.append((2 << 3) | 2)
        _put_varint(shp, len(dim))
        shp += dim
    out.append((2 << 3) | 2)
    _put_varint(out, len(shp))
    out += shp
    if offset:
        out.append(4 << 3)
        _put_varint(out, offset)
    out.append(5 << 3)
    _put_varint(out, size)
    out.append((6 << 3) | 5)
    out += struct.pack("<I", crc)
    return bytes(out)


_NP2TF = {np.dtype(np.float32): 1, np.dtype(np.float64): 2,
          np.dtype(np.int32): 3, np.dtype(np.int64): 9,
          np.dtype(np.float16): 19}


def write_tf_checkpoint(prefix: str, tensors: Dict[str, np.ndarray]):
    """Write ``prefix.index`` + ``prefix.data-00000-of-00001`` in the TF
    tensor-bundle V2 layout (single shard, uncompressed blocks)."""
    data = bytearray()
    kv: List[Tuple[bytes, bytes]] = []
    header = bytearray()
    header.append(1 << 3)
    _put_varint(header, 1)                   # num_shards = 1
    kv.append((b"", bytes(header)))
    for name in sorted(tensors):
        shape = list(np.shape(tensors[name]))   # ascontiguousarray 1-d's 0-d
        arr = np.ascontiguousarray(tensors[name])
        tf_dtype = _NP2TF.get(arr.dtype)
        if tf_dtype is None:
            raise ValueError(f"{name}: unsupported dtype {arr.dtype}")
        raw = arr.tobytes()
        off = len(data)
        data += raw
        kv.append((name.encode(), _entry_proto(
            tf_dtype, shape, off, len(raw), _masked_crc(raw))))

    out = bytearray()
    data_blk = _encode_block(kv)
    out += data_blk
    out.append(0)                            # no compression
    out += struct.pack("<I", _masked_crc(data_blk + b"\x00"))
    data_off, data_size = 0, len(data_blk)

    meta_blk = _encode_block([])
    meta_off = len(out)
    out += meta_blk
    out.append(0)
    out += struct.pack("<I", _masked_crc(meta_blk + b"\x00"))

    handle = bytearray()
    _put_varint(handle, data_off)
    _put_varint(handle, data_size)
    last_key = kv[-1][0]
    idx_blk = _encode_block([(last_key, bytes(handle))])
    idx_off = len(out)
    out += idx_blk
    out.append(0)
    out += struct.pack("<I", _masked_crc(idx_blk + b"\x00"))

    footer = bytearray()
    _put_varint(footer, meta_off)
    _put_varint(footer, len(meta_blk))
    _put_varint(footer, idx_off)
    _put_varint(footer, len(idx_blk))
    footer += b"\x00" * (_FOOTER_SIZE - 8 - len(footer))
    footer += struct.pack("<Q", _TABLE_MAGIC)
    out += footer

    with open(prefix + ".index", "wb") as f:
        f.write(out)
    with open(prefix + ".data-00000-of-00001", "wb") as f:
        f.write(data)
