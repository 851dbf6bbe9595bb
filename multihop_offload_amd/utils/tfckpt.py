"""Minimal TensorFlow checkpoint-bundle reader (no TF dependency).

Reads the reference's shipped trained models
(``model/model_ChebConv_*/cp-0000.ckpt.{index,data-00000-of-00001}``,
Keras ``save_weights`` TF format) so they can be loaded into our
``ChebConvStack`` for direct parity evaluation.

Format: the ``.index`` file is a LevelDB-style SSTable whose entries map
tensor keys (``layer_with_weights-<i>/kernel/.ATTRIBUTES/VARIABLE_VALUE``)
to serialized ``BundleEntryProto`` messages {dtype, shape, shard, offset,
size, crc32c}; tensor bytes live in the ``.data-00000-of-00001`` shard.
Only the features the reference checkpoints use are implemented
(no compression, single shard, fp32/fp64 dtypes).
"""

from __future__ import annotations

import struct
from typing import Dict, Tuple

import numpy as np

_MAGIC = 0xDB4775248B80FB57


def _varint(buf: bytes, pos: int) -> Tuple[int, int]:
    out = shift = 0
    while True:
        b = buf[pos]
        pos += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, pos
        shift += 7


def _read_block(data: bytes, offset: int, size: int) -> bytes:
    comp = data[offset + size]
    if comp != 0:
        raise NotImplementedError("compressed TF index blocks not supported")
    return data[offset:offset + size]


def _iter_block_entries(block: bytes):
    n_restarts = struct.unpack("<I", block[-4:])[0]
    end = len(block) - 4 - 4 * n_restarts
    pos = 0
    key = b""
    while pos < end:
        shared, pos = _varint(block, pos)
        non_shared, pos = _varint(block, pos)
        vlen, pos = _varint(block, pos)
        key = key[:shared] + block[pos:pos + non_shared]
        pos += non_shared
        value = block[pos:pos + vlen]
        pos += vlen
        yield key.decode("utf-8", "replace"), value


def _parse_bundle_entry(value: bytes) -> dict:
    """BundleEntryProto: 1 dtype, 2 shape(TensorShapeProto: 2 dim{1 size}),
    3 shard_id, 4 offset, 5 size, 6 crc32c (fixed32)."""
    out = {"dtype": 0, "shape": [], "shard": 0, "offset": 0, "size": 0}
    pos = 0
    while pos < len(value):
        tag, pos = _varint(value, pos)
        field, wt = tag >> 3, tag & 7
        if wt == 0:
            v, pos = _varint(value, pos)
            if field == 1:
                out["dtype"] = v
            elif field == 3:
                out["shard"] = v
            elif field == 4:
                out["offset"] = v
            elif field == 5:
                out["size"] = v
        elif wt == 2:
            ln, pos = _varint(value, pos)
            sub = value[pos:pos + ln]
            pos += ln
            if field == 2:                       # TensorShapeProto
                spos = 0
                while spos < len(sub):
                    stag, spos = _varint(sub, spos)
                    if stag >> 3 == 2 and stag & 7 == 2:   # dim
                        dln, spos = _varint(sub, spos)
                        dim = sub[spos:spos + dln]
                        spos += dln
                        dpos = 0
                        size = 0
                        while dpos < len(dim):
                            dtag, dpos = _varint(dim, dpos)
                            if dtag >> 3 == 1 and dtag & 7 == 0:
                                size, dpos = _varint(dim, dpos)
                            else:
                                break
                        out["shape"].append(size)
                    else:
                        break
        elif wt == 5:
            pos += 4
        elif wt == 1:
            pos += 8
    return out


_DTYPES = {1: np.float32, 2: np.float64, 3: np.int32, 9: np.int64}


def read_bundle(prefix: str) -> Dict[str, np.ndarray]:
    """Read all tensors of a TF bundle given its prefix
    (e.g. ``.../cp-0000.ckpt``)."""
    index = open(prefix + ".index", "rb").read()
    if struct.unpack("<Q", index[-8:])[0] != _MAGIC:
        raise ValueError("not a TF bundle index (bad magic)")
    # footer: metaindex handle + index handle (varint offset,size each)
    footer = index[-48:]
    pos = 0
    _mi_off, pos = _varint(footer, pos)
    _mi_sz, pos = _varint(footer, pos)
    ix_off, pos = _varint(footer, pos)
    ix_sz, pos = _varint(footer, pos)
    index_block = _read_block(index, ix_off, ix_sz)
    # index block entries point at data blocks
    data_blocks = []
    for _key, value in _iter_block_entries(index_block):
        off, p = _varint(value, 0)
        sz, p = _varint(value, p)
        data_blocks.append((off, sz))

    shard = open(prefix + ".data-00000-of-00001", "rb").read()
    tensors = {}
    for off, sz in data_blocks:
        for key, value in _iter_block_entries(_read_block(index, off, sz)):
            if not key or key.startswith("_CHECKPOINTABLE"):
                continue
            ent = _parse_bundle_entry(value)
            dt = _DTYPES.get(ent["dtype"])
            if dt is None:
                continue
            arr = np.frombuffer(
                shard, dtype=dt, count=max(ent["size"] // dt().itemsize, 0),
                offset=ent["offset"]).reshape(ent["shape"])
            tensors[key] = arr
    return tensors


# --------------------------------------------------------------------------
# Bundle WRITER — emits a TF-format checkpoint (.index SSTable +
# .data-00000-of-00001 shard) from a ChebConvStack, with the reference's
# key layout and TF-masked crc32c per tensor, so the round-trip back into
# the reference's TF stack (gnn_offloading_agent.py:125-132 /
# tf.train.load_checkpoint) is possible without TF installed here.
# --------------------------------------------------------------------------

_CRC32C_TABLE = None


def _crc32c(data: bytes) -> int:
    """CRC-32C (Castagnoli), as TF's tensor-bundle uses."""
    global _CRC32C_TABLE
    if _CRC32C_TABLE is None:
        poly = 0x82F63B78
        tbl = []
        for i in range(256):
            c = i
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            tbl.append(c)
        _CRC32C_TABLE = tbl
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC32C_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc32c(data: bytes) -> int:
    c = _crc32c(data)
    return ((c >> 15) | (c << 17)) + 0xA282EAD8 & 0xFFFFFFFF


def _enc_varint(v: int) -> bytes:
    out = b""
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _enc_block(entries) -> bytes:
    """One SSTable block, no prefix compression (shared=0 every entry,
    single restart point)."""
    body = b""
    for key, value in entries:
        kb = key.encode()
        body += _enc_varint(0) + _enc_varint(len(kb)) \
            + _enc_varint(len(value)) + kb + value
    body += struct.pack("<I", 0)        # restart offset 0
    body += struct.pack("<I", 1)        # n_restarts
    return body


def _block_trailer(block: bytes) -> bytes:
    # compression byte 0 + masked crc32c over (block || type byte)
    return b"\x00" + struct.pack("<I", _masked_crc32c(block + b"\x00"))


def _entry_proto(dtype_code: int, shape, offset: int, size: int,
                 crc: int) -> bytes:
    shp = b""
    for d in shape:
        dim = _enc_varint(1 << 3 | 0) + _enc_varint(d)   # Dim.size
        shp += _enc_varint(2 << 3 | 2) + _enc_varint(len(dim)) + dim
    out = _enc_varint(1 << 3 | 0) + _enc_varint(dtype_code)
    out += _enc_varint(2 << 3 | 2) + _enc_varint(len(shp)) + shp
    if offset:
        out += _enc_varint(4 << 3 | 0) + _enc_varint(offset)
    out += _enc_varint(5 << 3 | 0) + _enc_varint(size)
    out += _enc_varint(6 << 3 | 5) + struct.pack("<I", crc)
    return out


def write_bundle(prefix: str, tensors: Dict[str, np.ndarray]):
    """Write `{key: array}` as a single-shard TF bundle at `prefix`
    (``prefix.index`` + ``prefix.data-00000-of-00001``)."""
    np_to_code = {np.dtype(np.float32): 1, np.dtype(np.float64): 2,
                  np.dtype(np.int32): 3, np.dtype(np.int64): 9}
    keys = sorted(tensors)
    shard = b""
    entries = []
    # header entry (key ""): BundleHeaderProto {num_shards=1, version{producer=1}}
    ver = _enc_varint(1 << 3 | 0) + _enc_varint(1)
    header = _enc_varint(1 << 3 | 0) + _enc_varint(1) \
        + _enc_varint(3 << 3 | 2) + _enc_varint(len(ver)) + ver
    entries.append(("", header))
    for k in keys:
        a = np.ascontiguousarray(tensors[k])
        code = np_to_code[a.dtype]
        data = a.tobytes()
        entries.append((k, _entry_proto(code, a.shape, len(shard),
                                        len(data), _masked_crc32c(data))))
        shard += data

    data_block = _enc_block(entries)
    index = data_block + _block_trailer(data_block)
    data_handle = _enc_varint(0) + _enc_varint(len(data_block))

    meta_off = len(index)
    meta_block = _enc_block([])
    index += meta_block + _block_trailer(meta_block)
    meta_handle = _enc_varint(meta_off) + _enc_varint(len(meta_block))

    ix_off = len(index)
    last_key = keys[-1] if keys else ""
    ix_block = _enc_block([(last_key, data_handle)])
    index += ix_block + _block_trailer(ix_block)
    ix_handle = _enc_varint(ix_off) + _enc_varint(len(ix_block))

    footer = meta_handle + ix_handle
    footer += b"\x00" * (40 - len(footer))
    footer += struct.pack("<Q", _MAGIC)
    index += footer

    with open(prefix + ".index", "wb") as f:
        f.write(index)
    with open(prefix + ".data-00000-of-00001", "wb") as f:
        f.write(shard)


def save_reference_weights(model, prefix: str, dtype=np.float64):
    """Write a ``ChebConvStack``'s weights as a TF bundle with the
    reference's key layout (``layer_with_weights-<i>/{kernel,bias}/
    .ATTRIBUTES/VARIABLE_VALUE``, float64 like the reference's Keras
    model), plus a ``checkpoint`` manifest so
    ``tf.train.latest_checkpoint`` finds it.  Name-based TF restore
    (``tf.train.load_checkpoint``) reads it directly; Keras
    object-graph-based ``load_weights`` additionally wants the
    ``_CHECKPOINTABLE_OBJECT_GRAPH`` entry, which name-based loaders
    ignore — use ``tf.train.load_checkpoint`` on the TF side."""
    tensors = {}
    for i, layer in enumerate(model.layers):
        base = f"layer_with_weights-{i}"
        tensors[f"{base}/kernel/.ATTRIBUTES/VARIABLE_VALUE"] = \
            layer.weight.detach().cpu().numpy().astype(dtype)
        tensors[f"{base}/bias/.ATTRIBUTES/VARIABLE_VALUE"] = \
            layer.bias.detach().cpu().numpy().astype(dtype)
    write_bundle(prefix, tensors)
    import os
    d, name = os.path.split(prefix)
    with open(os.path.join(d or ".", "checkpoint"), "w") as f:
        f.write(f'model_checkpoint_path: "{name}"\n'
                f'all_model_checkpoint_paths: "{name}"\n')


def load_reference_weights(model, prefix: str):
    """Load a reference Keras checkpoint into a ``ChebConvStack``.  Keys:
    ``layer_with_weights-<i>/{kernel,bias}/.ATTRIBUTES/VARIABLE_VALUE``.
    The checkpoint's Chebyshev order (kernel dim 0) must match the model."""
    import torch
    tensors = read_bundle(prefix)
    with torch.no_grad():
        for i, layer in enumerate(model.layers):
            k = tensors[f"layer_with_weights-{i}/kernel/.ATTRIBUTES/"
                        f"VARIABLE_VALUE"]
            b = tensors[f"layer_with_weights-{i}/bias/.ATTRIBUTES/"
                        f"VARIABLE_VALUE"]
            if tuple(k.shape) != tuple(layer.weight.shape):
                raise ValueError(
                    f"kernel {i}: checkpoint {k.shape} vs model "
                    f"{tuple(layer.weight.shape)} — instantiate the model "
                    f"with K={k.shape[0]}")
            layer.weight.copy_(torch.from_numpy(k.copy())
                               .to(layer.weight.dtype))
            layer.bias.copy_(torch.from_numpy(b.copy()).to(layer.bias.dtype))
    return model
