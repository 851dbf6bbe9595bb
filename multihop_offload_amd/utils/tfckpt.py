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
