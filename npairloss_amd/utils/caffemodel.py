"""Minimal .caffemodel (binary protobuf NetParameter) reader/writer.

The reference's training snapshots are Caffe `.caffemodel` files
(solver.prototxt:15-16 -> NetParameter protos with per-layer blob
payloads, loaded by layer NAME).  This is a dependency-free protobuf
wire-format codec for exactly the NetParameter subset needed:

    NetParameter { name=1 (string); layers=2 (V1LayerParameter, repeated);
                   layer=100 (LayerParameter, repeated) }
    LayerParameter (V2) { name=1; type=2 (string); blobs=7 }
    V1LayerParameter { name=4; type=5 (enum); blobs=6 }
    BlobProto { num=1 channels=2 height=3 width=4 (legacy shape);
                data=5 (packed/unpacked float); shape=7 (BlobShape) }
    BlobShape { dim=1 (packed/unpacked int64) }

`load_caffemodel_into` copies blobs into a model whose backbone exposes
`caffe_names()` (Caffe layer name -> conv/linear module), matching by
name like Caffe's Net::CopyTrainedLayersFrom.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import Dict, List, Tuple

import numpy as np
import torch

# ---------------------------------------------------------------------------
# wire primitives
# ---------------------------------------------------------------------------


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def _write_varint(out: bytearray, value: int) -> None:
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return


def _iter_fields(buf: bytes):
    pos = 0
    n = len(buf)
    while pos < n:
        key, pos = _read_varint(buf, pos)
        fno, wt = key >> 3, key & 7
        if wt == 0:
            v, pos = _read_varint(buf, pos)
            yield fno, wt, v
        elif wt == 1:
            yield fno, wt, buf[pos : pos + 8]
            pos += 8
        elif wt == 2:
            ln, pos = _read_varint(buf, pos)
            yield fno, wt, buf[pos : pos + ln]
            pos += ln
        elif wt == 5:
            yield fno, wt, buf[pos : pos + 4]
            pos += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")


def _key(fno: int, wt: int) -> bytes:
    out = bytearray()
    _write_varint(out, (fno << 3) | wt)
    return bytes(out)


def _len_delim(fno: int, payload: bytes) -> bytes:
    out = bytearray(_key(fno, 2))
    _write_varint(out, len(payload))
    out += payload
    return bytes(out)


# ---------------------------------------------------------------------------
# NetParameter
# ---------------------------------------------------------------------------


@dataclass
class CaffeLayer:
    name: str
    type: str = ""
    blobs: List[np.ndarray] = field(default_factory=list)


def _parse_blobshape(buf: bytes) -> List[int]:
    dims: List[int] = []
    for fno, wt, v in _iter_fields(buf):
        if fno == 1:
            if wt == 0:
                dims.append(int(v))
            elif wt == 2:  # packed
                pos = 0
                while pos < len(v):
                    d, pos = _read_varint(v, pos)
                    dims.append(int(d))
    return dims


def _parse_blob(buf: bytes) -> np.ndarray:
    data_chunks: List[bytes] = []
    scalars: List[float] = []
    legacy = {}
    shape: List[int] = []
    for fno, wt, v in _iter_fields(buf):
        if fno == 5:  # data
            if wt == 2:
                data_chunks.append(v)
            elif wt == 5:
                scalars.append(struct.unpack("<f", v)[0])
        elif fno == 7 and wt == 2:
            shape = _parse_blobshape(v)
        elif fno in (1, 2, 3, 4) and wt == 0:
            legacy[fno] = int(v)
    if data_chunks:
        arr = np.frombuffer(b"".join(data_chunks), dtype="<f4").astype(np.float32)
    else:
        arr = np.array(scalars, dtype=np.float32)
    if not shape and legacy:
        shape = [legacy.get(i, 1) for i in (1, 2, 3, 4)]
    if shape and int(np.prod(shape)) == arr.size:
        arr = arr.reshape(shape)
    return arr


_V1_TYPE_NAMES = {4: "Convolution", 14: "InnerProduct"}  # partial, name-match is primary


def _parse_layer(buf: bytes, v1: bool) -> CaffeLayer:
    name = ""
    ltype = ""
    blobs: List[np.ndarray] = []
    name_f = 4 if v1 else 1
    type_f = 5 if v1 else 2
    blobs_f = 6 if v1 else 7
    for fno, wt, v in _iter_fields(buf):
        if fno == name_f and wt == 2:
            name = v.decode("utf-8", "replace")
        elif fno == type_f:
            if v1 and wt == 0:
                ltype = _V1_TYPE_NAMES.get(int(v), str(int(v)))
            elif not v1 and wt == 2:
                ltype = v.decode("utf-8", "replace")
        elif fno == blobs_f and wt == 2:
            blobs.append(_parse_blob(v))
    return CaffeLayer(name=name, type=ltype, blobs=blobs)


def read_caffemodel(path_or_bytes) -> Dict[str, CaffeLayer]:
    """Parse a .caffemodel into {layer_name: CaffeLayer} (layers with blobs)."""
    if isinstance(path_or_bytes, (bytes, bytearray)):
        buf = bytes(path_or_bytes)
    else:
        with open(path_or_bytes, "rb") as fh:
            buf = fh.read()
    layers: Dict[str, CaffeLayer] = {}
    for fno, wt, v in _iter_fields(buf):
        if fno == 100 and wt == 2:  # layer (V2)
            l = _parse_layer(v, v1=False)
            if l.blobs:
                layers[l.name] = l
        elif fno == 2 and wt == 2:  # layers (V1)
            l = _parse_layer(v, v1=True)
            if l.blobs:
                layers[l.name] = l
    return layers


def _encode_blob(arr: np.ndarray) -> bytes:
    out = bytearray()
    shape = bytearray()
    for d in arr.shape:
        shape += _key(1, 0)
        _write_varint(shape, int(d))
    out += _len_delim(7, bytes(shape))
    out += _len_delim(5, arr.astype("<f4").tobytes())
    return bytes(out)


def write_caffemodel(path: str, layers: List[CaffeLayer], net_name: str = "net") -> None:
    """Write a V2-layer NetParameter (for tests / interop round-trips)."""
    out = bytearray()
    out += _len_delim(1, net_name.encode())
    for l in layers:
        payload = bytearray()
        payload += _len_delim(1, l.name.encode())
        if l.type:
            payload += _len_delim(2, l.type.encode())
        for b in l.blobs:
            payload += _len_delim(7, _encode_blob(np.asarray(b, dtype=np.float32)))
        out += _len_delim(100, bytes(payload))
    with open(path, "wb") as fh:
        fh.write(bytes(out))


# ---------------------------------------------------------------------------
# SolverState (.solverstate): mid-training resume artifacts
#
#   SolverState { iter=1 (int32); learned_net=2 (string);
#                 history=3 (BlobProto, repeated); current_step=4 (int32) }
#
# `history` holds one blob per learnable parameter — the previous SGD
# update vector (Caffe's momentum state; identical quantity to CaffeSGD's
# per-param "v") — in net learnable-parameter order: for each layer of
# caffe_names(), weight then bias.
# ---------------------------------------------------------------------------


@dataclass
class CaffeSolverState:
    iter: int = 0
    learned_net: str = ""
    history: List[np.ndarray] = field(default_factory=list)
    current_step: int = 0


def read_solverstate(path_or_bytes) -> CaffeSolverState:
    if isinstance(path_or_bytes, (bytes, bytearray)):
        buf = bytes(path_or_bytes)
    else:
        with open(path_or_bytes, "rb") as fh:
            buf = fh.read()
    st = CaffeSolverState()
    for fno, wt, v in _iter_fields(buf):
        if fno == 1 and wt == 0:
            st.iter = int(v)
        elif fno == 2 and wt == 2:
            st.learned_net = v.decode("utf-8", "replace")
        elif fno == 3 and wt == 2:
            st.history.append(_parse_blob(v))
        elif fno == 4 and wt == 0:
            st.current_step = int(v)
    return st


def write_solverstate(path: str, state: CaffeSolverState) -> None:
    out = bytearray()
    out += _key(1, 0)
    _write_varint(out, int(state.iter))
    if state.learned_net:
        out += _len_delim(2, state.learned_net.encode())
    for h in state.history:
        out += _len_delim(3, _encode_blob(np.asarray(h, dtype=np.float32)))
    out += _key(4, 0)
    _write_varint(out, int(state.current_step))
    with open(path, "wb") as fh:
        fh.write(bytes(out))


def caffe_param_order(model: torch.nn.Module) -> List[Tuple[str, torch.Tensor]]:
    """Canonical learnable-parameter order for solverstate history blobs:
    caffe_names() layer order, weight then bias per layer."""
    target = model
    if not hasattr(target, "caffe_names") and hasattr(target, "backbone"):
        target = target.backbone
    if not hasattr(target, "caffe_names"):
        raise TypeError("model does not expose caffe_names()")
    out: List[Tuple[str, torch.Tensor]] = []
    for name, mod in target.caffe_names().items():
        out.append((f"{name}/weight", mod.weight))
        if getattr(mod, "bias", None) is not None:
            out.append((f"{name}/bias", mod.bias))
    return out


# ---------------------------------------------------------------------------
# model loading
# ---------------------------------------------------------------------------


def save_caffemodel(model: torch.nn.Module, path: str, net_name: str = "net") -> int:
    """Write the model's caffe-named weights as a .caffemodel (the inverse
    of load_caffemodel_into; Caffe itself can read the result)."""
    target = model
    if not hasattr(target, "caffe_names") and hasattr(target, "backbone"):
        target = target.backbone
    if not hasattr(target, "caffe_names"):
        raise TypeError("model does not expose caffe_names()")
    layers = []
    for name, mod in target.caffe_names().items():
        blobs = [mod.weight.detach().cpu().numpy()]
        if getattr(mod, "bias", None) is not None:
            blobs.append(mod.bias.detach().cpu().numpy())
        layers.append(CaffeLayer(name, "Convolution", blobs))
    write_caffemodel(path, layers, net_name=net_name)
    return len(layers)


def load_caffemodel_into(model: torch.nn.Module, path_or_bytes,
                         strict: bool = False) -> Tuple[List[str], List[str]]:
    """Copy blobs into `model` by Caffe layer name.

    The model (or model.backbone) must expose `caffe_names() ->
    {caffe_layer_name: module}` where each module has .weight (+ optional
    .bias).  Returns (loaded_names, skipped_names)."""
    target = model
    if not hasattr(target, "caffe_names") and hasattr(target, "backbone"):
        target = target.backbone
    if not hasattr(target, "caffe_names"):
        raise TypeError("model does not expose caffe_names()")
    name_map = target.caffe_names()
    layers = read_caffemodel(path_or_bytes)
    loaded, skipped = [], []
    with torch.no_grad():
        for name, layer in layers.items():
            mod = name_map.get(name)
            if mod is None:
                skipped.append(name)
                continue
            w = torch.from_numpy(np.ascontiguousarray(layer.blobs[0]))
            if w.shape != mod.weight.shape:
                w = w.reshape(mod.weight.shape)
            mod.weight.copy_(w)
            if len(layer.blobs) > 1 and getattr(mod, "bias", None) is not None:
                mod.bias.copy_(torch.from_numpy(np.ascontiguousarray(layer.blobs[1])).reshape(mod.bias.shape))
            loaded.append(name)
    if strict and skipped:
        raise KeyError(f"unmatched caffemodel layers: {skipped}")
    missing = [n for n in name_map if n not in layers]
    if strict and missing:
        raise KeyError(f"model layers absent from caffemodel: {missing}")
    return loaded, skipped
