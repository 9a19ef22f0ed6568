"""Wire format for control-plane payloads.

The reference moves every payload as ``pickle.dumps`` of a dict of CPU
tensors (/root/reference/manager.py:77-85, worker.py:111-117)
— remote code execution by design for anyone who can reach the port (defect
D6). This module keeps the SAME dict schemas ("Baton's checkpoint layout")
but serializes them safely:

    [8B magic b'BTNWIRE1'][u32 LE meta length][meta JSON][safetensors blob]

* meta: every non-tensor field of the payload (update_name, n_epoch,
  n_samples, loss_history, ...) as JSON.
* tensors: the ``state_dict`` as a safetensors blob (pure data — no code
  execution on load). Non-contiguous tensors are made contiguous; dtypes and
  shapes round-trip exactly, including 0-dim and integer buffers (defect D4's
  BatchNorm ``num_batches_tracked``).

Payload schemas preserved from the reference:
  round_start: {'state_dict': ..., 'update_name': str, 'n_epoch': int}
  update:      {'state_dict': ..., 'n_samples': int, 'update_name': str,
                'loss_history': [float]}
"""

from __future__ import annotations

import io
import json
import struct
from collections import OrderedDict
from typing import Any, Dict, Optional, Tuple

import torch

MAGIC = b"BTNWIRE1"

# torch dtype <-> stable string names (covers everything a state_dict holds).
_DTYPE_TO_STR = {
    torch.float32: "f32",
    torch.float64: "f64",
    torch.float16: "f16",
    torch.bfloat16: "bf16",
    torch.int64: "i64",
    torch.int32: "i32",
    torch.int16: "i16",
    torch.int8: "i8",
    torch.uint8: "u8",
    torch.bool: "bool",
}
_STR_TO_DTYPE = {v: k for k, v in _DTYPE_TO_STR.items()}


def _save_tensors(tensors: "OrderedDict[str, torch.Tensor]") -> bytes:
    """Serialize tensors: JSON header (name -> dtype/shape/offsets) + raw
    little-endian bytes. Same layout idea as safetensors, implemented
    directly so the wire format is self-contained and version-pinned."""
    header: Dict[str, Any] = {}
    blobs = []
    offset = 0
    for name, t in tensors.items():
        t = t.detach()
        if t.device.type != "cpu":
            t = t.cpu()
        t = t.contiguous()
        # bool tensors serialize as u8
        storage = t.view(torch.uint8) if t.dtype == torch.bool else t
        raw = storage.numpy().tobytes() if t.dtype != torch.bfloat16 else (
            t.view(torch.uint16).numpy().tobytes()
        )
        header[name] = {
            "dtype": _DTYPE_TO_STR[t.dtype],
            "shape": list(t.shape),
            "data_offsets": [offset, offset + len(raw)],
        }
        blobs.append(raw)
        offset += len(raw)
    hjson = json.dumps(header, separators=(",", ":")).encode()
    out = io.BytesIO()
    out.write(struct.pack("<Q", len(hjson)))
    out.write(hjson)
    for b in blobs:
        out.write(b)
    return out.getvalue()


_ITEMSIZE = {
    "f32": 4, "f64": 8, "f16": 2, "bf16": 2,
    "i64": 8, "i32": 4, "i16": 2, "i8": 1, "u8": 1, "bool": 1,
}


def _load_tensors(data: bytes) -> "OrderedDict[str, torch.Tensor]":
    """Header geometry is attacker-controllable: every shape/offset is
    validated against the actual blob BEFORE any tensor is built, so a
    hostile body can neither request absurd allocations nor alias outside
    its own bytes."""
    if len(data) < 8:
        raise ValueError("truncated tensor blob")
    (hlen,) = struct.unpack_from("<Q", data, 0)
    if hlen > len(data) - 8:
        raise ValueError("tensor header length exceeds payload")
    header = json.loads(data[8 : 8 + hlen].decode())
    if not isinstance(header, dict):
        raise ValueError("tensor header must be an object")
    base = 8 + hlen
    blob_len = len(data) - base
    out: "OrderedDict[str, torch.Tensor]" = OrderedDict()
    for name, info in header.items():
        dtype_str = info["dtype"]
        if dtype_str not in _STR_TO_DTYPE:
            raise ValueError(f"unknown dtype {dtype_str!r}")
        dtype = _STR_TO_DTYPE[dtype_str]
        shape = info["shape"]
        if not isinstance(shape, list) or not all(
            isinstance(d, int) and 0 <= d for d in shape
        ):
            raise ValueError(f"bad shape for {name!r}")
        lo, hi = info["data_offsets"]
        if not (isinstance(lo, int) and isinstance(hi, int)
                and 0 <= lo <= hi <= blob_len):
            raise ValueError(f"data_offsets out of range for {name!r}")
        numel = 1
        for d in shape:
            numel *= d
        if numel * _ITEMSIZE[dtype_str] != hi - lo:
            raise ValueError(f"shape/bytes mismatch for {name!r}")
        raw = data[base + lo : base + hi]
        if hi == lo:
            t = torch.empty(0, dtype=dtype)
        elif dtype == torch.bool:
            t = torch.frombuffer(bytearray(raw), dtype=torch.uint8).view(torch.bool)
        elif dtype == torch.bfloat16:
            t = torch.frombuffer(bytearray(raw), dtype=torch.uint16).view(torch.bfloat16)
        else:
            t = torch.frombuffer(bytearray(raw), dtype=dtype)
        out[name] = t.reshape(shape)
    return out


def encode_payload(
    meta: Dict[str, Any],
    state_dict: Optional["OrderedDict[str, torch.Tensor]"] = None,
) -> bytes:
    """Encode a control-plane payload. ``meta`` must be JSON-serializable."""
    meta_json = json.dumps(meta, separators=(",", ":")).encode()
    body = _save_tensors(state_dict if state_dict is not None else OrderedDict())
    return MAGIC + struct.pack("<I", len(meta_json)) + meta_json + body


def decode_payload(data: bytes) -> Tuple[Dict[str, Any], "OrderedDict[str, torch.Tensor]"]:
    """Decode a payload produced by :func:`encode_payload`.

    Raises ``ValueError`` on bad magic/framing — a malformed or hostile body
    can fail but never execute code (unlike the reference's pickle.loads at
    manager.py:98 / worker.py:92).
    """
    if len(data) < len(MAGIC) + 4 or data[: len(MAGIC)] != MAGIC:
        raise ValueError("bad payload magic")
    try:
        (mlen,) = struct.unpack_from("<I", data, len(MAGIC))
        start = len(MAGIC) + 4
        meta = json.loads(data[start : start + mlen].decode())
        tensors = _load_tensors(data[start + mlen :])
    except ValueError:
        raise
    except Exception as e:
        # truncated tensor blobs surface as struct.error / RuntimeError
        # (torch.frombuffer) / UnicodeDecodeError — normalize so HTTP
        # handlers map every malformed body to 400, never a 500
        raise ValueError(f"malformed payload: {type(e).__name__}: {e}")
    return meta, tensors
