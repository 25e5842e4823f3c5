"""safetensors container parsing — header only, zero-copy by design.

Format: u64-LE header length, then a JSON header mapping tensor name ->
{"dtype", "shape", "data_offsets": [begin, end)} relative to the data
section, then the raw data section.  A blob landed contiguously in HBM
therefore becomes tensors by *viewing* byte ranges — no per-tensor copy at
all (the MI355X replacement for a host-side parse+upload loop).
"""

from __future__ import annotations

import json
import struct
from dataclasses import dataclass

_DTYPES = {
    "BOOL": ("bool", 1), "U8": ("uint8", 1), "I8": ("int8", 1),
    "F8_E4M3": ("float8_e4m3fn", 1), "F8_E5M2": ("float8_e5m2", 1),
    "I16": ("int16", 2), "U16": ("uint16", 2), "F16": ("float16", 2),
    "BF16": ("bfloat16", 2), "I32": ("int32", 4), "U32": ("uint32", 4),
    "F32": ("float32", 4), "F64": ("float64", 8), "I64": ("int64", 8),
    "U64": ("uint64", 8),
}


@dataclass
class TensorInfo:
    name: str
    st_dtype: str          # safetensors dtype tag, e.g. "BF16"
    torch_dtype: str       # torch name, e.g. "bfloat16"
    itemsize: int
    shape: tuple[int, ...]
    begin: int             # byte offset into the DATA section
    end: int

    @property
    def nbytes(self) -> int:
        return self.end - self.begin


@dataclass
class SafetensorsHeader:
    header_bytes: int      # 8 + json length
    data_offset: int       # where the data section starts in the file
    tensors: list[TensorInfo]
    metadata: dict

    @property
    def data_bytes(self) -> int:
        return max((t.end for t in self.tensors), default=0)


def parse_header(prefix: bytes) -> SafetensorsHeader:
    """Parse from the first bytes of a safetensors file.

    Raises ValueError if `prefix` is too short (caller feeds more bytes) —
    len required is 8 + header_len, available via header_len_needed().
    """
    if len(prefix) < 8:
        raise ValueError("need at least 8 bytes")
    (hlen,) = struct.unpack("<Q", prefix[:8])
    if hlen > 100 << 20:
        raise ValueError(f"implausible safetensors header length {hlen}")
    if len(prefix) < 8 + hlen:
        raise ValueError(f"need {8 + hlen} header bytes, have {len(prefix)}")
    obj = json.loads(prefix[8:8 + hlen])
    meta = obj.pop("__metadata__", {})
    tensors = []
    for name, spec in obj.items():
        tag = spec["dtype"]
        if tag not in _DTYPES:
            raise ValueError(f"unsupported safetensors dtype {tag!r}")
        tname, isz = _DTYPES[tag]
        b, e = spec["data_offsets"]
        # untrusted input: negative offsets would make the python-slice
        # views silently read from the blob's END
        if not (isinstance(b, int) and isinstance(e, int)
                and 0 <= b <= e):
            raise ValueError(f"bad data_offsets for {name!r}: {b}, {e}")
        tensors.append(TensorInfo(
            name=name, st_dtype=tag, torch_dtype=tname, itemsize=isz,
            shape=tuple(spec["shape"]), begin=b, end=e))
    tensors.sort(key=lambda t: t.begin)
    return SafetensorsHeader(header_bytes=8 + hlen, data_offset=8 + hlen,
                             tensors=tensors, metadata=meta)


def header_len_needed(first8: bytes) -> int:
    (hlen,) = struct.unpack("<Q", first8[:8])
    return 8 + hlen


def build_header(tensors: dict[str, tuple[str, tuple[int, ...], int]],
                 ) -> tuple[bytes, int]:
    """Serialize a header for synthetic blob generation.

    tensors: name -> (st_dtype, shape, nbytes); offsets assigned in order.
    Returns (header_bytes_blob, data_section_bytes).
    """
    obj = {}
    off = 0
    for name, (tag, shape, nbytes) in tensors.items():
        obj[name] = {"dtype": tag, "shape": list(shape),
                     "data_offsets": [off, off + nbytes]}
        off += nbytes
    js = json.dumps(obj).encode()
    pad = (8 - (len(js) % 8)) % 8  # spec: header often padded with spaces
    js += b" " * pad
    return struct.pack("<Q", len(js)) + js, off


def torch_views(header: SafetensorsHeader, blob_u8):
    """Zero-copy tensor views over a landed blob (torch uint8 1-D tensor
    covering the WHOLE file, header included)."""
    import torch

    out = {}
    data0 = header.data_offset
    for t in header.tensors:
        raw = blob_u8[data0 + t.begin: data0 + t.end]
        dt = getattr(torch, t.torch_dtype)
        out[t.name] = raw.view(dt).view(t.shape)
    return out
