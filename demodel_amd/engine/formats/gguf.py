"""GGUF container parsing, synthesis, and GPU dequantization.

The Ollama model layer (mediaType application/vnd.ollama.image.model,
reference CONTRIBUTING.md:141-146) is a GGUF file.  This module parses the
header from a landed blob's head bytes, sizes each quantized tensor, and
drives the CDNA4 dequant kernels (csrc/gguf_dequant.hip) straight from the
blob's HBM region into fresh bf16 tensors — the blob never round-trips to
host.

Only the container layout + the quant types the kernels support are
implemented; unknown quant types fail loudly.
"""

from __future__ import annotations

import os
import struct
from dataclasses import dataclass

MAGIC = 0x46554747  # 'GGUF' little-endian

# ggml type id -> (name, block_elems, block_bytes)
GGML_TYPES = {
    0: ("f32", 1, 4),
    1: ("f16", 1, 2),
    2: ("q4_0", 32, 18),
    3: ("q4_1", 32, 20),
    6: ("q5_0", 32, 22),
    7: ("q5_1", 32, 24),
    8: ("q8_0", 32, 34),
    10: ("q2_K", 256, 84),
    11: ("q3_K", 256, 110),
    12: ("q4_K", 256, 144),
    13: ("q5_K", 256, 176),
    14: ("q6_K", 256, 210),
    30: ("bf16", 1, 2),
}

_VT_FMT = {0: "<B", 1: "<b", 2: "<H", 3: "<h", 4: "<I", 5: "<i",
           6: "<f", 7: "<B", 10: "<Q", 11: "<q", 12: "<d"}


@dataclass
class GGUFTensor:
    name: str
    dims: tuple[int, ...]     # ggml order: dims[0] fastest-varying
    type_id: int
    offset: int               # into the data section

    @property
    def type_name(self) -> str:
        return GGML_TYPES[self.type_id][0]

    @property
    def n_elems(self) -> int:
        n = 1
        for d in self.dims:
            n *= d
        return n

    @property
    def nbytes(self) -> int:
        _, be, bb = GGML_TYPES[self.type_id]
        assert self.n_elems % be == 0, (self.name, self.dims)
        return self.n_elems // be * bb

    @property
    def n_blocks(self) -> int:
        _, be, _ = GGML_TYPES[self.type_id]
        return self.n_elems // be


@dataclass
class GGUFModel:
    version: int
    kv: dict
    tensors: list[GGUFTensor]
    data_offset: int          # absolute file offset of the data section
    alignment: int
    blob: object = None       # LandedBlob when parsed from one

    def tensor(self, name: str) -> GGUFTensor:
        for t in self.tensors:
            if t.name == name:
                return t
        raise KeyError(name)


class _Reader:
    def __init__(self, data: bytes):
        self.d = data
        self.o = 0

    def take(self, fmt: str):
        v = struct.unpack_from(fmt, self.d, self.o)
        self.o += struct.calcsize(fmt)
        return v[0] if len(v) == 1 else v

    def _need(self, n: int) -> None:
        # untrusted input: a declared length beyond the remaining bytes
        # must fail loudly, not hang a billion-element loop or walk
        # struct.unpack into the weeds
        if n < 0 or n > len(self.d) - self.o:
            raise ValueError(
                f"gguf: declared length {n} exceeds remaining "
                f"{len(self.d) - self.o} bytes")

    def take_str(self) -> str:
        n = self.take("<Q")
        self._need(n)
        s = self.d[self.o:self.o + n].decode("utf-8", "replace")
        self.o += n
        return s

    def take_value(self, vt: int):
        if vt in _VT_FMT:
            return self.take(_VT_FMT[vt])
        if vt == 8:
            return self.take_str()
        if vt == 9:
            et = self.take("<I")
            n = self.take("<Q")
            self._need(n)  # every element is >= 1 byte
            return [self.take_value(et) for _ in range(n)]
        raise ValueError(f"unknown gguf value type {vt}")


def parse_bytes(head: bytes) -> GGUFModel:
    r = _Reader(head)
    magic = r.take("<I")
    if magic != MAGIC:
        raise ValueError(f"not a GGUF file (magic {magic:#x})")
    version = r.take("<I")
    n_tensors = r.take("<Q")
    n_kv = r.take("<Q")
    # every entry consumes multiple bytes; counts beyond the data size
    # are crafted (would loop for ages before running dry)
    if n_tensors > len(head) or n_kv > len(head):
        raise ValueError(
            f"gguf: implausible counts ({n_tensors} tensors, {n_kv} kv "
            f"in {len(head)} bytes)")
    kv = {}
    for _ in range(n_kv):
        k = r.take_str()
        vt = r.take("<I")
        kv[k] = r.take_value(vt)
    tensors = []
    for _ in range(n_tensors):
        name = r.take_str()
        nd = r.take("<I")
        dims = tuple(r.take("<Q") for _ in range(nd))
        tid = r.take("<I")
        off = r.take("<Q")
        if tid not in GGML_TYPES:
            raise ValueError(
                f"unsupported ggml tensor type {tid} for {name!r}")
        tensors.append(GGUFTensor(name, dims, tid, off))
    align = int(kv.get("general.alignment", 32))
    data_offset = (r.o + align - 1) // align * align
    return GGUFModel(version=version, kv=kv, tensors=tensors,
                     data_offset=data_offset, alignment=align)


def parse(blob) -> GGUFModel:
    """Parse from a LandedBlob's head bytes."""
    gg = parse_bytes(blob.head)
    gg.blob = blob
    return gg


# ------------------------------------------------------------------ #
# synthesis (tests + synthetic benchmarks)

def _build_header(tensors: list[tuple[str, tuple[int, ...], int]],
                  kv: dict | None = None):
    """Serialize a GGUF v3 header for (name, dims, type_id) tensors.
    Returns (head_bytes, infos, data_offset, align, payload_bytes)."""
    kv = dict(kv or {})
    kv.setdefault("general.architecture", "llama")
    kv.setdefault("general.alignment", 32)

    def enc_str(s: str) -> bytes:
        b = s.encode()
        return struct.pack("<Q", len(b)) + b

    def enc_kv(k: str, v) -> bytes:
        if isinstance(v, bool):
            return enc_str(k) + struct.pack("<IB", 7, int(v))
        if isinstance(v, int):
            return enc_str(k) + struct.pack("<Iq", 11, v)
        if isinstance(v, float):
            return enc_str(k) + struct.pack("<If", 6, v)
        if isinstance(v, str):
            return enc_str(k) + struct.pack("<I", 8) + enc_str(v)
        raise TypeError(type(v))

    head = struct.pack("<IIQQ", MAGIC, 3, len(tensors), len(kv))
    for k, v in kv.items():
        head += enc_kv(k, v)
    infos = []
    off = 0
    align = int(kv["general.alignment"])
    for name, dims, tid in tensors:
        t = GGUFTensor(name, dims, tid, off)
        infos.append(t)
        off += (t.nbytes + align - 1) // align * align
    for t in infos:
        head += enc_str(t.name)
        head += struct.pack("<I", len(t.dims))
        for d in t.dims:
            head += struct.pack("<Q", d)
        head += struct.pack("<IQ", t.type_id, t.offset)
    data_offset = (len(head) + align - 1) // align * align
    return head, infos, data_offset, align, off


def build_virtual(tensors: list[tuple[str, tuple[int, ...], int]],
                  kv: dict | None = None) -> tuple[bytes, int]:
    """A GGUF blob for virtual origins: (prefix_bytes, total_size).

    The prefix is the REAL serialized header (padded to data_offset);
    payloads are left to the origin's tiled pattern — any bytes are
    numerically valid quant blocks, so dequant-at-scale benchmarks
    (llama3:70b, BASELINE config 4 at nameplate size) run without a
    41 GB file on disk."""
    head, infos, data_offset, align, payload = _build_header(tensors, kv)
    prefix = head + b"\0" * (data_offset - len(head))
    return prefix, data_offset + payload


def build_file(path: str, tensors: list[tuple[str, tuple[int, ...], int]],
               kv: dict | None = None, rng=None) -> GGUFModel:
    """Write a synthetic GGUF file: (name, dims, type_id) with random
    quant payloads."""
    import numpy as np

    rng = rng or np.random.default_rng(0)
    head, infos, data_offset, align, _ = _build_header(tensors, kv)
    # payloads tile a fixed random block so multi-GB synthesis is IO-bound,
    # not RNG-bound
    tile = rng.integers(0, 256, size=16 << 20, dtype=np.uint8).tobytes()
    with open(path, "wb") as f:
        f.write(head)
        f.write(b"\0" * (data_offset - len(head)))
        for t in infos:
            if t.type_id in (0, 1, 30):
                # keep float payloads finite so dequant comparisons work
                vals = rng.standard_normal(t.n_elems)
                if t.type_id == 0:
                    payload = vals.astype(np.float32).tobytes()
                elif t.type_id == 1:
                    payload = vals.astype(np.float16).tobytes()
                else:
                    payload = (vals.astype(np.float32).view(np.uint32)
                               >> 16).astype(np.uint16).tobytes()
                f.write(payload)
            else:
                left = t.nbytes
                while left > 0:
                    take = min(left, len(tile))
                    f.write(tile[:take])
                    left -= take
            pad = (t.nbytes + align - 1) // align * align - t.nbytes
            f.write(b"\0" * pad)
    gg = parse_bytes(open(path, "rb").read(min(
        os.path.getsize(path), 8 << 20)))
    return gg


# ------------------------------------------------------------------ #
# CPU reference dequantization (tests; mirrors csrc/gguf_dequant.hip)

def dequant_cpu(type_id: int, raw: bytes, n_elems: int):
    """Reference dequant -> float32 numpy array (GGML layouts)."""
    import numpy as np

    # synthetic test payloads contain random f16 scales (NaN/inf): the
    # products are still bit-comparable against the GPU kernel, so the
    # IEEE warnings are noise
    with np.errstate(invalid="ignore", over="ignore"):
        return _dequant_cpu_impl(type_id, raw, n_elems)


def _dequant_cpu_impl(type_id: int, raw: bytes, n_elems: int):
    import numpy as np

    if type_id == 0:
        return np.frombuffer(raw, np.float32)[:n_elems].copy()
    if type_id == 1:
        return np.frombuffer(raw, np.float16)[:n_elems].astype(np.float32)
    if type_id == 30:
        u = np.frombuffer(raw, np.uint16)[:n_elems].astype(np.uint32) << 16
        return u.view(np.float32)
    b = np.frombuffer(raw, np.uint8)
    if type_id == 2:  # q4_0
        nb = n_elems // 32
        blk = b[:nb * 18].reshape(nb, 18)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)
        qs = blk[:, 2:18]
        lo = (qs & 0xF).astype(np.int32) - 8
        hi = (qs >> 4).astype(np.int32) - 8
        out = np.concatenate([lo, hi], axis=1).astype(np.float32)
        return (out * d).reshape(-1)
    if type_id == 8:  # q8_0
        nb = n_elems // 32
        blk = b[:nb * 34].reshape(nb, 34)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)
        q = blk[:, 2:34].copy().view(np.int8).astype(np.float32)
        return (q * d).reshape(-1)
    if type_id == 12:  # q4_K
        nb = n_elems // 256
        blk = b[:nb * 144].reshape(nb, 144)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)[:, 0]
        dmin = blk[:, 2:4].copy().view(np.float16).astype(np.float32)[:, 0]
        scales = blk[:, 4:16]
        qs = blk[:, 16:144]
        out = np.empty((nb, 256), np.float32)
        for j in range(8):  # 32-elem sub-blocks
            if j < 4:
                sc = (scales[:, j] & 63).astype(np.float32)
                mn = (scales[:, j + 4] & 63).astype(np.float32)
            else:
                sc = ((scales[:, j + 4] & 0xF)
                      | ((scales[:, j - 4] >> 6) << 4)).astype(np.float32)
                mn = ((scales[:, j + 4] >> 4)
                      | ((scales[:, j] >> 6) << 4)).astype(np.float32)
            pair = j // 2            # 64-elem group
            hi_nib = j % 2
            q8 = qs[:, pair * 32:(pair + 1) * 32]
            nib = (q8 >> 4) if hi_nib else (q8 & 0xF)
            out[:, j * 32:(j + 1) * 32] = (
                d[:, None] * sc[:, None] * nib.astype(np.float32)
                - dmin[:, None] * mn[:, None])
        return out.reshape(-1)
    if type_id == 3:  # q4_1
        nb = n_elems // 32
        blk = b[:nb * 20].reshape(nb, 20)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)
        m = blk[:, 2:4].copy().view(np.float16).astype(np.float32)
        qs = blk[:, 4:20]
        lo = (qs & 0xF).astype(np.float32)
        hi = (qs >> 4).astype(np.float32)
        out = np.concatenate([lo, hi], axis=1)
        return (out * d + m).reshape(-1)
    if type_id == 6:  # q5_0
        nb = n_elems // 32
        blk = b[:nb * 22].reshape(nb, 22)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)
        qh = blk[:, 2:6].copy().view(np.uint32)[:, 0]
        qs = blk[:, 6:22]
        j = np.arange(16)
        xh0 = ((qh[:, None] >> j) << 4) & 0x10
        xh1 = (qh[:, None] >> (j + 12)) & 0x10
        lo = ((qs & 0xF) | xh0).astype(np.int32) - 16
        hi = ((qs >> 4) | xh1).astype(np.int32) - 16
        out = np.concatenate([lo, hi], axis=1).astype(np.float32)
        return (out * d).reshape(-1)
    if type_id == 7:  # q5_1
        nb = n_elems // 32
        blk = b[:nb * 24].reshape(nb, 24)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)
        m = blk[:, 2:4].copy().view(np.float16).astype(np.float32)
        qh = blk[:, 4:8].copy().view(np.uint32)[:, 0]
        qs = blk[:, 8:24]
        j = np.arange(16)
        xh0 = ((qh[:, None] >> j) << 4) & 0x10
        xh1 = (qh[:, None] >> (j + 12)) & 0x10
        lo = ((qs & 0xF) | xh0).astype(np.float32)
        hi = ((qs >> 4) | xh1).astype(np.float32)
        out = np.concatenate([lo, hi], axis=1)
        return (out * d + m).reshape(-1)
    if type_id == 13:  # q5_K
        nb = n_elems // 256
        blk = b[:nb * 176].reshape(nb, 176)
        d = blk[:, 0:2].copy().view(np.float16).astype(np.float32)[:, 0]
        dmin = blk[:, 2:4].copy().view(np.float16).astype(np.float32)[:, 0]
        scales = blk[:, 4:16]
        qh = blk[:, 16:48]
        qs = blk[:, 48:176]
        out = np.empty((nb, 256), np.float32)
        for j in range(8):  # 32-elem sub-blocks
            if j < 4:
                sc = (scales[:, j] & 63).astype(np.float32)
                mn = (scales[:, j + 4] & 63).astype(np.float32)
            else:
                sc = ((scales[:, j + 4] & 0xF)
                      | ((scales[:, j - 4] >> 6) << 4)).astype(np.float32)
                mn = ((scales[:, j + 4] >> 4)
                      | ((scales[:, j] >> 6) << 4)).astype(np.float32)
            pair = j // 2
            hi_nib = j % 2
            q8 = qs[:, pair * 32:(pair + 1) * 32]
            nib = (q8 >> 4) if hi_nib else (q8 & 0xF)
            q5 = nib.astype(np.float32) + (
                ((qh >> j) & 1) << 4).astype(np.float32)
            out[:, j * 32:(j + 1) * 32] = (
                d[:, None] * sc[:, None] * q5
                - dmin[:, None] * mn[:, None])
        return out.reshape(-1)
    if type_id == 11:  # q3_K
        nb = n_elems // 256
        blk = b[:nb * 110].reshape(nb, 110)
        hmask = blk[:, 0:32]
        qs = blk[:, 32:96]
        sp = blk[:, 96:108].copy().view(np.uint32)  # (nb, 3)
        d = blk[:, 108:110].copy().view(np.float16).astype(np.float32)[:, 0]
        km1, km2 = 0x03030303, 0x0f0f0f0f
        w0, w1, w2 = sp[:, 0], sp[:, 1], sp[:, 2]
        aux = np.stack([
            (w0 & km2) | (((w2 >> 0) & km1) << 4),
            (w1 & km2) | (((w2 >> 2) & km1) << 4),
            ((w0 >> 4) & km2) | (((w2 >> 4) & km1) << 4),
            ((w1 >> 4) & km2) | (((w2 >> 6) & km1) << 4),
        ], axis=1)  # (nb, 4) u32 -> 16 bytes = 16 scales
        sc16 = aux.view(np.uint8).reshape(nb, 16).astype(np.int32) - 32
        out = np.empty((nb, 256), np.float32)
        for nh in range(2):
            for jj in range(4):
                byte = qs[:, nh * 32:(nh + 1) * 32]
                q2 = ((byte >> (jj * 2)) & 3).astype(np.int32)
                bit = nh * 4 + jj
                hm = ((hmask >> bit) & 1).astype(np.int32)
                qv = q2 - np.where(hm == 1, 0, 4)
                for half16 in range(2):
                    is_ = nh * 8 + jj * 2 + half16
                    sl = slice(half16 * 16, half16 * 16 + 16)
                    col = nh * 128 + jj * 32 + half16 * 16
                    out[:, col:col + 16] = (
                        d[:, None] * sc16[:, is_][:, None]
                        * qv[:, sl].astype(np.float32))
        return out.reshape(-1)
    if type_id == 10:  # q2_K
        nb = n_elems // 256
        blk = b[:nb * 84].reshape(nb, 84)
        scales = blk[:, 0:16]
        qs = blk[:, 16:80]
        d = blk[:, 80:82].copy().view(np.float16).astype(np.float32)[:, 0]
        dmin = blk[:, 82:84].copy().view(np.float16).astype(np.float32)[:, 0]
        out = np.empty((nb, 256), np.float32)
        for nh in range(2):
            for jj in range(4):
                byte = qs[:, nh * 32:(nh + 1) * 32]
                q2 = ((byte >> (jj * 2)) & 3).astype(np.float32)
                for half16 in range(2):
                    is_ = nh * 8 + jj * 2 + half16
                    sc = scales[:, is_]
                    sl = slice(half16 * 16, half16 * 16 + 16)
                    col = nh * 128 + jj * 32 + half16 * 16
                    out[:, col:col + 16] = (
                        d[:, None] * (sc & 0xF)[:, None].astype(np.float32)
                        * q2[:, sl]
                        - dmin[:, None]
                        * (sc >> 4)[:, None].astype(np.float32))
        return out.reshape(-1)
    if type_id == 14:  # q6_K
        nb = n_elems // 256
        blk = b[:nb * 210].reshape(nb, 210)
        ql = blk[:, 0:128]
        qh = blk[:, 128:192]
        sc = blk[:, 192:208].copy().view(np.int8)
        d = blk[:, 208:210].copy().view(np.float16).astype(np.float32)[:, 0]
        out = np.empty((nb, 256), np.float32)
        for n in range(2):
            qln = ql[:, n * 64:(n + 1) * 64]
            qhn = qh[:, n * 32:(n + 1) * 32]
            scn = sc[:, n * 8:(n + 1) * 8]
            for half in range(4):
                if half == 0:
                    q = (qln[:, 0:32] & 0xF) | (((qhn >> 0) & 3) << 4)
                elif half == 1:
                    q = (qln[:, 32:64] & 0xF) | (((qhn >> 2) & 3) << 4)
                elif half == 2:
                    q = (qln[:, 0:32] >> 4) | (((qhn >> 4) & 3) << 4)
                else:
                    q = (qln[:, 32:64] >> 4) | (((qhn >> 6) & 3) << 4)
                qv = q.astype(np.int32) - 32
                scale_idx = half * 2  # + l//16
                s0 = scn[:, scale_idx].astype(np.float32)
                s1 = scn[:, scale_idx + 1].astype(np.float32)
                seg = np.empty((nb, 32), np.float32)
                seg[:, :16] = (qv[:, :16] * s0[:, None])
                seg[:, 16:] = (qv[:, 16:] * s1[:, None])
                out[:, n * 128 + half * 32: n * 128 + (half + 1) * 32] = \
                    d[:, None] * seg
        return out.reshape(-1)
    raise ValueError(f"unsupported type {type_id}")


# ------------------------------------------------------------------ #
# GPU dequantization from a landed blob

class ProgressiveDequant:
    """Dequantize a GGUF blob WHILE it lands.

    Feed `on_range` events from a (segmented) pull
    (pull_hf(..., on_range=pd.on_range)): the header parses from the
    first range's head bytes, and each quant tensor's dequant kernel
    launches the moment the landed contiguous prefix covers its byte
    range — so the dequant (tens of ms of GPU work) hides entirely
    under the download instead of running after it.  `finish(blob)`
    launches whatever remains, syncs, and returns {name: bf16 tensor}.

    Thread-safe: on_range arrives from segment worker threads.  If the
    head bytes don't hold the whole header (huge embedded tokenizer
    metadata), everything simply launches at finish() — correctness
    never depends on the overlap.
    """

    QUANT_IDS = (2, 3, 6, 7, 8, 10, 11, 12, 13, 14)

    def __init__(self, device_index: int = 0, buffer_pool=None):
        import threading

        self._pool = buffer_pool
        self._arena_bytes = 0
        self._lock = threading.Lock()
        self._ranges: list = []     # merged completed [lo, hi)
        self._prefix = 0            # contiguous bytes landed from 0
        self._gg: GGUFModel | None = None
        self._buf = None            # DeviceBuffer being filled
        self._h = None
        self._stream = None
        self._arena = None
        self._offsets: dict = {}
        self._quants: list = []
        self._next = 0

    # -- range bookkeeping -------------------------------------------- #
    def _add_range(self, lo: int, hi: int) -> None:
        rs = self._ranges
        rs.append((lo, hi))
        rs.sort()
        merged = [rs[0]]
        for a, b in rs[1:]:
            if a <= merged[-1][1]:
                merged[-1] = (merged[-1][0], max(merged[-1][1], b))
            else:
                merged.append((a, b))
        self._ranges = merged
        self._prefix = merged[0][1] if merged[0][0] == 0 else 0

    def _init_plan(self, head: bytes) -> None:
        from ...gpu import hip

        try:
            self._gg = parse_bytes(bytes(head))
        except Exception:
            return  # header bigger than head capture: finish() covers it
        self._h = hip()
        self._stream = self._h.Stream(0)
        self._quants = [t for t in self._gg.tensors
                        if t.type_id in self.QUANT_IDS]
        out_bytes = sum(t.n_elems * 2 for t in self._quants)
        self._arena_bytes = max(out_bytes, 1)
        self._arena = (self._pool.take(self._arena_bytes)
                       if self._pool is not None else None)
        if self._arena is None:
            self._arena = self._h.DeviceBuffer(self._arena_bytes)
        off = 0
        for t in self._quants:
            self._offsets[t.name] = off
            off += t.n_elems * 2

    def _launch_covered(self) -> None:
        if self._gg is None or self._buf is None:
            return
        while self._next < len(self._quants):
            t = self._quants[self._next]
            end = self._gg.data_offset + t.offset + t.nbytes
            if end > self._prefix:
                break
            self._h.gguf_dequant(
                t.type_id, self._buf.ptr + self._gg.data_offset + t.offset,
                self._arena.ptr + self._offsets[t.name], t.n_blocks,
                self._stream.handle)
            self._next += 1

    # -- pull hooks ---------------------------------------------------- #
    def on_range(self, name, lo, hi, buf, head) -> None:
        """Pull progress hook: [lo, hi) of `name` is resident in `buf`;
        head is the first bytes (non-None when the range covers 0)."""
        with self._lock:
            if head is not None and self._gg is None:
                self._init_plan(head)
            if getattr(buf, "ptr", None) is not None:
                self._buf = buf
            self._add_range(lo, hi)
            self._launch_covered()

    def finish(self, blob) -> dict:
        """Launch any remaining tensors, sync, return {name: bf16}."""
        import torch

        with self._lock:
            if self._gg is None:
                self._init_plan(blob.head)
                if self._gg is None:
                    raise ValueError("GGUF header did not parse")
            self._early = self._next
            self._buf = blob.buffer
            self._prefix = blob.nbytes
            self._launch_covered()
            assert self._next == len(self._quants)
        self._stream.sync()
        self._gg.blob = blob
        u8 = torch.from_dlpack(self._arena.to_dlpack())
        out = {}
        for t in self._gg.tensors:
            if t.type_id in self.QUANT_IDS:
                o = self._offsets[t.name]
                out[t.name] = (u8[o:o + t.n_elems * 2]
                               .view(torch.bfloat16).view(t.dims[::-1]))
            else:
                out[t.name] = dequant_tensor_gpu(self._gg, t,
                                                 stream=self._stream)
        return out

    @property
    def launched_early(self) -> int:
        """How many tensors launched before finish() (observability)."""
        return getattr(self, "_early", self._next if self._gg else 0)

    def recycle_arena(self) -> None:
        """Hand the bf16 output arena back to the buffer pool (near
        device capacity a fresh hipMalloc of just-freed pages costs
        SECONDS of driver page reclaim — BufferPool doc).  Caller must
        have dropped every tensor view returned by finish()."""
        if self._pool is not None and self._arena is not None:
            self._pool.put(self._arena, self._arena_bytes)
        self._arena = None


def _check_extent(gg: GGUFModel, t: GGUFTensor) -> None:
    """A crafted GGUF from an untrusted registry must not drive an
    out-of-bounds HBM read: validate the tensor's byte extent against
    the landed blob before any kernel launch."""
    end = gg.data_offset + t.offset + t.nbytes
    if t.offset < 0 or end > gg.blob.nbytes:
        raise ValueError(
            f"GGUF tensor {t.name!r} extent [{gg.data_offset + t.offset},"
            f" {end}) exceeds blob size {gg.blob.nbytes}")


def dequant_all_gpu(gg: GGUFModel, stream=None) -> dict:
    """Dequantize every tensor of a landed GGUF blob -> {name: bf16
    torch tensor}.  One output arena, all launches async on one stream
    (avoids per-tensor allocation serialization)."""
    import torch

    from ...gpu import hip

    h = hip()
    own = stream is None
    stream = stream or h.Stream(0)
    quants = [t for t in gg.tensors
              if t.type_id in ProgressiveDequant.QUANT_IDS]
    for t in quants:
        _check_extent(gg, t)
    out_bytes = sum(t.n_elems * 2 for t in quants)
    arena = h.DeviceBuffer(max(out_bytes, 1))
    offsets = {}
    off = 0
    for t in quants:
        h.gguf_dequant(t.type_id,
                       gg.blob.buffer.ptr + gg.data_offset + t.offset,
                       arena.ptr + off, t.n_blocks, stream.handle)
        offsets[t.name] = off
        off += t.n_elems * 2
    if own:
        stream.sync()
    u8 = torch.from_dlpack(arena.to_dlpack())
    out = {}
    for t in gg.tensors:
        if t.type_id in ProgressiveDequant.QUANT_IDS:
            o = offsets[t.name]
            out[t.name] = (u8[o:o + t.n_elems * 2]
                           .view(torch.bfloat16).view(t.dims[::-1]))
        else:
            out[t.name] = dequant_tensor_gpu(gg, t, stream=stream)
    return out


def dequant_tensor_gpu(gg: GGUFModel, t: GGUFTensor, stream=None):
    """Dequantize one tensor from the landed blob -> torch bf16 tensor
    (shape reversed from ggml dims: torch shape = dims[::-1])."""
    import torch

    from ...gpu import hip

    h = hip()
    blob = gg.blob
    assert blob is not None and blob.device != "cpu", \
        "GPU dequant needs a GPU-landed blob"
    _check_extent(gg, t)
    src = blob.buffer.ptr + gg.data_offset + t.offset
    own_stream = stream is None
    stream = stream or h.Stream(0)
    if t.type_id in (1, 0, 30):
        # f16/f32/bf16: view + cast via torch (no custom kernel needed)
        u8 = blob.torch_u8()
        raw = u8[gg.data_offset + t.offset:
                 gg.data_offset + t.offset + t.nbytes]
        if t.type_id == 30:
            return raw.view(torch.bfloat16).view(t.dims[::-1])
        if t.type_id == 1:
            return raw.view(torch.float16).view(t.dims[::-1]).to(
                torch.bfloat16)
        return raw.view(torch.float32).view(t.dims[::-1]).to(torch.bfloat16)
    out = h.DeviceBuffer(t.n_elems * 2)
    h.gguf_dequant(t.type_id, src, out.ptr, t.n_blocks, stream.handle)
    if own_stream:
        stream.sync()
    u8 = torch.from_dlpack(out.to_dlpack())
    return u8.view(torch.bfloat16).view(t.dims[::-1])
