"""Parquet page extraction for GPU decompression (BASELINE.json config 5:
c4-en parquet streaming).

A parquet file's column chunks are sequences of pages, each preceded by a
Thrift compact-protocol PageHeader giving the compressed/uncompressed
sizes.  The heavy bytes are the page payloads; the headers are tiny.  So:
CPU walks the headers (this module — a minimal Thrift compact reader, no
parquet library in the data path), and the GPU decompresses every page
payload in one wave-parallel zstd/deflate launch into the HBM ring.

Scope: the full pyarrow codec matrix — ZSTD (csrc/zstd_kernel.hip),
SNAPPY (csrc/snappy.hip, parquet's default codec), GZIP (member header
parsed on CPU, raw DEFLATE on csrc/inflate.hip) and UNCOMPRESSED
(device copy) — for BOTH v1 and v2 data pages (v2 rep+def level
prefixes device-copy; only the values region feeds the codec kernel).
Unknown codecs fail loudly.
"""

from __future__ import annotations

from dataclasses import dataclass

# Thrift compact protocol type ids
_CT_STOP = 0
_CT_TRUE = 1
_CT_FALSE = 2
_CT_BYTE = 3
_CT_I16 = 4
_CT_I32 = 5
_CT_I64 = 6
_CT_DOUBLE = 7
_CT_BINARY = 8
_CT_LIST = 9
_CT_SET = 10
_CT_MAP = 11
_CT_STRUCT = 12


class _Compact:
    def __init__(self, data, pos=0):
        self.d = data
        self.p = pos

    def byte(self):
        b = self.d[self.p]
        self.p += 1
        return b

    def _need(self, n: int) -> None:
        # untrusted input: a declared element count beyond the
        # remaining bytes would spin a near-infinite skip loop
        if n < 0 or n > len(self.d) - self.p:
            raise ValueError(
                f"thrift: declared {n} elements with "
                f"{len(self.d) - self.p} bytes left")

    def skip_elem(self, ctype):
        """Skip one LIST/SET/MAP element: unlike struct FIELDS, bool
        elements are encoded as one byte in the compact protocol."""
        if ctype in (_CT_TRUE, _CT_FALSE):
            self.byte()
        else:
            self.skip(ctype)

    def varint(self):
        out = 0
        shift = 0
        while True:
            b = self.byte()
            out |= (b & 0x7F) << shift
            if not b & 0x80:
                return out
            shift += 7

    def zigzag(self):
        v = self.varint()
        return (v >> 1) ^ -(v & 1)

    def skip(self, ctype):
        if ctype in (_CT_TRUE, _CT_FALSE):
            return
        if ctype == _CT_BYTE:
            self.byte()
        elif ctype in (_CT_I16, _CT_I32, _CT_I64):
            self.varint()
        elif ctype == _CT_DOUBLE:
            self.p += 8
        elif ctype == _CT_BINARY:
            n = self.varint()  # NB: varint() moves p; don't fold into +=
            self.p += n
        elif ctype in (_CT_LIST, _CT_SET):
            b = self.byte()
            n = b >> 4
            et = b & 0xF
            if n == 15:
                n = self.varint()
            self._need(n)  # every element is >= 1 byte
            for _ in range(n):
                self.skip_elem(et)
        elif ctype == _CT_MAP:
            n = self.varint()
            self._need(n)
            if n:
                kv = self.byte()
                for _ in range(n):
                    self.skip_elem(kv >> 4)
                    self.skip_elem(kv & 0xF)
        elif ctype == _CT_STRUCT:
            last = 0
            while True:
                b = self.byte()
                if b == _CT_STOP:
                    return
                delta = b >> 4
                ft = b & 0xF
                last = last + delta if delta else self.zigzag()
                self.skip(ft)
        else:
            raise ValueError(f"unknown thrift compact type {ctype}")

    def read_struct_fields(self):
        """Yield (field_id, ctype) and leave .p at each value start;
        caller must consume or .skip(ctype)."""
        last = 0
        while True:
            b = self.byte()
            if b == _CT_STOP:
                return
            delta = b >> 4
            ft = b & 0xF
            if delta:
                last = last + delta
            else:
                last = self.zigzag()
            yield last, ft


@dataclass
class PageInfo:
    page_type: int          # 0 data, 2 dictionary, 3 data_v2
    comp_offset: int        # absolute offset of the page payload
    comp_size: int
    uncomp_size: int
    # v2 data pages: rep+def levels are stored UNCOMPRESSED as the first
    # lvl_bytes of the payload; only the rest is codec-compressed (and
    # only when is_compressed)
    lvl_bytes: int = 0
    is_compressed: bool = True
    # first bytes of the VALUES region (payload after any v2 level
    # prefix), captured during the header walk — GZIP pages need the
    # member header parsed on CPU (RFC 1952)
    head: bytes = b""


def parse_page_header(data, pos: int) -> tuple[PageInfo, int]:
    """Parse one PageHeader at `pos`; returns (info, payload_offset)."""
    c = _Compact(data, pos)
    page_type = None
    comp = uncomp = None
    lvl = 0
    is_comp = True

    def parse_v2(cc):
        nonlocal lvl, is_comp
        for fid2, ft2 in cc.read_struct_fields():
            if fid2 == 5:           # definition_levels_byte_length
                lvl += cc.zigzag()
            elif fid2 == 6:         # repetition_levels_byte_length
                lvl += cc.zigzag()
            elif fid2 == 7 and ft2 in (_CT_TRUE, _CT_FALSE):
                is_comp = ft2 == _CT_TRUE
            else:
                cc.skip(ft2)

    for fid, ft in c.read_struct_fields():
        if fid == 1 and ft in (_CT_I32, _CT_BYTE, _CT_I16):
            page_type = c.zigzag() if ft != _CT_BYTE else c.byte()
        elif fid == 2:
            uncomp = c.zigzag()
        elif fid == 3:
            comp = c.zigzag()
        elif fid == 8 and ft == _CT_STRUCT:
            parse_v2(c)             # DataPageHeaderV2
        else:
            c.skip(ft)
    if page_type is None or comp is None or uncomp is None:
        raise ValueError("malformed parquet PageHeader")
    if page_type != 3:
        lvl, is_comp = 0, True
    return PageInfo(page_type, c.p, comp, uncomp, lvl, is_comp), c.p


CODEC_UNCOMPRESSED = 0
CODEC_SNAPPY = 1
CODEC_GZIP = 2
CODEC_LZ4 = 5
CODEC_ZSTD = 6
CODEC_LZ4_RAW = 7


def column_chunk_pages(data, start: int, total_compressed: int
                       ) -> list[PageInfo]:
    """Walk all pages of one column chunk (headers live in `data`, a
    buffer covering [start, start+total_compressed))."""
    pages = []
    pos = start
    end = start + total_compressed
    while pos < end:
        info, payload = parse_page_header(data, pos)
        lv = payload + info.lvl_bytes
        info.head = bytes(data[lv:lv + 64])
        pages.append(info)
        pos = payload + info.comp_size
    if pos != end:
        raise ValueError(f"column chunk walk overran: {pos} != {end}")
    return pages


@dataclass
class ChunkMeta:
    codec: int
    start: int               # first page offset (dictionary if present)
    total_compressed: int


def parse_footer(footer: bytes) -> list[ChunkMeta]:
    """Parse a parquet FileMetaData (Thrift compact) for column-chunk
    locations and codecs — self-contained, no parquet library.

    `footer` is the thrift blob (file[-8-len:-8])."""
    chunks: list[ChunkMeta] = []
    c = _Compact(footer, 0)

    def parse_column_meta(cc: _Compact):
        codec = start = total = dict_off = data_off = None
        for fid, ft in cc.read_struct_fields():
            if fid == 4:
                codec = cc.zigzag()
            elif fid == 7:
                total = cc.zigzag()
            elif fid == 9:
                data_off = cc.zigzag()
            elif fid == 11:
                dict_off = cc.zigzag()
            else:
                cc.skip(ft)
        start = dict_off if dict_off not in (None, 0) else data_off
        if codec is None or total is None or start is None:
            raise ValueError("incomplete ColumnMetaData")
        return ChunkMeta(codec, start, total)

    def parse_column_chunk(cc: _Compact):
        meta = None
        for fid, ft in cc.read_struct_fields():
            if fid == 3 and ft == _CT_STRUCT:
                meta = parse_column_meta(cc)
            else:
                cc.skip(ft)
        if meta is None:
            raise ValueError("ColumnChunk without meta_data")
        return meta

    def parse_row_group(cc: _Compact):
        out = []
        for fid, ft in cc.read_struct_fields():
            if fid == 1 and ft == _CT_LIST:
                b = cc.byte()
                n = b >> 4
                et = b & 0xF
                if n == 15:
                    n = cc.varint()
                assert et == _CT_STRUCT
                for _ in range(n):
                    out.append(parse_column_chunk(cc))
            else:
                cc.skip(ft)
        return out

    for fid, ft in c.read_struct_fields():
        if fid == 4 and ft == _CT_LIST:
            b = c.byte()
            n = b >> 4
            et = b & 0xF
            if n == 15:
                n = c.varint()
            assert et == _CT_STRUCT
            for _ in range(n):
                chunks.extend(parse_row_group(c))
        else:
            c.skip(ft)
    return chunks


def footer_span(tail8: bytes) -> int:
    """Footer thrift length from a file's last 8 bytes
    (u32 LE length + b'PAR1')."""
    import struct as _struct

    if tail8[4:] != b"PAR1":
        raise ValueError("not a parquet file (missing PAR1 trailer)")
    return _struct.unpack("<I", tail8[:4])[0]


def blob_pages(blob) -> list[tuple[int, "PageInfo"]]:
    """Page locations of a parquet blob LANDED IN HBM: footer + page
    headers are read back via small staged D2H copies (payload bytes
    never leave the device).  Returns [(codec, PageInfo)]."""
    import ctypes

    from ...gpu import hip

    h = hip()
    s = h.Stream(0)

    def d2h(off: int, n: int) -> bytes:
        n = min(n, blob.nbytes - off)
        out = bytearray(n)
        addr = ctypes.addressof((ctypes.c_char * n).from_buffer(out))
        h.d2h_async(addr, blob.buffer.ptr + off, n, s.handle)
        s.sync()
        return bytes(out)

    flen = footer_span(d2h(blob.nbytes - 8, 8))
    footer = d2h(blob.nbytes - 8 - flen, flen)
    out = []
    for cm in parse_footer(footer):
        # walk headers with a sliding 64 KiB host window
        pos = cm.start
        end = cm.start + cm.total_compressed
        win_off = -1
        win = b""
        while pos < end:
            if win_off < 0 or pos + 512 > win_off + len(win):
                win_off = pos
                win = d2h(pos, 64 << 10)
            info, payload = parse_page_header(win, pos - win_off)
            abs_payload = payload + win_off
            lv = payload + info.lvl_bytes
            head = bytes(win[lv:lv + 64])
            if len(head) < min(64, info.comp_size - info.lvl_bytes):
                head = d2h(abs_payload + info.lvl_bytes, 64)
            info = PageInfo(info.page_type, abs_payload,
                            info.comp_size, info.uncomp_size,
                            info.lvl_bytes, info.is_compressed, head)
            out.append((cm.codec, info))
            pos = info.comp_offset + info.comp_size
        if pos != end:
            raise ValueError("page walk overran column chunk")
    return out


def file_pages(path: str):
    """All (column-chunk codec, PageInfo) of a parquet file, using
    pyarrow only for the FOOTER metadata (offsets/codecs), never for page
    payloads.  Returns (file_bytes, [(codec, PageInfo)])."""
    import pyarrow.parquet as pq

    meta = pq.ParquetFile(path).metadata
    raw = open(path, "rb").read()
    out = []
    codec_names = {"UNCOMPRESSED": CODEC_UNCOMPRESSED,
                   "SNAPPY": CODEC_SNAPPY, "GZIP": CODEC_GZIP,
                   "ZSTD": CODEC_ZSTD,
                   # arrow writes both as raw LZ4 blocks in pages
                   "LZ4": CODEC_LZ4, "LZ4_RAW": CODEC_LZ4_RAW}
    for rg in range(meta.num_row_groups):
        for col in range(meta.num_columns):
            cc = meta.row_group(rg).column(col)
            codec = codec_names.get(cc.compression.upper())
            if codec is None:
                raise ValueError(f"unsupported codec {cc.compression}")
            start = cc.dictionary_page_offset
            if start is None or start <= 0:
                start = cc.data_page_offset
            for info in column_chunk_pages(raw, start,
                                           cc.total_compressed_size):
                out.append((codec, info))
    return raw, out


def prep_pages_gpu(blob, pages, ring=None):
    """Build the decode plan for a landed parquet blob's pages: returns
    (frames, snappy, deflate, copies, ring, spans) — zstd / snappy /
    raw-DEFLATE (gzip pages, member header parsed from PageInfo.head)
    kernel inputs plus (dst, src, n) device copies for UNCOMPRESSED
    pages and v2 level prefixes."""
    from ...gpu import hip
    from .compress import gzip_deflate_offset

    h = hip()
    total = sum(p.uncomp_size for _, p in pages)
    ring = ring or h.DeviceBuffer(max(total, 1))
    frames = []
    snappy = []
    deflate = []
    lz4 = []
    copies = []
    spans = []
    off = 0
    for codec, p in pages:
        lvl = p.lvl_bytes
        if lvl:
            # v2: rep+def levels pass through uncompressed
            copies.append((ring.ptr + off,
                           blob.buffer.ptr + p.comp_offset, lvl))
        src = blob.buffer.ptr + p.comp_offset + lvl
        dst = ring.ptr + off + lvl
        clen = p.comp_size - lvl
        ulen = p.uncomp_size - lvl
        if not p.is_compressed or codec == CODEC_UNCOMPRESSED:
            copies.append((dst, src, clen))
        elif codec == CODEC_ZSTD:
            frames.append((src, clen, dst, ulen))
        elif codec == CODEC_SNAPPY:
            snappy.append((src, clen, dst, ulen))
        elif codec in (CODEC_LZ4, CODEC_LZ4_RAW):
            lz4.append((src, clen, dst, ulen))
        elif codec == CODEC_GZIP:
            gz = gzip_deflate_offset(p.head)
            # raw DEFLATE stream between the member header and the
            # 8-byte crc32+isize trailer
            deflate.append((src + gz, clen - gz - 8, dst, ulen))
        else:
            raise ValueError(f"GPU path supports ZSTD/SNAPPY/GZIP/LZ4/"
                             f"UNCOMPRESSED, got codec {codec}")
        spans.append((off, p.uncomp_size))
        off += p.uncomp_size
    return frames, snappy, deflate, lz4, copies, ring, spans


def launch_pages_gpu(blob, pages, ring=None):
    """Queue GPU decompression of a landed parquet blob's ZSTD pages
    into an HBM ring, ASYNC on the job's own stream.  Returns (job,
    ring, spans); call job.wait() and check the results before reading
    the ring."""
    from ...gpu import hip
    from .compress import ZstdJob

    h = hip()
    frames, snappy, deflate, lz4, copies, ring, spans = prep_pages_gpu(
        blob, pages, ring)

    def pre(handle):
        for dst, src, n in copies:
            h.d2d_async(dst, src, n, handle)

    # 16 KiB LDS window: page batches run as concurrent jobs in
    # stream_dataset, so occupancy beats far-match locality
    job = ZstdJob(frames, pre_launch=pre, window=16 << 10,
                  snappy_frames=snappy, deflate_frames=deflate,
                  lz4_frames=lz4)
    return job, ring, spans


def decompress_pages_gpu(blob, pages, ring=None):
    """Synchronous wrapper around launch_pages_gpu: returns
    (ring_buffer, [(out_offset, size)]) covering every page."""
    job, ring, spans = launch_pages_gpu(blob, pages, ring)
    results = (job.wait() + job.snappy_results + job.deflate_results
               + job.lz4_results)
    bad = [(i, r) for i, r in enumerate(results) if not r.ok]
    if bad:
        raise IOError(f"GPU page decompress failed: {bad[:3]}")
    return ring, spans
