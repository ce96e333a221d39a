"""Minimal BAM/BGZF reader + writer (pure Python, zlib-based).

The reference uses pysam for all BAM I/O (pre_lib.py:50-91 SubreadGrouper,
quick_inference.py:447-482 stream_bam, :738-760 BAM output). pysam is not
available in this environment, so this module implements the subset of the
BAM format (SAM spec v1.6 section 4) the pipeline needs:

* BGZF block framing (gzip members with the BC extra field), read and write;
* header parsing (text + reference list);
* alignment record decode/encode: flag, pos, mapq, cigar, seq, qual, and the
  aux tags used by DeepConsensus (zm, pw, ip, sn, ec, np, rq, RG, wl);
* a pysam-like BamRead API surface: qname, is_unmapped/is_reverse/
  is_supplementary, pos, reference_name, cigartuples, seq, query_qualities,
  get_tag/has_tag, query_alignment_start/end, get_aligned_pairs.

Random access (.bai) is intentionally not implemented; "fetch by reference
name" for truth-to-CCS lookups is provided by a sequential scan into a dict
(fetch_index), which matches how DeepConsensus actually uses the index
(pre_lib.py:1001-1014: one alignment per CCS read name).
"""
from __future__ import annotations

import os
import struct
import zlib
from typing import Any, Dict, Iterator, List, Optional, Sequence, Tuple

import numpy as np

from deepconsensus_amd.utils import constants

BAM_MAGIC = b"BAM\x01"
_CIGAR_CHARS = "MIDNSHP=XB"
_SEQ_NT16 = "=ACMGRSVTWYHKDBN"
_NT16_OF = {c: i for i, c in enumerate(_SEQ_NT16)}
_NT16_U8 = np.frombuffer(_SEQ_NT16.encode("ascii"), np.uint8)

# Flag bits.
FUNMAP = 0x4
FREVERSE = 0x10
FSECONDARY = 0x100
FSUPPLEMENTARY = 0x800


# ---------------------------------------------------------------------------
# BGZF framing
# ---------------------------------------------------------------------------

BGZF_EOF = bytes.fromhex(
    "1f8b08040000000000ff0600424302001b0003000000000000000000"
)


def _read_bgzf_blocks(fh) -> Iterator[bytes]:
    """Yields decompressed BGZF block payloads."""
    while True:
        header = fh.read(12)
        if len(header) == 0:
            return
        if len(header) < 12:
            raise ValueError("truncated BGZF header")
        magic, _mtime, _xfl, _os, xlen = struct.unpack("<4sLBBH", header)
        if magic[:2] != b"\x1f\x8b":
            raise ValueError("not a gzip stream")
        extra = fh.read(xlen)
        bsize = None
        off = 0
        while off + 4 <= len(extra):
            si1, si2, slen = struct.unpack_from("<BBH", extra, off)
            if si1 == 66 and si2 == 67 and slen == 2:
                bsize = struct.unpack_from("<H", extra, off + 4)[0]
            off += 4 + slen
        if bsize is None:
            raise ValueError("missing BGZF BC extra field")
        cdata_len = bsize - xlen - 19
        cdata = fh.read(cdata_len)
        _crc, isize = struct.unpack("<LL", fh.read(8))
        data = zlib.decompress(cdata, -15)
        if len(data) != isize:
            raise ValueError("BGZF ISIZE mismatch")
        if data:
            yield data


def _bgzf_compress_block(data: bytes, level: int = 6) -> bytes:
    co = zlib.compressobj(level, zlib.DEFLATED, -15)
    cdata = co.compress(data) + co.flush()
    # BSIZE field = total block length - 1 = 12 + xlen(6) + cdata + 8 - 1.
    bsize = len(cdata) + 25
    header = struct.pack(
        "<4sLBBHBBHH",
        b"\x1f\x8b\x08\x04",
        0,
        0,
        255,
        6,
        66,
        67,
        2,
        bsize,
    )
    footer = struct.pack("<LL", zlib.crc32(data) & 0xFFFFFFFF, len(data))
    return header + cdata + footer


class BgzfWriter:
    """Streams data into BGZF blocks (<=64 KiB payload each)."""

    def __init__(self, fh, level: int = 6):
        self.fh = fh
        self.level = level
        self.buf = bytearray()

    def write(self, data: bytes) -> None:
        self.buf += data
        while len(self.buf) >= 65000:
            chunk = bytes(self.buf[:65000])
            del self.buf[:65000]
            self.fh.write(_bgzf_compress_block(chunk, self.level))

    def close(self) -> None:
        if self.buf:
            self.fh.write(_bgzf_compress_block(bytes(self.buf), self.level))
            self.buf = bytearray()
        self.fh.write(BGZF_EOF)
        self.fh.flush()


# ---------------------------------------------------------------------------
# Alignment record
# ---------------------------------------------------------------------------


class BamRead:
    """One BAM alignment record with a pysam-like API subset."""

    __slots__ = (
        "qname", "flag", "ref_id", "pos", "mapq", "cigartuples", "seq",
        "query_qualities", "tags", "_header", "next_ref_id", "next_pos",
        "tlen",
    )

    def __init__(
        self,
        qname: str = "",
        flag: int = 0,
        ref_id: int = -1,
        pos: int = -1,
        mapq: int = 255,
        cigartuples: Optional[List[Tuple[int, int]]] = None,
        seq: str = "",
        query_qualities: Optional[Sequence[int]] = None,
        tags: Optional[Dict[str, Any]] = None,
        header: Optional["BamHeader"] = None,
    ):
        self.qname = qname
        self.flag = flag
        self.ref_id = ref_id
        self.pos = pos
        self.mapq = mapq
        self.cigartuples = cigartuples or []
        self.seq = seq
        self.query_qualities = (
            np.asarray(query_qualities, dtype=np.int16)
            if query_qualities is not None
            else None
        )
        self.tags = tags or {}
        self._header = header
        self.next_ref_id = -1
        self.next_pos = -1
        self.tlen = 0

    # pysam-compatible surface --------------------------------------------
    @property
    def query_sequence(self) -> str:
        return self.seq

    @query_sequence.setter
    def query_sequence(self, v: str) -> None:
        self.seq = v

    @property
    def is_unmapped(self) -> bool:
        return bool(self.flag & FUNMAP)

    @property
    def is_reverse(self) -> bool:
        return bool(self.flag & FREVERSE)

    @property
    def is_supplementary(self) -> bool:
        return bool(self.flag & FSUPPLEMENTARY)

    @property
    def reference_name(self) -> Optional[str]:
        if self._header is None or self.ref_id < 0:
            return None
        return self._header.references[self.ref_id][0]

    @property
    def reference_start(self) -> int:
        return self.pos

    @property
    def cigarstring(self) -> str:
        return "".join(f"{n}{_CIGAR_CHARS[op]}" for op, n in self.cigartuples)

    @property
    def cigar(self) -> List[Tuple[int, int]]:
        return self.cigartuples

    def get_tag(self, name: str) -> Any:
        return self.tags[name]

    def has_tag(self, name: str) -> bool:
        return name in self.tags

    def set_tag(self, name: str, value: Any) -> None:
        self.tags[name] = value

    @property
    def query_alignment_start(self) -> int:
        qpos = 0
        for op, n in self.cigartuples:
            if op == constants.CSOFT_CLIP:
                qpos += n
            elif op == constants.CHARD_CLIP:
                continue
            else:
                break
        return qpos

    @property
    def query_alignment_end(self) -> int:
        qlen = len(self.seq)
        clip = 0
        for op, n in reversed(self.cigartuples):
            if op == constants.CHARD_CLIP:
                continue
            if op == constants.CSOFT_CLIP:
                clip += n
            else:
                break
        return qlen - clip

    def get_aligned_pairs(self) -> List[Tuple[Optional[int], Optional[int]]]:
        """(query_pos, ref_pos) pairs per aligned column (SAM semantics)."""
        pairs: List[Tuple[Optional[int], Optional[int]]] = []
        qpos, rpos = 0, self.pos
        for op, n in self.cigartuples:
            if op in (constants.CMATCH, constants.CEQUAL, constants.CDIFF):
                for _ in range(n):
                    pairs.append((qpos, rpos))
                    qpos += 1
                    rpos += 1
            elif op in (constants.CINS, constants.CSOFT_CLIP):
                for _ in range(n):
                    pairs.append((qpos, None))
                    qpos += 1
            elif op in (constants.CDEL, constants.CREF_SKIP):
                for _ in range(n):
                    pairs.append((None, rpos))
                    rpos += 1
            # CHARD_CLIP / CPAD consume nothing.
        return pairs

    def aligned_index_arrays(self) -> Tuple[np.ndarray, np.ndarray]:
        """Vectorized get_aligned_pairs: (read_idx, ccs_idx) int32 arrays
        per aligned column, -1 where the side is absent. Loops over cigar
        OPS, not bases."""
        total = 0
        for op, n in self.cigartuples:
            if op not in (constants.CHARD_CLIP, constants.CPAD):
                total += n
        read_idx = np.full(total, -1, np.int32)
        ccs_idx = np.full(total, -1, np.int32)
        qpos, rpos, out = 0, self.pos, 0
        for op, n in self.cigartuples:
            if op in (constants.CMATCH, constants.CEQUAL, constants.CDIFF):
                read_idx[out:out + n] = np.arange(qpos, qpos + n)
                ccs_idx[out:out + n] = np.arange(rpos, rpos + n)
                qpos += n
                rpos += n
            elif op in (constants.CINS, constants.CSOFT_CLIP):
                read_idx[out:out + n] = np.arange(qpos, qpos + n)
                qpos += n
            elif op in (constants.CDEL, constants.CREF_SKIP):
                ccs_idx[out:out + n] = np.arange(rpos, rpos + n)
                rpos += n
            else:  # CHARD_CLIP / CPAD consume nothing
                continue
            out += n
        return read_idx, ccs_idx

    def infer_query_length(self) -> int:
        return len(self.seq)

    def __repr__(self):
        return f"BamRead({self.qname} flag={self.flag} pos={self.pos})"


class BamHeader:
    def __init__(self, text: str = "", references=None):
        self.text = text
        self.references: List[Tuple[str, int]] = references or []

    @property
    def ref_index(self) -> Dict[str, int]:
        return {name: i for i, (name, _l) in enumerate(self.references)}


# ---------------------------------------------------------------------------
# Tag codec
# ---------------------------------------------------------------------------

_TAG_FMT = {"c": "b", "C": "B", "s": "h", "S": "H", "i": "i", "I": "I",
            "f": "f"}
_ARRAY_DTYPE = {"c": np.int8, "C": np.uint8, "s": np.int16, "S": np.uint16,
                "i": np.int32, "I": np.uint32, "f": np.float32}


def _parse_tags(buf: bytes) -> Dict[str, Any]:
    tags: Dict[str, Any] = {}
    off = 0
    n = len(buf)
    while off + 3 <= n:
        name = buf[off:off + 2].decode()
        typ = chr(buf[off + 2])
        off += 3
        if typ == "A":
            tags[name] = chr(buf[off])
            off += 1
        elif typ in _TAG_FMT:
            fmt = _TAG_FMT[typ]
            size = struct.calcsize(fmt)
            tags[name] = struct.unpack_from("<" + fmt, buf, off)[0]
            off += size
        elif typ == "Z":
            end = buf.index(0, off)
            tags[name] = buf[off:end].decode()
            off = end + 1
        elif typ == "H":
            end = buf.index(0, off)
            tags[name] = buf[off:end].decode()
            off = end + 1
        elif typ == "B":
            sub = chr(buf[off])
            count = struct.unpack_from("<I", buf, off + 1)[0]
            dt = _ARRAY_DTYPE[sub]
            arr = np.frombuffer(
                buf, dtype=dt, count=count, offset=off + 5
            ).copy()
            tags[name] = arr
            off += 5 + count * arr.dtype.itemsize
        else:
            raise ValueError(f"unknown tag type {typ!r}")
    return tags


def _encode_tags(tags: Dict[str, Any]) -> bytes:
    out = bytearray()
    for name, val in tags.items():
        key = name.encode()
        if isinstance(val, str) and len(val) == 1 and name in ("rs",):
            out += key + b"A" + val.encode()
        elif isinstance(val, (bool, int, np.integer)):
            v = int(val)
            if -2147483648 <= v <= 2147483647:
                out += key + b"i" + struct.pack("<i", v)
            else:
                out += key + b"I" + struct.pack("<I", v)
        elif isinstance(val, (float, np.floating)):
            out += key + b"f" + struct.pack("<f", float(val))
        elif isinstance(val, str):
            out += key + b"Z" + val.encode() + b"\x00"
        elif isinstance(val, (list, tuple, np.ndarray)):
            arr = np.asarray(val)
            if arr.dtype.kind == "f":
                sub, dt = "f", np.float32
            elif arr.dtype.kind in "iu" and arr.min(initial=0) >= 0 and arr.max(initial=0) <= 255:
                sub, dt = "C", np.uint8
            elif arr.dtype.kind in "iu" and abs(arr).max(initial=0) <= 32767:
                sub, dt = "s", np.int16
            else:
                sub, dt = "i", np.int32
            arr = arr.astype(dt)
            out += (
                key + b"B" + sub.encode()
                + struct.pack("<I", arr.size) + arr.tobytes()
            )
        else:
            raise ValueError(f"cannot encode tag {name}={val!r}")
    return bytes(out)


# ---------------------------------------------------------------------------
# Reader / writer
# ---------------------------------------------------------------------------


def _read_bgzf_blocks_parallel(fh, threads: int = 4, window: int = 64
                               ) -> Iterator[bytes]:
    """Decompresses BGZF blocks on a thread pool, yielding in file order.

    zlib.decompress releases the GIL, so block decompression scales with
    real threads; the serial part left on the caller is only the raw
    file reads and the yield. This is the feeder-side answer to the
    single-core BGZF decode ceiling (ROADMAP: host pipeline item 1).
    """
    import collections as _collections
    import concurrent.futures as _futures

    def read_compressed():
        """Yields (cdata, isize) without decompressing."""
        while True:
            header = fh.read(12)
            if len(header) == 0:
                return
            if len(header) < 12:
                raise ValueError("truncated BGZF header")
            magic, _mtime, _xfl, _os, xlen = struct.unpack(
                "<4sLBBH", header
            )
            if magic[:2] != b"\x1f\x8b":
                raise ValueError("not a gzip stream")
            extra = fh.read(xlen)
            bsize = None
            off = 0
            while off + 4 <= len(extra):
                si1, si2, slen = struct.unpack_from("<BBH", extra, off)
                if si1 == 66 and si2 == 67 and slen == 2:
                    bsize = struct.unpack_from("<H", extra, off + 4)[0]
                off += 4 + slen
            if bsize is None:
                raise ValueError("missing BGZF BC extra field")
            cdata = fh.read(bsize - xlen - 19)
            _crc, isize = struct.unpack("<LL", fh.read(8))
            yield cdata, isize

    def decompress(job):
        cdata, isize = job
        data = zlib.decompress(cdata, -15)
        if len(data) != isize:
            raise ValueError("BGZF ISIZE mismatch")
        return data

    with _futures.ThreadPoolExecutor(threads) as pool:
        pending: "_collections.deque" = _collections.deque()
        src = read_compressed()
        for job in src:
            pending.append(pool.submit(decompress, job))
            if len(pending) >= window:
                data = pending.popleft().result()
                if data:
                    yield data
        while pending:
            data = pending.popleft().result()
            if data:
                yield data


def _threaded_blocks(gen: Iterator[bytes], depth: int = 64
                     ) -> Iterator[bytes]:
    """Readahead thread: BGZF read+decompress runs ahead of the record
    parser (the serial ZMW feeder's floor — profiles/r01_perf_journal)."""
    import queue
    import threading

    q: "queue.Queue" = queue.Queue(maxsize=depth)
    _END = object()

    def pump():
        try:
            for block in gen:
                q.put(block)
            q.put(_END)
        except BaseException as e:  # surface decode errors to the consumer
            q.put(e)

    t = threading.Thread(target=pump, daemon=True)
    t.start()
    while True:
        item = q.get()
        if item is _END:
            return
        if isinstance(item, BaseException):
            raise item
        yield item


def _parse_bam_header(stream: "_ConcatStream", path: str) -> "BamHeader":
    magic = stream.read(4)
    if magic != BAM_MAGIC:
        raise ValueError(f"{path} is not a BAM file")
    (l_text,) = struct.unpack("<l", stream.read(4))
    text = stream.read(l_text).decode(errors="replace").rstrip("\x00")
    (n_ref,) = struct.unpack("<l", stream.read(4))
    refs = []
    for _ in range(n_ref):
        (l_name,) = struct.unpack("<l", stream.read(4))
        name = stream.read(l_name)[:-1].decode()
        (l_ref,) = struct.unpack("<l", stream.read(4))
        refs.append((name, l_ref))
    return BamHeader(text, refs)


class BamReader:
    """Sequential BAM reader (with a decompression readahead thread)."""

    def __init__(self, path: str, reader_threads: int = 1):
        self.path = path
        self._fh = open(path, "rb")
        blocks = _read_bgzf_blocks(self._fh)
        if reader_threads and reader_threads > 0:
            blocks = _threaded_blocks(blocks)
        self._stream = _ConcatStream(blocks)
        self.header = _parse_bam_header(self._stream, path)

    def __iter__(self) -> Iterator[BamRead]:
        return self

    def __next__(self) -> BamRead:
        head = self._stream.read(4)
        if len(head) < 4:
            raise StopIteration
        (block_size,) = struct.unpack("<l", head)
        buf = self._stream.read(block_size)
        return self._decode(buf)

    def _decode(self, buf: bytes) -> BamRead:
        return decode_record(buf, self.header)

    def close(self):
        self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def decode_record(buf: bytes, header: Optional["BamHeader"]) -> BamRead:
    """Decodes one raw BAM record buffer (everything after block_size)."""
    (ref_id, pos, l_read_name, mapq, _bin, n_cigar, flag, l_seq,
     next_ref, next_pos, tlen) = struct.unpack_from("<llBBHHHllll", buf, 0)
    off = 32
    qname = buf[off:off + l_read_name - 1].decode()
    off += l_read_name
    # Vectorized cigar + 4-bit seq decode (the per-base python loops
    # were the hottest lines of the whole serial feeder path).
    cig = np.frombuffer(buf, "<u4", n_cigar, off)
    cigartuples = list(zip((cig & 0xF).tolist(), (cig >> 4).tolist()))
    off += 4 * n_cigar
    nbytes = (l_seq + 1) // 2
    packed = np.frombuffer(buf, np.uint8, nbytes, off)
    chars = np.empty(2 * nbytes, np.uint8)
    chars[0::2] = _NT16_U8[packed >> 4]
    chars[1::2] = _NT16_U8[packed & 0xF]
    seq = chars[:l_seq].tobytes().decode("ascii")
    off += nbytes
    quals = np.frombuffer(buf, np.uint8, l_seq, off).astype(np.int16)
    if l_seq and quals.size and quals[0] == 0xFF:
        qual_arr = None
    else:
        qual_arr = quals
    off += l_seq
    tags = _parse_tags(buf[off:])
    read = BamRead(
        qname=qname, flag=flag, ref_id=ref_id, pos=pos, mapq=mapq,
        cigartuples=cigartuples, seq=seq, query_qualities=qual_arr,
        tags=tags, header=header,
    )
    read.next_ref_id, read.next_pos, read.tlen = next_ref, next_pos, tlen
    return read


# ---------------------------------------------------------------------------
# Raw-record peeks (no full decode)
# ---------------------------------------------------------------------------
# Fixed-offset layout of a raw record buffer: ref_id@0, pos@4,
# l_read_name@8, mapq@9, bin@10, n_cigar@12, flag@14, l_seq@16,
# next_ref@20, next_pos@24, tlen@28, qname@32.


def raw_flag(buf: bytes) -> int:
    return struct.unpack_from("<H", buf, 14)[0]


def raw_ref_id(buf: bytes) -> int:
    return struct.unpack_from("<l", buf, 0)[0]


def raw_qname(buf: bytes) -> str:
    l_read_name = buf[8]
    return buf[32:32 + l_read_name - 1].decode()


def raw_tag(buf: bytes, name: str, default: Any = None) -> Any:
    """Minimal tag scan: walks tag headers, decoding only `name`.

    ~10x cheaper than a full decode_record when only the zm group key
    (or the wl smart-window widths) is needed on the serial feeder path.
    """
    l_read_name = buf[8]
    n_cigar = struct.unpack_from("<H", buf, 12)[0]
    l_seq = struct.unpack_from("<l", buf, 16)[0]
    off = 32 + l_read_name + 4 * n_cigar + (l_seq + 1) // 2 + l_seq
    want = name.encode()
    n = len(buf)
    while off + 3 <= n:
        key = buf[off:off + 2]
        typ = chr(buf[off + 2])
        off += 3
        hit = key == want
        if typ == "A":
            if hit:
                return chr(buf[off])
            off += 1
        elif typ in _TAG_FMT:
            fmt = _TAG_FMT[typ]
            size = struct.calcsize(fmt)
            if hit:
                return struct.unpack_from("<" + fmt, buf, off)[0]
            off += size
        elif typ in ("Z", "H"):
            end = buf.index(0, off)
            if hit:
                return buf[off:end].decode()
            off = end + 1
        elif typ == "B":
            sub = chr(buf[off])
            count = struct.unpack_from("<I", buf, off + 1)[0]
            dt = _ARRAY_DTYPE[sub]
            size = count * np.dtype(dt).itemsize
            if hit:
                return np.frombuffer(
                    buf, dtype=dt, count=count, offset=off + 5
                ).copy()
            off += 5 + size
            continue
        else:
            raise ValueError(f"unknown tag type {typ!r}")
    return default


class RawBamReader:
    """Sequential reader yielding RAW record buffers (no record decode).

    The serial feeder only needs the zm group key, the flag and the
    reference id of each subread record; full decode_record runs in the
    worker pool instead (feeder.RawZmwJob). BGZF decompression runs on
    a thread pool (decompress_threads) since zlib releases the GIL.

    ``start=(compressed_offset, within_block_offset)`` resumes reading
    mid-file at a record boundary taken from a ZMW index
    (build_zmw_index) — this is how N-shard runs each decompress only
    1/N of the stream instead of all of it.
    """

    def __init__(self, path: str, decompress_threads: int = 8,
                 start: Optional[Tuple[int, int]] = None):
        self.path = path
        # 4 MB read buffering: a BGZF block is ~64 KB, the default 8 KB
        # buffer turns every block into several syscalls.
        self._fh = open(path, "rb", buffering=4 << 20)
        if start is not None:
            # Header lives at file start; fetch it with a tiny separate
            # read, then jump straight to the shard's first block.
            self.header = read_bam_header(path)
            self._fh.seek(start[0])
        if decompress_threads > 1:
            blocks = _read_bgzf_blocks_parallel(
                self._fh, threads=decompress_threads
            )
        else:
            blocks = _threaded_blocks(_read_bgzf_blocks(self._fh))
        self._stream = _ConcatStream(blocks)
        if start is not None:
            if start[1]:
                self._stream.read(start[1])
        else:
            self.header = _parse_bam_header(self._stream, path)

    def __iter__(self) -> Iterator[bytes]:
        return self

    def __next__(self) -> bytes:
        head = self._stream.read(4)
        if len(head) < 4:
            raise StopIteration
        (block_size,) = struct.unpack("<l", head)
        return self._stream.read(block_size)

    def close(self):
        self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def read_bam_header(path: str) -> BamHeader:
    """Reads just the BAM header (decompressing only the leading blocks)."""
    with open(path, "rb") as fh:
        stream = _ConcatStream(_read_bgzf_blocks(fh))
        return _parse_bam_header(stream, path)


class _OffsetStream:
    """Byte stream over (compressed_offset, payload) blocks that can
    report the BGZF virtual offset of its current read position."""

    def __init__(self, blocks: Iterator[Tuple[int, bytes]]):
        self._blocks = blocks
        self._buf = bytearray()
        self._g = 0  # global (decompressed) position of _buf[0]
        self._end_g = 0  # global position just past the appended data
        import collections as _c

        self._spans: "_c.deque" = _c.deque()  # (start_g, coffset, length)

    def _pull(self) -> bool:
        try:
            coffset, payload = next(self._blocks)
        except StopIteration:
            return False
        self._spans.append((self._end_g, coffset, len(payload)))
        self._end_g += len(payload)
        self._buf += payload
        return True

    def read(self, n: int) -> bytes:
        while len(self._buf) < n:
            if not self._pull():
                break
        out = bytes(self._buf[:n])
        del self._buf[:n]
        self._g += len(out)
        return out

    def voffset(self) -> Optional[Tuple[int, int]]:
        """(compressed_offset, within_block_offset) of the current
        position, or None at EOF."""
        while self._spans and (
            self._spans[0][0] + self._spans[0][2] <= self._g
        ):
            self._spans.popleft()
        if not self._spans:
            if not self._pull():
                return None
        start_g, coffset, _length = self._spans[0]
        return coffset, self._g - start_g


ZMW_INDEX_SUFFIX = ".zmi.npz"


def build_zmw_index(path: str, out_path: Optional[str] = None,
                    decompress_threads: int = 8) -> str:
    """One fast pass over a ZMW-sorted BAM writing a group index sidecar.

    For every run of consecutive records sharing a zm tag, stores
    (zmw, compressed_offset, within_block_offset) of the first record.
    Shard i/N then seeks straight to its contiguous ZMW range and
    decompresses only ~1/N of the stream (ROADMAP: the coarse index
    pass that breaks the serial per-rank BGZF floor). The pass itself
    runs at raw-feeder speed: parallel decompress + zm peeks only.

    Written via np.savez: zmw[G] int64, coffset[G] int64,
    uoffset[G] uint16, sorted_flag (1 if zmw ids are nondecreasing).
    """
    out_path = out_path or path + ZMW_INDEX_SUFFIX
    zmws: List[int] = []
    coffs: List[int] = []
    uoffs: List[int] = []
    is_sorted = True
    with open(path, "rb", buffering=4 << 20) as fh:
        stream = _OffsetStream(
            _read_bgzf_blocks_parallel_offsets(fh, decompress_threads)
        )
        _parse_bam_header(stream, path)
        prev_zm = None
        while True:
            vo = stream.voffset()
            head = stream.read(4)
            if len(head) < 4:
                break
            (block_size,) = struct.unpack("<l", head)
            rec = stream.read(block_size)
            zm = raw_tag(rec, "zm")
            if zm is None:
                raise ValueError(
                    f"{path}: record without zm tag; cannot index"
                )
            if zm != prev_zm:
                if prev_zm is not None and zm < prev_zm:
                    is_sorted = False
                zmws.append(int(zm))
                coffs.append(vo[0])
                uoffs.append(vo[1])
                prev_zm = zm
    tmp = out_path + ".tmp"
    with open(tmp, "wb") as f:
        np.savez(
            f,
            zmw=np.asarray(zmws, np.int64),
            coffset=np.asarray(coffs, np.int64),
            uoffset=np.asarray(uoffs, np.uint16),
            sorted_flag=np.array([1 if is_sorted else 0], np.int8),
            bam_size=np.array([os.path.getsize(path)], np.int64),
            # mtime (ns) guards against a rewritten BAM of identical byte
            # size passing the staleness check.
            bam_mtime_ns=np.array([os.stat(path).st_mtime_ns], np.int64),
        )
    os.replace(tmp, out_path)
    return out_path


def load_zmw_index(path: str) -> Optional[Dict[str, np.ndarray]]:
    """Loads a sidecar written by build_zmw_index.

    Returns None if absent, unreadable, or stale (the BAM's size no
    longer matches the one recorded at build time — a stale index
    would silently mis-shard)."""
    idx_path = path + ZMW_INDEX_SUFFIX
    if not os.path.exists(idx_path):
        return None
    try:
        with np.load(idx_path) as z:
            idx = {k: z[k] for k in ("zmw", "coffset", "uoffset",
                                     "sorted_flag", "bam_size")}
            # Older sidecars lack the mtime field; treat as unknown.
            mtime = (int(z["bam_mtime_ns"][0])
                     if "bam_mtime_ns" in z.files else None)
    except (OSError, KeyError, ValueError):
        return None
    stale = idx["bam_size"][0] != os.path.getsize(path) or (
        mtime is not None and mtime != os.stat(path).st_mtime_ns
    )
    if stale:
        import logging

        logging.getLogger(__name__).warning(
            "%s is stale (BAM size or mtime changed); ignoring — rerun "
            "`deepconsensus index`", idx_path,
        )
        return None
    return idx


def _read_bgzf_blocks_parallel_offsets(
    fh, threads: int = 8, window: int = 64
) -> Iterator[Tuple[int, bytes]]:
    """Like _read_bgzf_blocks_parallel but yields
    (compressed_file_offset, payload) so callers can build indexes."""
    import collections as _collections
    import concurrent.futures as _futures

    def read_compressed():
        while True:
            coffset = fh.tell()
            header = fh.read(12)
            if len(header) == 0:
                return
            if len(header) < 12:
                raise ValueError("truncated BGZF header")
            magic, _mtime, _xfl, _os, xlen = struct.unpack(
                "<4sLBBH", header
            )
            if magic[:2] != b"\x1f\x8b":
                raise ValueError("not a gzip stream")
            extra = fh.read(xlen)
            bsize = None
            off = 0
            while off + 4 <= len(extra):
                si1, si2, slen = struct.unpack_from("<BBH", extra, off)
                if si1 == 66 and si2 == 67 and slen == 2:
                    bsize = struct.unpack_from("<H", extra, off + 4)[0]
                off += 4 + slen
            if bsize is None:
                raise ValueError("missing BGZF BC extra field")
            cdata = fh.read(bsize - xlen - 19)
            _crc, isize = struct.unpack("<LL", fh.read(8))
            yield coffset, cdata, isize

    def decompress(job):
        coffset, cdata, isize = job
        data = zlib.decompress(cdata, -15)
        if len(data) != isize:
            raise ValueError("BGZF ISIZE mismatch")
        return coffset, data

    with _futures.ThreadPoolExecutor(threads) as pool:
        pending: "_collections.deque" = _collections.deque()
        for job in read_compressed():
            pending.append(pool.submit(decompress, job))
            if len(pending) >= window:
                coffset, data = pending.popleft().result()
                if data:
                    yield coffset, data
        while pending:
            coffset, data = pending.popleft().result()
            if data:
                yield coffset, data


class _ConcatStream:
    """Byte stream over an iterator of chunks."""

    def __init__(self, chunks: Iterator[bytes]):
        self._chunks = chunks
        self._buf = bytearray()

    def read(self, n: int) -> bytes:
        while len(self._buf) < n:
            try:
                self._buf += next(self._chunks)
            except StopIteration:
                break
        out = bytes(self._buf[:n])
        del self._buf[:n]
        return out


class BamWriter:
    """Sequential BAM writer (BGZF)."""

    def __init__(self, path: str, header: BamHeader):
        self.header = header
        self._fh = open(path, "wb")
        self._w = BgzfWriter(self._fh)
        text = header.text
        if text and not text.endswith("\n"):
            text += "\n"
        payload = BAM_MAGIC + struct.pack("<l", len(text)) + text.encode()
        payload += struct.pack("<l", len(header.references))
        for name, length in header.references:
            nb = name.encode() + b"\x00"
            payload += struct.pack("<l", len(nb)) + nb
            payload += struct.pack("<l", length)
        self._w.write(payload)
        self._ref_index = header.ref_index

    def write(self, read: BamRead) -> None:
        qname_b = read.qname.encode() + b"\x00"
        l_seq = len(read.seq)
        cig = b"".join(
            struct.pack("<I", (n << 4) | op) for op, n in read.cigartuples
        )
        seq_b = bytearray((l_seq + 1) // 2)
        for i, ch in enumerate(read.seq):
            code = _NT16_OF.get(ch, 15)
            if i % 2 == 0:
                seq_b[i // 2] = code << 4
            else:
                seq_b[i // 2] |= code
        if read.query_qualities is None:
            qual_b = b"\xff" * l_seq
        else:
            qual_b = np.asarray(
                read.query_qualities, dtype=np.uint8
            ).tobytes()
        tag_b = _encode_tags(read.tags)
        rec = struct.pack(
            "<llBBHHHllll",
            read.ref_id, read.pos, len(qname_b), read.mapq, 0,
            len(read.cigartuples), read.flag, l_seq,
            read.next_ref_id, read.next_pos, read.tlen,
        ) + qname_b + cig + bytes(seq_b) + qual_b + tag_b
        self._w.write(struct.pack("<l", len(rec)) + rec)

    def close(self):
        self._w.close()
        self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def fetch_index(path: str) -> Dict[str, List[BamRead]]:
    """Sequentially scans a BAM, grouping reads by reference name.

    Replaces indexed `fetch(ref_name)` for the truth-to-CCS lookup
    (pre_lib.py:1001-1014): DeepConsensus fetches exactly one primary
    alignment per CCS read name.
    """
    out: Dict[str, List[BamRead]] = {}
    with BamReader(path) as reader:
        for read in reader:
            name = read.reference_name
            if name is None:
                continue
            out.setdefault(name, []).append(read)
    return out
