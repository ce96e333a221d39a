"""TFRecord container format reader/writer (no TensorFlow dependency).

Format per record: uint64 length, masked crc32c(length), payload, masked
crc32c(payload). The reference writes gzip-compressed TFRecords
(preprocess.py:158-169); pass a ".gz"-suffixed path (or compression="gzip")
for the same framing inside a gzip stream.

CRC32C (Castagnoli) is implemented with a numpy-vectorized table; this is the
only place the framework needs it.
"""
from __future__ import annotations

import glob as globlib
import gzip
import struct
from typing import Iterator, List, Optional

import numpy as np

_POLY = 0x82F63B78


def _make_table() -> np.ndarray:
    tbl = np.zeros(256, dtype=np.uint32)
    for i in range(256):
        c = i
        for _ in range(8):
            c = (c >> 1) ^ (_POLY if (c & 1) else 0)
        tbl[i] = c
    return tbl


_TABLE = _make_table()


def crc32c(data: bytes) -> int:
    crc = np.uint32(0xFFFFFFFF)
    arr = np.frombuffer(data, dtype=np.uint8)
    tbl = _TABLE
    c = int(crc)
    for b in arr.tobytes():
        c = tbl[(c ^ b) & 0xFF] ^ (c >> 8)
        c = int(c)
    return c ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    c = crc32c(data)
    return ((c >> 15) | (c << 17)) + 0xA282EAD8 & 0xFFFFFFFF


def _open(path: str, mode: str, compression: Optional[str]):
    gz = compression == "gzip" or (
        compression is None and path.endswith(".gz")
    )
    if gz:
        return gzip.open(path, mode)
    return open(path, mode)


class TFRecordWriter:
    def __init__(self, path: str, compression: Optional[str] = None):
        self._fh = _open(path, "wb", compression)

    def write(self, record: bytes) -> None:
        length = struct.pack("<Q", len(record))
        self._fh.write(length)
        self._fh.write(struct.pack("<I", _masked_crc(length)))
        self._fh.write(record)
        self._fh.write(struct.pack("<I", _masked_crc(record)))

    def close(self):
        self._fh.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def read_tfrecords(
    path: str, compression: Optional[str] = None, check_crc: bool = False
) -> Iterator[bytes]:
    """Yields raw record payloads from one TFRecord file."""
    with _open(path, "rb", compression) as fh:
        while True:
            head = fh.read(12)
            if len(head) < 12:
                return
            (length,) = struct.unpack("<Q", head[:8])
            payload = fh.read(length)
            crc = fh.read(4)
            if len(payload) < length or len(crc) < 4:
                raise ValueError("truncated TFRecord")
            if check_crc:
                (want,) = struct.unpack("<I", crc)
                if _masked_crc(payload) != want:
                    raise ValueError("TFRecord CRC mismatch")
            yield payload


def read_tfrecords_glob(
    pattern: str, compression: Optional[str] = None
) -> Iterator[bytes]:
    """Yields records from all files matching a glob pattern, sorted."""
    files = sorted(globlib.glob(pattern))
    for f in files:
        yield from read_tfrecords(f, compression)


def list_files(pattern: str) -> List[str]:
    return sorted(globlib.glob(pattern))
