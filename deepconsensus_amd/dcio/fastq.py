"""FASTQ reader/writer (plain or gzip)."""
from __future__ import annotations

import dataclasses
import gzip
from typing import Iterator, List


@dataclasses.dataclass
class FastqRecord:
    name: str
    sequence: str
    quality: str

    def get_quality_array(self) -> List[int]:
        return [ord(c) - 33 for c in self.quality]

    def __str__(self) -> str:
        return f"@{self.name}\n{self.sequence}\n+\n{self.quality}"


def _open(path: str, mode: str):
    if path.endswith(".gz"):
        return gzip.open(path, mode + "t")
    return open(path, mode)


def read_fastq(path: str) -> Iterator[FastqRecord]:
    with _open(path, "r") as fh:
        while True:
            header = fh.readline()
            if not header:
                return
            seq = fh.readline().rstrip("\n")
            fh.readline()  # '+'
            qual = fh.readline().rstrip("\n")
            yield FastqRecord(header.rstrip("\n")[1:], seq, qual)


def write_fastq(path: str, records) -> None:
    with _open(path, "w") as fh:
        for r in records:
            fh.write(str(r) + "\n")
