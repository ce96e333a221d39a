"""FASTA reader (whole-file, dict-backed fetch)."""
from __future__ import annotations

import gzip
from typing import Dict, List


class FastaFile:
    """In-memory FASTA with a pysam-like fetch API."""

    def __init__(self, path: str):
        self.sequences: Dict[str, str] = {}
        opener = gzip.open if path.endswith(".gz") else open
        name = None
        parts: List[str] = []
        with opener(path, "rt") as fh:
            for line in fh:
                line = line.rstrip("\n")
                if line.startswith(">"):
                    if name is not None:
                        self.sequences[name] = "".join(parts)
                    name = line[1:].split()[0]
                    parts = []
                else:
                    parts.append(line)
        if name is not None:
            self.sequences[name] = "".join(parts)

    @property
    def references(self) -> List[str]:
        return list(self.sequences)

    def get_reference_length(self, contig: str) -> int:
        return len(self.sequences[contig])

    def fetch(self, contig: str, start: int = 0, stop: int = None) -> str:
        seq = self.sequences[contig]
        return seq[start:stop] if stop is not None else seq[start:]

    def close(self):
        pass


def write_fasta(path: str, sequences: Dict[str, str], width: int = 70):
    with open(path, "w") as fh:
        for name, seq in sequences.items():
            fh.write(f">{name}\n")
            for i in range(0, len(seq), width):
                fh.write(seq[i:i + width] + "\n")
