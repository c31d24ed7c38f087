"""Minimal FASTA reader/writer (the image has no Biopython).

The reference uses Bio.SeqIO for contig parsing and writing
(features.py:125-126, inference.py:149-154); this module provides the same
capability with zero dependencies.
"""

from __future__ import annotations

import gzip
import io
import os
from typing import Iterator, Tuple


def _open_text(path: str):
    if path.endswith(".gz"):
        return io.TextIOWrapper(gzip.open(path, "rb"))
    return open(path, "r")


def read_fasta(path: str) -> Iterator[Tuple[str, str]]:
    """Yield ``(name, sequence)`` per record. Name = header up to first space."""
    name = None
    parts: list[str] = []
    with _open_text(path) as fh:
        for line in fh:
            line = line.rstrip("\n")
            if not line:
                continue
            if line.startswith(">"):
                if name is not None:
                    yield name, "".join(parts)
                name = line[1:].split()[0]
                parts = []
            else:
                if name is None:
                    raise ValueError(f"{path}: sequence data before first header")
                parts.append(line.strip())
    if name is not None:
        yield name, "".join(parts)


def write_fasta(path: str, records, width: int = 80) -> None:
    """Write ``(name, sequence)`` pairs; creates parent dirs."""
    d = os.path.dirname(os.path.abspath(path))
    os.makedirs(d, exist_ok=True)
    with open(path, "w") as fh:
        for name, seq in records:
            fh.write(f">{name}\n")
            for i in range(0, len(seq), width):
                fh.write(seq[i : i + width])
                fh.write("\n")
