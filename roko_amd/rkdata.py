"""RKW — the framework's window-data container format.

Role-equivalent to the reference's HDF5 layout (roko/data.py:29-91: per-region
groups with ``positions``/``labels``/``examples`` datasets plus a ``contigs``
group carrying the draft sequences), but self-contained and memmap-friendly:
the image ships no h5py, and for feeding 8 GPUs a zero-copy mmap read path
beats HDF5 chunk decompression anyway (SURVEY.md §7 hard part (d)).

File layout::

    [8B magic b"RKWIN001"]
    [blob 0][blob 1]...            # raw little-endian arrays, 64-byte aligned
    [footer JSON (utf-8)]          # groups, contigs, blob offsets/dtypes/shapes
    [8B little-endian footer length][8B magic b"RKWEND01"]

Groups carry per-region window batches:
  ``positions``  int32  (N, W, 2)   — (reference position, insertion slot)
  ``examples``   uint8  (N, R, W)   — sampled-read base-id matrices
  ``labels``     uint8  (N, W)      — truth classes (training files only)

Contig records carry the draft assembly (name, length, sequence blob) so
inference can stitch the polished FASTA without re-reading the draft
(reference: data.py:84-91, inference.py:129-147).
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass
from typing import Iterable, List, Optional, Tuple

import numpy as np

MAGIC = b"RKWIN001"
END_MAGIC = b"RKWEND01"
ALIGN = 64


def _dtype_str(a: np.ndarray) -> str:
    return a.dtype.str  # e.g. '<i4', '|u1'


@dataclass
class BlobRef:
    offset: int
    dtype: str
    shape: Tuple[int, ...]

    def to_json(self):
        return {"o": self.offset, "d": self.dtype, "s": list(self.shape)}

    @staticmethod
    def from_json(j) -> "BlobRef":
        return BlobRef(j["o"], j["d"], tuple(j["s"]))


class RkwWriter:
    """Streaming writer. ``store()`` appends one region-group per call."""

    def __init__(self, path: str, inference: bool = False):
        d = os.path.dirname(os.path.abspath(path))
        os.makedirs(d, exist_ok=True)
        self.path = path
        self.inference = inference
        self._fh = open(path, "wb")
        self._fh.write(MAGIC)
        self._off = len(MAGIC)
        self._groups: List[dict] = []
        self._contigs: List[dict] = []
        self._closed = False

    def _append(self, arr: np.ndarray) -> BlobRef:
        pad = (-self._off) % ALIGN
        if pad:
            self._fh.write(b"\0" * pad)
            self._off += pad
        if arr.dtype.byteorder == ">":
            arr = arr.astype(arr.dtype.newbyteorder("<"))
        data = np.ascontiguousarray(arr)
        ref = BlobRef(self._off, _dtype_str(data), data.shape)
        self._fh.write(memoryview(data).cast("B"))  # zero-copy
        self._off += data.nbytes
        return ref

    def write_contigs(self, refs: Iterable[Tuple[str, str]]) -> None:
        for name, seq in refs:
            blob = self._append(np.frombuffer(seq.encode("ascii"), dtype=np.uint8))
            self._contigs.append({"name": name, "len": len(seq), "seq": blob.to_json()})

    def store(
        self,
        contig: str,
        start: int,
        end: int,
        positions: np.ndarray,
        examples: np.ndarray,
        labels: Optional[np.ndarray] = None,
    ) -> None:
        positions = np.asarray(positions)
        examples = np.asarray(examples, dtype=np.uint8)
        n = len(positions)
        if n == 0:
            return
        if examples.shape[0] != n:
            raise ValueError("positions/examples length mismatch")
        if positions.ndim != 3 or positions.shape[2] != 2:
            raise ValueError(f"positions must be (N, W, 2), got {positions.shape}")
        g = {
            "contig": contig,
            "start": int(start),
            "end": int(end),
            "size": int(n),
            "positions": self._append(positions.astype(np.int32)).to_json(),
            "examples": self._append(examples).to_json(),
        }
        if labels is not None:
            labels = np.asarray(labels, dtype=np.uint8)
            if labels.shape[0] != n:
                raise ValueError("labels length mismatch")
            g["labels"] = self._append(labels).to_json()
        elif not self.inference:
            raise ValueError("training writer requires labels")
        self._groups.append(g)

    def close(self) -> None:
        if self._closed:
            return
        footer = json.dumps(
            {
                "version": 1,
                "inference": self.inference,
                "groups": self._groups,
                "contigs": self._contigs,
            }
        ).encode("utf-8")
        self._fh.write(footer)
        self._fh.write(len(footer).to_bytes(8, "little"))
        self._fh.write(END_MAGIC)
        self._fh.close()
        self._closed = True

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()


class RkwFile:
    """Memory-mapped reader. Safe to open lazily per DataLoader worker."""

    def __init__(self, path: str):
        self.path = path
        with open(path, "rb") as fh:
            head = fh.read(8)
            if head != MAGIC:
                raise ValueError(f"{path}: not an RKW file")
            fh.seek(-16, os.SEEK_END)
            flen = int.from_bytes(fh.read(8), "little")
            if fh.read(8) != END_MAGIC:
                raise ValueError(f"{path}: truncated or corrupt (bad end magic)")
            fh.seek(-16 - flen, os.SEEK_END)
            meta = json.loads(fh.read(flen).decode("utf-8"))
        self.meta = meta
        self.inference = bool(meta.get("inference", False))
        self.groups: List[dict] = meta["groups"]
        self._mm = np.memmap(path, dtype=np.uint8, mode="r")
        self._cum = np.cumsum([0] + [g["size"] for g in self.groups])

    # -- low-level ---------------------------------------------------------
    def _blob(self, j) -> np.ndarray:
        ref = BlobRef.from_json(j)
        dt = np.dtype(ref.dtype)
        count = int(np.prod(ref.shape)) if ref.shape else 1
        a = self._mm[ref.offset : ref.offset + count * dt.itemsize].view(dt)
        return a.reshape(ref.shape)

    # -- contigs -----------------------------------------------------------
    def contig_names(self) -> List[str]:
        return [c["name"] for c in self.meta["contigs"]]

    def contig_seq(self, name: str) -> str:
        for c in self.meta["contigs"]:
            if c["name"] == name:
                return self._blob(c["seq"]).tobytes().decode("ascii")
        raise KeyError(name)

    # -- windows -----------------------------------------------------------
    @property
    def num_windows(self) -> int:
        return int(self._cum[-1])

    def group_arrays(self, gi: int):
        g = self.groups[gi]
        pos = self._blob(g["positions"])
        ex = self._blob(g["examples"])
        lab = self._blob(g["labels"]) if "labels" in g else None
        return g, pos, ex, lab

    def locate(self, idx: int) -> Tuple[int, int]:
        """Global window index -> (group index, offset inside group)."""
        if idx < 0 or idx >= self.num_windows:
            raise IndexError(idx)
        gi = int(np.searchsorted(self._cum, idx, side="right") - 1)
        return gi, idx - int(self._cum[gi])

    def window(self, idx: int):
        gi, off = self.locate(idx)
        g, pos, ex, lab = self.group_arrays(gi)
        return (
            g["contig"],
            pos[off],
            ex[off],
            (lab[off] if lab is not None else None),
        )


def list_rkw_files(path: str) -> List[str]:
    """Accept a file or a directory of ``.rkw`` files (reference:
    datasets.py:9-18 accepts both)."""
    if os.path.isdir(path):
        out = sorted(
            os.path.join(path, f) for f in os.listdir(path) if f.endswith(".rkw")
        )
        if not out:
            raise FileNotFoundError(f"no .rkw files under {path}")
        return out
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    return [path]
