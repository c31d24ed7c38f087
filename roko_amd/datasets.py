"""Torch datasets over RKW feature files.

Role-equivalent to the reference's datasets layer (roko/datasets.py:20-125 and
the InferenceDataset in roko/inference.py:27-87): a flat global index over all
window groups across one file or a directory of files, with lazy per-worker
re-opening (mmaps must not cross a DataLoader fork).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import numpy as np
import torch
from torch.utils.data import Dataset

from .rkdata import RkwFile, list_rkw_files


class _LazyFiles:
    """Per-process lazy RkwFile cache (reference: datasets.py:58-60)."""

    def __init__(self, paths: List[str]):
        self.paths = paths
        self._pid: Optional[int] = None
        self._files: List[Optional[RkwFile]] = [None] * len(paths)

    def get(self, i: int) -> RkwFile:
        pid = os.getpid()
        if pid != self._pid:
            self._files = [None] * len(self.paths)
            self._pid = pid
        f = self._files[i]
        if f is None:
            f = RkwFile(self.paths[i])
            self._files[i] = f
        return f


class TrainDataset(Dataset):
    """(example int64 (R,W), label int64 (W)) pairs across training files."""

    def __init__(self, path: str):
        self.paths = list_rkw_files(path)
        self._lazy = _LazyFiles(self.paths)
        counts = []
        for i, p in enumerate(self.paths):
            f = RkwFile(p)
            if f.inference:
                raise ValueError(f"{p} is an inference file (no labels)")
            counts.append(f.num_windows)
        self._file_cum = np.cumsum([0] + counts)

    def __len__(self) -> int:
        return int(self._file_cum[-1])

    def _locate(self, idx: int) -> Tuple[int, int]:
        fi = int(np.searchsorted(self._file_cum, idx, side="right") - 1)
        return fi, idx - int(self._file_cum[fi])

    def __getitem__(self, idx: int):
        fi, off = self._locate(idx)
        f = self._lazy.get(fi)
        _, _, ex, lab = f.group_arrays(f.locate(off)[0])
        j = f.locate(off)[1]
        x = torch.from_numpy(np.ascontiguousarray(ex[j])).long()
        y = torch.from_numpy(np.ascontiguousarray(lab[j])).long()
        return x, y


class InMemoryTrainDataset(Dataset):
    """Everything pre-loaded into RAM (reference: datasets.py:82-119)."""

    def __init__(self, path: str):
        xs, ys = [], []
        for p in list_rkw_files(path):
            f = RkwFile(p)
            if f.inference:
                raise ValueError(f"{p} is an inference file (no labels)")
            for gi in range(len(f.groups)):
                _, _, ex, lab = f.group_arrays(gi)
                xs.append(np.array(ex))
                ys.append(np.array(lab))
        self.x = torch.from_numpy(np.concatenate(xs)).long()
        self.y = torch.from_numpy(np.concatenate(ys)).long()

    def __len__(self) -> int:
        return self.x.shape[0]

    def __getitem__(self, idx: int):
        return self.x[idx], self.y[idx]


class InferenceDataset(Dataset):
    """Yields (group index, window offset, example) over one inference file.

    Contig/positions lookups go through group metadata on the consumer side
    (reference: inference.py:27-87 returns (contig, positions, X); we return
    indices instead so the collated batch stays tensor-only and the voter
    fetches positions zero-copy from the mmap).
    """

    def __init__(self, path: str, groups: Optional[List[int]] = None):
        self.path = path
        f = RkwFile(path)
        self.group_meta = f.groups
        self.groups = (list(range(len(f.groups))) if groups is None
                       else list(groups))
        # cumulative window counts instead of a per-window (gi, j) list:
        # a whole-genome file has ~1e8 windows and a materialised index
        # would cost tens of GB of host RAM (BASELINE config 5)
        self._cum = np.cumsum(
            [0] + [int(f.groups[gi]["size"]) for gi in self.groups])
        self._lazy = _LazyFiles([path])

    def __len__(self) -> int:
        return int(self._cum[-1])

    def __getitem__(self, idx: int):
        if idx < 0 or idx >= len(self):
            raise IndexError(idx)
        k = int(np.searchsorted(self._cum, idx, side="right") - 1)
        gi = self.groups[k]
        j = idx - int(self._cum[k])
        f = self._lazy.get(0)
        _, _, ex, _ = f.group_arrays(gi)
        x = torch.from_numpy(np.array(ex[j]))
        return gi, j, x
