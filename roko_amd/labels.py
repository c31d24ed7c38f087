"""Truth-to-draft label generation (Medaka-style).

Role-equivalent to the reference's roko/labels.py:24-189: fetch truth-contig
alignments against the draft from a BAM, resolve overlapping alignments, then
walk the aligned pairs mapping every draft (position, insertion) to a truth
base. Uses the framework's own BAM reader (roko_amd.ops._pileup.fetch_records)
instead of pysam.

Deviation from the reference (documented): aligned pairs here cover only
M/=/X/I/D CIGAR ops — soft-clipped query tails are NOT emitted as insertion
labels (pysam's get_aligned_pairs includes them, which lets clip artefacts
leak into labels at alignment ends; reference labels.py:121-189).
"""

from __future__ import annotations

import itertools
from dataclasses import dataclass
from typing import Iterator, List, Optional, Tuple

import numpy as np

from . import config as C

Region = Tuple[str, int, Optional[int]]  # (name, start, end)


@dataclass
class TruthAlign:
    """One truth alignment with clip-adjustable [start, end) bounds."""

    qname: str
    flag: int
    pos: int
    mapq: int
    cigar: np.ndarray  # uint32, len<<4|op
    seq: str
    start: int = 0
    end: int = 0
    keep: bool = True

    @property
    def reference_start(self) -> int:
        return self.pos

    @property
    def reference_end(self) -> int:
        p = self.pos
        for c in self.cigar:
            op = c & 0xF
            if op in (0, 2, 3, 7, 8):  # M D N = X
                p += int(c) >> 4
        return p

    @property
    def reference_length(self) -> int:
        return self.reference_end - self.pos


def get_aligns(bam: str, ref_name: str, start: int = 0, end: Optional[int] = None) -> List[TruthAlign]:
    """Filtered, start-sorted truth alignments overlapping the region
    (reference: labels.py:24-50 — drops unmapped and secondary)."""
    from .ops import _pileup

    out = []
    qend = end if end is not None else 1 << 60
    for qname, flag, pos, mapq, cigar, seq in _pileup.fetch_records(bam, ref_name, start, qend):
        if flag & 0x4 or flag & 0x100:
            continue
        a = TruthAlign(qname, flag, pos, mapq, np.asarray(cigar, dtype=np.uint32), seq)
        if a.reference_end <= start or (end is not None and a.reference_start >= end):
            continue
        a.start, a.end = a.reference_start, a.reference_end
        out.append(a)
    out.sort(key=lambda a: a.reference_start)
    return out


def filter_aligns(
    aligns: List[TruthAlign],
    len_threshold: float = 2.0,
    ol_threshold: float = 0.5,
    min_len: int = 1000,
) -> List[TruthAlign]:
    """Resolve overlapping truth alignments (reference: labels.py:60-118).

    For each overlapping pair, by length ratio LR = longer/shorter and overlap
    fraction OF = overlap/shorter:
      LR <  t, OF >= t: drop both (ambiguous repeat)
      LR <  t, OF <  t: split the overlap between the two
      LR >= t, OF >= t: drop the shorter
      LR >= t, OF <  t: clip the shorter past the overlap
    then drop every alignment shorter than min_len after clipping.
    """
    for i, j in itertools.combinations(aligns, 2):
        first, second = sorted((i, j), key=lambda a: a.reference_start)
        if second.start >= first.end:
            continue
        ol_start, ol_end = second.start, first.end
        shorter, longer = sorted((i, j), key=lambda a: a.reference_length)
        if shorter.reference_length == 0:
            shorter.keep = False
            continue
        len_ratio = longer.reference_length / shorter.reference_length
        ol_fraction = (ol_end - ol_start) / shorter.reference_length
        if len_ratio < len_threshold:
            if ol_fraction >= ol_threshold:
                shorter.keep = False
                longer.keep = False
            else:
                first.end = ol_start
                second.start = ol_end
        else:
            if ol_fraction >= ol_threshold:
                shorter.keep = False
            else:
                second.start = ol_end

    filtered = [a for a in aligns if a.keep and a.end - a.start >= min_len]
    filtered.sort(key=lambda a: a.start)
    return filtered


def aligned_pairs(a: TruthAlign) -> Iterator[Tuple[Optional[int], Optional[int]]]:
    """Yield (query_pos, ref_pos) pairs over M/=/X/I/D ops (no soft clips)."""
    q, r = 0, a.pos
    for c in a.cigar:
        op = int(c) & 0xF
        ln = int(c) >> 4
        if op in (0, 7, 8):  # M = X
            for k in range(ln):
                yield q + k, r + k
            q += ln
            r += ln
        elif op == 1:  # I
            for k in range(ln):
                yield q + k, None
            q += ln
        elif op in (2, 3):  # D N
            for k in range(ln):
                yield None, r + k
            r += ln
        elif op == 4:  # S
            q += ln
        # H, P: nothing


def get_pos_and_labels(
    a: TruthAlign, region_start: int, region_end: Optional[int]
) -> Tuple[List[Tuple[int, int]], List[int]]:
    """((pos, ins), label) stream for one truth alignment clipped to the
    region (reference: labels.py:141-189)."""
    start = max(region_start, a.start)
    end = min(region_end if region_end is not None else 1 << 60, a.end)

    all_pos: List[Tuple[int, int]] = []
    all_labels: List[int] = []
    cur_pos: Optional[int] = None
    ins_count = 0

    pairs = itertools.dropwhile(
        lambda p: p[1] is None or p[1] < start, aligned_pairs(a)
    )
    for qpos, rpos in pairs:
        if rpos is not None and rpos >= end:
            break
        if rpos is None:
            if cur_pos is None:
                continue  # insertion before the first in-region aligned base
            ins_count += 1
        else:
            ins_count = 0
            cur_pos = rpos
        qbase = a.seq[qpos].upper() if qpos is not None else "*"
        label = C.LABEL_ENCODING.get(qbase, C.LABEL_UNKNOWN)
        all_pos.append((cur_pos, ins_count))
        all_labels.append(label)

    return all_pos, all_labels
