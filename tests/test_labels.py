"""Label-generation semantics tests."""

import numpy as np
import pytest

from roko_amd import config as C
from roko_amd.io.bamio import SamRecord, write_bam
from roko_amd.labels import (TruthAlign, aligned_pairs, filter_aligns,
                             get_aligns, get_pos_and_labels)


def mk_align(pos, cigar, seq, qname="t", flag=0):
    import struct
    cig = np.array([(l << 4) | "MIDNSHP=X".index(op) for l, op in cigar],
                   dtype=np.uint32)
    a = TruthAlign(qname, flag, pos, 60, cig, seq)
    a.start, a.end = a.reference_start, a.reference_end
    return a


def test_aligned_pairs_match_ins_del():
    a = mk_align(10, [(3, "M"), (2, "I"), (2, "D"), (3, "M")], "ACGTTACG")
    got = list(aligned_pairs(a))
    assert got == [
        (0, 10), (1, 11), (2, 12),          # 3M
        (3, None), (4, None),                # 2I
        (None, 13), (None, 14),              # 2D
        (5, 15), (6, 16), (7, 17),           # 3M
    ]


def test_soft_clips_not_in_pairs():
    a = mk_align(5, [(2, "S"), (3, "M"), (2, "S")], "TTACGTT")
    got = list(aligned_pairs(a))
    assert got == [(2, 5), (3, 6), (4, 7)]


def test_get_pos_and_labels_basic():
    # truth ACG TT ACG against draft starting at 10: ins labels at (12,1),(12,2)
    a = mk_align(10, [(3, "M"), (2, "I"), (3, "M")], "ACGTTACG")
    pos, lab = get_pos_and_labels(a, 0, None)
    assert pos == [(10, 0), (11, 0), (12, 0), (12, 1), (12, 2),
                   (13, 0), (14, 0), (15, 0)]
    dec = [C.LABEL_DECODING[l] for l in lab]
    assert dec == list("ACGTTACG")


def test_get_pos_and_labels_deletion_is_gap():
    a = mk_align(0, [(2, "M"), (2, "D"), (2, "M")], "ACGT")
    pos, lab = get_pos_and_labels(a, 0, None)
    assert pos == [(0, 0), (1, 0), (2, 0), (3, 0), (4, 0), (5, 0)]
    dec = [C.LABEL_DECODING[l] for l in lab]
    assert dec == ["A", "C", "*", "*", "G", "T"]


def test_get_pos_and_labels_region_clip():
    a = mk_align(0, [(10, "M")], "ACGTACGTAC")
    pos, lab = get_pos_and_labels(a, 3, 7)
    assert pos == [(3, 0), (4, 0), (5, 0), (6, 0)]
    assert [C.LABEL_DECODING[l] for l in lab] == list("TACG")


def test_get_pos_and_labels_ambiguous_is_unknown():
    a = mk_align(0, [(4, "M")], "ACNT")
    _, lab = get_pos_and_labels(a, 0, None)
    assert lab[2] == C.LABEL_UNKNOWN


def test_filter_aligns_drop_both_on_similar_overlap():
    # two similar-length aligns overlapping > 50% of the shorter: both dropped
    a = mk_align(0, [(2000, "M")], "A" * 2000, "a")
    b = mk_align(500, [(1900, "M")], "A" * 1900, "b")
    got = filter_aligns([a, b])
    assert got == []


def test_filter_aligns_clip_on_small_overlap():
    a = mk_align(0, [(2000, "M")], "A" * 2000, "a")
    b = mk_align(1900, [(2100, "M")], "A" * 2100, "b")
    got = filter_aligns([a, b])
    assert len(got) == 2
    first = min(got, key=lambda x: x.start)
    second = max(got, key=lambda x: x.start)
    assert first.end == 1900  # clipped to overlap start
    assert second.start == 2000  # clipped past overlap end


def test_filter_aligns_drop_short_keep_long():
    a = mk_align(0, [(9000, "M")], "A" * 9000, "a")
    b = mk_align(1000, [(2000, "M")], "A" * 2000, "b")  # fully inside a
    got = filter_aligns([a, b])
    assert [x.qname for x in got] == ["a"]


def test_filter_aligns_min_len():
    a = mk_align(0, [(800, "M")], "A" * 800, "a")
    assert filter_aligns([a]) == []


def test_get_aligns_from_bam(tmp_path):
    recs = [
        SamRecord("t1", 0, 0, 100, 60, [(1500, "M")], "A" * 1500),
        SamRecord("sec", 0x100, 0, 150, 60, [(1500, "M")], "A" * 1500),
        SamRecord("unmapped", 0x4, 0, 200, 0, [(1500, "M")], "A" * 1500),
    ]
    path = str(tmp_path / "truth.bam")
    write_bam(path, [("c", 5000)], recs)
    got = get_aligns(path, "c", 0, 5000)
    assert [a.qname for a in got] == ["t1"]
    assert got[0].reference_end == 1600
