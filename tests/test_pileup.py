"""Window-builder semantics tests against hand-crafted pileups.

Small geometries (cols/stride/rows shrunk) let expected matrices be written
by hand; semantics under test mirror the reference's emitted features
(SURVEY.md §3.1): column keys, deletion GAPs, insertion slots, strand offset,
bounds-based GAP/UNKNOWN defaults, sampling determinism.
"""

import numpy as np
import pytest

from roko_amd.io.bamio import SamRecord, write_bam
from roko_amd.ops import _pileup


def gen(path, contig="c", start=0, end=100, **kw):
    args = dict(rows=8, cols=4, stride=2, max_ins=3, filter_flag=0xF04,
                min_mapq=10, seed=7)
    args.update(kw)
    return _pileup.generate_features(path, contig, start, end, **args)


def test_simple_columns_and_positions(tmp_path):
    # one read covering 0..12, no indels -> columns (0,0)..(11,0)
    recs = [SamRecord("r", 0, 0, 0, 60, [(12, "M")], "ACGTACGTACGT")]
    path = str(tmp_path / "a.bam")
    write_bam(path, [("c", 100)], recs)
    pos, X = gen(path)
    # 12 columns, cols=4 stride=2 -> windows at 0,2,4,6,8 -> 5 windows
    assert pos.shape == (5, 4, 2)
    assert X.shape == (5, 8, 4)
    assert [tuple(p) for p in pos[0]] == [(0, 0), (1, 0), (2, 0), (3, 0)]
    assert [tuple(p) for p in pos[1]] == [(2, 0), (3, 0), (4, 0), (5, 0)]
    # single forward read: rows are all that read; bases = ACGT -> 0,1,2,3
    assert np.array_equal(X[0], np.tile([0, 1, 2, 3], (8, 1)))


def test_reverse_strand_offset_applies_to_all(tmp_path):
    recs = [SamRecord("r", 16, 0, 0, 60, [(12, "M")], "ACGTACGTACGT")]
    path = str(tmp_path / "b.bam")
    write_bam(path, [("c", 100)], recs)
    _, X = gen(path)
    assert np.array_equal(X[0], np.tile([6, 7, 8, 9], (8, 1)))


def test_deletion_emits_gap(tmp_path):
    # read: 4M 2D 4M starting at 0 -> columns 4,5 are deletions (GAP=4)
    recs = [SamRecord("r", 0, 0, 0, 60, [(4, "M"), (2, "D"), (4, "M")], "ACGTACGT")]
    path = str(tmp_path / "c.bam")
    write_bam(path, [("c", 100)], recs)
    pos, X = gen(path)
    # columns 0..9; window starting at col 4 = positions 4,5,6,7
    w = [tuple(p) for p in pos[2]]
    assert w == [(4, 0), (5, 0), (6, 0), (7, 0)]
    assert np.array_equal(X[2], np.tile([4, 4, 0, 1], (8, 1)))  # GAP GAP A C


def test_insertion_columns(tmp_path):
    # read1: 3M 2I 3M at 0 (ins after pos 2); read2 plain 6M at 0
    recs = [
        SamRecord("r1", 0, 0, 0, 60, [(3, "M"), (2, "I"), (3, "M")], "ACGTTACG"),
        SamRecord("r2", 0, 0, 0, 60, [(6, "M")], "ACGACG"),
    ]
    path = str(tmp_path / "d.bam")
    write_bam(path, [("c", 100)], recs)
    pos, X = gen(path, rows=64)
    w = [tuple(p) for p in pos[0]]
    assert w == [(0, 0), (1, 0), (2, 0), (2, 1)]
    # second window starts after stride 2 -> cols (2,0),(2,1),(2,2),(3,0)
    w1 = [tuple(p) for p in pos[1]]
    assert w1 == [(2, 0), (2, 1), (2, 2), (3, 0)]
    # rows are one of the two reads: r1 shows G,T at (2,0),(2,1); r2 shows
    # G,GAP (within bounds, no entry at insertion slot)
    rows = {tuple(r) for r in X[0]}
    assert rows <= {(0, 1, 2, 3), (0, 1, 2, 4)}  # ACG+T(ins)  /  ACG+GAP
    assert len(rows) == 2  # with 64 samples both reads appear w.h.p.


def test_unknown_outside_bounds(tmp_path):
    # two disjoint-ish reads; window spanning both shows UNKNOWN where a read
    # doesn't reach
    recs = [
        SamRecord("r1", 0, 0, 0, 60, [(4, "M")], "ACGT"),
        SamRecord("r2", 0, 0, 2, 60, [(6, "M")], "ACGTAC"),
    ]
    path = str(tmp_path / "e.bam")
    write_bam(path, [("c", 100)], recs)
    pos, X = gen(path, rows=64)
    # columns 0..7; window 2 = positions 4..7: r1 ended at 4 (exclusive) but
    # the reference counts pos == ref_end as inside (GAP); 5..7 are UNKNOWN
    w2 = [tuple(p) for p in pos[2]]
    assert w2 == [(4, 0), (5, 0), (6, 0), (7, 0)]
    rows = {tuple(int(v) for v in r) for r in X[2]}
    # r1 row: GAP at 4 (== ref_end quirk), UNKNOWN after; r2 row: GTAC->2,3,0,1
    assert rows <= {(4, 5, 5, 5), (2, 3, 0, 1)}
    assert (2, 3, 0, 1) in rows


def test_filters_respected(tmp_path):
    recs = [
        SamRecord("ok", 0, 0, 0, 60, [(8, "M")], "ACGTACGT"),
        SamRecord("dup", 0x400, 0, 0, 60, [(8, "M")], "TTTTTTTT"),
        SamRecord("lowmq", 0, 0, 0, 5, [(8, "M")], "TTTTTTTT"),
        SamRecord("secondary", 0x100, 0, 0, 60, [(8, "M")], "TTTTTTTT"),
    ]
    path = str(tmp_path / "f.bam")
    write_bam(path, [("c", 100)], recs)
    _, X = gen(path)
    assert set(np.unique(X)) <= {0, 1, 2, 3}  # only the ok ACGT read sampled


def test_seed_determinism(tmp_path, rng):
    n = 30
    recs = []
    for i in range(n):
        s = int(rng.integers(0, 50))
        seq = "".join(rng.choice(list("ACGT"), 40))
        recs.append(SamRecord(f"r{i}", 0, 0, s, 60, [(40, "M")], seq))
    recs.sort(key=lambda r: r.pos)
    path = str(tmp_path / "g.bam")
    write_bam(path, [("c", 200)], recs)
    p1, x1 = gen(path, seed=42)
    p2, x2 = gen(path, seed=42)
    p3, x3 = gen(path, seed=43)
    assert np.array_equal(x1, x2) and np.array_equal(p1, p2)
    assert not np.array_equal(x1, x3)  # different sampling
    assert np.array_equal(p1, p3)  # but identical columns


def test_region_bounds_columns_only_inside(tmp_path):
    recs = [SamRecord("r", 0, 0, 0, 60, [(50, "M")], "A" * 50)]
    path = str(tmp_path / "h.bam")
    write_bam(path, [("c", 100)], recs)
    pos, _ = gen(path, start=10, end=20)
    assert pos[:, :, 0].min() >= 10
    assert pos[:, :, 0].max() < 20


def test_max_ins_caps_insertions(tmp_path):
    recs = [
        SamRecord("r1", 0, 0, 0, 60, [(3, "M"), (6, "I"), (3, "M")], "ACGTTTTTTACG"),
        SamRecord("r2", 0, 0, 0, 60, [(6, "M")], "ACGACG"),
    ]
    path = str(tmp_path / "i.bam")
    write_bam(path, [("c", 100)], recs)
    pos, _ = gen(path)
    assert pos[:, :, 1].max() == 3  # never beyond max_ins


def test_zero_coverage_gap_in_columns(tmp_path):
    # two reads with a coverage hole between them: hole positions yield no
    # columns; windows span the hole seamlessly
    recs = [
        SamRecord("r1", 0, 0, 0, 60, [(6, "M")], "ACGTAC"),
        SamRecord("r2", 0, 0, 20, 60, [(6, "M")], "GTACGT"),
    ]
    path = str(tmp_path / "j.bam")
    write_bam(path, [("c", 100)], recs)
    pos, X = gen(path)
    allpos = pos[:, :, 0].ravel()
    assert not np.any((allpos >= 6) & (allpos < 20))
