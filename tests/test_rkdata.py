"""RKW container round-trip tests."""

import numpy as np
import pytest

from roko_amd.rkdata import RkwFile, RkwWriter, list_rkw_files


def test_roundtrip_train(tmp_path):
    path = str(tmp_path / "t.rkw")
    pos = np.arange(2 * 90 * 2, dtype=np.int32).reshape(2, 90, 2)
    ex = np.arange(2 * 200 * 90, dtype=np.uint8).reshape(2, 200, 90)
    lab = np.ones((2, 90), dtype=np.uint8)
    with RkwWriter(path, inference=False) as w:
        w.write_contigs([("ctg", "ACGT" * 10)])
        w.store("ctg", 0, 1000, pos, ex, lab)
    f = RkwFile(path)
    assert not f.inference
    assert f.num_windows == 2
    assert f.contig_names() == ["ctg"]
    assert f.contig_seq("ctg") == "ACGT" * 10
    g, p, e, l = f.group_arrays(0)
    assert g["contig"] == "ctg" and g["size"] == 2
    assert np.array_equal(p, pos) and np.array_equal(e, ex) and np.array_equal(l, lab)


def test_roundtrip_inference_multiple_groups(tmp_path):
    path = str(tmp_path / "i.rkw")
    with RkwWriter(path, inference=True) as w:
        w.write_contigs([("a", "AAAA"), ("b", "CCCC")])
        for gi, contig in enumerate(["a", "a", "b"]):
            n = gi + 1
            w.store(contig, gi * 100, gi * 100 + 100,
                    np.zeros((n, 4, 2), np.int32), np.zeros((n, 8, 4), np.uint8))
    f = RkwFile(path)
    assert f.num_windows == 6
    assert len(f.groups) == 3
    # locate: global window index -> (group, offset)
    assert f.locate(0) == (0, 0)
    assert f.locate(1) == (1, 0)
    assert f.locate(2) == (1, 1)
    assert f.locate(5) == (2, 2)
    contig, p, e, l = f.window(5)
    assert contig == "b" and l is None


def test_writer_requires_labels_for_training(tmp_path):
    with RkwWriter(str(tmp_path / "x.rkw"), inference=False) as w:
        with pytest.raises(ValueError):
            w.store("c", 0, 10, np.zeros((1, 4, 2), np.int32),
                    np.zeros((1, 8, 4), np.uint8))
        w.store("c", 0, 10, np.zeros((1, 4, 2), np.int32),
                np.zeros((1, 8, 4), np.uint8), np.zeros((1, 4), np.uint8))


def test_corrupt_file_rejected(tmp_path):
    p = tmp_path / "bad.rkw"
    p.write_bytes(b"RKWIN001 not really a full file")
    with pytest.raises(ValueError):
        RkwFile(str(p))


def test_list_rkw_files(tmp_path):
    for n in ["b.rkw", "a.rkw"]:
        with RkwWriter(str(tmp_path / n), inference=True) as w:
            pass
    got = list_rkw_files(str(tmp_path))
    assert [g.split("/")[-1] for g in got] == ["a.rkw", "b.rkw"]
    with pytest.raises(FileNotFoundError):
        list_rkw_files(str(tmp_path / "nope"))
