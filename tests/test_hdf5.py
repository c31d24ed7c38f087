"""HDF5 interop: self-contained reader/writer for the reference layout
(roko/data.py:29-91) and the HDF5<->RKW converters.

h5py is absent in this image, so the primary tests round-trip through our
own spec-level writer; when h5py IS importable (user machines) the
cross-validation tests run too.
"""

import numpy as np
import pytest
import torch

from roko_amd import config as C
from roko_amd.io.hdf5 import H5File, H5Writer
from roko_amd.io.hdf5_compat import hdf5_to_rkw, rkw_to_hdf5
from roko_amd.rkdata import RkwFile, RkwWriter

try:
    import h5py  # noqa: F401
    HAVE_H5PY = True
except ImportError:
    HAVE_H5PY = False


def _reference_layout_file(path, n=6, train=True, rng=None):
    """Write a file shaped exactly like the reference's DataWriter output."""
    rng = rng or np.random.default_rng(0)
    pos = np.zeros((n, C.WINDOW_COLS, 2), dtype=np.int64)
    pos[..., 0] = (np.arange(n)[:, None] * C.WINDOW_STRIDE
                   + np.arange(C.WINDOW_COLS)[None, :] // 2)
    pos[..., 1] = np.arange(C.WINDOW_COLS)[None, :] % 2
    ex = rng.integers(0, 12, (n, C.WINDOW_ROWS, C.WINDOW_COLS)).astype(np.uint8)
    lab = rng.integers(0, 5, (n, C.WINDOW_COLS)).astype(np.int64)
    seq = "".join(rng.choice(list("ACGT"), 500))
    with H5Writer(path) as f:
        g = f.create_group(f"ctg_a_0-{int(pos[-1, -1, 0])}")
        g["positions"] = pos
        if train:
            g["labels"] = lab
        g.create_dataset("examples", ex)
        g.attrs["contig"] = "ctg_a"
        g.attrs["size"] = n
        cg = f.create_group("contigs")
        sub = cg.create_group("ctg_a")
        sub.attrs["name"] = "ctg_a"
        sub.attrs["seq"] = seq
        sub.attrs["len"] = len(seq)
    return pos, ex, lab, seq


def test_h5_writer_reader_roundtrip(tmp_path):
    path = str(tmp_path / "ref.hdf5")
    pos, ex, lab, seq = _reference_layout_file(path)
    f = H5File(path)
    assert set(f.keys()) == {f"ctg_a_0-{int(pos[-1,-1,0])}", "contigs"}
    g = f[f"ctg_a_0-{int(pos[-1,-1,0])}"]
    assert np.array_equal(np.asarray(g["positions"]), pos)
    assert np.array_equal(np.asarray(g["examples"]), ex)
    assert np.array_equal(np.asarray(g["labels"]), lab)
    assert g.attrs["contig"] == "ctg_a"
    assert g.attrs["size"] == 6
    sub = f["contigs"]["ctg_a"]
    assert sub.attrs["seq"] == seq
    assert sub.attrs["len"] == len(seq)


def test_h5_many_groups(tmp_path):
    """> 8 children exercises the SNOD chunking + B-tree build."""
    path = str(tmp_path / "many.hdf5")
    with H5Writer(path) as f:
        for i in range(40):
            g = f.create_group(f"g{i:03d}")
            g["d"] = np.arange(i + 1, dtype=np.int32)
            g.attrs["i"] = i
    f = H5File(path)
    assert len(f.keys()) == 40
    for i in (0, 7, 8, 23, 39):
        g = f[f"g{i:03d}"]
        assert g.attrs["i"] == i
        assert np.array_equal(np.asarray(g["d"]),
                              np.arange(i + 1, dtype=np.int32))


def test_hdf5_to_rkw_feeds_training(tmp_path):
    """A reference-layout file converts to RKW and round-trips through the
    training dataset and inference voting (the interop contract)."""
    from roko_amd.datasets import InMemoryTrainDataset

    h5 = str(tmp_path / "ref.hdf5")
    pos, ex, lab, seq = _reference_layout_file(h5)
    rkw = str(tmp_path / "conv.rkw")
    n = hdf5_to_rkw(h5, rkw)
    assert n == len(pos)

    f = RkwFile(rkw)
    assert not f.inference
    assert f.contig_names() == ["ctg_a"]
    assert f.contig_seq("ctg_a") == seq
    _, rpos, rex, rlab = f.group_arrays(0)
    assert np.array_equal(np.asarray(rpos), pos.astype(np.int32))
    assert np.array_equal(np.asarray(rex), ex)
    assert np.array_equal(np.asarray(rlab), lab.astype(np.uint8))

    ds = InMemoryTrainDataset(rkw)
    assert len(ds) == n
    x, y = ds[2]
    assert torch.equal(x, torch.from_numpy(ex[2].astype(np.int64)))
    assert torch.equal(y, torch.from_numpy(lab[2]))


def test_rkw_to_hdf5_roundtrip(tmp_path):
    """RKW -> reference HDF5 -> RKW is lossless (export interop)."""
    rkw1 = str(tmp_path / "a.rkw")
    rng = np.random.default_rng(3)
    w = RkwWriter(rkw1, inference=True)
    P = np.zeros((5, C.WINDOW_COLS, 2), dtype=np.int32)
    P[..., 0] = np.arange(C.WINDOW_COLS)[None, :]
    X = rng.integers(0, 12, (5, C.WINDOW_ROWS, C.WINDOW_COLS)).astype(np.uint8)
    w.store("c_1", 0, 89, P, X)
    w.write_contigs([("c_1", "ACGT" * 100)])
    w.close()

    h5 = str(tmp_path / "b.hdf5")
    assert rkw_to_hdf5(rkw1, h5) == 5
    rkw2 = str(tmp_path / "c.rkw")
    assert hdf5_to_rkw(h5, rkw2) == 5
    f1, f2 = RkwFile(rkw1), RkwFile(rkw2)
    assert f2.inference
    for gi in range(len(f1.groups)):
        _, p1, x1, _ = f1.group_arrays(gi)
        _, p2, x2, _ = f2.group_arrays(gi)
        assert np.array_equal(np.asarray(p1), np.asarray(p2))
        assert np.array_equal(np.asarray(x1), np.asarray(x2))
    assert f2.contig_seq("c_1") == f1.contig_seq("c_1")


@pytest.mark.skipif(not HAVE_H5PY, reason="h5py not installed in this image")
def test_h5py_reads_our_file(tmp_path):
    """Cross-validation when real h5py is present."""
    import h5py

    path = str(tmp_path / "ours.hdf5")
    pos, ex, lab, seq = _reference_layout_file(path)
    with h5py.File(path, "r") as f:
        g = f[list(k for k in f.keys() if k != "contigs")[0]]
        assert np.array_equal(g["positions"][()], pos)
        assert np.array_equal(g["examples"][()], ex)
        assert g.attrs["size"] == 6


@pytest.mark.skipif(not HAVE_H5PY, reason="h5py not installed in this image")
def test_we_read_h5py_file(tmp_path):
    import h5py

    path = str(tmp_path / "theirs.hdf5")
    rng = np.random.default_rng(1)
    ex = rng.integers(0, 12, (4, 200, 90)).astype(np.uint8)
    with h5py.File(path, "w") as f:
        g = f.create_group("ctg_0-100")
        g["positions"] = np.zeros((4, 90, 2), dtype=np.int64)
        g["labels"] = np.zeros((4, 90), dtype=np.int64)
        g.create_dataset("examples", data=ex, chunks=(1, 200, 90))
        g.attrs["contig"] = "ctg"
        g.attrs["size"] = 4
    f = H5File(path)
    g = f["ctg_0-100"]
    assert np.array_equal(np.asarray(g["examples"]), ex)
    assert g.attrs["contig"] == "ctg"


def test_chunked_dataset_roundtrip(tmp_path):
    """Chunked writing (the reference uses chunks=(1,200,90) for examples)
    exercises the chunk-B-tree writer AND reader, incl. multi-level trees
    (>64 chunks) and partial edge chunks."""
    rng = np.random.default_rng(9)
    path = str(tmp_path / "chunked.hdf5")
    a = rng.integers(0, 250, (70, 20, 9)).astype(np.uint8)   # 70 chunks > 64
    b = rng.standard_normal((13, 7)).astype(np.float32)      # partial chunks
    with H5Writer(path) as f:
        g = f.create_group("g")
        g.create_dataset("a", a, chunks=(1, 20, 9))
        g.create_dataset("b", b, chunks=(4, 4))
        g.create_dataset("c", a[:3], chunks=(2, 20, 9))      # edge chunk
    f = H5File(path)
    assert np.array_equal(np.asarray(f["g"]["a"]), a)
    assert np.array_equal(np.asarray(f["g"]["b"]), b)
    assert np.array_equal(np.asarray(f["g"]["c"]), a[:3])


def test_reader_rejects_garbage(tmp_path):
    bad = tmp_path / "x.hdf5"
    bad.write_bytes(b"not an hdf5 file at all")
    with pytest.raises(ValueError, match="not an HDF5"):
        H5File(str(bad))
    trunc = tmp_path / "t.hdf5"
    trunc.write_bytes(b"\x89HDF\r\n\x1a\n\x00\x00")
    with pytest.raises(ValueError, match="truncated"):
        H5File(str(trunc))


def test_hdf5_compat_cli(tmp_path, capsys):
    from roko_amd.io.hdf5_compat import main as compat_main

    h5 = str(tmp_path / "a.hdf5")
    _reference_layout_file(h5)
    rkw = str(tmp_path / "a.rkw")
    compat_main([h5, rkw])
    assert "converted 6 windows" in capsys.readouterr().out
    h52 = str(tmp_path / "b.hdf5")
    compat_main([rkw, h52])
    assert "converted 6 windows" in capsys.readouterr().out
    f = H5File(h52)
    assert any(k != "contigs" for k in f.keys())
