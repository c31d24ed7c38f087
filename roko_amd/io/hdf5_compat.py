"""Converters between the reference's HDF5 dataset layout and RKW.

The reference's on-disk contract (roko/data.py:29-48) is an HDF5 file of
groups ``{contig}_{start}-{end}`` holding ``positions`` (N, W, 2),
``examples`` (N, R, W) uint8 and — for training files — ``labels`` (N, W),
plus a ``contigs`` group with per-contig subgroups whose attrs carry
``name``/``seq``/``len``. This module round-trips that layout against RKW
using the self-contained HDF5 reader/writer in ``roko_amd.io.hdf5`` (no
h5py in this environment), so existing roko feature files feed this
framework and RKW files can be exported back.

CLI:  python -m roko_amd.io.hdf5_compat <in.(hdf5|rkw)> <out.(rkw|hdf5)>
"""

from __future__ import annotations

import sys
from typing import List, Tuple

import numpy as np

from .hdf5 import H5File, H5Writer
from ..rkdata import RkwFile, RkwWriter


def _parse_span(group_name: str) -> Tuple[int, int]:
    """'{contig}_{start}-{end}' -> (start, end); contig may contain '_'."""
    tail = group_name.rsplit("_", 1)[-1]
    s, e = tail.split("-")
    return int(s), int(e)


def hdf5_to_rkw(h5_path: str, rkw_path: str) -> int:
    """Convert a reference-layout HDF5 feature file to RKW; returns the
    number of windows converted."""
    f = H5File(h5_path)
    names = [k for k in f.keys() if k not in ("contigs", "info")]
    # training file iff every window group carries labels
    has_labels = all("labels" in f[k] for k in names) and bool(names)
    w = RkwWriter(rkw_path, inference=not has_labels)
    n_total = 0
    for k in sorted(names):
        g = f[k]
        pos = np.asarray(g["positions"])
        ex = np.asarray(g["examples"]).astype(np.uint8)
        lab = np.asarray(g["labels"]) if "labels" in g else None
        contig = g.attrs.get("contig")
        if isinstance(contig, bytes):
            contig = contig.decode("utf-8")
        if contig is None:
            contig = k.rsplit("_", 1)[0]
        try:
            start, end = _parse_span(k)
        except ValueError:
            start, end = int(pos[0, 0, 0]), int(pos[-1, -1, 0])
        w.store(contig, start, end, pos.astype(np.int32), ex,
                labels=None if lab is None else lab.astype(np.uint8))
        n_total += len(pos)
    contigs: List[Tuple[str, str]] = []
    if "contigs" in f:
        cg = f["contigs"]
        for name in cg.keys():
            sub = cg[name]
            seq = sub.attrs.get("seq", "")
            if isinstance(seq, bytes):
                seq = seq.decode("utf-8")
            contigs.append((name, seq))
    w.write_contigs(contigs)
    w.close()
    return n_total


def rkw_to_hdf5(rkw_path: str, h5_path: str) -> int:
    """Export an RKW feature file to the reference HDF5 layout (readable by
    the reference's datasets.py/inference.py through h5py)."""
    f = RkwFile(rkw_path)
    n_total = 0
    with H5Writer(h5_path) as out:
        for gi in range(len(f.groups)):
            g, pos, ex, lab = f.group_arrays(gi)
            grp = out.create_group(f"{g['contig']}_{g['start']}-{g['end']}")
            # reference stores positions/labels as int64 (python-int arrays)
            grp["positions"] = np.asarray(pos).astype(np.int64)
            if lab is not None:
                grp["labels"] = np.asarray(lab).astype(np.int64)
            grp.create_dataset("examples", np.asarray(ex).astype(np.uint8),
                               chunks=(1, ex.shape[1], ex.shape[2]))
            grp.attrs["contig"] = g["contig"]
            grp.attrs["size"] = int(g["size"])
            n_total += int(g["size"])
        cg = out.create_group("contigs")
        for name in f.contig_names():
            sub = cg.create_group(name)
            seq = f.contig_seq(name)
            sub.attrs["name"] = name
            sub.attrs["seq"] = seq
            sub.attrs["len"] = len(seq)
    return n_total


def main(argv=None) -> None:
    args = sys.argv[1:] if argv is None else argv
    if len(args) != 2:
        raise SystemExit(__doc__)
    src, dst = args
    if src.endswith((".hdf5", ".h5")) and dst.endswith(".rkw"):
        n = hdf5_to_rkw(src, dst)
    elif src.endswith(".rkw") and dst.endswith((".hdf5", ".h5")):
        n = rkw_to_hdf5(src, dst)
    else:
        raise SystemExit("expected <in.(hdf5|rkw)> <out.(rkw|hdf5)>")
    print(f"converted {n} windows: {src} -> {dst}")


if __name__ == "__main__":
    main()
