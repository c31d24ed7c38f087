"""Feature-generation CLI: draft FASTA + reads BAM -> RKW window files.

Role-equivalent to the reference's roko/features.py:113-157: split every
contig into overlapping regions, fan regions out over worker processes, run
the C++ window builder per region, and (for training) join windows with
truth labels before writing.

Usage:
  python -m roko_amd.features <draft.fasta> <reads.bam> <out.rkw>
      [--Y truth.bam] [--t workers] [--seed N]
"""

from __future__ import annotations

import argparse
import time
from multiprocessing import Pool
from typing import Iterator, Optional, Tuple

import numpy as np

from . import config as C
from .config import FeatureConfig
from .io.fasta import read_fasta
from .rkdata import RkwWriter


def generate_regions(
    ref_len: int, size: int = C.REGION_SIZE, overlap: int = C.REGION_OVERLAP
) -> Iterator[Tuple[int, int]]:
    """(start, end) spans of `size` with `overlap` between neighbours
    (reference: features.py:16-27)."""
    i = 0
    while i < ref_len:
        end = min(i + size, ref_len)
        yield i, end
        if i + size >= ref_len:
            break
        i = i + size - overlap


def _features_for_region(bam, contig, start, end, cfg: FeatureConfig):
    from .ops import _pileup

    return _pileup.generate_features(
        bam,
        contig,
        start,
        end,
        rows=cfg.window_rows,
        cols=cfg.window_cols,
        stride=cfg.window_stride,
        max_ins=cfg.max_ins,
        filter_flag=cfg.filter_flag,
        min_mapq=cfg.min_mapq,
        seed=cfg.seed,
    )


def generate_infer(args):
    """Inference worker: windows only (reference: features.py:97-110).
    Failures are returned, not raised: one bad region must not kill the
    whole run (the reference dies on any worker exception).
    """
    bam_x, contig, start, end, cfg = args
    try:
        positions, examples = _features_for_region(bam_x, contig, start, end, cfg)
    except Exception as e:  # noqa: BLE001 — deliberate region-level fence
        return ("__error__", f"{contig}:{start}-{end}", repr(e))
    return contig, start, end, positions, examples, None


def generate_train(args):
    """Training worker: windows joined with truth labels
    (reference: features.py:37-94). Failures are returned, not raised."""
    try:
        return _generate_train(args)
    except Exception as e:  # noqa: BLE001 — deliberate region-level fence
        _, _, contig, start, end, _ = args
        return ("__error__", f"{contig}:{start}-{end}", repr(e))


def _generate_train(args):
    bam_x, bam_y, contig, start, end, cfg = args
    from .labels import filter_aligns, get_aligns, get_pos_and_labels

    aligns = get_aligns(bam_y, contig, start, end)
    filtered = filter_aligns(aligns)
    if not filtered:
        return None

    def in_region(pos: int) -> bool:
        return any(a.start <= pos < a.end for a in filtered)

    out_pos, out_x, out_y = [], [], []
    for a in filtered:
        t_pos, t_labels = get_pos_and_labels(a, start, end)
        pos_labels = {}
        n_pos = set()
        for p, l in zip(t_pos, t_labels):
            if l == C.LABEL_UNKNOWN:
                n_pos.add(p)
            else:
                pos_labels[p] = l
        if not pos_labels:
            continue
        pos_sorted = sorted(pos_labels)
        # feature columns strictly inside the labeled span (reference:
        # features.py:62-63 builds "{first+1}-{last}" which htslib reads as
        # the 0-based half-open [first, last))
        sub_start, sub_end = pos_sorted[0][0], pos_sorted[-1][0]
        if sub_end <= sub_start:
            continue
        positions, examples = _features_for_region(bam_x, contig, sub_start, sub_end, cfg)

        for w in range(positions.shape[0]):
            P = positions[w]
            keep = True
            Y = np.empty(cfg.window_cols, dtype=np.uint8)
            for s in range(cfg.window_cols):
                p = (int(P[s, 0]), int(P[s, 1]))
                assert in_region(p[0]), f"window position {p} outside filtered truth aligns"
                if p in n_pos:
                    keep = False
                    break
                try:
                    Y[s] = pos_labels[p]
                except KeyError:
                    if p[1] != 0:
                        Y[s] = C.LABEL_GAP  # un-labeled insertion slot
                    else:
                        raise KeyError(f"no label for draft position {p}") from None
            if keep:
                out_pos.append(P)
                out_x.append(examples[w])
                out_y.append(Y)

    if not out_pos:
        return None
    return (
        contig,
        start,
        end,
        np.stack(out_pos),
        np.stack(out_x),
        np.stack(out_y),
    )


def run(
    ref_path: str,
    bam_x: str,
    out_path: str,
    bam_y: Optional[str] = None,
    workers: int = 1,
    cfg: Optional[FeatureConfig] = None,
    log=print,
) -> int:
    cfg = cfg or FeatureConfig()
    inference = bam_y is None
    refs = list(read_fasta(ref_path))

    jobs = []
    for name, seq in refs:
        for start, end in generate_regions(len(seq), cfg.region_size, cfg.region_overlap):
            if inference:
                jobs.append((bam_x, name, start, end, cfg))
            else:
                jobs.append((bam_x, bam_y, name, start, end, cfg))
    func = generate_infer if inference else generate_train
    log(f"feature generation: {len(jobs)} region jobs, {workers} workers")

    n_windows = 0
    t0 = time.time()
    with RkwWriter(out_path, inference=inference) as writer:
        writer.write_contigs(refs)

        n_errors = 0

        def consume(result, job):
            nonlocal n_windows, n_errors
            if result is None:
                return
            if result[0] == "__error__":
                # one in-parent retry before giving the region up: transient
                # faults (an fs hiccup, a worker killed mid-region) should
                # not cost coverage (SURVEY.md §5.3 — the reference dies on
                # the first worker exception and has no retry at all)
                log(f"WARNING: region {result[1]} failed: {result[2]} "
                    "(retrying once)")
                result = func(job)
                if result is None:
                    return
                if result[0] == "__error__":
                    n_errors += 1
                    log(f"WARNING: region {result[1]} failed twice: "
                        f"{result[2]} (skipped)")
                    return
            contig, start, end, positions, examples, labs = result
            writer.store(contig, start, end, positions, examples, labs)
            n_windows += len(positions)

        if workers <= 1:
            for job in jobs:
                consume(func(job), job)
        else:
            with Pool(processes=workers) as pool:
                for result, job in zip(pool.imap(func, jobs), jobs):
                    consume(result, job)
    dt = time.time() - t0
    log(f"wrote {n_windows} windows to {out_path} in {dt:.1f}s "
        f"({n_windows / max(dt, 1e-9):.0f} windows/s)"
        + (f"; {n_errors} regions FAILED and were skipped" if n_errors else ""))
    return n_windows


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("ref", help="draft assembly FASTA")
    p.add_argument("X", help="reads-to-draft BAM (indexed)")
    p.add_argument("o", help="output .rkw path")
    p.add_argument("--Y", default=None, help="truth-to-draft BAM (training mode)")
    p.add_argument("--t", type=int, default=1, help="worker processes")
    p.add_argument("--seed", type=int, default=0)
    a = p.parse_args(argv)
    cfg = FeatureConfig(seed=a.seed)
    run(a.ref, a.X, a.o, bam_y=a.Y, workers=a.t, cfg=cfg)


if __name__ == "__main__":
    main()
