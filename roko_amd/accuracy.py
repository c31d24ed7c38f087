"""Assembly-quality assessment: draft/polished vs truth error breakdown.

The reference's published value is its error table — total error, mismatch,
insertion and deletion rates plus Q score on the S. aureus test set
(reference README.md:97-112), produced externally with pomoxis
``assess_assembly``. This module measures the same quantities in-repo with a
banded unit-cost global aligner (C++ ``_pileup.align_stats``), so the test
suite can assert that polishing actually reduces error on synthetic
truth/draft pairs (tests/test_accuracy.py) instead of only checking output
shape.

Rates follow the assess_assembly convention: errors / aligned truth length.
Q score = -10 log10(total error rate).
"""

from __future__ import annotations

import math
from typing import Dict


def _align_stats(query: str, target: str, band: int) -> Dict[str, int]:
    from .ops import pileup_ext

    return pileup_ext().align_stats(query, target, band=band)


def seq_stats(query: str, truth: str, band: int = 0) -> Dict[str, float]:
    """Alignment error stats of `query` against `truth`.

    band=0 picks ceil(5% of len)+32 and doubles on band overflow (the
    optimal path must fit inside the band for exact counts).
    """
    if not truth:
        raise ValueError("empty truth sequence")
    b = band if band > 0 else min(len(truth), 32 + len(truth) // 20)
    while True:
        try:
            s = _align_stats(query, truth, b)
            break
        except RuntimeError:
            if b >= max(len(truth), len(query)):
                raise
            b = min(2 * b, max(len(truth), len(query)))
    n = len(truth)
    err = s["edit_distance"] / n
    return {
        "edit_distance": int(s["edit_distance"]),
        "total_error": err,
        "mismatch": s["mismatches"] / n,
        "insertion": s["insertions"] / n,
        "deletion": s["deletions"] / n,
        "qscore": (-10.0 * math.log10(err)) if err > 0 else float("inf"),
        "band": b,
    }


def main(argv=None) -> None:
    """CLI: score an assembly (and optionally the pre-polish draft) against
    a truth FASTA — the in-repo equivalent of pomoxis assess_assembly.

    Usage: python -m roko_amd.accuracy polished.fasta truth.fasta
               [--draft draft.fasta] [--band N]
    Contigs are matched by name; the summary aggregates over matched pairs.
    """
    import argparse

    from .io.fasta import read_fasta

    p = argparse.ArgumentParser(description=main.__doc__)
    p.add_argument("assembly", help="polished (or any) assembly FASTA")
    p.add_argument("truth", help="truth FASTA (contig names must match)")
    p.add_argument("--draft", default=None,
                   help="pre-polish draft FASTA: also report error reduction")
    p.add_argument("--band", type=int, default=0,
                   help="alignment band (0 = auto, grows on overflow)")
    a = p.parse_args(argv)

    asm = dict(read_fasta(a.assembly))
    tru = dict(read_fasta(a.truth))
    dra = dict(read_fasta(a.draft)) if a.draft else {}
    names = sorted(set(asm) & set(tru))
    if not names:
        raise SystemExit("no contig names shared between assembly and truth")
    tot_e = tot_n = 0
    for name in names:
        st = seq_stats(asm[name], tru[name], a.band)
        line = (f"{name}: err {st['total_error']:.4%} "
                f"(mm {st['mismatch']:.4%} ins {st['insertion']:.4%} "
                f"del {st['deletion']:.4%}) Q{st['qscore']:.2f}")
        if name in dra:
            d = seq_stats(dra[name], tru[name], a.band)
            red = (1 - st["total_error"] / d["total_error"]
                   if d["total_error"] > 0 else 0.0)
            line += f"  [draft err {d['total_error']:.4%}, reduction {red:.1%}]"
        print(line)
        tot_e += st["edit_distance"]
        tot_n += len(tru[name])
    err = tot_e / max(tot_n, 1)
    q = (-10.0 * math.log10(err)) if err > 0 else float("inf")
    print(f"TOTAL: err {err:.4%} Q{q:.2f} over {len(names)} contig(s)")


def assess_polishing(draft: str, polished: str, truth: str,
                     band: int = 0) -> Dict[str, object]:
    """Compare draft-vs-truth and polished-vs-truth error; the headline
    number is `error_reduction` (1 = perfect polish, 0 = no improvement,
    negative = polishing made it worse)."""
    d = seq_stats(draft, truth, band)
    p = seq_stats(polished, truth, band)
    red = (1.0 - p["total_error"] / d["total_error"]
           if d["total_error"] > 0 else 0.0)
    return {"draft": d, "polished": p, "error_reduction": red}


if __name__ == "__main__":
    main()
