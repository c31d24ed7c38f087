"""Synthetic-alignment helpers for tests: build a draft from a truth genome
via a tracked edit script, so exact CIGARs for truth->draft and
read->draft alignments can be constructed without an aligner."""

from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np

from roko_amd.io.bamio import SamRecord

BASES = "ACGT"


class EditScript:
    """Mutate `truth` into a draft while recording the edit script.

    dstart[t]  — draft coordinate of truth base t, or -1 if deleted
    ins_after[t] — number of draft-only bases inserted after truth base t
    """

    def __init__(self, rng: np.random.Generator, truth: str,
                 sub_rate=0.01, ins_rate=0.005, del_rate=0.005):
        self.truth = truth
        dstart = np.full(len(truth), -1, dtype=np.int64)
        ins_after = np.zeros(len(truth), dtype=np.int64)
        out: List[str] = []
        for t, ch in enumerate(truth):
            r = rng.random()
            if r < del_rate:
                pass  # deleted from draft
            else:
                dstart[t] = len(out)
                if r < del_rate + sub_rate:
                    ch2 = BASES[int(rng.integers(4))]
                    out.append(ch2)
                else:
                    out.append(ch)
            while rng.random() < ins_rate:
                out.append(BASES[int(rng.integers(4))])
                ins_after[t] += 1
        self.draft = "".join(out)
        self.dstart = dstart
        self.ins_after = ins_after

    def align_substring(self, qname: str, s: int, e: int, flag: int = 0,
                        tid: int = 0, mapq: int = 60) -> Optional[SamRecord]:
        """SamRecord aligning truth[s:e) against the draft."""
        seq = self.truth[s:e]
        # first aligned truth base
        t0 = s
        while t0 < e and self.dstart[t0] < 0:
            t0 += 1
        if t0 >= e:
            return None
        # last aligned truth base
        t1 = e - 1
        while t1 >= t0 and self.dstart[t1] < 0:
            t1 -= 1

        cig: List[Tuple[int, str]] = []

        def add(n: int, op: str):
            if n <= 0:
                return
            if cig and cig[-1][1] == op:
                cig[-1] = (cig[-1][0] + n, op)
            else:
                cig.append((n, op))

        add(t0 - s, "S")
        for t in range(t0, t1 + 1):
            if self.dstart[t] >= 0:
                add(1, "M")
            else:
                add(1, "I")
            if t < t1:
                add(int(self.ins_after[t]), "D")
        add(e - 1 - t1, "S")
        return SamRecord(qname, flag, tid, int(self.dstart[t0]), mapq, cig, seq)


def mutate_seq(rng: np.random.Generator, seq: str, err: float) -> str:
    """Apply sequencing-style noise: substitutions, insertions and
    deletions at err/3 each (total per-base error ~err)."""
    if err <= 0:
        return seq
    out: List[str] = []
    third = err / 3.0
    for ch in seq:
        r = rng.random()
        if r < third:
            continue  # deletion
        if r < 2 * third:
            out.append(BASES[int(rng.integers(4))])  # substitution
        else:
            out.append(ch)
        while rng.random() < third:
            out.append(BASES[int(rng.integers(4))])  # insertion
    return "".join(out)


def noisy_record(rng: np.random.Generator, es: EditScript, qname: str,
                 s: int, e: int, err: float, px, flag: int = 0,
                 mapq: int = 60) -> Optional[SamRecord]:
    """SamRecord of a NOISY read (sequencing error `err` on a truth
    substring) aligned against the draft with the in-repo banded aligner
    (`px.align_cigar`) — the realistic path: read noise + draft errors both
    present, CIGAR produced by alignment rather than by construction."""
    t0 = s
    while t0 < e and es.dstart[t0] < 0:
        t0 += 1
    t1 = e - 1
    while t1 >= t0 and es.dstart[t1] < 0:
        t1 -= 1
    if t1 < t0:
        return None
    p0 = int(es.dstart[t0])
    p1 = int(es.dstart[t1]) + 1
    seq = mutate_seq(rng, es.truth[t0:t1 + 1], err)
    if not seq:
        return None
    band = max(64, int((t1 - t0) * (err + 0.05)) + 32)
    res = px.align_cigar(seq, es.draft[p0:p1], band=band)
    cig: List[Tuple[int, str]] = []
    num = 0
    for ch in res["cigar"]:
        if ch.isdigit():
            num = num * 10 + int(ch)
        else:
            cig.append((num, ch))
            num = 0
    # normalize to a valid BAM CIGAR: leading/trailing deletions shift the
    # mapping coordinates; terminal insertions become soft clips
    pos = p0
    while cig and cig[0][1] == "D":
        pos += cig.pop(0)[0]
    while cig and cig[-1][1] == "D":
        cig.pop()
    if cig and cig[0][1] == "I":
        cig[0] = (cig[0][0], "S")
    if cig and cig[-1][1] == "I":
        cig[-1] = (cig[-1][0], "S")
    if not any(op == "M" for _, op in cig):
        return None
    return SamRecord(qname, flag, 0, pos, mapq, cig, seq)


def build_assembly(rng: np.random.Generator, outdir, *, length=3000, cov=20,
                   read_len=400, read_err=0.0, sub=0.01, ins=0.003,
                   dl=0.003):
    """Full synthetic polishing scenario on disk: truth genome, draft with
    the given error profile, reads at `cov`x (error-free exact-CIGAR reads
    when read_err=0, otherwise noisy reads aligned with the banded
    aligner), plus the truth-to-draft BAM for labeling. Returns the same
    dict shape as the `tiny_assembly` fixture."""
    import os

    from roko_amd.io.bamio import write_bam
    from roko_amd.io.fasta import write_fasta
    from roko_amd.ops import pileup_ext

    truth = "".join(BASES[int(b)] for b in rng.integers(0, 4, length))
    es = EditScript(rng, truth, sub_rate=sub, ins_rate=ins, del_rate=dl)
    draft = es.draft
    outdir = str(outdir)
    os.makedirs(outdir, exist_ok=True)
    draft_fasta = os.path.join(outdir, "draft.fasta")
    write_fasta(draft_fasta, [("ctg1", draft)])
    refs = [("ctg1", len(draft))]

    px = pileup_ext()
    reads = []
    n_reads = max(1, cov * len(truth) // read_len)
    for i in range(n_reads):
        s = int(rng.integers(0, max(1, len(truth) - read_len)))
        flag = 16 if i % 2 else 0
        if read_err > 0:
            rec = noisy_record(rng, es, f"read{i}", s, s + read_len,
                               read_err, px, flag=flag)
        else:
            rec = es.align_substring(f"read{i}", s, s + read_len, flag=flag)
        if rec is not None:
            reads.append(rec)
    reads.sort(key=lambda r: (r.tid, r.pos))
    reads_bam = os.path.join(outdir, "reads.bam")
    write_bam(reads_bam, refs, reads)

    trec = es.align_substring("truth_ctg1", 0, len(truth), flag=0)
    truth_bam = os.path.join(outdir, "truth.bam")
    write_bam(truth_bam, refs, [trec])
    return {
        "draft_fasta": draft_fasta,
        "reads_bam": reads_bam,
        "truth_bam": truth_bam,
        "truth": truth,
        "draft": draft,
        "edit_script": es,
    }
