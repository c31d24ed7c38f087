"""BAM/BGZF/BAI *writer* in pure Python (stdlib zlib only).

The framework's C++ side only ever reads BAM; writing is needed for test
fixtures and synthetic-data simulation (the image has no samtools/pysam).
Implements the BAM, BGZF and BAI on-disk formats per the SAM specification
(samtools/hts-specs SAMv1).
"""

from __future__ import annotations

import struct
import zlib
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

# CIGAR op chars in spec order; op code = index
CIGAR_OPS = "MIDNSHP=X"
_SEQ_CODE = {c: i for i, c in enumerate("=ACMGRSVTWYHKDBN")}

BGZF_EOF = bytes.fromhex(
    "1f8b08040000000000ff0600424302001b0003000000000000000000"
)


@dataclass
class SamRecord:
    qname: str
    flag: int
    tid: int
    pos: int  # 0-based
    mapq: int
    cigar: Sequence[Tuple[int, str]]  # [(length, op char)]
    seq: str
    qual: Optional[bytes] = None  # phred values; defaults to 30s

    def ref_span(self) -> int:
        return sum(l for l, op in self.cigar if op in "MDN=X")

    def query_len(self) -> int:
        return sum(l for l, op in self.cigar if op in "MIS=X")

    def encode(self) -> bytes:
        name = self.qname.encode() + b"\0"
        n_cigar = len(self.cigar)
        cig = b"".join(
            struct.pack("<I", (l << 4) | CIGAR_OPS.index(op)) for l, op in self.cigar
        )
        l_seq = len(self.seq)
        seq4 = bytearray((l_seq + 1) // 2)
        for i, ch in enumerate(self.seq):
            code = _SEQ_CODE.get(ch.upper(), 15)
            if i % 2 == 0:
                seq4[i // 2] |= code << 4
            else:
                seq4[i // 2] |= code
        qual = self.qual if self.qual is not None else bytes([30]) * l_seq
        end = self.pos + max(self.ref_span(), 1)
        body = struct.pack(
            "<iiBBHHHiiii",
            self.tid,
            self.pos,
            len(name),
            self.mapq,
            reg2bin(self.pos, end),
            n_cigar,
            self.flag,
            l_seq,
            -1,  # next_refID
            -1,  # next_pos
            0,  # tlen
        )
        body += name + cig + bytes(seq4) + bytes(qual)
        return struct.pack("<I", len(body)) + body


def reg2bin(beg: int, end: int) -> int:
    end -= 1
    if beg >> 14 == end >> 14:
        return ((1 << 15) - 1) // 7 + (beg >> 14)
    if beg >> 17 == end >> 17:
        return ((1 << 12) - 1) // 7 + (beg >> 17)
    if beg >> 20 == end >> 20:
        return ((1 << 9) - 1) // 7 + (beg >> 20)
    if beg >> 23 == end >> 23:
        return ((1 << 6) - 1) // 7 + (beg >> 23)
    if beg >> 26 == end >> 26:
        return ((1 << 3) - 1) // 7 + (beg >> 26)
    return 0


class _BgzfWriter:
    def __init__(self, fh):
        self.fh = fh
        self.buf = bytearray()
        self.file_offset = 0

    @property
    def voffset(self) -> int:
        return (self.file_offset << 16) | len(self.buf)

    def write(self, data: bytes) -> None:
        self.buf += data
        while len(self.buf) >= 60000:
            self._flush_block(self.buf[:60000])
            del self.buf[:60000]

    def _flush_block(self, payload: bytes) -> None:
        co = zlib.compressobj(6, zlib.DEFLATED, -15)
        cdata = co.compress(bytes(payload)) + co.flush()
        bsize = len(cdata) + 12 + 6 + 8  # hdr + extra + crc/isize
        block = (
            bytes([0x1F, 0x8B, 8, 4]) + b"\0\0\0\0" + bytes([0, 0xFF])
            + struct.pack("<H", 6) + b"BC" + struct.pack("<HH", 2, bsize - 1)
            + cdata
            + struct.pack("<II", zlib.crc32(bytes(payload)) & 0xFFFFFFFF, len(payload))
        )
        self.fh.write(block)
        self.file_offset += len(block)

    def finish(self) -> None:
        if self.buf:
            self._flush_block(bytes(self.buf))
            self.buf.clear()
        self.fh.write(BGZF_EOF)


def write_bam(
    path: str,
    references: Sequence[Tuple[str, int]],
    records: Sequence[SamRecord],
    write_index: bool = True,
) -> None:
    """Write a coordinate-sorted BAM (+ .bai when requested).

    `records` must already be sorted by (tid, pos); asserts otherwise.
    """
    order = [(r.tid, r.pos) for r in records if r.tid >= 0]
    assert order == sorted(order), "records must be coordinate-sorted"

    header_text = "@HD\tVN:1.6\tSO:coordinate\n" + "".join(
        f"@SQ\tSN:{n}\tLN:{l}\n" for n, l in references
    )

    # per-ref index accumulators
    bins: List[dict] = [dict() for _ in references]  # bin -> [ [beg,end], ... ]
    linear: List[dict] = [dict() for _ in references]  # intv -> min voffset

    with open(path, "wb") as fh:
        w = _BgzfWriter(fh)
        w.write(b"BAM\1")
        ht = header_text.encode()
        w.write(struct.pack("<i", len(ht)) + ht)
        w.write(struct.pack("<i", len(references)))
        for name, length in references:
            nb = name.encode() + b"\0"
            w.write(struct.pack("<i", len(nb)) + nb + struct.pack("<i", length))
        for rec in records:
            beg_v = w.voffset
            w.write(rec.encode())
            end_v = w.voffset
            if rec.tid < 0:
                continue
            rend = rec.pos + max(rec.ref_span(), 1)
            b = reg2bin(rec.pos, rend)
            chunks = bins[rec.tid].setdefault(b, [])
            if chunks and chunks[-1][1] == beg_v:
                chunks[-1][1] = end_v
            else:
                chunks.append([beg_v, end_v])
            for iv in range(rec.pos >> 14, (rend - 1 >> 14) + 1):
                cur = linear[rec.tid].get(iv)
                if cur is None or beg_v < cur:
                    linear[rec.tid][iv] = beg_v
        w.finish()

    if write_index:
        _write_bai(path + ".bai", bins, linear)


def _write_bai(path: str, bins: List[dict], linear: List[dict]) -> None:
    with open(path, "wb") as fh:
        fh.write(b"BAI\1")
        fh.write(struct.pack("<i", len(bins)))
        for rb, rl in zip(bins, linear):
            fh.write(struct.pack("<i", len(rb)))
            for bin_id in sorted(rb):
                chunks = rb[bin_id]
                fh.write(struct.pack("<Ii", bin_id, len(chunks)))
                for beg, end in chunks:
                    fh.write(struct.pack("<QQ", beg, end))
            n_intv = (max(rl) + 1) if rl else 0
            fh.write(struct.pack("<i", n_intv))
            last = 0
            for iv in range(n_intv):
                if iv in rl:
                    last = rl[iv]
                fh.write(struct.pack("<Q", last))
