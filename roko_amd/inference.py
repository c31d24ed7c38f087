"""Inference CLI: RKW windows + .pth checkpoint -> polished FASTA.

Role-equivalent to the reference's roko/inference.py:90-154, with two
deliberate re-designs for MI355X-scale throughput:

  * the per-(pos, ins) majority vote is a vectorised numpy group-by over all
    windows of a contig instead of a Python dict-of-Counters per base (the
    reference's CPU bottleneck, inference.py:119-124);
  * contig groups are sharded across ranks (one process per GPU); partial
    vote tables are gathered to rank 0, which merges (votes are associative
    counts) and stitches.

Stitch semantics match the reference (inference.py:129-147): sort voted
columns, drop leading insertion-only columns, take the draft prefix before
the first voted position, emit the majority base per column skipping GAP,
then append the draft suffix after the last voted position.
"""

from __future__ import annotations

import argparse
import time
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch
import torch.distributed as dist
from torch.utils.data import DataLoader

from . import config as C
from .datasets import InferenceDataset
from .io.fasta import write_fasta
from .model import RokoModel
from .parallel.ddp import init_distributed
from .rkdata import RkwFile
from .utils.metrics import Meter, trace_range

#: vote tables: contig -> (keys int64 (K,), counts int64 (K, NUM_CLASSES))
VoteTable = Dict[str, Tuple[np.ndarray, np.ndarray]]

_KEY_SHIFT = 3  # key = pos << 3 | ins  (ins <= MAX_INS < 8)


def accumulate_votes(
    positions: np.ndarray, preds: np.ndarray
) -> Tuple[np.ndarray, np.ndarray]:
    """Group (N, W, 2) positions + (N, W) class predictions into per-column
    class counts. Returns (unique keys sorted, counts (K, NUM_CLASSES))."""
    pos = positions.reshape(-1, 2).astype(np.int64)
    keys = (pos[:, 0] << _KEY_SHIFT) | pos[:, 1]
    cls = preds.reshape(-1).astype(np.int64)
    combo = keys * C.NUM_CLASSES + cls
    uniq, cnt = np.unique(combo, return_counts=True)
    ukeys = uniq // C.NUM_CLASSES
    ucls = uniq % C.NUM_CLASSES
    out_keys, inv = np.unique(ukeys, return_inverse=True)
    counts = np.zeros((len(out_keys), C.NUM_CLASSES), dtype=np.int64)
    counts[inv, ucls] += cnt
    return out_keys, counts


class StreamingVotes:
    """Bounded-memory vote accumulation (VERDICT r1 item 7 / BASELINE
    config 5): windows are buffered per contig and folded into a per-contig
    (keys, counts) table every `chunk_windows`, so peak host memory is the
    vote table (O(contig positions)) plus one chunk — NOT all windows'
    positions+predictions (~7 GB transient for a chr1-sized contig in the
    round-1 design; the reference's per-base Python Counter loop is
    inference.py:119-124)."""

    def __init__(self, chunk_windows: int = 4096):
        self.chunk = chunk_windows
        self.tables: VoteTable = {}
        self._pos: Dict[str, List[np.ndarray]] = defaultdict(list)
        self._pred: Dict[str, List[np.ndarray]] = defaultdict(list)

    def add(self, contig: str, positions: np.ndarray, preds: np.ndarray) -> None:
        """One window: positions (W, 2), preds (W,)."""
        self._pos[contig].append(positions)
        self._pred[contig].append(preds)
        if len(self._pos[contig]) >= self.chunk:
            self._flush(contig)

    def _flush(self, contig: str) -> None:
        if not self._pos[contig]:
            return
        t = accumulate_votes(np.stack(self._pos[contig]),
                             np.stack(self._pred[contig]))
        self._pos[contig].clear()
        self._pred[contig].clear()
        prev = self.tables.get(contig)
        self.tables[contig] = t if prev is None else merge_votes([prev, t])

    def buffered(self, contig: str) -> int:
        return len(self._pos[contig])

    def finalize(self) -> VoteTable:
        for contig in list(self._pos):
            self._flush(contig)
        return self.tables


def merge_votes(tables: List[Tuple[np.ndarray, np.ndarray]]) -> Tuple[np.ndarray, np.ndarray]:
    tables = [t for t in tables if len(t[0])]
    if not tables:
        return np.empty(0, dtype=np.int64), np.empty((0, C.NUM_CLASSES), dtype=np.int64)
    all_keys = np.concatenate([t[0] for t in tables])
    out_keys, inv = np.unique(all_keys, return_inverse=True)
    counts = np.zeros((len(out_keys), C.NUM_CLASSES), dtype=np.int64)
    off = 0
    for keys, cnts in tables:
        counts[inv[off : off + len(keys)]] += cnts
        off += len(keys)
    return out_keys, counts


def stitch_contig(draft: str, keys: np.ndarray, counts: np.ndarray) -> str:
    """Majority-vote consensus spliced into the draft (reference:
    inference.py:129-147). Ties resolve to the smallest class id
    (deterministic; the reference inherits Counter insertion order)."""
    if len(keys) == 0:
        return draft
    maj = counts.argmax(axis=1)
    pos = (keys >> _KEY_SHIFT).astype(np.int64)
    ins = (keys & ((1 << _KEY_SHIFT) - 1)).astype(np.int64)
    # drop leading insertion-only columns (reference: inference.py:134)
    first = 0
    while first < len(keys) and ins[first] != 0:
        first += 1
    if first == len(keys):
        return draft
    pos, ins, maj = pos[first:], ins[first:], maj[first:]
    base_chars = np.array(list(C.LABEL_ALPHABET[: C.NUM_CLASSES]))
    keep = maj != C.LABEL_GAP
    mid = "".join(base_chars[maj[keep]])
    return draft[: pos[0]] + mid + draft[pos[-1] + 1 :]


def infer(
    data_path: str,
    model_path: str,
    out_path: Optional[str],
    batch_size: int = C.BATCH_SIZE,
    workers: int = 0,
    device: Optional[torch.device] = None,
    model: Optional[RokoModel] = None,
    log=print,
) -> Dict[str, str]:
    """Polish the draft in `data_path` (RKW inference file) and return
    {contig: polished sequence}. Writes FASTA if `out_path` is given.

    Under torchrun (WORLD_SIZE > 1) the window groups are sharded across
    ranks round-robin and partial vote tables gathered to rank 0; only rank 0
    returns sequences / writes the FASTA.
    """
    rank, local_rank, world = init_distributed()
    if device is None:
        device = (
            torch.device("cuda", local_rank)
            if torch.cuda.is_available()
            else torch.device("cpu")
        )

    if model is None:
        model = RokoModel()
        model.load_reference_checkpoint(model_path)
    model = model.to(device).eval()

    rkw = RkwFile(data_path)
    my_groups = [gi for gi in range(len(rkw.groups)) if gi % world == rank]
    ds = InferenceDataset(data_path, groups=my_groups)
    dl = DataLoader(ds, batch_size=batch_size, num_workers=workers,
                    pin_memory=torch.cuda.is_available())

    # streaming per-contig vote accumulation (bounded memory)
    sv = StreamingVotes()

    def account(gis, js, preds):
        for k in range(len(gis)):
            g, pos_arr, _, _ = rkw.group_arrays(int(gis[k]))
            sv.add(g["contig"], np.asarray(pos_arr[js[k]]), preds[k])

    t0 = time.time()
    n_windows = 0
    meter = Meter("inference", rank=rank)
    pipe = None
    if device.type == "cuda":
        from .ops.forward import InferencePipeline

        pipe = InferencePipeline(model, batch_size, depth=48)
    with torch.no_grad():
        if pipe is not None:
            # pipelined: keep `depth` batches in flight; votes are harvested
            # one pipeline-depth behind submission
            pending: List[tuple] = []
            for gis, js, x in dl:
                if len(pending) >= pipe.depth:
                    tk, pgis, pjs = pending.pop(0)
                    account(pgis, pjs, tk().numpy())
                with trace_range("submit"):
                    ticket = pipe.submit(x, copy_out=True)
                pending.append((ticket, gis.numpy(), js.numpy()))
                n_windows += len(gis)
                meter.add(windows=len(gis), bases=len(gis) * C.WINDOW_STRIDE)
            for tk, pgis, pjs in pending:
                account(pgis, pjs, tk().numpy())
        else:
            for gis, js, x in dl:
                x = x.to(device, non_blocking=True)
                logits = model(x.long())
                preds = logits.argmax(dim=2).to(torch.uint8).cpu().numpy()
                n_windows += len(gis)
                meter.add(windows=len(gis), bases=len(gis) * C.WINDOW_STRIDE)
                account(gis.numpy(), js.numpy(), preds)
    meter.close()
    dt = time.time() - t0
    bases = n_windows * C.WINDOW_STRIDE
    log(
        f"rank {rank}: {n_windows} windows in {dt:.1f}s "
        f"({bases / max(dt, 1e-9):.0f} bases/s)"
    )

    votes: VoteTable = sv.finalize()

    if world > 1:
        gathered: List[Optional[VoteTable]] = [None] * world if rank == 0 else None
        dist.gather_object(votes, gathered, dst=0)
        if rank != 0:
            return {}
        merged: VoteTable = {}
        names = set()
        for t in gathered:
            names.update(t.keys())
        for name in names:
            merged[name] = merge_votes([t[name] for t in gathered if name in t])
        votes = merged

    out: Dict[str, str] = {}
    for name in rkw.contig_names():
        draft = rkw.contig_seq(name)
        if name in votes:
            out[name] = stitch_contig(draft, *votes[name])
        else:
            out[name] = draft
    if out_path:
        write_fasta(out_path, sorted(out.items()))
        log(f"wrote {len(out)} polished contigs to {out_path}")
    return out


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("data", help="inference .rkw file")
    p.add_argument("model", help=".pth checkpoint (reference format)")
    p.add_argument("out", help="output FASTA path")
    p.add_argument("--t", type=int, default=0, help="DataLoader workers")
    p.add_argument("--b", type=int, default=C.BATCH_SIZE, help="batch size")
    a = p.parse_args(argv)
    infer(a.data, a.model, a.out, batch_size=a.b, workers=a.t)


if __name__ == "__main__":
    main()
