"""Accuracy benchmark: the in-repo analogue of the reference's published
error table (reference README: draft vs polished error, assessed with
pomoxis). Sweeps read error rate x coverage on the synthetic scenario
(tests/simple_align.py — truth known exactly), trains briefly per cell,
polishes, and prints a markdown table of draft vs polished error.

CPU-only; ~1-2 min per cell. Usage: python scripts/accuracy_bench.py
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from roko_amd import features as F
from roko_amd.accuracy import assess_polishing
from roko_amd.config import TrainConfig
from roko_amd.inference import infer
from roko_amd.train import train
from tests.simple_align import build_assembly


def run_cell(read_err, cov, seed=0):
    rng = np.random.default_rng(seed)
    with tempfile.TemporaryDirectory() as td:
        asm = build_assembly(rng, os.path.join(td, "asm"), length=3000,
                             cov=cov, read_err=read_err)
        train_rkw = os.path.join(td, "train.rkw")
        F.run(asm["draft_fasta"], asm["reads_bam"], train_rkw,
              bam_y=asm["truth_bam"], workers=1,
              cfg=F.FeatureConfig(region_size=2000, region_overlap=300),
              log=lambda *a: None)
        infer_rkw = os.path.join(td, "infer.rkw")
        F.run(asm["draft_fasta"], asm["reads_bam"], infer_rkw, workers=1,
              cfg=F.FeatureConfig(region_size=2000, region_overlap=300),
              log=lambda *a: None)
        cfg = TrainConfig(batch_size=16, epochs=50, lr=2e-3, in_memory=True,
                          seed=0)
        model, _ = train(train_rkw, os.path.join(td, "out"), cfg=cfg,
                         log=lambda *a: None, max_steps=200)
        ckpt = os.path.join(td, "m.pth")
        torch.save(model.state_dict(), ckpt)
        seqs = infer(infer_rkw, ckpt, None, batch_size=32,
                     log=lambda *a: None)
        return assess_polishing(asm["draft"], seqs["ctg1"], asm["truth"])


def main():
    torch.manual_seed(0)
    print("| read err | coverage | draft err | polished err | reduction |")
    print("|---|---|---|---|---|")
    for read_err in (0.0, 0.03, 0.05, 0.10):
        for cov in (10, 20, 40):
            res = run_cell(read_err, cov)
            print(f"| {read_err:.0%} | {cov}x "
                  f"| {res['draft']['total_error']:.3%} "
                  f"| {res['polished']['total_error']:.3%} "
                  f"| {res['error_reduction']:.1%} |", flush=True)


if __name__ == "__main__":
    main()
