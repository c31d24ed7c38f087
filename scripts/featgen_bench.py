"""Feature-generation throughput: synthetic assembly -> windows/s.

The reference's practical wall time is dominated by this stage (pileup
walk + window emission over htslib); here it is the own C++ BAM/pileup
path. CPU-only. Usage: python scripts/featgen_bench.py [genome_kb] [cov]
"""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

from roko_amd import features as F
from roko_amd.rkdata import RkwFile
from tests.simple_align import build_assembly


def main():
    kb = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    cov = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    rng = np.random.default_rng(0)
    with tempfile.TemporaryDirectory() as td:
        t0 = time.perf_counter()
        asm = build_assembly(rng, os.path.join(td, "asm"), length=kb * 1000,
                             cov=cov, read_len=3000)
        t_asm = time.perf_counter() - t0
        bam_mb = os.path.getsize(asm["reads_bam"]) / 1e6
        for workers in (1, 4, 8):
            out = os.path.join(td, f"w{workers}.rkw")
            t0 = time.perf_counter()
            F.run(asm["draft_fasta"], asm["reads_bam"], out, workers=workers,
                  log=lambda *a: None)
            dt = time.perf_counter() - t0
            f = RkwFile(out)
            n = sum(g["size"] for g in f.groups)
            print(f"workers={workers}: {n} windows in {dt:.2f}s "
                  f"({n / dt:.0f} windows/s, {bam_mb / dt:.1f} MB BAM/s, "
                  f"{kb / dt:.0f} kb draft/s)")
        print(f"(scenario build {t_asm:.1f}s, BAM {bam_mb:.1f} MB, {cov}x)")


if __name__ == "__main__":
    main()
