"""roko-mi355x: MI355X-native nanopore consensus polisher.

Pipeline (mirrors the reference's capability surface):

  features  — draft FASTA + reads BAM -> RKW window files (C++ pileup core)
  train     — RKW (+labels) -> .pth checkpoint (HIP kernels, RCCL DP)
  inference — RKW + .pth -> polished FASTA (HIP kernels, contig sharding)
"""

__version__ = "0.1.0"

import os as _os

# Pipelined serving runs ~dozens of HIP streams; ROCm's default of 4 hardware
# queues serializes them (measured 2.4 concurrent kernels at depth 32 —
# profiles/PERF_HISTORY.md). 20 queues is the measured sweet spot on MI355X;
# beyond ~24 the HWS oversubscribes and throughput drops. Must be set before
# the HIP runtime initializes, hence here (and in bench.py before torch).
_os.environ.setdefault("GPU_MAX_HW_QUEUES", "20")

from . import config  # noqa: F401
