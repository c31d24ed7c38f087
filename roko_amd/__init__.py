"""roko-mi355x: MI355X-native nanopore consensus polisher.

Pipeline (mirrors the reference's capability surface):

  features  — draft FASTA + reads BAM -> RKW window files (C++ pileup core)
  train     — RKW (+labels) -> .pth checkpoint (HIP kernels, RCCL DP)
  inference — RKW + .pth -> polished FASTA (HIP kernels, contig sharding)
"""

__version__ = "0.1.0"

from . import config  # noqa: F401
