"""In-tree extension build for roko-mi355x.

  python setup.py build_ext --inplace

builds:
  * roko_amd.ops._pileup   — C++ data path (BGZF/BAM/BAI + window builder)
  * roko_amd.ops._hip_ops  — CDNA4 HIP kernels (gfx950), built whenever hipcc
    is available (cross-compiles fine on GPU-less hosts)

Built .so files live in-tree so that the gpurun snapshot carries them.
"""

import os
import shutil
import subprocess
import sys

from setuptools import setup
from pybind11.setup_helpers import Pybind11Extension, build_ext

ROOT = os.path.dirname(os.path.abspath(__file__))

# Sanitizer build mode (SURVEY.md §5.2 — the reference ships none):
#   ROKO_SANITIZE=address|undefined|thread python setup.py build_ext --inplace
# builds the C++ data path instrumented; run the data-path tests with
# LD_PRELOAD=$(gcc -print-file-name=libasan.so) (see scripts/sanitize.sh).
_SAN = os.environ.get("ROKO_SANITIZE", "")
_san_args = ([f"-fsanitize={_SAN}", "-fno-omit-frame-pointer", "-g", "-O1"]
             if _SAN else [])

ext_modules = [
    Pybind11Extension(
        "roko_amd.ops._pileup",
        ["roko_amd/ops/cpp/bam.cpp", "roko_amd/ops/cpp/pileup.cpp",
         "roko_amd/ops/cpp/align.cpp", "roko_amd/ops/cpp/module.cpp"],
        cxx_std=17,
        libraries=["z"],
        extra_compile_args=(["-O3", "-Wall"] + _san_args),
        extra_link_args=_san_args,
    ),
]


class BuildExt(build_ext):
    def run(self):
        super().run()
        if shutil.which("hipcc") and not os.environ.get("SKIP_HIP"):
            build_hip()


def build_hip():
    """Build the HIP kernel extension via torch's cpp_extension + hipcc."""
    script = os.path.join(ROOT, "roko_amd", "ops", "build_hip.py")
    if os.path.exists(script):
        subprocess.check_call([sys.executable, script])


setup(
    name="roko-mi355x",
    version="0.1.0",
    packages=["roko_amd", "roko_amd.io", "roko_amd.ops", "roko_amd.parallel",
              "roko_amd.utils"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExt},
)
