"""Build roko_amd.ops._hip_ops in-tree for gfx950.

Invoked by setup.py (and __graft_entry__.build()). Uses torch's cpp_extension
so all torch/ABI flags are correct, then places the .so next to this file so
the gpurun snapshot carries it.

Cross-compiles fine on GPU-less hosts (PYTORCH_ROCM_ARCH=gfx950).
"""

import glob
import os
import shutil

HERE = os.path.dirname(os.path.abspath(__file__))


def build():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    build_dir = os.path.join(HERE, "_build")
    os.makedirs(build_dir, exist_ok=True)
    names = ["bindings.cpp", "adam.hip", "ce.hip", "embed_mlp.hip", "embgrad.hip", "front_train.hip", "gemm.hip",
             "gru.hip", "head.hip", "probe.hip"]
    sources = [os.path.join(HERE, "hip", n) for n in names]
    load(
        name="_hip_ops",
        sources=sources,
        build_directory=build_dir,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=True,
        is_python_module=True,
        keep_intermediates=True,
    )
    so = os.path.join(build_dir, "_hip_ops.so")
    if not os.path.exists(so):
        cands = glob.glob(os.path.join(build_dir, "_hip_ops*.so"))
        if not cands:
            raise RuntimeError("HIP extension build produced no .so")
        so = cands[0]
    dst = os.path.join(HERE, "_hip_ops.so")
    shutil.copy2(so, dst)
    print(f"built {dst}")
    return dst


if __name__ == "__main__":
    build()
