"""Build the HIP kernel extension in-tree: ding/ops/_hiprl*.so (gfx950).

Run: PYTORCH_ROCM_ARCH=gfx950 python setup_ops.py
(also driven from __graft_entry__.build()). hipcc cross-compiles on
CPU-only hosts; the .so travels with repo snapshots to the GPU box.
"""
import glob
import os
import shutil
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.environ.setdefault("MAX_JOBS", "8")

from torch.utils.cpp_extension import BuildExtension, CUDAExtension
from setuptools import setup

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "ding", "ops", "csrc")


def main():
    sources = [os.path.join(CSRC, "hiprl.cpp")] + sorted(glob.glob(os.path.join(CSRC, "*.hip")))
    ext = CUDAExtension(
        name="ding.ops._hiprl",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3"],
            "nvcc": ["-O3", "--offload-arch=gfx950", "-std=c++17"],
        },
    )
    sys.argv = [sys.argv[0], "build_ext", "--inplace"]
    setup(
        name="ding-hiprl",
        ext_modules=[ext],
        cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
        script_args=["build_ext", "--inplace"],
    )


if __name__ == "__main__":
    main()
