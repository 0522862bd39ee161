"""Build script for the MI355X-native decision-intelligence engine.

Two native artifacts:
  * ``ding/ops/_hiprl`` -- HIP/CDNA4 fused RL kernels (gfx950 only, built via
    hipcc through torch.utils.cpp_extension; cross-compiles on CPU-only hosts).
  * ``ding/utils/_ctree`` -- C++ sum/min segment tree used by prioritized
    replay on the host side (replaces the reference's numba JIT,
    ding/utils/segment_tree.py in opendilab/DI-engine).

Use ``python setup.py build_ext --inplace`` so the .so files live in-tree and
travel with repo snapshots.
"""
import os
import sys

from setuptools import setup, find_packages

ROOT = os.path.dirname(os.path.abspath(__file__))


def _ext_modules():
    # C++ (host) extension: segment tree. Always buildable.
    from pybind11.setup_helpers import Pybind11Extension
    exts = [
        Pybind11Extension(
            "ding.utils._ctree",
            ["ding/utils/csrc/ctree.cpp"],
            cxx_std=17,
            extra_compile_args=["-O3"],
        )
    ]
    return exts


setup(
    name="ding-mi355x",
    version="0.1.0",
    description="MI355X-native decision intelligence engine (DI-engine capability parity)",
    packages=find_packages(include=["ding", "ding.*", "dizoo", "dizoo.*"]),
    python_requires=">=3.8",
    ext_modules=_ext_modules(),
    entry_points={
        "console_scripts": [
            "ding=ding.entry.cli:cli",
            "ditask=ding.entry.cli_ditask:cli_ditask",
        ]
    },
)
