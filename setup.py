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


def build_hip_extension(force: bool = False) -> str | None:
    """Compile the HIP extension in-tree with hipcc for gfx950.

    Returns the path of the built .so, or None when hipcc is unavailable.
    Deliberately bypasses torch's JIT extension cache: the .so must live
    in-tree so it travels with the repo snapshot to the GPU box.
    """
    hipcc = shutil.which("hipcc") or "/opt/rocm/bin/hipcc"
    if not os.path.exists(hipcc):
        return None
    import sysconfig
    ext_suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    src = os.path.join(ROOT, "csrc", "hip", "sgns_hip.hip")
    out = os.path.join(PKG, "_hip_native" + ext_suffix)
    if not force and os.path.exists(out) and \
            os.path.getmtime(out) > os.path.getmtime(src):
        return out
    cmd = [
        hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared", "-fvisibility=hidden",
        "-I", pybind11.get_include(),
        "-I", sysconfig.get_paths()["include"],
        "-x", "hip", src,
        "-o", out,
    ]
    print("building HIP extension:", " ".join(cmd))
    subprocess.check_call(cmd, cwd=ROOT)
    return out


class BuildExt(build_ext):
    def run(self):
        super().run()
        build_hip_extension()


setup(
    name="glint_word2vec_amd",
    version="0.1.0",
    packages=["glint_word2vec_amd", "glint_word2vec_amd.models",
              "glint_word2vec_amd.ops", "glint_word2vec_amd.parallel",
              "glint_word2vec_amd.utils"],
    ext_modules=[cpu_ext],
    cmdclass={"build_ext": BuildExt},
)
