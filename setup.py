"""Build for moolib_amd native extensions.

Two in-tree extensions:
  moolib_amd._core     — C++ runtime (RPC, services, accumulator, batcher, envpool)
  moolib_amd._kernels  — HIP/CDNA4 (gfx950) kernels; cross-compiled, loadable
                         only on a ROCm machine.

Usage: python setup.py build_ext --inplace
"""
import os
from pathlib import Path

from setuptools import setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils import cpp_extension

ROOT = Path(__file__).resolve().parent

CORE_SOURCES = sorted(str(p) for p in (ROOT / "csrc").glob("*.cc"))
HIP_SOURCES = sorted(str(p) for p in (ROOT / "hip").glob("*.hip"))

# The hipcc compile step emits no dependency file, so ninja cannot see the
# *.hip.inc includes — an edit there would silently ship a STALE kernels.so
# (bit us in round 2). Bump the including .hip's mtime whenever an include
# is newer.
for hip in HIP_SOURCES:
    hp = Path(hip)
    for inc in (ROOT / "hip").glob("*.hip.inc"):
        if inc.stat().st_mtime > hp.stat().st_mtime:
            hp.touch()
            break

common_args = [
    "-O2",
    "-std=c++17",
    "-fvisibility=hidden",
    "-Wno-unused-parameter",
]

ext_modules = [
    cpp_extension.CppExtension(
        name="moolib_amd._core",
        sources=CORE_SOURCES,
        extra_compile_args=common_args,
    )
]

if HIP_SOURCES:
    ext_modules.append(
        cpp_extension.CUDAExtension(  # drives hipcc under ROCm
            name="moolib_amd._kernels",
            sources=HIP_SOURCES,
            extra_compile_args={
                "cxx": common_args,
                "nvcc": ["-O3", "-std=c++17", "--offload-arch=gfx950"],
            },
        )
    )

setup(
    name="moolib_amd",
    version="0.1.0",
    packages=["moolib_amd"],
    ext_modules=ext_modules,
    cmdclass={"build_ext": cpp_extension.BuildExtension.with_options(use_ninja=True)},
)
