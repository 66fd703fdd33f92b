"""Build the _hipstore native extension in-tree for gfx950.

Deliberately drives hipcc directly (no hipify, no torch C++ ABI coupling —
the extension is pybind11 + raw HIP).  Usage:

    python setup.py build_ext --inplace

Produces torchstore_amd/_hipstore.cpython-*.so next to the package sources
so the .so travels with repo snapshots.
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

from setuptools import setup, find_packages
from setuptools.command.build_ext import build_ext
from setuptools.extension import Extension

ROOT = Path(__file__).parent.resolve()
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


class HipExtension(Extension):
    def __init__(self, name, sources):
        super().__init__(name, sources=sources)


class hipcc_build_ext(build_ext):
    def build_extension(self, ext):
        if not isinstance(ext, HipExtension):
            return super().build_extension(ext)
        import pybind11

        out = Path(self.get_ext_fullpath(ext.name))
        out.parent.mkdir(parents=True, exist_ok=True)
        cmd = [
            HIPCC,
            f"--offload-arch={ARCH}",
            "-O3",
            "-std=c++17",
            "-fPIC",
            "-shared",
            "-fvisibility=hidden",
            f"-I{pybind11.get_include()}",
            f"-I{sysconfig.get_path('include')}",
            *[str(ROOT / s) for s in ext.sources],
            "-o",
            str(out),
        ]
        print("+", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)
        # also drop a copy in the source tree for --inplace-style imports
        inplace = ROOT / "torchstore_amd" / out.name
        if self.inplace and out.resolve() != inplace.resolve():
            import shutil

            shutil.copy2(out, inplace)


setup(
    name="torchstore_amd",
    version="0.1.0",
    packages=find_packages(include=["torchstore_amd", "torchstore_amd.*"]),
    ext_modules=[
        HipExtension(
            "torchstore_amd._hipstore",
            sources=["torchstore_amd/csrc/hipstore.hip"],
        )
    ],
    cmdclass={"build_ext": hipcc_build_ext},
)
