"""Build the in-tree gfx950 HIP extension (accelerate_amd._C).

hipcc is driven directly (no hipify, no CUDA shims): every source under
accelerate_amd/ops/csrc is native HIP written for CDNA4. Cross-compiles on
GPU-less boxes; the resulting .so is committed-adjacent (git-ignored) and
travels with repo snapshots.
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

from setuptools import setup
from setuptools.command.build_ext import build_ext
from setuptools.extension import Extension

ROOT = Path(__file__).parent.resolve()
CSRC = ROOT / "accelerate_amd" / "ops" / "csrc"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch
    import torch.utils.cpp_extension as ce

    return ce.include_paths(), [str(Path(torch.__file__).parent / "lib")]


class HipBuildExt(build_ext):
    def build_extension(self, ext):
        import pybind11

        includes, libdirs = torch_paths()
        sources = sorted(str(p) for p in CSRC.glob("*.hip"))
        out = Path(self.get_ext_fullpath(ext.name))
        out.parent.mkdir(parents=True, exist_ok=True)
        hipcc = os.environ.get("HIPCC", "hipcc")
        cmd = [
            hipcc,
            f"--offload-arch={ARCH}",
            "-O3",
            "-std=c++17",
            "-fPIC",
            "-shared",
            "-DTORCH_EXTENSION_NAME=_C",
            "-DUSE_ROCM",
            "-DTORCH_API_INCLUDE_EXTENSION_H",
            f"-I{CSRC}",
            f"-I{pybind11.get_include()}",
            f"-I{sysconfig.get_paths()['include']}",
        ]
        cmd += [f"-I{inc}" for inc in includes]
        cmd += sources
        cmd += ["-o", str(out)]
        cmd += [f"-L{d}" for d in libdirs]
        cmd += ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10", "-lc10_hip", "-ltorch_hip", "-lamdhip64"]
        print(" ".join(cmd), flush=True)
        subprocess.check_call(cmd)


setup(
    name="accelerate_amd",
    version="0.1.0",
    description="MI355X-native training-loop framework (Accelerate-equivalent)",
    packages=["accelerate_amd"],
    ext_modules=[Extension("accelerate_amd._C", sources=[])],
    cmdclass={"build_ext": HipBuildExt},
)
