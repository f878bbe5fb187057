"""In-tree build of the mdi_llm_amd HIP extension for gfx950 (MI355X).

Two-step build, fully controlled (no hipify, no JIT cache):
  1. hipcc compiles the .hip kernel TUs with --offload-arch=gfx950
     (device code; no torch headers needed).
  2. the host bindings TU is built as a plain C++ torch extension and
     linked with the kernel objects + libamdhip64.

Usage:  python setup.py build_ext --inplace
The resulting mdi_llm_amd/_hip_ops*.so ships with the repo snapshot to the
GPU box (it is git-ignored but NOT gpurun-ignored).
"""

import os
import subprocess
import sys
from pathlib import Path

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension
import torch

ROOT = Path(__file__).resolve().parent
HIP_DIR = ROOT / "mdi_llm_amd" / "ops" / "hip"
BUILD_DIR = ROOT / "build" / "hip_objs"

ROCM_HOME = os.environ.get("ROCM_HOME", os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = os.path.join(ROCM_HOME, "bin", "hipcc")
OFFLOAD_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950").split(";")[0]

KERNEL_SOURCES = ["decode_kernels.hip"]


def compile_hip_objects() -> list:
    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    objs = []
    for src in KERNEL_SOURCES:
        src_path = HIP_DIR / src
        obj_path = BUILD_DIR / (src_path.stem + ".o")
        if (
            not obj_path.exists()
            or obj_path.stat().st_mtime < src_path.stat().st_mtime
            or obj_path.stat().st_mtime < (HIP_DIR / "decode_kernels.h").stat().st_mtime
        ):
            cmd = [
                HIPCC,
                f"--offload-arch={OFFLOAD_ARCH}",
                "-O3",
                "-std=c++17",
                "-fPIC",
                "-c",
                str(src_path),
                "-o",
                str(obj_path),
            ]
            print("[hipcc]", " ".join(cmd), flush=True)
            subprocess.check_call(cmd)
        objs.append(str(obj_path))
    return objs


def torch_lib_dir() -> str:
    return str(Path(torch.__file__).parent / "lib")


ext = CppExtension(
    name="mdi_llm_amd._hip_ops",
    sources=[str(HIP_DIR / "bindings.cpp")],
    extra_objects=compile_hip_objects(),
    include_dirs=[os.path.join(ROCM_HOME, "include")],
    library_dirs=[os.path.join(ROCM_HOME, "lib"), torch_lib_dir()],
    libraries=["amdhip64", "c10_hip", "torch_hip"],
    extra_compile_args=["-D__HIP_PLATFORM_AMD__=1", "-O2"],
)

setup(
    name="mdi_llm_amd_ops",
    version="0.1.0",
    ext_modules=[ext],
    cmdclass={"build_ext": BuildExtension.with_options(no_python_abi_suffix=False)},
)
