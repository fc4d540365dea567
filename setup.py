"""In-tree build of the native extensions (gfx950 only).

  python setup.py build_ext --inplace

Steps:
  1. hipcc --offload-arch=gfx950 compiles every csrc/kernels/*.hip to .o
     (mtime-cached, parallel).  Device code never sees torch headers.
  2. Torch CppExtensions compile the pybind bindings (csrc/bind) and the
     native comm engine (csrc/comm, host-side HIP + RCCL) and link the
     kernel objects.  The resulting .so files land inside the package so
     the repo snapshot carries them to GPU boxes.
"""
import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

from setuptools import find_packages, setup

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

from torch.utils.cpp_extension import BuildExtension, CppExtension  # noqa: E402

ROOT = Path(__file__).parent.resolve()
KERNEL_DIR = ROOT / "csrc" / "kernels"
OBJ_DIR = ROOT / "build" / "hip_obj"
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
OFFLOAD_ARCH = os.environ.get("DDL_OFFLOAD_ARCH", "gfx950")

HIPCC_FLAGS = [
    f"--offload-arch={OFFLOAD_ARCH}",
    "-O3", "-std=c++17", "-fPIC", "-ffast-math",
    "-Wno-unused-result",
]


def compile_hip_objects() -> list:
    OBJ_DIR.mkdir(parents=True, exist_ok=True)
    srcs = sorted(KERNEL_DIR.glob("*.hip"))
    objs = []
    jobs = []
    hdrs = list(KERNEL_DIR.glob("*.h")) + list(KERNEL_DIR.glob("*.hpp"))
    hdr_mtime = max((h.stat().st_mtime for h in hdrs), default=0.0)
    for src in srcs:
        obj = OBJ_DIR / (src.stem + ".o")
        objs.append(str(obj))
        if obj.exists() and obj.stat().st_mtime > max(src.stat().st_mtime, hdr_mtime):
            continue
        jobs.append((src, obj))

    def run(job):
        src, obj = job
        cmd = [HIPCC, *HIPCC_FLAGS, "-c", str(src), "-o", str(obj)]
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            sys.stderr.write(r.stdout + r.stderr)
            raise RuntimeError(f"hipcc failed for {src}")
        return obj

    if jobs:
        with ThreadPoolExecutor(max_workers=min(16, len(jobs))) as ex:
            list(ex.map(run, jobs))
        print(f"hipcc: compiled {len(jobs)} kernel objects -> {OBJ_DIR}")
    return objs


kernel_objs = compile_hip_objects()

common = dict(
    include_dirs=["/opt/rocm/include", str(ROOT / "csrc")],
    library_dirs=["/opt/rocm/lib"],
    extra_compile_args=["-D__HIP_PLATFORM_AMD__", "-O2", "-std=c++17"],
)

ext_modules = [
    CppExtension(
        "distributeddeeplearning_amd._C_kernels",
        ["csrc/bind/kernels_binding.cpp"],
        extra_objects=kernel_objs,
        libraries=["amdhip64"],
        **common,
    ),
    CppExtension(
        "distributeddeeplearning_amd._C_comm",
        ["csrc/comm/rccl_engine.cpp", "csrc/bind/comm_binding.cpp"],
        libraries=["amdhip64", "rccl"],
        **common,
    ),
]

setup(
    name="distributeddeeplearning_amd",
    version="0.1.0",
    packages=find_packages(include=["distributeddeeplearning_amd",
                                    "distributeddeeplearning_amd.*"]),
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension},
)
