"""In-tree build of the CDNA4 HIP kernel extension (fms_fsdp_amd._C).

The .hip kernel sources are compiled DIRECTLY with hipcc for gfx950 (no
hipify pass — they are native HIP/CDNA4 code) and linked into a plain
C++ pybind extension. `PYTORCH_ROCM_ARCH=gfx950 python setup.py
build_ext --inplace` puts the .so inside fms_fsdp_amd/ so it travels with
the repo snapshot to the GPU box.
"""

import os
import subprocess

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
ROCM = os.environ.get("ROCM_HOME", "/opt/rocm")
REPO = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(REPO, "fms_fsdp_amd", "ops", "hip")
BUILD_DIR = os.path.join(REPO, "build", "hip_objs")

HIP_SOURCES = ["rmsnorm.hip", "rope.hip", "swiglu.hip", "cross_entropy.hip",
               "adamw.hip", "attention.hip", "causal_conv1d.hip", "ssd.hip",
               "gemm_nt.hip"]


def compile_hip_objects():
    os.makedirs(BUILD_DIR, exist_ok=True)
    objs = []
    for src in HIP_SOURCES:
        src_path = os.path.join(HIP_DIR, src)
        obj_path = os.path.join(BUILD_DIR, src.replace(".hip", ".o"))
        if (not os.path.exists(obj_path)
                or os.path.getmtime(obj_path) < os.path.getmtime(src_path)
                or os.path.getmtime(obj_path) < os.path.getmtime(
                    os.path.join(HIP_DIR, "common.h"))):
            cmd = [os.path.join(ROCM, "bin", "hipcc"),
                   f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
                   "-c", src_path, "-o", obj_path]
            print(" ".join(cmd))
            subprocess.check_call(cmd)
        objs.append(obj_path)
    return objs


class HipBuildExt(BuildExtension):
    def build_extensions(self):
        objs = compile_hip_objects()
        for ext in self.extensions:
            ext.extra_objects = objs + list(ext.extra_objects or [])
        super().build_extensions()


setup(
    name="fms_fsdp_amd",
    packages=["fms_fsdp_amd"],
    ext_modules=[
        CppExtension(
            name="fms_fsdp_amd._C",
            sources=[os.path.join("fms_fsdp_amd", "ops", "hip", "bindings.cpp")],
            extra_compile_args=["-O3", "-std=c++17",
                                "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1"],
            include_dirs=[os.path.join(ROCM, "include")],
            library_dirs=[os.path.join(ROCM, "lib")],
            libraries=["amdhip64", "c10_hip", "torch_hip"],
        )
    ],
    cmdclass={"build_ext": HipBuildExt},
)
