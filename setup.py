"""Build the KungFu-AMD native extensions in-tree.

Two extensions:
  kungfu_amd._core — C++17 control plane + CPU collective engine (g++).
  kungfu_amd._hip  — hand-written gfx950 HIP kernels (hipcc, pybind11 host
                     bindings; tensors are passed as raw device pointers and
                     HIP streams as integers, so no torch headers and no
                     hipify are involved).

Usage: python setup.py build_ext --inplace
"""
import os
import subprocess
import sys

from setuptools import setup, Extension
from setuptools.command.build_ext import build_ext

import pybind11

ROOT = os.path.dirname(os.path.abspath(__file__))

CORE_SOURCES = [
    "csrc/core/reduce.cpp",
    "csrc/core/graph.cpp",
    "csrc/core/plan.cpp",
    "csrc/net/transport.cpp",
    "csrc/net/endpoints.cpp",
    "csrc/session/session.cpp",
    "csrc/peer/peer.cpp",
    "csrc/peer/aux.cpp",
    "csrc/pybind/module.cpp",
]

HIP_SOURCES = [
    "csrc/hip/kernels.hip",
    "csrc/hip/fused_bn.hip",
    "csrc/hip/fused_ln.hip",
    "csrc/hip/module_hip.cpp",
]

HIP_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def build_hip_ext(out_dir):
    """Compile the HIP extension with hipcc directly (gfx950 only)."""
    import sysconfig

    py_inc = sysconfig.get_paths()["include"]
    pb_inc = pybind11.get_include()
    out = os.path.join(out_dir, "_hip" + sysconfig.get_config_var("EXT_SUFFIX"))
    objs = []
    os.makedirs(os.path.join(ROOT, "build", "hip"), exist_ok=True)
    for src in HIP_SOURCES:
        obj = os.path.join(
            ROOT, "build", "hip",
            os.path.basename(src).replace(".", "_") + ".o")
        cmd = [
            HIPCC, "-c", os.path.join(ROOT, src), "-o", obj,
            f"--offload-arch={HIP_ARCH}", "-O3", "-std=c++17", "-fPIC",
            f"-I{py_inc}", f"-I{pb_inc}",
        ]
        print("+", " ".join(cmd))
        subprocess.check_call(cmd)
        objs.append(obj)
    link = [HIPCC, "-shared", "-fPIC", "-o", out] + objs + [
        f"--offload-arch={HIP_ARCH}"
    ]
    print("+", " ".join(link))
    subprocess.check_call(link)
    return out


class BuildExt(build_ext):
    def run(self):
        super().run()
        if os.environ.get("KUNGFU_SKIP_HIP", "0") != "1":
            if os.path.exists(HIPCC):
                dest = os.path.join(ROOT, "kungfu_amd")
                build_hip_ext(dest)
            else:
                print("hipcc not found; skipping kungfu_amd._hip")


core_ext = Extension(
    "kungfu_amd._core",
    sources=CORE_SOURCES,
    include_dirs=[pybind11.get_include()],
    language="c++",
    extra_compile_args=["-O3", "-std=c++17", "-pthread", "-fvisibility=hidden"],
    extra_link_args=["-pthread"],
)

setup(
    name="kungfu_amd",
    version="0.1.0",
    description="MI355X-native adaptive distributed training runtime",
    packages=[
        "kungfu_amd", "kungfu_amd.ops", "kungfu_amd.optimizers",
        "kungfu_amd.parallel", "kungfu_amd.models", "kungfu_amd.utils",
        "kungfu_amd.launcher", "kungfu_amd.cmd", "kungfu_amd.benchmarks",
    ],
    ext_modules=[core_ext],
    cmdclass={"build_ext": BuildExt},
)
