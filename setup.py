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

RCCL_SOURCES = [
    "csrc/rccl/rccl_layer.cpp",
    "csrc/rccl/module_rccl.cpp",
]

HIP_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _hipcc_ext(name, sources, out_dir, extra_link=()):
    """Compile one hipcc extension in-tree (gfx950 only)."""
    import sysconfig

    py_inc = sysconfig.get_paths()["include"]
    pb_inc = pybind11.get_include()
    out = os.path.join(out_dir, name + sysconfig.get_config_var("EXT_SUFFIX"))
    objs = []
    os.makedirs(os.path.join(ROOT, "build", name), exist_ok=True)
    for src in sources:
        obj = os.path.join(
            ROOT, "build", name,
            os.path.basename(src).replace(".", "_") + ".o")
        src_path = os.path.join(ROOT, src)
        import glob

        deps = [src_path] + glob.glob(os.path.join(ROOT, "csrc", "**",
                                                   "*.h*"), recursive=True)
        if (os.path.exists(obj) and
                os.path.getmtime(obj) > max(map(os.path.getmtime, deps))):
            objs.append(obj)
            continue
        cmd = [
            HIPCC, "-c", src_path, "-o", obj,
            f"--offload-arch={HIP_ARCH}", "-O3", "-std=c++17", "-fPIC",
            f"-I{py_inc}", f"-I{pb_inc}", "-I/opt/rocm/include",
        ]
        print("+", " ".join(cmd))
        subprocess.check_call(cmd)
        objs.append(obj)
    link = [HIPCC, "-shared", "-fPIC", "-o", out] + objs + [
        f"--offload-arch={HIP_ARCH}"
    ] + list(extra_link)
    print("+", " ".join(link))
    subprocess.check_call(link)
    return out


def build_hip_ext(out_dir):
    return _hipcc_ext("_hip", HIP_SOURCES, out_dir)


def build_rccl_ext(out_dir):
    return _hipcc_ext("_rccl", RCCL_SOURCES, out_dir,
                      extra_link=["-L/opt/rocm/lib", "-lrccl",
                                  "-Wl,-rpath,/opt/rocm/lib"])


class BuildExt(build_ext):
    def run(self):
        super().run()
        if os.environ.get("KUNGFU_SKIP_HIP", "0") != "1":
            if os.path.exists(HIPCC):
                dest = os.path.join(ROOT, "kungfu_amd")
                build_hip_ext(dest)
                build_rccl_ext(dest)
            else:
                print("hipcc not found; skipping kungfu_amd._hip/_rccl")


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
