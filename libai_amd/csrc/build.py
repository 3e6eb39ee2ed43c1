#!/usr/bin/env python3
"""In-tree build of the libai_amd HIP extension for gfx950.

Two-stage build (kept out of torch's JIT cache so the .so travels with the
repo snapshot to GPU boxes):
  1. hipcc --offload-arch=gfx950 compiles each csrc/kernels/*.hip (pure HIP,
     no torch headers -> seconds per file)
  2. g++ compiles csrc/bindings.cpp against the torch/pybind11 headers
  3. g++ links everything into libai_amd/_C.so

Usage: python libai_amd/csrc/build.py [--force]
Incremental: skips compiles when the object is newer than its sources.
"""

import os
import subprocess
import sys
import sysconfig

CSRC = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.dirname(CSRC)
KERNELS = os.path.join(CSRC, "kernels")
BUILD = os.path.join(CSRC, "_build")
OUT_SO = os.path.join(PKG, "_C.so")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _newer(target, *sources):
    if not os.path.exists(target):
        return False
    t = os.path.getmtime(target)
    return all(os.path.getmtime(s) <= t for s in sources if os.path.exists(s))


def _run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build(force=False, verbose=True):
    import torch.utils.cpp_extension as cpp_ext

    os.makedirs(BUILD, exist_ok=True)
    common_h = os.path.join(KERNELS, "common.h")
    objs = []

    for fname in sorted(os.listdir(KERNELS)):
        if not fname.endswith(".hip"):
            continue
        src = os.path.join(KERNELS, fname)
        obj = os.path.join(BUILD, fname[:-4] + ".o")
        objs.append(obj)
        if not force and _newer(obj, src, common_h):
            continue
        _run(
            [
                HIPCC,
                f"--offload-arch={ARCH}",
                "-O3",
                "-std=c++17",
                "-fPIC",
            ]
            + os.environ.get("EXTRA_HIP_FLAGS", "").split()
            + [
                "-c",
                src,
                "-o",
                obj,
            ]
        )

    torch_includes = cpp_ext.include_paths()
    py_include = sysconfig.get_paths()["include"]
    for cpp_name in ("bindings.cpp", "gemm_lt.cpp"):
        cpp_src = os.path.join(CSRC, cpp_name)
        cpp_obj = os.path.join(BUILD, cpp_name[:-4] + ".o")
        if force or not _newer(cpp_obj, cpp_src):
            cmd = [
                "g++",
                "-O2",
                "-std=c++17",
                "-fPIC",
                "-D__HIP_PLATFORM_AMD__=1",
                "-DUSE_ROCM=1",
                "-DTORCH_EXTENSION_NAME=_C",
                "-DTORCH_API_INCLUDE_EXTENSION_H",
            ]
            for p in torch_includes:
                cmd += ["-isystem", p]
            cmd += ["-isystem", "/opt/rocm/include", "-isystem", py_include]
            cmd += ["-c", cpp_src, "-o", cpp_obj]
            _run(cmd)
        objs.append(cpp_obj)

    # standalone pybind11 data-helpers extension (no torch dependency)
    dh_src = os.path.join(CSRC, "data_helpers.cpp")
    dh_so = os.path.join(PKG, "_data_helpers.so")
    if os.path.exists(dh_src) and (force or not _newer(dh_so, dh_src)):
        import pybind11

        ext_suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
        _run(
            [
                "g++", "-O3", "-std=c++17", "-fPIC", "-shared",
                "-isystem", pybind11.get_include(),
                "-isystem", py_include,
                dh_src, "-o", dh_so,
            ]
        )

    torch_lib = cpp_ext.library_paths()[0]
    if force or not _newer(OUT_SO, *objs):
        _run(
            ["g++", "-shared"]
            + objs
            + [
                "-L" + torch_lib,
                "-lc10",
                "-lc10_hip",
                "-ltorch_cpu",
                "-ltorch_hip",
                "-ltorch",
                "-ltorch_python",
                "-L/opt/rocm/lib",
                "-lamdhip64",
                "-lhipblaslt",
                "-o",
                OUT_SO,
            ]
        )
    print(f"built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
