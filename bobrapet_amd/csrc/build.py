"""In-tree build of the gfx950 HIP extension.

Builds bobrapet_amd/_hipops.so with explicit hipcc invocations (no hipify,
no CUDA shims — the sources are native HIP/CDNA4).  The .so lands inside
the package so it travels to GPU boxes with the repo snapshot.

Usage: python -m bobrapet_amd.csrc.build [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC_DIR = os.path.join(PKG_DIR, "csrc", "hip")
OUT_SO = os.path.join(PKG_DIR, "_hipops.so")
CORE_SO = os.path.join(PKG_DIR, "_core.so")
CORE_DIR = os.path.join(PKG_DIR, "csrc", "core")
BUILD_DIR = os.path.join(PKG_DIR, "csrc", "build")

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = ["elementwise.hip", "attention.hip", "gemm.hip", "gemm256.hip", "gemm256b.hip", "gemm256w.hip", "gemmsk.hip", "debug.hip"]
BINDING_SOURCE = "bindings.cpp"


def _torch_flags():
    import torch
    import torch.utils.cpp_extension as ce

    includes = [f"-I{p}" for p in ce.include_paths()]
    includes.append(f"-I{sysconfig.get_paths()['include']}")
    libs = [f"-L{p}" for p in ce.library_paths()]
    libs += [f"-Wl,-rpath,{p}" for p in ce.library_paths()]
    libs += ["-ltorch", "-ltorch_python", "-lc10", "-lc10_hip", "-ltorch_hip"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    defines = [
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_hipops",
        "-DUSE_ROCM",
        "-D__HIP_PLATFORM_AMD__",
    ]
    return includes, libs, defines


def _newer(path: str, than: str) -> bool:
    if not os.path.exists(than):
        return True
    return os.path.getmtime(path) > os.path.getmtime(than)


def _run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build(force: bool = False, verbose: bool = True) -> str:
    os.makedirs(BUILD_DIR, exist_ok=True)
    includes, libs, defines = _torch_flags()
    common = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-Wno-unused-result",
    ]
    objs = []
    header = os.path.join(SRC_DIR, "common.h")
    rebuilt = False
    for src in KERNEL_SOURCES:
        src_path = os.path.join(SRC_DIR, src)
        if not os.path.exists(src_path):
            continue
        obj = os.path.join(BUILD_DIR, src.rsplit(".", 1)[0] + ".o")
        objs.append(obj)
        if force or _newer(src_path, obj) or _newer(header, obj):
            _run(common + ["-c", src_path, "-o", obj])
            rebuilt = True
    core_hdrs = [
        os.path.join(PKG_DIR, "csrc", "core", "jvalue.h"),
        os.path.join(PKG_DIR, "csrc", "core", "native_lane.h"),
    ]
    for src in (BINDING_SOURCE, "native_engrams.cpp"):
        src_path = os.path.join(SRC_DIR, src)
        obj = os.path.join(BUILD_DIR, src.rsplit(".", 1)[0] + ".o")
        if force or _newer(src_path, obj) or any(_newer(h, obj) for h in core_hdrs):
            _run(common + includes + defines + ["-x", "hip", "-c", src_path, "-o", obj])
            rebuilt = True
        objs.append(obj)
    if force or rebuilt or not os.path.exists(OUT_SO):
        _run(common + ["-shared", *objs, *libs, "-o", OUT_SO])
    return OUT_SO


def build_core(force: bool = False) -> str:
    """Build the bobraccel native DAG core (plain C++, no torch headers)."""
    import pybind11

    os.makedirs(BUILD_DIR, exist_ok=True)
    srcs = [os.path.join(CORE_DIR, f) for f in ("engine.cpp", "pybind.cpp")]
    hdrs = [os.path.join(CORE_DIR, f) for f in ("jvalue.h", "expr.h", "engine.h")]
    deps = srcs + hdrs
    if not force and os.path.exists(CORE_SO) and not any(
        _newer(d, CORE_SO) for d in deps
    ):
        return CORE_SO
    py_inc = sysconfig.get_paths()["include"]
    ext_suffix = ""
    cmd = [
        "g++", "-O2", "-std=c++17", "-shared", "-fPIC",
        f"-I{pybind11.get_include()}", f"-I{py_inc}", f"-I{CORE_DIR}",
        *srcs, "-o", CORE_SO, "-pthread",
    ]
    _run(cmd)
    return CORE_SO


def build_all(force: bool = False) -> None:
    build(force=force)
    build_core(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
    print(f"built {OUT_SO} and {CORE_SO}")
