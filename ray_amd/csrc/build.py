"""Build the ray_amd HIP extension in-tree with hipcc for gfx950.

Direct hipcc invocation (no hipify, no CUDA path): compiles
csrc/ops.hip -> ray_amd/_hip_ops.so. The .so is git-ignored but travels
to GPU boxes with the repo snapshot.
"""
from __future__ import annotations

import os
import subprocess
import sys

import torch
from torch.utils import cpp_extension

CSRC = os.path.dirname(os.path.abspath(__file__))
PKG = os.path.dirname(CSRC)
OUT = os.path.join(PKG, "_hip_ops.so")
OUT_SHM = os.path.join(PKG, "_shm_native.so")


def _python_include() -> str:
    import sysconfig

    return sysconfig.get_paths()["include"]


def needs_rebuild() -> bool:
    if not os.path.exists(OUT):
        return True
    out_m = os.path.getmtime(OUT)
    for root, _, files in os.walk(CSRC):
        for f in files:
            if f.endswith((".hip", ".h")) and os.path.getmtime(
                os.path.join(root, f)
            ) > out_m:
                return True
    return False


def _sanitize_flags() -> list:
    """RAY_AMD_SANITIZE=address|thread|undefined builds the native C++
    with the matching sanitizer (reference: bazel --config=asan/tsan
    presets; see docs/SANITIZERS.md). Run python with the sanitizer
    runtime preloaded (LD_PRELOAD=$(g++ -print-file-name=libasan.so))."""
    san = os.environ.get("RAY_AMD_SANITIZE")
    if not san:
        return []
    return [f"-fsanitize={san}", "-fno-omit-frame-pointer", "-g", "-O1"]


def build_shm(verbose: bool = True, force: bool = False) -> str:
    """Native shm store data path: plain C++ (g++), no HIP needed."""
    src = os.path.join(CSRC, "shm_store.cpp")
    if (not force and os.path.exists(OUT_SHM)
            and os.path.getmtime(OUT_SHM) >= os.path.getmtime(src)
            and not os.environ.get("RAY_AMD_SANITIZE")):
        return OUT_SHM
    import pybind11

    cmd = [
        "g++", "-O3", "-std=c++17", "-fPIC", "-shared", src, "-o", OUT_SHM,
        f"-I{pybind11.get_include()}", f"-I{_python_include()}",
        "-pthread",
    ] + _sanitize_flags()
    if verbose:
        print("[ray_amd build]", " ".join(cmd), file=sys.stderr)
    subprocess.check_call(cmd)
    return OUT_SHM


def _hipcc_torch_ext(src: str, out: str, name: str, extra_libs=(),
                     verbose: bool = True):
    hipcc = os.path.join(
        os.environ.get("ROCM_PATH", "/opt/rocm"), "bin", "hipcc"
    )
    includes = cpp_extension.include_paths() + [_python_include(), CSRC]
    lib_dirs = cpp_extension.library_paths()
    abi = int(torch.compiled_with_cxx11_abi())
    cmd = [
        hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared", src, "-o", out,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        f"-DTORCH_EXTENSION_NAME={name}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1", "-fno-gpu-rdc", "-Wno-unused-result",
    ]
    for i in includes:
        cmd.append(f"-I{i}")
    for l in lib_dirs:
        cmd.append(f"-L{l}")
        cmd.append(f"-Wl,-rpath,{l}")
    cmd += ["-ltorch", "-ltorch_python", "-lc10", "-ltorch_hip", "-lc10_hip",
            "-lamdhip64"]
    cmd += list(extra_libs)
    if verbose:
        print("[ray_amd build]", " ".join(cmd), file=sys.stderr)
    subprocess.check_call(cmd)
    return out


OUT_RCCL = os.path.join(PKG, "_rccl_comm.so")


def build_rccl(verbose: bool = True, force: bool = False) -> str:
    """Native RCCL communicator extension (rccl_comm.hip)."""
    src = os.path.join(CSRC, "rccl_comm.hip")
    if (not force and os.path.exists(OUT_RCCL)
            and os.path.getmtime(OUT_RCCL) >= os.path.getmtime(src)):
        return OUT_RCCL
    return _hipcc_torch_ext(src, OUT_RCCL, "_rccl_comm",
                            extra_libs=["-lrccl"], verbose=verbose)


def build(verbose: bool = True, force: bool = False) -> str:
    build_shm(verbose, force)
    build_rccl(verbose, force)
    if not force and not needs_rebuild():
        return OUT
    hipcc = os.path.join(
        os.environ.get("ROCM_PATH", "/opt/rocm"), "bin", "hipcc"
    )
    includes = cpp_extension.include_paths() + [_python_include(), CSRC]
    lib_dirs = cpp_extension.library_paths()
    abi = int(torch.compiled_with_cxx11_abi())
    cmd = [
        hipcc,
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        os.path.join(CSRC, "ops.hip"),
        "-o",
        OUT,
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    for i in includes:
        cmd.append(f"-I{i}")
    for l in lib_dirs:
        cmd.append(f"-L{l}")
        cmd.append(f"-Wl,-rpath,{l}")
    cmd += ["-ltorch", "-ltorch_python", "-lc10", "-ltorch_hip", "-lc10_hip",
            "-lamdhip64"]
    if verbose:
        print("[ray_amd build]", " ".join(cmd), file=sys.stderr)
    subprocess.check_call(cmd)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(OUT)
