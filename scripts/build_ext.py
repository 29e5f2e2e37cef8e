#!/usr/bin/env python3
"""Build the in-tree gfx950 HIP extension: hipcc compiles csrc/*.hip and
csrc/bindings.cpp directly (no hipify, no CUDAExtension shim) and links
against libtorch -> chinesener_amd/ops/_hip_ops.so.

The built .so lives in-tree so the gpurun snapshot carries it to the GPU
box (round-end loaded-native-code check needs it there)."""
from __future__ import annotations

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(REPO, "csrc")
OUT_DIR = os.path.join(REPO, "chinesener_amd", "ops")
BUILD = os.path.join(REPO, "build")

HIP_SOURCES = ["elementwise.hip", "crf.hip", "softlexicon.hip", "adam.hip",
               "attention.hip", "tener.hip", "lstm.hip", "probe.hip",
               "wgrad.hip", "gemm_nt.hip", "embed.hip"]
CPP_SOURCES = ["bindings.cpp"]


def torch_flags():
    import torch
    import torch.utils.cpp_extension as ce
    from pybind11 import get_include as pybind_inc
    includes = ce.include_paths() + [pybind_inc()]
    import sysconfig
    includes.append(sysconfig.get_paths()["include"])
    libdirs = ce.library_paths()
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    return includes, libdirs, abi


def run(cmd):
    print(" ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build(verbose: bool = True) -> str:
    os.makedirs(BUILD, exist_ok=True)
    includes, libdirs, abi = torch_flags()
    inc_flags = [f"-I{i}" for i in includes] + [f"-I{CSRC}"]
    common = ["-O3", "-std=c++17", "-fPIC",
              f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
              "-DTORCH_EXTENSION_NAME=_hip_ops",
              "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
              "-fno-gpu-rdc", "-Wno-unused-result"]
    objs = []
    procs = []
    for src in HIP_SOURCES + CPP_SOURCES:
        obj = os.path.join(BUILD, src.rsplit(".", 1)[0] + ".o")
        objs.append(obj)
        src_path = os.path.join(CSRC, src)
        if (os.path.exists(obj)
                and os.path.getmtime(obj) > os.path.getmtime(src_path)
                and os.path.getmtime(obj) > os.path.getmtime(
                    os.path.join(CSRC, "common.h"))):
            continue
        cmd = ["hipcc", "--offload-arch=gfx950", "-c", src_path, "-o", obj] \
            + common + inc_flags
        if src.endswith(".cpp"):
            cmd = ["hipcc", "-x", "c++", "-c", src_path, "-o", obj] \
                + common + inc_flags
        print(" ".join(cmd), flush=True)
        procs.append(subprocess.Popen(cmd))
    for p in procs:
        if p.wait() != 0:
            raise RuntimeError("hipcc compile failed")
    so_path = os.path.join(OUT_DIR, "_hip_ops.so")
    link = (["hipcc", "-shared", "-fPIC", "-o", so_path] + objs
            + [f"-L{d}" for d in libdirs]
            + ["-ltorch", "-ltorch_python", "-ltorch_hip", "-lc10",
               "-lc10_hip", "-lamdhip64"]
            + [f"-Wl,-rpath,{d}" for d in libdirs])
    run(link)
    return so_path


if __name__ == "__main__":
    print(build())
