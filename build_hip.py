#!/usr/bin/env python3
"""In-tree HIP extension build: drives hipcc --offload-arch=gfx950 directly
(no hipify, no CUDA shims). Produces quda_amd_hip.so at the repo root, which
travels to GPU boxes with the source snapshot.

Usage: python build_hip.py [--force]
(also invoked by setup.py build_ext --inplace and __graft_entry__.build()).
"""

from __future__ import annotations

import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "csrc")
BUILD = os.path.join(ROOT, "build", "hip")
OUT_SO = os.path.join(ROOT, "quda_amd_hip.so")

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = [
    "blas.hip",
    "clover.hip",
    "dslash_wilson_d.hip",
    "dslash_wilson_s.hip",
    "dslash_wilson_h.hip",
    "dslash_wilson_q.hip",
    "dslash_staggered.hip",
    "dslash_dwf.hip",
    "dslash_wilson_mrhs.hip",
    "coarse.hip",
    "heatbath.hip",
]
BINDING_SOURCES = ["bindings.cpp"]


def torch_flags():
    import torch
    from torch.utils import cpp_extension as ce
    inc = [f"-I{p}" for p in ce.include_paths()]
    import sysconfig
    inc.append(f"-I{sysconfig.get_paths()['include']}")
    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    defs = [
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=quda_amd_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
    ]
    libdir = os.path.join(os.path.dirname(torch.__file__), "lib")
    link = [f"-L{libdir}", f"-Wl,-rpath,{libdir}",
            "-lc10", "-ltorch", "-ltorch_cpu", "-ltorch_python",
            "-lc10_hip", "-ltorch_hip", "-lamdhip64"]
    return inc + defs, link


COMMON = ["-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
          "-Wall", "-Wno-unused-function", f"-I{CSRC}"]


def _newer(a: str, b: str) -> bool:
    return not os.path.exists(b) or os.path.getmtime(a) > os.path.getmtime(b)


def _headers_mtime() -> float:
    newest = 0.0
    for dirp, _, files in os.walk(CSRC):
        for f in files:
            if f.endswith((".h", ".inl")):
                newest = max(newest, os.path.getmtime(os.path.join(dirp, f)))
    return newest


def build(force: bool = False, verbose: bool = True) -> str:
    os.makedirs(BUILD, exist_ok=True)
    # regenerate projector header
    subprocess.run([sys.executable, os.path.join(CSRC, "generate_proj.py")],
                   check=True, capture_output=True)
    tinc, tlink = torch_flags()
    hmt = _headers_mtime()
    jobs = []
    objs = []
    for src in KERNEL_SOURCES + BINDING_SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace("/", "_") + ".o")
        objs.append(op)
        if force or _newer(sp, op) or os.path.getmtime(sp) < hmt and _newer_f(hmt, op):
            extra = tinc if src in BINDING_SOURCES else []
            cmd = [HIPCC] + COMMON + extra + ["-c", sp, "-o", op]
            jobs.append(cmd)

    def run(cmd):
        if verbose:
            print("  " + os.path.basename(cmd[-3]))
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"compile failed: {' '.join(cmd)}\n{r.stdout}\n{r.stderr}")
        return r

    if jobs:
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(run, jobs))
    if jobs or force or not os.path.exists(OUT_SO):
        cmd = [HIPCC, "-shared", "-fPIC", f"--offload-arch={ARCH}",
               "-o", OUT_SO] + objs + tlink
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
        if verbose:
            print(f"linked {OUT_SO}")
    build_c_api(force=force, verbose=verbose)
    return OUT_SO


C_API_SO = os.path.join(ROOT, "libquda_amd_c.so")


def build_c_api(force: bool = False, verbose: bool = True) -> str:
    """extern \"C\" ABI library (include/quda_amd.h): embeds the engine via
    libpython; no torch C++ linkage, so it builds with plain g++."""
    import sysconfig

    import pybind11
    src = os.path.join(CSRC, "quda_c_api.cpp")
    hdr = os.path.join(ROOT, "include", "quda_amd.h")
    if not (force or _newer(src, C_API_SO) or _newer(hdr, C_API_SO)):
        return C_API_SO
    py_inc = sysconfig.get_paths()["include"]
    libdir = sysconfig.get_config_var("LIBDIR")
    pyver = f"python{sys.version_info.major}.{sys.version_info.minor}"
    fsrc = os.path.join(CSRC, "quda_fortran_api.cpp")
    cmd = ["g++", "-O2", "-std=c++17", "-fPIC", "-shared",
           f"-I{pybind11.get_include()}", f"-I{py_inc}",
           src, fsrc, "-o", C_API_SO, f"-L{libdir}", f"-l{pyver}"]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"C API build failed:\n{r.stdout}\n{r.stderr}")
    if verbose:
        print(f"linked {C_API_SO}")
    return C_API_SO


def _newer_f(mtime: float, path: str) -> bool:
    return not os.path.exists(path) or mtime > os.path.getmtime(path)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
