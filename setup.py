#!/usr/bin/env python3
"""In-tree build of the npairloss_amd HIP extension for gfx950.

Drives hipcc directly (no hipify source rewriting, no JIT cache outside the
tree): each csrc/*.hip (and bindings.cpp) compiles to an .o next to it,
linked into npairloss_amd/_C.so.  The .so travels to the GPU box with the
repo snapshot.

Usage:  python setup.py build_ext --inplace   (or just: python setup.py)
        python setup.py clean
"""

import concurrent.futures
import os
import shutil
import subprocess
import sys

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "npairloss_amd", "ops", "csrc")
OUT_SO = os.path.join(ROOT, "npairloss_amd", "_C.so")
BUILD = os.path.join(ROOT, "build", "csrc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

SOURCES = [
    "bindings.cpp",
    "npair_rows.hip",
    "sort_select.hip",
    "l2norm.hip",
    "gemm_f32.hip",
    "vision.hip",
    "gemm_lowp.hip",
    "biasrelu.hip",
    "conv1x1.hip",
]


def torch_paths():
    import torch

    tdir = os.path.dirname(torch.__file__)
    return tdir


def compile_flags(tdir):
    import sysconfig

    py_inc = sysconfig.get_paths()["include"]
    return [
        "-DWITH_HIP",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-isystem{tdir}/include",
        f"-isystem{tdir}/include/torch/csrc/api/include",
        f"-isystem{tdir}/include/THH",
        "-isystem/opt/rocm/include",
        f"-isystem{py_inc}",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-fPIC",
        "-std=c++17",
        "-O3",
        f"--offload-arch={ARCH}",
        "-fno-gpu-rdc",
    ]


def needs_build(src, obj):
    if not os.path.exists(obj):
        return True
    dep = [src, os.path.join(CSRC, "common.h"), os.path.abspath(__file__)]
    omt = os.path.getmtime(obj)
    return any(os.path.getmtime(d) > omt for d in dep if os.path.exists(d))


def build(verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    tdir = torch_paths()
    flags = compile_flags(tdir)

    objs = []
    jobs = []
    for src in SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace("/", "_") + ".o")
        objs.append(op)
        if needs_build(sp, op):
            jobs.append((sp, op))

    def compile_one(sp_op):
        sp, op = sp_op
        cmd = [HIPCC] + flags + ["-c", sp, "-o", op]
        if verbose:
            print("[hipcc]", os.path.basename(sp))
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed for {sp}:\n{r.stdout}\n{r.stderr}")
        return op

    if jobs:
        with concurrent.futures.ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(compile_one, jobs))

    if jobs or not os.path.exists(OUT_SO):
        link = (
            ["c++", "-shared"]
            + objs
            + [
                f"-L{tdir}/lib",
                "-lc10",
                "-lc10_hip",
                "-ltorch_cpu",
                "-ltorch_hip",
                "-ltorch",
                "-ltorch_python",
                "-L/opt/rocm/lib",
                "-lamdhip64",
                "-o",
                OUT_SO,
            ]
        )
        if verbose:
            print("[link]", os.path.relpath(OUT_SO, ROOT))
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    elif verbose:
        print("npairloss_amd/_C.so up to date")
    return OUT_SO


def clean():
    shutil.rmtree(BUILD, ignore_errors=True)
    if os.path.exists(OUT_SO):
        os.remove(OUT_SO)


if __name__ == "__main__":
    if "clean" in sys.argv:
        clean()
    else:
        build()
