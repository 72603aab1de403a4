#!/usr/bin/env python3
"""In-tree hipcc build of the CDNA4 kernel extension (_hipops.so).

Drives hipcc directly (no hipify, no CUDA shims): every TU is native HIP
written for gfx950. Incremental: .o files are cached under build/ keyed on
mtimes. The resulting .so lands inside the package so the repo snapshot
carries it to GPU boxes (gpurun ships in-tree .so files).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "distributed_vgg_f_amd", "csrc")
BUILD = os.path.join(ROOT, "build", "hipops")
OUT = os.path.join(ROOT, "distributed_vgg_f_amd", "_hipops.so")

SOURCES = ["gemm_ops.hip", "mem.hip", "lrnpool.hip", "softbn.hip",
           "hipops.cpp"]
HEADERS = ["common.h", "gemm_tile.h", "gemm8.h", "wgrad_fused.h",
           "conv_ring.h"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch
    import torch.utils.cpp_extension as ce
    inc = ce.include_paths()
    lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return inc, lib, abi


def build(verbose: bool = True) -> str:
    os.makedirs(BUILD, exist_ok=True)
    inc, libdir, abi = torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    cflags = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        # fp32 atomicAdd must lower to global_atomic_add_f32, not a CAS
        # loop (split-K accumulation + bias partials depend on it)
        "-munsafe-fp-atomics",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-DTORCH_EXTENSION_NAME=_hipops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-Wno-unused-result",
    ] + [f"-I{d}" for d in inc] + [f"-I{py_inc}", f"-I{CSRC}"]

    hdr_mtime = max(os.path.getmtime(os.path.join(CSRC, h)) for h in HEADERS)
    objs = []
    procs = []
    for src in SOURCES:
        sp = os.path.join(CSRC, src)
        op = os.path.join(BUILD, src.replace(".", "_") + ".o")
        objs.append(op)
        if (os.path.exists(op)
                and os.path.getmtime(op) > max(os.path.getmtime(sp), hdr_mtime)
                and os.path.getmtime(op) > os.path.getmtime(__file__)):
            continue
        cmd = ["hipcc", "-c", sp, "-o", op] + cflags
        if verbose:
            print("[build_hip]", src, flush=True)
        procs.append((src, subprocess.Popen(cmd)))
    fail = False
    for src, p in procs:
        if p.wait() != 0:
            print(f"[build_hip] FAILED: {src}", file=sys.stderr)
            fail = True
    if fail:
        raise RuntimeError("hipcc compilation failed")

    if (not os.path.exists(OUT)
            or any(os.path.getmtime(o) > os.path.getmtime(OUT) for o in objs)):
        link = ["hipcc", "-shared", "-o", OUT] + objs + [
            f"-L{libdir}", "-ltorch", "-ltorch_python", "-ltorch_hip",
            "-lc10", "-lc10_hip", "-lamdhip64",
            f"-Wl,-rpath,{libdir}",
        ]
        if verbose:
            print("[build_hip] linking", OUT, flush=True)
        subprocess.check_call(link)
    return OUT


if __name__ == "__main__":
    build()
