"""In-tree build of the gfx950 HIP extension.

Drives hipcc directly (no hipify, no CUDA shims): every .hip/.cpp under
csrc/ is compiled for --offload-arch=gfx950 and linked into
agentainer_amd/ops/_hip_ops.so. Built in-tree so the .so travels with the
gpurun snapshot (a JIT cache under ~/.cache would not).

hipcc cross-compiles on a GPU-less box, so `python -m agentainer_amd.ops.build`
doubles as the does-it-build check.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

import torch
from torch.utils import cpp_extension

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
# AGENTAINER_ASAN=1 builds a host-AddressSanitized variant (SURVEY.md §5
# race-detection row): the C++ binding layer (all torch interop, tensor
# lifetime, host bookkeeping) compiles with -fsanitize=address; device
# .hip kernels stay unsanitized (device ASAN needs xnack+, not a serving
# config). Run it with LD_PRELOAD=$(clang -print-file-name=libclang_rt.asan-x86_64.so).
ASAN = os.environ.get("AGENTAINER_ASAN", "") == "1"
OUT = os.path.join(HERE, "_hip_ops_asan.so" if ASAN else "_hip_ops.so")
OBJ_DIR = os.path.join(HERE, "_build_asan" if ASAN else "_build")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

SOURCES = [
    "bindings.cpp",
    "rmsnorm.hip",
    "elementwise.hip",
    "rope.hip",
    "kvcache.hip",
    "decode_attn.hip",
    "prefill_attn.hip",
    "sampling.hip",
    "skinny_gemm.hip",
]


def _newest(paths):
    return max(os.path.getmtime(p) for p in paths if os.path.exists(p))


def needs_build() -> bool:
    if not os.path.exists(OUT):
        return True
    srcs = [os.path.join(CSRC, s) for s in SOURCES]
    srcs.append(os.path.join(CSRC, "common.h"))
    srcs.append(os.path.abspath(__file__))
    return _newest(srcs) > os.path.getmtime(OUT)


def build(verbose: bool = True, force: bool = False) -> str:
    if not force and not needs_build():
        return OUT
    os.makedirs(OBJ_DIR, exist_ok=True)
    torch_inc = cpp_extension.include_paths()
    torch_lib = cpp_extension.library_paths()[0]
    py_inc = sysconfig.get_paths()["include"]
    inc_flags = [f"-I{p}" for p in torch_inc + [py_inc, CSRC]]
    cxx_flags = [
        "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    objs = []
    procs = []
    for src in SOURCES:
        src_path = os.path.join(CSRC, src)
        obj = os.path.join(OBJ_DIR, os.path.splitext(src)[0] + ".o")
        objs.append(obj)
        if (os.path.exists(obj)
                and os.path.getmtime(obj) > _newest([src_path, os.path.join(CSRC, "common.h")])
                and not force):
            continue
        san = (["-fsanitize=address", "-shared-libsan"]
               if ASAN and src.endswith(".cpp") else [])
        cmd = (["hipcc", "-c", src_path, "-o", obj] + cxx_flags + san
               + inc_flags + (["-x", "hip"] if src.endswith(".cpp") else []))
        if verbose:
            print("[ops.build]", " ".join(cmd), flush=True)
        procs.append((src, subprocess.Popen(cmd, stdout=subprocess.PIPE,
                                            stderr=subprocess.STDOUT)))
    failed = False
    for src, p in procs:
        out, _ = p.communicate()
        if p.returncode != 0:
            failed = True
            print(f"[ops.build] FAILED {src}:\n{out.decode()}", file=sys.stderr)
    if failed:
        raise RuntimeError("HIP extension build failed")
    link = (["hipcc", "-shared", "-fPIC", "-o", OUT] + objs
            + (["-fsanitize=address", "-shared-libsan"] if ASAN else [])
            + [f"-L{torch_lib}", "-ltorch", "-ltorch_python", "-lc10",
               "-ltorch_hip", "-lc10_hip", f"-Wl,-rpath,{torch_lib}"])
    if verbose:
        print("[ops.build]", " ".join(link), flush=True)
    res = subprocess.run(link, capture_output=True)
    if res.returncode != 0:
        raise RuntimeError(f"link failed:\n{res.stdout.decode()}\n{res.stderr.decode()}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {OUT}")
