"""Direct hipcc build of the in-tree HIP extension (no hipify, no nvcc).

torch.utils.cpp_extension runs HIPIFY over source files on ROCm — this
framework writes native HIP/CDNA4 code, so we drive hipcc ourselves:
device TUs (*.hip) compile standalone against the HIP runtime only;
bindings.cpp compiles as host C++ against the torch/ATen headers; the link
pulls in the torch + HIP runtime libraries. Output .so lands in-tree so it
travels with the repo snapshot (gpurun) instead of a JIT cache.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG, "csrc")
OUT = os.path.join(PKG, "_r2d2dpg_hip.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

HIP_SOURCES = ["lstm.hip", "lstm_px.hip", "td_polyak.hip", "sumtree.hip", "replay_pack.hip"]
CPP_SOURCES = ["bindings.cpp"]


def _run(cmd):
    print("[build]", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def _newer(src, obj):
    if not os.path.exists(obj):
        return True
    st = os.path.getmtime
    deps = [src, os.path.join(CSRC, "common.h")]
    return any(st(d) > st(obj) for d in deps if os.path.exists(d))


def build(force: bool = False, verbose: bool = False) -> str:
    import torch
    from torch.utils import cpp_extension as ce

    build_dir = os.path.join(os.path.dirname(PKG), "..", "build", "hip")
    build_dir = os.path.abspath(build_dir)
    os.makedirs(build_dir, exist_ok=True)

    hipcc = os.path.join(ce.ROCM_HOME or "/opt/rocm", "bin", "hipcc")
    abi = "1" if torch._C._GLIBCXX_USE_CXX11_ABI else "0"

    objs = []
    # ---- device TUs: pure HIP, no torch headers
    dev_flags = [f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC", "-c"]
    for s in HIP_SOURCES:
        src = os.path.join(CSRC, s)
        obj = os.path.join(build_dir, s.replace(".hip", ".o"))
        objs.append(obj)
        if force or _newer(src, obj):
            _run([hipcc, *dev_flags, src, "-o", obj])

    # ---- host binding TU: torch/ATen headers, HIP platform defines
    inc = [f"-I{p}" for p in ce.include_paths("cuda")]
    inc.append(f"-I{sysconfig.get_paths()['include']}")
    host_flags = [
        "-O2", "-std=c++17", "-fPIC", "-c", "-x", "c++",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_r2d2dpg_hip",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
    ]
    for s in CPP_SOURCES:
        src = os.path.join(CSRC, s)
        obj = os.path.join(build_dir, s.replace(".cpp", ".o"))
        objs.append(obj)
        if force or _newer(src, obj):
            _run([hipcc, *host_flags, *inc, src, "-o", obj])

    # ---- link
    libdirs = [f"-L{p}" for p in ce.library_paths("cuda")]
    libs = ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
            "-ltorch_hip", "-lc10_hip", "-lamdhip64"]
    need_link = (not os.path.exists(OUT) or force
                 or any(os.path.getmtime(o) > os.path.getmtime(OUT) for o in objs))
    if need_link:
        _run([hipcc, "-shared", "-fPIC", *objs, *libdirs, *libs, "-o", OUT])
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv, verbose=True)
