"""In-tree build of the gfx950 HIP extension (`_hip_ops.so`).

Drives hipcc directly (native HIP — no hipify, no CUDA shims):
  - each csrc/*.hip is device code compiled with --offload-arch=gfx950
  - csrc/bindings.cpp is host code including torch/extension.h
  - linked into alpa_amd/ops/_hip_ops.so (travels to the GPU box with the
    repo snapshot; kept out of git by .gitignore)
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

HERE = Path(__file__).resolve().parent
CSRC = HERE / "csrc"
BUILD = HERE / "_build"
OUT = HERE / "_hip_ops.so"

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

TORCH_HIP_DEFS = [
    "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
    "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
    "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
]


def _torch_paths():
    import torch
    import torch.utils.cpp_extension as ce
    includes = ce.include_paths()
    libdirs = ce.library_paths()
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return includes, libdirs, abi


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"build command failed ({' '.join(cmd[:3])}...):\n{r.stdout}\n{r.stderr}")
    return r


def _needs_rebuild(srcs) -> bool:
    if not OUT.exists():
        return True
    out_mtime = OUT.stat().st_mtime
    for s in srcs:
        if s.stat().st_mtime > out_mtime:
            return True
    for extra in [Path(__file__)]:
        if extra.stat().st_mtime > out_mtime:
            return True
    return False


def build_all(verbose: bool = True, force: bool = False) -> Path:
    hip_srcs = sorted(CSRC.glob("*.hip"))
    cpp_srcs = sorted(CSRC.glob("*.cpp"))
    srcs = hip_srcs + cpp_srcs + sorted(CSRC.glob("*.h"))
    if not force and not _needs_rebuild(srcs):
        if verbose:
            print(f"[build_ext] {OUT.name} up to date")
        return OUT

    includes, libdirs, abi = _torch_paths()
    BUILD.mkdir(exist_ok=True)
    py_include = sysconfig.get_paths()["include"]

    common = [
        "-O3", "-std=c++17", "-fPIC", f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        *TORCH_HIP_DEFS,
    ]
    objs = []
    for src in hip_srcs:
        obj = BUILD / (src.stem + ".o")
        cmd = [HIPCC, f"--offload-arch={ARCH}", *common, "-c", str(src),
               "-o", str(obj), "-I", str(CSRC)]
        if verbose:
            print(f"[build_ext] hipcc {src.name}")
        _run(cmd)
        objs.append(obj)

    for src in cpp_srcs:
        obj = BUILD / (src.stem + ".o")
        cmd = [HIPCC, f"--offload-arch={ARCH}", *common,
               "-DTORCH_API_INCLUDE_EXTENSION_H",
               "-DTORCH_EXTENSION_NAME=_hip_ops",
               "-c", str(src), "-o", str(obj), "-I", str(CSRC),
               "-I", py_include]
        for inc in includes:
            cmd += ["-I", inc]
        if verbose:
            print(f"[build_ext] hipcc {src.name} (host/bindings)")
        _run(cmd)
        objs.append(obj)

    link = [HIPCC, "-shared", "-fPIC", "-o", str(OUT)]
    link += [str(o) for o in objs]
    for ld in libdirs:
        link += ["-L", ld, f"-Wl,-rpath,{ld}"]
    link += ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
             "-ltorch_python", "-lamdhip64"]
    if verbose:
        print(f"[build_ext] link -> {OUT.name}")
    _run(link)
    return OUT


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
