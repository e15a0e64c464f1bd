"""In-tree build of the gfx950 HIP extension (melgan_amd._C).

Drives hipcc directly (no hipify, no CUDA shims): each .hip translation
unit is compiled for --offload-arch=gfx950 and linked against libtorch
into melgan_amd/_C.so, which travels with the repo snapshot to GPU boxes.

    python -m melgan_amd.kernels.build          # build
    python -m melgan_amd.kernels.build --check  # report status
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
KERNEL_DIR = os.path.join(PKG_DIR, "kernels")
OUT_SO = os.path.join(PKG_DIR, "_C.so")

SOURCES = [
    "kernels.hip",
    "conv1d_mfma.hip",
    "stft.hip",
    "gemm.hip",
    "xstage.hip",
    "pqmf.hip",
    "gemm256.hip",
    "cin1.hip",
]


def _torch_paths():
    import torch
    import torch.utils.cpp_extension as ce

    includes = ce.include_paths()
    lib_dir = os.path.join(os.path.dirname(torch.__file__), "lib")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return includes, lib_dir, abi


def needs_rebuild() -> bool:
    if not os.path.exists(OUT_SO):
        return True
    so_mtime = os.path.getmtime(OUT_SO)
    for src in SOURCES + ["build.py"]:
        p = os.path.join(KERNEL_DIR, src)
        if os.path.exists(p) and os.path.getmtime(p) > so_mtime:
            return True
    return False


def build(verbose: bool = True, force: bool = False) -> str:
    if not force and not needs_rebuild():
        if verbose:
            print(f"[melgan_amd.build] up to date: {OUT_SO}")
        return OUT_SO
    includes, lib_dir, abi = _torch_paths()
    py_include = sysconfig.get_paths()["include"]
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    srcs = [os.path.join(KERNEL_DIR, s) for s in SOURCES]
    cmd = [
        hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
        "-shared",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-D__HIP_NO_HALF_OPERATORS__=1",  # keep at::Half conversions
        "-DUSE_ROCM", "-DTORCH_HIP_VERSION=702",
        "-Wno-unused-result", "-Wno-deprecated-declarations",
        "-fno-gpu-rdc", "-parallel-jobs=8",
    ]
    for inc in includes:
        cmd += ["-I", inc]
    cmd += ["-I", py_include]
    cmd += srcs
    cmd += [
        "-o", OUT_SO, f"-L{lib_dir}", "-ltorch", "-ltorch_python",
        "-ltorch_hip", "-lc10", "-lc10_hip", "-lamdhip64",
        f"-Wl,-rpath,{lib_dir}",
    ]
    if verbose:
        print("[melgan_amd.build]", " ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout[-4000:] + "\n" + r.stderr[-8000:] + "\n")
        raise RuntimeError("hipcc build of melgan_amd._C failed")
    if verbose and r.stderr.strip():
        print(r.stderr[-2000:])
    if verbose:
        print(f"[melgan_amd.build] built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    if "--check" in sys.argv:
        print("rebuild needed" if needs_rebuild() else "up to date")
    else:
        build(force="--force" in sys.argv)
