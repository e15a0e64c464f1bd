#!/usr/bin/env python3
"""Clang static analyzer over the HIP kernel sources' host pass
(SURVEY.md §5.2 race-detection/sanitizer row: develop kernels under
host-code sanitizers where possible — there is no GPU-side ASan on this
stack, so the host pass gets the clang analyzer and device code is
covered by the numerics suite + soak runs).

    python scripts/static_analysis.py        # all TUs; nonzero exit on findings
"""
import os
import subprocess
import sys
import sysconfig

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from melgan_amd.kernels.build import KERNEL_DIR, SOURCES, _torch_paths  # noqa: E402

includes, _lib, abi = _torch_paths()
base = ["/opt/rocm/lib/llvm/bin/clang++", "-x", "hip", "--offload-host-only",
        "--rocm-path=/opt/rocm", "-std=c++17",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}", "-DTORCH_EXTENSION_NAME=_C",
        "-D__HIP_NO_HALF_OPERATORS__=1", "-DUSE_ROCM", "-DTORCH_HIP_VERSION=702"]
for inc in includes:
    base += ["-I", inc]
base += ["-I", sysconfig.get_paths()["include"]]

total = 0
for src in SOURCES:
    r = subprocess.run(base + ["--analyze", "-Xclang", "-analyzer-output=text",
                               os.path.join(KERNEL_DIR, src)],
                       capture_output=True, text=True)
    ours = [l for l in (r.stdout + r.stderr).splitlines()
            if "melgan_amd" in l and "warning:" in l]
    total += len(ours)
    print(f"{src}: rc={r.returncode} findings={len(ours)}")
    for l in ours:
        print("  ", l)
print(f"TOTAL analyzer findings in melgan_amd sources: {total}")
sys.exit(1 if total else 0)
