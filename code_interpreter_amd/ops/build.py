"""In-tree build of the _hipops extension for gfx950.

hipcc cross-compiles without a GPU, and the resulting .so lives next to
this file (in-tree, so it travels to GPU boxes with the repo snapshot).

Usage: python -m code_interpreter_amd.ops.build [--force]
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
HIP_DIR = OPS_DIR / "hip"
OUTPUT = OPS_DIR / "_hipops.so"

SOURCES = [
    HIP_DIR / "hipops.cpp",
    HIP_DIR / "kernels_ew.hip",
    HIP_DIR / "gemm_f32.hip",
    HIP_DIR / "gemm_f64.hip",
    HIP_DIR / "gemm_bf16.hip",
    HIP_DIR / "gemm_bf16_256.hip",
]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def needs_build() -> bool:
    if not OUTPUT.exists():
        return True
    out_mtime = OUTPUT.stat().st_mtime
    deps = list(SOURCES) + [HIP_DIR / "common.h", Path(__file__)]
    return any(p.stat().st_mtime > out_mtime for p in deps)


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_build():
        return OUTPUT
    hipcc = os.environ.get("HIPCC", "hipcc")
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        hipcc,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-x", "hip",
        *[str(s) for s in SOURCES],
        f"-I{py_include}",
        f"-I{HIP_DIR}",
        "-o",
        str(OUTPUT),
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return OUTPUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {OUTPUT}")
