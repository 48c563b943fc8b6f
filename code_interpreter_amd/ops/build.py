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

GEMM_SOURCES = [
    HIP_DIR / "gemm_f32.hip",
    HIP_DIR / "gemm_f64.hip",
    HIP_DIR / "gemm_bf16.hip",
    HIP_DIR / "gemm_bf16_256.hip",
]
SOURCES = [
    HIP_DIR / "hipops.cpp",
    HIP_DIR / "kernels_ew.hip",
    HIP_DIR / "sort.hip",
    *GEMM_SOURCES,
]
# _hipgemm: the torch-interop module. Built SEPARATELY because zygote
# children fork with _hipops pre-imported, and a fat-binary registered
# pre-fork cannot be launched after the child re-initializes HIP (torch
# init) -- the torch routing layer dlopens this fresh per child.
GEMM_OUTPUT = OPS_DIR / "_hipgemm.so"
GEMM_MODULE_SOURCES = [HIP_DIR / "hipgemm.cpp", *GEMM_SOURCES]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _needs(output: Path, sources) -> bool:
    if not output.exists():
        return True
    out_mtime = output.stat().st_mtime
    deps = list(sources) + [HIP_DIR / "common.h", Path(__file__)]
    return any(p.stat().st_mtime > out_mtime for p in deps)


def needs_build() -> bool:
    return _needs(OUTPUT, SOURCES) or _needs(GEMM_OUTPUT, GEMM_MODULE_SOURCES)


def _compile(output: Path, sources, verbose: bool) -> None:
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
        *[str(s) for s in sources],
        f"-I{py_include}",
        f"-I{HIP_DIR}",
        "-o",
        str(output),
    ]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build(force: bool = False, verbose: bool = True) -> Path:
    if force or _needs(OUTPUT, SOURCES):
        _compile(OUTPUT, SOURCES, verbose)
    if force or _needs(GEMM_OUTPUT, GEMM_MODULE_SOURCES):
        _compile(GEMM_OUTPUT, GEMM_MODULE_SOURCES, verbose)
    return OUTPUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {OUTPUT}")
