"""In-tree build of the gfx950 HIP extension.

Invokes hipcc directly (no hipify, no CUDA shims) and links against the
installed PyTorch-ROCm. The resulting _kaito_C.so lives next to this file so
it travels with the repo snapshot to GPU boxes.
"""
from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
SO_PATH = OPS_DIR / "_kaito_C.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")

SOURCES = sorted([*CSRC.glob("*.hip"), *CSRC.glob("*.cpp")])


def _torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    return ce.include_paths(), ce.library_paths(), torch.compiled_with_cxx11_abi()


def needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    return any(s.stat().st_mtime > so_mtime for s in SOURCES)


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not needs_build():
        return SO_PATH
    includes, libdirs, cxx11 = _torch_paths()
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-DUSE_ROCM",
        "-D__HIP_PLATFORM_AMD__=1",
        f"-D_GLIBCXX_USE_CXX11_ABI={1 if cxx11 else 0}",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    for inc in includes:
        cmd += ["-I", inc]
    cmd += [str(s) for s in SOURCES]
    for ld in libdirs:
        cmd += ["-L", ld, f"-Wl,-rpath,{ld}"]
    cmd += ["-ltorch", "-ltorch_cpu", "-lc10", "-ltorch_hip", "-lc10_hip",
            "-L/opt/rocm/lib", "-lamdhip64", "-o", str(SO_PATH)]
    if verbose:
        print("[kaito_amd] building HIP extension:", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
