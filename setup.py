"""Build the bloombee_amd gfx950 HIP extension in-tree.

Invokes hipcc directly (native HIP source — no hipify pass, no CUDA
compatibility layer). The built .so lands at bloombee_amd/ops/_hip_ops.so so
it travels with the repo snapshot to the GPU box (gpurun ships in-tree .so
files; a JIT cache under ~/.cache would not).

Usage:
    python setup.py build_ext --inplace     # or: python setup.py hip
"""
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
HIP_DIR = ROOT / "bloombee_amd" / "ops" / "hip"
OUT_SO = ROOT / "bloombee_amd" / "ops" / "_hip_ops.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch

    tdir = Path(torch.__file__).parent
    return tdir / "include", tdir / "lib"


def build(verbose: bool = True) -> Path:
    tinc, tlib = torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    src = HIP_DIR / "ext.hip"
    obj = HIP_DIR / "ext.o"
    compile_cmd = [
        "/opt/rocm/bin/hipcc",
        f"--offload-arch={ARCH}",
        "-O3", "-std=c++17", "-fPIC", "-fno-gpu-rdc",
        "-I", str(tinc),
        "-I", str(tinc / "torch" / "csrc" / "api" / "include"),
        "-I", "/opt/rocm/include",
        "-I", py_inc,
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DCUDA_HAS_FP16=1", "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H", "-DTORCH_EXTENSION_NAME=_hip_ops",
        "-c", str(src), "-o", str(obj),
    ]
    link_cmd = [
        "g++", "-shared", str(obj),
        "-L", str(tlib), "-L", "/opt/rocm/lib",
        "-lc10", "-ltorch", "-ltorch_cpu", "-ltorch_python",
        "-lamdhip64", "-lc10_hip", "-ltorch_hip", "-lz",
        "-o", str(OUT_SO),
    ]
    for cmd in (compile_cmd, link_cmd):
        if verbose:
            print("+", " ".join(cmd), file=sys.stderr)
        subprocess.run(cmd, check=True)
    obj.unlink(missing_ok=True)
    return OUT_SO


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] in ("hip", "build_ext"):
        build()
    else:
        print(__doc__)
