"""In-tree build of the gfx950 HIP extension.

`python setup.py build_ext --inplace` produces
sentio_amd/ops/_sentio_hip.so by driving hipcc directly (no hipify — the
sources are native HIP/CDNA4).  Kernels compile in their own TUs (fast);
bindings.hip is the only TU touching torch headers.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).parent
CSRC = ROOT / "sentio_amd" / "ops" / "csrc"
OUT = ROOT / "sentio_amd" / "ops" / "_sentio_hip.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_TUS = ["elementwise.hip", "retrieval.hip", "attention.hip", "gemm.hip", "blaslt.hip"]
BINDING_TU = "bindings.hip"


def torch_paths():
    from torch.utils import cpp_extension

    return cpp_extension.include_paths(), cpp_extension.library_paths()


def build(verbose: bool = True) -> Path:
    includes, libdirs = torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    build_dir = ROOT / "build"
    build_dir.mkdir(exist_ok=True)

    common = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DNDEBUG",
    ]
    # SENTIO_SANITIZE=1: host-side ASan+UBSan test build (SURVEY §5 race/
    # sanitizer row).  Run with LD_PRELOAD of the ASan runtime, e.g.
    #   LD_PRELOAD=$(hipcc -print-file-name=libclang_rt.asan-x86_64.so) pytest
    if os.environ.get("SENTIO_DECODE_GLDS") == "1":
        # experimental LDS-DMA K staging in decode attention
        common += ["-DSENTIO_DECODE_GLDS"]
    if os.environ.get("SENTIO_SANITIZE") == "1":
        common += ["-fsanitize=address,undefined",
                   "-fno-omit-frame-pointer", "-shared-libasan"]
    objs = []

    def compile_tu(src: Path, extra: list[str]) -> Path:
        obj = build_dir / (src.stem + ".o")
        if obj.exists() and obj.stat().st_mtime > max(
            src.stat().st_mtime, (CSRC / "common.h").stat().st_mtime
        ):
            return obj
        cmd = common + extra + ["-c", str(src), "-o", str(obj)]
        if verbose:
            print("[sentio build]", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)
        return obj

    for tu in KERNEL_TUS:
        objs.append(compile_tu(CSRC / tu, []))

    torch_flags = (
        [f"-I{p}" for p in includes]
        + [f"-I{py_inc}"]
        + ["-DTORCH_EXTENSION_NAME=_sentio_hip", "-D__HIP_PLATFORM_AMD__=1",
           "-DUSE_ROCM=1"]
    )
    objs.append(compile_tu(CSRC / BINDING_TU, torch_flags))

    link = (
        common
        + ["-shared", "-o", str(OUT)]
        + [str(o) for o in objs]
        + [f"-L{d}" for d in libdirs]
        + ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python", "-lhipblaslt"]
        + [f"-Wl,-rpath,{d}" for d in libdirs]
    )
    if verbose:
        print("[sentio build]", " ".join(link), flush=True)
    subprocess.run(link, check=True)
    return OUT


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "build_ext":
        build()
    else:
        build()
