"""In-tree build of the kukeon_amd HIP extension for gfx950.

Drives hipcc directly (native .hip sources — no hipify, no CUDA paths) and
links against libtorch so the resulting ``_C.so`` lives inside the package
tree and travels with repo snapshots.  Cross-compiles fine on a GPU-less box.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent.parent  # kukeon_amd/
CSRC = PKG_DIR / "ops" / "csrc"
OBJ_DIR = PKG_DIR / "ops" / "_build"
SO_PATH = PKG_DIR / "_C.so"
ARCH = os.environ.get("KUKEON_GFX_ARCH", "gfx950")

SOURCES = [
    "rmsnorm.hip",
    "activation.hip",
    "rope_kv.hip",
    "paged_attn.hip",
    "prefill_attn.hip",
    "sampling.hip",
    "moe.hip",
    "skinny_gemm.hip",
    "bindings.cpp",
]


def _torch_paths():
    from torch.utils import cpp_extension

    return cpp_extension.include_paths(), cpp_extension.library_paths()


def _run(cmd: list[str]) -> None:
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        sys.stderr.write(" ".join(cmd) + "\n")
        sys.stderr.write(proc.stdout[-4000:] + "\n" + proc.stderr[-8000:] + "\n")
        raise RuntimeError(f"build command failed (rc={proc.returncode})")


def build(verbose: bool = True, force: bool = False) -> Path:
    inc, libdirs = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    OBJ_DIR.mkdir(parents=True, exist_ok=True)

    common_flags = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-DNDEBUG",
        "-D__HIP_NO_HALF_OPERATORS__=1",
        "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DUSE_ROCM=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    common_flags += [f"-I{p}" for p in inc]
    common_flags += [f"-I{py_inc}", f"-I{CSRC}"]

    hdr_mtime = max(p.stat().st_mtime for p in CSRC.glob("*.h"))
    objs = []
    for src in SOURCES:
        sp = CSRC / src
        op = OBJ_DIR / (src.rsplit(".", 1)[0] + ".o")
        objs.append(op)
        if (not force and op.exists()
                and op.stat().st_mtime > max(sp.stat().st_mtime, hdr_mtime)):
            continue
        if verbose:
            print(f"[kukeon_amd.build] hipcc -c {src}")
        cmd = ["hipcc", "-c", str(sp), "-o", str(op)] + common_flags
        if src.endswith(".cpp"):
            cmd.append("-x")
            cmd.append("c++")
        _run(cmd)

    if force or not SO_PATH.exists() or any(
            o.stat().st_mtime > SO_PATH.stat().st_mtime for o in objs):
        if verbose:
            print(f"[kukeon_amd.build] linking {SO_PATH.name}")
        link = (
            ["hipcc", "-shared", "-fPIC", "-o", str(SO_PATH)]
            + [str(o) for o in objs]
            + [f"-L{d}" for d in libdirs]
            + [
                "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
                "-ltorch_python", "-lamdhip64",
            ]
            + [f"-Wl,-rpath,{d}" for d in libdirs]
        )
        _run(link)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(f"built {SO_PATH}")
