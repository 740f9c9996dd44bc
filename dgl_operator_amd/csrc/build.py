"""In-tree build of the native extension: hipcc --offload-arch=gfx950.

Produces dgl_operator_amd/_C.so next to the package sources so the built
artifact travels with repo snapshots (gpurun) without a JIT cache.

Usage:  python -m dgl_operator_amd.csrc.build  [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

CSRC = Path(__file__).resolve().parent
PKG = CSRC.parent
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "hipcc")

SOURCES = [
    "gnn_ops.hip",
    "sampling.hip",
    "adagrad.hip",
    "kge.hip",
    "gather_mm.hip",
    "ldg.cpp",
    "bindings.cpp",
]


def _torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    includes = ce.include_paths()
    lib_dir = os.path.join(os.path.dirname(torch.__file__), "lib")
    return includes, lib_dir


def _needs_build(obj: Path, src: Path) -> bool:
    if not obj.exists():
        return True
    deps = [src, CSRC / "common.h", CSRC / "build.py"]
    return any(d.exists() and d.stat().st_mtime > obj.stat().st_mtime for d in deps)


def build(force: bool = False, verbose: bool = True) -> Path:
    includes, torch_lib = _torch_paths()
    py_include = sysconfig.get_paths()["include"]
    out_so = PKG / "_C.so"
    objdir = CSRC / ".obj"
    objdir.mkdir(exist_ok=True)

    cflags = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-fno-gpu-rdc",
        "-munsafe-fp-atomics",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DTORCH_EXTENSION_NAME=_C",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-Wno-unused-result",
    ] + [f"-I{p}" for p in includes + [py_include, str(CSRC)]]

    def compile_one(src_name: str):
        src = CSRC / src_name
        if not src.exists():
            return None
        obj = objdir / (src_name.replace(".", "_") + ".o")
        if not force and not _needs_build(obj, src):
            return obj
        cmd = [HIPCC, "-c", str(src), "-o", str(obj)] + cflags + ["-x", "hip"]
        if verbose:
            print("[build]", src_name)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed on {src_name}:\n{r.stdout}\n{r.stderr}")
        return obj

    failures = []

    def compile_safe(name):
        try:
            return compile_one(name)
        except Exception as e:  # noqa: BLE001
            failures.append(str(e))
            return None

    with ThreadPoolExecutor(max_workers=8) as pool:
        objs = [o for o in pool.map(compile_safe, SOURCES) if o is not None]
    if failures:
        raise RuntimeError(
            "native build FAILED:\n" + "\n\n".join(failures)
        )
    missing = [s for s in SOURCES if (CSRC / s).exists()] 
    if len(objs) != len(missing):
        raise RuntimeError("native build: object count mismatch")

    if force or not out_so.exists() or any(
        o.stat().st_mtime > out_so.stat().st_mtime for o in objs
    ):
        link = (
            [HIPCC, "-shared", "-fPIC", "-o", str(out_so)]
            + [str(o) for o in objs]
            + [
                f"-L{torch_lib}",
                f"-Wl,-rpath,{torch_lib}",
                "-ltorch",
                "-ltorch_cpu",
                "-lc10",
                "-ltorch_hip",
                "-lc10_hip",
                "-ltorch_python",
            ]
        )
        if verbose:
            print("[link]", out_so.name)
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    return out_so


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print("built", PKG / "_C.so")
