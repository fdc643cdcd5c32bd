"""Ahead-of-time build of the HIP ops extension (gfx950).

Compiles megatron_amd/ops/csrc/*.{hip,cpp} with hipcc (cross-compiles fine on
machines without a GPU) and links megatron_amd/ops/_C.so in-tree so the .so
travels with repo snapshots. Run: python -m megatron_amd.ops.build
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
BUILD = os.path.join(OPS_DIR, "build")
OUT = os.path.join(OPS_DIR, "_C.so")

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch.utils.cpp_extension as ce

    return ce.include_paths(), ce.library_paths()


def build(verbose: bool = True, force: bool = False) -> str:
    os.makedirs(BUILD, exist_ok=True)
    inc_paths, lib_paths = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]

    sources = sorted(
        os.path.join(CSRC, f)
        for f in os.listdir(CSRC)
        if f.endswith(".hip") or f.endswith(".cpp")
    )
    common_h = os.path.join(CSRC, "common.h")

    cflags = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DTORCH_EXTENSION_NAME=_C", "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1", "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-Wno-unused-result",
    ]
    for p in inc_paths:
        cflags.append(f"-I{p}")
    cflags.append(f"-I{py_inc}")

    objs = []
    jobs = []
    for src in sources:
        obj = os.path.join(
            BUILD, os.path.basename(src).rsplit(".", 1)[0] + ".o"
        )
        objs.append(obj)
        need = force or not os.path.exists(obj) or (
            os.path.getmtime(obj) < os.path.getmtime(src)
            or os.path.getmtime(obj) < os.path.getmtime(common_h)
        )
        if need:
            cmd = ["hipcc", "-c", "-x", "hip", src, "-o", obj] + cflags
            jobs.append(cmd)

    def run(cmd):
        if verbose:
            print("[ops.build]", " ".join(cmd[:4]), "...", flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(
                f"hipcc failed for {cmd[4] if len(cmd) > 4 else cmd}:\n"
                + r.stdout[-4000:] + r.stderr[-4000:]
            )
        return cmd

    with ThreadPoolExecutor(max_workers=min(8, max(1, len(jobs)))) as tp:
        list(tp.map(run, jobs))

    if jobs or force or not os.path.exists(OUT):
        link = (
            ["hipcc", "-shared", "-fPIC", f"--offload-arch={ARCH}"]
            + objs
            + [f"-L{p}" for p in lib_paths]
            + ["-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
               "-ltorch_python", "-lhipblas", "-lhipblaslt", "-lrocblas", "-lamdhip64", "-o", OUT]
        )
        if verbose:
            print("[ops.build] linking _C.so ...", flush=True)
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError("link failed:\n" + r.stdout[-4000:]
                               + r.stderr[-4000:])
    if verbose:
        print(f"[ops.build] built {OUT}", flush=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
