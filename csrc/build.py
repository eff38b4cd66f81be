#!/usr/bin/env python3
"""Build the gfx950 HIP extension (hyperspot._C) in-tree.

Drives hipcc directly — sources are native HIP/CDNA4, no hipify pass, one
offload arch (gfx950).  The built .so lands in hyperspot/ so it travels with
the repo snapshot to GPU boxes (JIT caches under ~/.cache do not).

Usage: python csrc/build.py [--force] [--verbose]
"""

from __future__ import annotations

import argparse
import concurrent.futures as cf
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

import torch
from torch.utils import cpp_extension as ce

ROOT = Path(__file__).resolve().parent.parent
CSRC = ROOT / "csrc"
OUT_DIR = CSRC / "build"
SO_PATH = ROOT / "hyperspot" / ("_C" + sysconfig.get_config_var("EXT_SUFFIX"))

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

COMMON = [
    "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
    "-DNDEBUG", "-Wno-unused-result",
]
TORCH_DEFS = [
    "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1", "-DHIPBLAS_V2",
    "-DTORCH_EXTENSION_NAME=_C", "-DTORCH_API_INCLUDE_EXTENSION_H",
    f"-D_GLIBCXX_USE_CXX11_ABI={int(torch.compiled_with_cxx11_abi())}",
]


def _includes():
    incs = ce.include_paths("cuda") + [sysconfig.get_paths()["include"]]
    return [f"-I{p}" for p in incs]


def _needs_build(src: Path, obj: Path) -> bool:
    if not obj.exists():
        return True
    deps = [src, CSRC / "common.h", Path(__file__)]
    return any(d.stat().st_mtime > obj.stat().st_mtime for d in deps)


def _compile(src: Path, extra, verbose=False) -> Path:
    obj = OUT_DIR / (src.stem + ".o")
    if not _needs_build(src, obj):
        return obj
    cmd = [HIPCC, "-c", str(src), "-o", str(obj)] + COMMON + extra
    if verbose:
        print(" ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout + r.stderr)
        raise SystemExit(f"hipcc failed on {src.name}")
    return obj


def build(force: bool = False, verbose: bool = False) -> Path:
    OUT_DIR.mkdir(exist_ok=True)
    if force:
        for f in OUT_DIR.glob("*.o"):
            f.unlink()
    kernels = sorted(CSRC.glob("*.hip"))
    incs = _includes()
    with cf.ThreadPoolExecutor(max_workers=8) as ex:
        futs = [ex.submit(_compile, k, [], verbose) for k in kernels]
        futs.append(ex.submit(_compile, CSRC / "bindings.cpp",
                              TORCH_DEFS + incs, verbose))
        objs = [f.result() for f in futs]
    lib_dir = ce.library_paths("cuda")[0]
    if SO_PATH.exists() and not force and \
            all(o.stat().st_mtime <= SO_PATH.stat().st_mtime for o in objs):
        return SO_PATH
    cmd = ([HIPCC, "-shared", "-fPIC"] + [str(o) for o in objs] +
           [f"-L{lib_dir}", "-ltorch", "-ltorch_cpu", "-ltorch_hip",
            "-lc10", "-lc10_hip", "-ltorch_python",
            f"-Wl,-rpath,{lib_dir}", "-L/opt/rocm/lib", "-lamdhip64",
            "-o", str(SO_PATH)])
    if verbose:
        print(" ".join(cmd))
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout + r.stderr)
        raise SystemExit("link failed")
    return SO_PATH


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--force", action="store_true")
    ap.add_argument("--verbose", action="store_true")
    a = ap.parse_args()
    print(build(force=a.force, verbose=a.verbose))
