#!/usr/bin/env python3
"""Build mi355._C in-tree for gfx950 (MI355X).

Drives hipcc directly (one TU per kernel file, parallel compile, then link)
so the built .so lives at mi355/_C.so and travels with the repo snapshot to
GPU boxes. Run: python tools/build_ext.py [--force]
"""
from __future__ import annotations

import os
import subprocess
import sys
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
CSRC = REPO / "mi355" / "csrc"
BUILD = REPO / "mi355" / "_ext_build"
OUT = REPO / "mi355" / "_C.so"

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def torch_paths():
    import torch

    t = Path(torch.__file__).parent
    return t / "include", t / "include" / "torch" / "csrc" / "api" / "include", t / "lib"


def build(force: bool = False) -> Path:
    inc1, inc2, libdir = torch_paths()
    import sysconfig

    pyinc = sysconfig.get_paths()["include"]
    sources = sorted(CSRC.glob("kernels/*.hip")) + [CSRC / "bindings.cpp"]
    BUILD.mkdir(parents=True, exist_ok=True)

    common = [
        "-O3", "-std=c++17", "-fPIC", f"--offload-arch={ARCH}",
        "-DTORCH_EXTENSION_NAME=_C", "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-D_GLIBCXX_USE_CXX11_ABI=1", "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        f"-I{inc1}", f"-I{inc2}", f"-I{pyinc}", f"-I{CSRC}",
        "-Wno-unused-result", "-Wno-switch-bool",
    ]

    objs = []
    jobs = []
    for src in sources:
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        if not force and obj.exists() and obj.stat().st_mtime > src.stat().st_mtime \
                and obj.stat().st_mtime > (CSRC / "common.h").stat().st_mtime:
            continue
        # bindings.cpp is host-only but hipcc handles it uniformly
        jobs.append(["hipcc", "-x", "hip", *common, "-c", str(src), "-o", str(obj)])

    def run(cmd):
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            sys.stderr.write(" ".join(cmd) + "\n" + r.stdout[-4000:] + r.stderr[-8000:])
            raise SystemExit(f"compile failed: {cmd[-3]}")
        return cmd[-3]

    if jobs:
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            for done in ex.map(run, jobs):
                print(f"  compiled {Path(done).name}")

    if force or not OUT.exists() or any(o.stat().st_mtime > OUT.stat().st_mtime for o in objs):
        link = [
            "hipcc", "-shared", "-fPIC", *[str(o) for o in objs], "-o", str(OUT),
            f"-L{libdir}", "-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
            "-lc10_hip", "-ltorch_hip", "-lamdhip64", f"-Wl,-rpath,{libdir}",
        ]
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            sys.stderr.write(r.stdout[-2000:] + r.stderr[-8000:])
            raise SystemExit("link failed")
        print(f"  linked {OUT.relative_to(REPO)}")
    return OUT


def build_store(force: bool = False) -> Path:
    """Torch-free native shm store core (g++ + pybind11) -> mi355/_store.so."""
    import sysconfig

    import pybind11

    src = CSRC / "store" / "shmstore.cpp"
    out = REPO / "mi355" / "_store.so"
    if not force and out.exists() and out.stat().st_mtime > src.stat().st_mtime:
        return out
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
           f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}",
           str(src), "-o", str(out), "-lrt"]
    # sanitizer CI target for the native store
    # (MI355_SANITIZE=address or thread)
    san = os.environ.get("MI355_SANITIZE")
    if san:
        cmd[1:1] = [f"-fsanitize={san}", "-g", "-fno-omit-frame-pointer"]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout[-2000:] + r.stderr[-6000:])
        raise SystemExit("store build failed")
    print(f"  built {out.relative_to(REPO)}")
    return out


def build_cpp_module(src_rel: str, name: str, force: bool = False) -> Path:
    import sysconfig

    import pybind11

    src = CSRC / src_rel
    out = REPO / "mi355" / f"{name}.so"
    if not force and out.exists() and out.stat().st_mtime > src.stat().st_mtime:
        return out
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
           f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}",
           str(src), "-o", str(out)]
    san = os.environ.get("MI355_SANITIZE")
    if san:
        cmd[1:1] = [f"-fsanitize={san}", "-g", "-fno-omit-frame-pointer"]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout[-2000:] + r.stderr[-6000:])
        raise SystemExit(f"{name} build failed")
    print(f"  built {out.relative_to(REPO)}")
    return out


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    build_store(force="--force" in sys.argv)
    build_cpp_module("tokenize/unigram_core.cpp", "_tokenize",
                     force="--force" in sys.argv)
    print("OK")
