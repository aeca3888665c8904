#!/usr/bin/env python3
"""In-tree build of the gfx950 HIP extension.

Compiles every distributed_training_guide_amd/_hip/*.hip kernel file and the
torch bindings with hipcc (no hipify, no CUDA path), links them into
distributed_training_guide_amd/_C.so.  hipcc cross-compiles gfx950 without a
GPU present, so this runs anywhere the ROCm toolchain exists.

Incremental: objects are rebuilt only when their source (or common.h) is
newer.  Kernel files compile in parallel.
"""
import concurrent.futures as cf
import os
import subprocess
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
PKG = REPO / "distributed_training_guide_amd"
HIP_DIR = PKG / "_hip"
BUILD = REPO / "build" / "hip"
OUT_SO = PKG / "_C.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch  # noqa: F401
    import torch.utils.cpp_extension as ce

    return ce.include_paths(), ce.library_paths()


def _common_flags(includes):
    import sysconfig

    flags = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-DUSE_ROCM=1",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=_C",
    ]
    if os.environ.get("DTGA_HIP_FLAGS"):
        flags += os.environ["DTGA_HIP_FLAGS"].split()
    for inc in includes:
        flags.append(f"-I{inc}")
    flags.append(f"-I{sysconfig.get_paths()['include']}")
    return flags


def _needs_build(src: Path, obj: Path, extra_deps=()):
    if not obj.exists():
        return True
    mt = obj.stat().st_mtime
    deps = [src, HIP_DIR / "common.h", *extra_deps]
    return any(d.exists() and d.stat().st_mtime > mt for d in deps)


def _compile(src: Path, obj: Path, flags):
    cmd = ["hipcc", "-c", str(src), "-o", str(obj)] + flags
    print("  [hipcc]", src.name, flush=True)
    subprocess.run(cmd, check=True)


def build(verbose=True):
    BUILD.mkdir(parents=True, exist_ok=True)
    includes, libpaths = _torch_paths()
    flags = _common_flags(includes)

    sources = sorted(HIP_DIR.glob("*.hip")) + [HIP_DIR / "bindings.cpp"]
    jobs = []
    objs = []
    for src in sources:
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        if _needs_build(src, obj):
            jobs.append((src, obj))

    if jobs:
        nproc = min(len(jobs), os.cpu_count() or 4)
        with cf.ThreadPoolExecutor(nproc) as ex:
            futs = [ex.submit(_compile, s, o, flags) for s, o in jobs]
            for f in futs:
                f.result()

    if jobs or not OUT_SO.exists():
        link = ["hipcc", "-shared", "-fPIC", "-o", str(OUT_SO)]
        link += [str(o) for o in objs]
        for lp in libpaths:
            link += [f"-L{lp}", f"-Wl,-rpath,{lp}"]
        link += ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip",
                 "-ltorch_python", "-lamdhip64"]
        print("  [link]", OUT_SO.name, flush=True)
        subprocess.run(link, check=True)
    if verbose:
        print(f"built {OUT_SO}")
    return OUT_SO


if __name__ == "__main__":
    build()
