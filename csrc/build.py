"""In-tree build driver for paddle_amd._C (gfx950 HIP extension).

Compiles kernels/*.hip with hipcc --offload-arch=gfx950 (cross-compiles
without a GPU), bindings.cpp with g++ against the torch headers, and
links paddle_amd/_C.so.  Timestamp-cached; `python csrc/build.py` or
paddle_amd._ext.build() both drive it.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

CSRC = Path(__file__).resolve().parent
REPO = CSRC.parent
OUT = CSRC / "build"
TARGET = REPO / "paddle_amd" / "_C.so"
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
HIPCC = str(ROCM / "bin" / "hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    from torch.utils import cpp_extension as ce

    includes = ce.include_paths(device_type="cuda")
    lib_dir = str(Path(torch.__file__).parent / "lib")
    return includes, lib_dir


def _newer(src: Path, obj: Path, deps=()) -> bool:
    if not obj.exists():
        return True
    ot = obj.stat().st_mtime
    if src.stat().st_mtime > ot:
        return True
    return any(Path(d).stat().st_mtime > ot for d in deps)


def _run(cmd):
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(
            f"build command failed:\n{' '.join(str(c) for c in cmd)}\n{r.stdout}\n{r.stderr}"
        )
    return r


def build(verbose: bool = False) -> Path:
    OUT.mkdir(exist_ok=True)
    includes, torch_lib = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    deps = [CSRC / "kernels" / "api.h", CSRC / "common.h"]

    hip_sources = sorted((CSRC / "kernels").glob("*.hip"))
    objs = []
    jobs = []
    for src in hip_sources:
        obj = OUT / (src.stem + ".hip.o")
        objs.append(obj)
        if _newer(src, obj, deps):
            cmd = [HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
                   "-ffast-math", f"-I{CSRC}", "-c", str(src), "-o", str(obj)]
            jobs.append(cmd)

    for cpp_name in ("bindings.cpp", "blaslt.cpp"):
        cpp_src = CSRC / cpp_name
        cpp_obj = OUT / (cpp_name[:-4] + ".o")
        objs.append(cpp_obj)
        if _newer(cpp_src, cpp_obj, deps):
            cmd = ["g++", "-O2", "-std=c++17", "-fPIC", "-D__HIP_PLATFORM_AMD__=1",
                   "-DUSE_ROCM=1", "-DTORCH_EXTENSION_NAME=_C",
                   "-D_GLIBCXX_USE_CXX11_ABI=1",
                   f"-I{CSRC}", f"-I{py_inc}", f"-I{ROCM / 'include'}"]
            cmd += [f"-I{i}" for i in includes]
            cmd += ["-c", str(cpp_src), "-o", str(cpp_obj)]
            jobs.append(cmd)

    if jobs:
        with ThreadPoolExecutor(max_workers=min(8, len(jobs))) as ex:
            list(ex.map(_run, jobs))

    if any(_newer(o, TARGET) for o in objs):
        link = [HIPCC, "-shared", "-fPIC", "-o", str(TARGET)]
        link += [str(o) for o in objs]
        link += [f"-L{torch_lib}", "-ltorch", "-ltorch_python", "-lc10",
                 "-ltorch_hip", "-lc10_hip", f"-Wl,-rpath,{torch_lib}",
                 f"-L{ROCM / 'lib'}", "-lamdhip64", "-lhipblaslt"]
        _run(link)
    if verbose:
        print(f"built {TARGET}")
    return TARGET


if __name__ == "__main__":
    build(verbose=True)
