"""In-tree build of the native extensions (no JIT cache outside the repo).

Builds:
  grove_amd/scheduler/_sched.so   — C++ gang-placement core (pybind11, CPU)
  grove_amd/topology/_topo.so     — rocm_smi/KFD topology discovery (pybind11, CPU)
  grove_amd/ops/_gpuwork.so       — HIP MFMA/stream kernels (hipcc, gfx950, torch ext)

hipcc cross-compiles gfx950 without a GPU, so this runs in the CPU container; the .so
files travel with the repo snapshot to GPU boxes.
"""
from __future__ import annotations

import os
import shutil
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG = Path(__file__).resolve().parent.parent
REPO = PKG.parent
GFX_ARCH = "gfx950"


def _run(cmd, **kw):
    print("+", " ".join(str(c) for c in cmd), flush=True)
    subprocess.run([str(c) for c in cmd], check=True, **kw)


def _pybind_includes():
    import pybind11
    return [f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}"]


def _newer(target: Path, sources) -> bool:
    if not target.exists():
        return False
    t = target.stat().st_mtime
    return all(Path(s).stat().st_mtime <= t for s in sources)


def build_sched(force: bool = False) -> Path:
    src = PKG / "scheduler" / "core.cpp"
    out = PKG / "scheduler" / "_sched.so"
    if not force and _newer(out, [src]):
        return out
    _run(["g++", "-O3", "-shared", "-fPIC", "-std=c++17", *_pybind_includes(),
          src, "-o", out])
    return out


def build_topo(force: bool = False) -> Path:
    src = PKG / "topology" / "topo.cpp"
    out = PKG / "topology" / "_topo.so"
    if not force and _newer(out, [src]):
        return out
    _run(["g++", "-O2", "-shared", "-fPIC", "-std=c++17", *_pybind_includes(),
          src, "-ldl", "-o", out])
    return out


def build_gpuwork(force: bool = False) -> Path:
    src = PKG / "ops" / "gpuwork.hip"
    out = PKG / "ops" / "_gpuwork.so"
    if not force and _newer(out, [src]):
        return out
    os.environ.setdefault("PYTORCH_ROCM_ARCH", GFX_ARCH)
    build_dir = PKG / "ops" / "_build"
    build_dir.mkdir(exist_ok=True)
    from torch.utils import cpp_extension
    cpp_extension.load(
        name="_gpuwork",
        sources=[str(src)],
        build_directory=str(build_dir),
        extra_cuda_cflags=[f"--offload-arch={GFX_ARCH}", "-O3"],
        is_python_module=False,
        verbose=True,
    )
    built = build_dir / "_gpuwork.so"
    shutil.copy2(built, out)
    return out


def build_all(force: bool = False) -> None:
    build_sched(force)
    build_topo(force)
    build_gpuwork(force)
    print("native extensions built:",
          [str(p.relative_to(REPO)) for p in
           [PKG / "scheduler" / "_sched.so", PKG / "topology" / "_topo.so",
            PKG / "ops" / "_gpuwork.so"]])


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
