"""In-tree native build: C++ ledger ext + gfx950 HIP ops ext.

Everything is built into the package directory (not a JIT cache) so the
built .so files travel with the repo snapshot to the GPU box.
hipcc cross-compiles gfx950 on a CPU-only machine, so build() works
everywhere.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
PKG = REPO / "bflc_amd"
CSRC = REPO / "csrc"
EXT_SUFFIX = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _newer(target: Path, sources: list[Path]) -> bool:
    if not target.exists():
        return False
    t = target.stat().st_mtime
    return all(s.stat().st_mtime < t for s in sources)


def _run(cmd: list[str]) -> None:
    print("[bflc build]", " ".join(str(c) for c in cmd), flush=True)
    subprocess.check_call([str(c) for c in cmd])


def _py_includes() -> list[str]:
    import pybind11
    return [
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{pybind11.get_include()}",
    ]


def build_ledger(force: bool = False) -> Path:
    """Compile csrc/ledger.cpp -> bflc_amd/_ledger<EXT_SUFFIX>."""
    src = CSRC / "ledger.cpp"
    out = PKG / f"_ledger{EXT_SUFFIX}"
    if not force and _newer(out, [src]):
        return out
    cmd = (
        ["g++", "-O2", "-g0", "-shared", "-fPIC", "-std=c++17",
         "-fvisibility=hidden"]
        + _py_includes()
        + [str(src), "-o", str(out)]
    )
    _run(cmd)
    return out


def _torch_paths() -> tuple[list[str], list[str], list[str]]:
    """(include flags, lib dir flags, libs) for linking a torch extension."""
    import torch
    from torch.utils import cpp_extension as ce

    incs = [f"-I{p}" for p in ce.include_paths(device_type="cuda")]
    libdirs = []
    for p in ce.library_paths(device_type="cuda"):
        libdirs += [f"-L{p}", f"-Wl,-rpath,{p}"]
    libs = ["-ltorch", "-ltorch_cpu", "-ltorch_python", "-lc10",
            "-ltorch_hip", "-lc10_hip", "-lamdhip64"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    incs.append(f"-D_GLIBCXX_USE_CXX11_ABI={abi}")
    return incs, libdirs, libs


HIP_SOURCES = [
    "elementwise.hip",
    "softmax_ce.hip",
    "gemm_bf16.hip",
    "conv2d.hip",
    "batchnorm.hip",
    "reduce.hip",
    "bindings.cpp",
]


def build_hip_ops(force: bool = False) -> Path | None:
    """Compile csrc/hip/*.hip -> bflc_amd/_hip_ops<EXT_SUFFIX> (gfx950).

    Uses hipcc for every translation unit (hipcc is clang and compiles the
    pybind11/torch binding .cpp too). Returns None if the HIP sources do
    not exist yet.
    """
    hipdir = CSRC / "hip"
    srcs = [hipdir / s for s in HIP_SOURCES if (hipdir / s).exists()]
    if not srcs:
        return None
    out = PKG / f"_hip_ops{EXT_SUFFIX}"
    if not force and _newer(out, srcs):
        return out

    incs, libdirs, libs = _torch_paths()
    objdir = REPO / "build" / "hip"
    objdir.mkdir(parents=True, exist_ok=True)

    common = (
        [f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17", "-fPIC",
         "-fvisibility=hidden", "-DNDEBUG",
         "-DTORCH_EXTENSION_NAME=_hip_ops",
         "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
         "-DHIPBLAS_V2", "-fno-gpu-rdc",
         "-Wno-unused-result", "-Wno-switch-bool"]
        + incs
        + _py_includes()
    )
    objs = []
    for s in srcs:
        o = objdir / (s.name + ".o")
        objs.append(o)
        if _newer(o, [s]) and not force:
            continue
        lang = [] if s.suffix == ".hip" else ["-x", "c++"]
        _run([HIPCC, "-c", *lang, str(s), "-o", str(o)] + common)

    _run([HIPCC, "-shared", "-fPIC", f"--offload-arch={GFX_ARCH}"]
         + [str(o) for o in objs]
         + libdirs + libs + ["-o", str(out)])
    return out


def build_all(force: bool = False) -> None:
    build_ledger(force=force)
    build_hip_ops(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
