"""In-tree build of the k3samd native components.

Two build products:
  1. ``k3samd/_C*.so`` — the torch HIP extension with the CDNA4 kernels
     (STREAM suite + MFMA), compiled with hipcc for gfx950 only.
  2. ``native/`` C++ binaries/libraries (device plugin, OCI hook, topology
     lib, mi355x-smi) via its Makefile.

Everything is built *in-tree* so the artifacts travel to the GPU box with
the repo snapshot (JIT caches under ~/.cache do not).
"""

from __future__ import annotations

import os
import subprocess
import sys
from pathlib import Path

REPO_ROOT = Path(__file__).resolve().parent.parent
PKG_DIR = Path(__file__).resolve().parent
HIP_SRC = PKG_DIR / "ops" / "hip" / "k3samd_kernels.hip"
EXT_NAME = "_C"
GFX_ARCH = "gfx950"


def _ext_path() -> Path:
    import sysconfig

    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return PKG_DIR / f"{EXT_NAME}{suffix}"


def _needs_rebuild() -> bool:
    so = _ext_path()
    if not so.exists():
        return True
    so_mtime = so.stat().st_mtime
    deps = [HIP_SRC, HIP_SRC.parent / "stream_kernels.h"]
    return any(d.exists() and d.stat().st_mtime > so_mtime for d in deps)


def build_extension(verbose: bool = True) -> Path:
    """Compile the HIP torch extension for gfx950 into k3samd/_C*.so."""
    so = _ext_path()
    if not _needs_rebuild():
        return so

    # torch's generated build.ninja tracks only the .hip source, NOT the
    # included headers — a header-only change leaves ninja convinced the
    # object is fresh and silently re-ships a stale kernel (this bit us:
    # an empty mx_gemm16_kernel survived two rebuilds). Touch the source
    # so ninja always recompiles when we decided a rebuild is needed.
    HIP_SRC.touch()

    os.environ.setdefault("PYTORCH_ROCM_ARCH", GFX_ARCH)
    from torch.utils.cpp_extension import load

    build_dir = REPO_ROOT / "build" / "torch_ext"
    build_dir.mkdir(parents=True, exist_ok=True)
    # name must equal the final import name ("_C") so the .so's PyInit symbol
    # matches `import k3samd._C` after we copy it into the package.
    load(
        name=EXT_NAME,
        sources=[str(HIP_SRC)],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=str(build_dir),
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
    )
    # torch's load() with is_python_module=False loads the .so; we want the
    # artifact in-tree under the package so imports and the gpurun snapshot
    # find it.  Find the built library and copy it.
    built = build_dir / f"{EXT_NAME}.so"
    if not built.exists():
        raise RuntimeError(f"extension build produced no {built}")
    import shutil

    shutil.copy2(built, so)
    return so


def build_native(verbose: bool = True) -> None:
    """Build the native C++ tree (device plugin, hook, topology, smi)."""
    makefile = REPO_ROOT / "native" / "Makefile"
    if not makefile.exists():
        return
    jobs = os.cpu_count() or 4
    subprocess.run(
        ["make", "-C", str(REPO_ROOT / "native"), f"-j{jobs}"],
        check=True,
        stdout=None if verbose else subprocess.DEVNULL,
    )


def build_all(verbose: bool = True) -> None:
    build_extension(verbose=verbose)
    build_native(verbose=verbose)


if __name__ == "__main__":
    build_all()
    print("k3samd build complete:", _ext_path())
