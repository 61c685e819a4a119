"""In-tree build of triton_dist_amd._C with hipcc for gfx950.

Direct hipcc invocation (no torch cpp_extension): the module exchanges
tensors with torch via DLPack, so it links only against amdhip64 and builds
in seconds without a GPU (cross-compile). The resulting .so lives in-tree so
it travels to GPU boxes with the repo snapshot.
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_DIR = Path(__file__).resolve().parent
CSRC = PKG_DIR / "csrc"
ARCH = os.environ.get("TD_GPU_ARCH", "gfx950")


def _sources() -> list[Path]:
    return [CSRC / "module.hip"] + sorted((CSRC / "kernels").glob("*.hip"))


def _so_path() -> Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return PKG_DIR / f"_C{suffix}"


def _needs_build(so: Path, srcs: list[Path]) -> bool:
    if not so.exists():
        return True
    so_mtime = so.stat().st_mtime
    deps = srcs + list((CSRC / "include" / "td").glob("*.hpp")) + [Path(__file__)]
    return any(p.stat().st_mtime > so_mtime for p in deps)


def build(force: bool = False, verbose: bool = True) -> Path:
    import pybind11

    srcs = _sources()
    so = _so_path()
    if not force and not _needs_build(so, srcs):
        return so
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        hipcc,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        "-Wno-unused-result",
        f"-I{CSRC / 'include'}",
        f"-I{py_include}",
        f"-I{pybind11.get_include()}",
        *[str(s) for s in srcs],
        "-o",
        str(so),
    ]
    if verbose:
        print("[triton_dist_amd.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return so


if __name__ == "__main__":
    build(force="--force" in sys.argv)
