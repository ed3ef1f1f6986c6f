"""Build the in-tree native extensions (no JIT cache — the .so files live
in the package directory so they travel with the source tree).

- ``_amdhal``: plain C++ (g++/amdclang++), links libamd_smi.
- ``_hiphealth``: HIP, compiled for gfx950 only (hipcc cross-compiles
  without a GPU present).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CPP_DIR = os.path.join(PKG_DIR, "cpp")
ROCM = os.environ.get("ROCM_PATH", "/opt/rocm")
GPU_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _ext_suffix() -> str:
    return sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _pybind_includes() -> list:
    import pybind11

    return [f"-I{pybind11.get_include()}", f"-I{sysconfig.get_paths()['include']}"]


def _needs_build(src: str, out: str) -> bool:
    return not os.path.exists(out) or os.path.getmtime(src) > os.path.getmtime(out)


def _run(cmd: list) -> None:
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)


def build_amdhal(force: bool = False) -> str:
    src = os.path.join(CPP_DIR, "amdhal.cpp")
    out = os.path.join(PKG_DIR, f"_amdhal{_ext_suffix()}")
    if force or _needs_build(src, out):
        _run(
            ["g++", "-shared", "-fPIC", "-O2", "-std=c++17"]
            + _pybind_includes()
            + [
                f"-I{ROCM}/include",
                src,
                f"-L{ROCM}/lib",
                "-lamd_smi",
                f"-Wl,-rpath,{ROCM}/lib",
                "-o",
                out,
            ]
        )
    return out


def build_amdhal_asan(force: bool = False) -> str:
    """AddressSanitizer flavor of the amdsmi binding (the
    cgo-shim ASan build the reference's pure-Go stack never needed).
    Run it with:
        LD_PRELOAD=$(gcc -print-file-name=libasan.so) \
        ASAN_OPTIONS=detect_leaks=0 python -c 'import ... _amdhal_asan'
    (leak detection off: the Python interpreter itself 'leaks' at exit).
    """
    src = os.path.join(CPP_DIR, "amdhal.cpp")
    out = os.path.join(PKG_DIR, f"_amdhal_asan{_ext_suffix()}")
    if force or _needs_build(src, out):
        _run(
            [
                "g++",
                "-shared",
                "-fPIC",
                "-g",
                "-O1",
                "-std=c++17",
                "-fsanitize=address",
                "-fno-omit-frame-pointer",
                "-DPYBIND11_MODULE_NAME=_amdhal_asan",
            ]
            + _pybind_includes()
            + [
                f"-I{ROCM}/include",
                src,
                f"-L{ROCM}/lib",
                "-lamd_smi",
                f"-Wl,-rpath,{ROCM}/lib",
                "-o",
                out,
            ]
        )
    return out


def build_hiphealth(force: bool = False) -> str:
    src = os.path.join(CPP_DIR, "hiphealth.hip")
    out = os.path.join(PKG_DIR, f"_hiphealth{_ext_suffix()}")
    if force or _needs_build(src, out):
        hipcc = os.path.join(ROCM, "bin", "hipcc")
        if not os.path.exists(hipcc):
            hipcc = "hipcc"
        _run(
            [
                hipcc,
                f"--offload-arch={GPU_ARCH}",
                "-shared",
                "-fPIC",
                "-O3",
                "-std=c++17",
            ]
            + _pybind_includes()
            + [src, "-o", out]
        )
    return out


def build(force: bool = False) -> None:
    build_amdhal(force)
    build_hiphealth(force)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
