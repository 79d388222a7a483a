"""In-tree builds of the native extensions.

Built .so files live next to the sources (they travel with gpurun snapshots
and are git-ignored), so a GPU box needs no JIT cache.  Used by
__graft_entry__.build() and the Makefile.
"""

from __future__ import annotations

import os
import subprocess
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _python_includes() -> list:
    import pybind11

    return [
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{pybind11.get_include()}",
    ]


def _needs_build(src: str, out: str) -> bool:
    if not os.path.exists(out):
        return True
    return os.path.getmtime(src) > os.path.getmtime(out)


def _run(cmd: list) -> None:
    proc = subprocess.run(cmd, capture_output=True, text=True)
    if proc.returncode != 0:
        raise RuntimeError(
            f"build failed: {' '.join(cmd)}\n--- stdout ---\n{proc.stdout}"
            f"\n--- stderr ---\n{proc.stderr}"
        )


def build_drmctl(force: bool = False) -> str:
    src = os.path.join(PKG_DIR, "drmctl.cpp")
    out = os.path.join(PKG_DIR, "_drmctl.so")
    if force or _needs_build(src, out):
        _run(
            ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", src, "-o", out]
            + _python_includes()
        )
    return out


def build_healthprobe(force: bool = False) -> str:
    src = os.path.join(PKG_DIR, "health_probe.hip")
    out = os.path.join(PKG_DIR, "_healthprobe.so")
    if force or _needs_build(src, out):
        _run(
            [
                HIPCC,
                f"--offload-arch={GFX_ARCH}",
                "-O3",
                "-std=c++17",
                "-shared",
                "-fPIC",
                src,
                "-o",
                out,
            ]
            + _python_includes()
        )
    return out


def build_fastserver(force: bool = False) -> str:
    src = os.path.join(PKG_DIR, "fastserver.cpp")
    out = os.path.join(PKG_DIR, "_fastserver.so")
    hdr = os.path.join(PKG_DIR, "nghttp2_abi.h")
    if force or _needs_build(src, out) or _needs_build(hdr, out):
        _run(
            ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", src, "-o", out,
             "-ldl", "-pthread"]
            + _python_includes()
        )
    return out


def build_h2tool(force: bool = False) -> str:
    src = os.path.join(PKG_DIR, "h2tool.cpp")
    out = os.path.join(PKG_DIR, "_h2tool.so")
    hdr = os.path.join(PKG_DIR, "nghttp2_abi.h")
    if force or _needs_build(src, out) or _needs_build(hdr, out):
        _run(
            ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", src, "-o", out,
             "-ldl"]
            + _python_includes()
        )
    return out


def build_all(force: bool = False) -> list:
    return [build_drmctl(force), build_healthprobe(force),
            build_fastserver(force), build_h2tool(force)]


if __name__ == "__main__":
    import sys

    force = "--force" in sys.argv
    for so in build_all(force):
        print(so)
