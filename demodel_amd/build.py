"""In-tree builds of the native extensions.

Two extensions, built straight with the toolchain (no setuptools overhead),
.so files landing inside the package so they travel with a repo snapshot:

* ``demodel_amd/_native.so``  — CPU helpers (libcrypto cert minting).
  Plain C++, no torch, builds and runs anywhere.
* ``demodel_amd/_hip.so``     — the GPU pipeline: pinned-ring allocator,
  HIP streams/events, and the CDNA4 kernels (SHA-256, inflate, scatter,
  GGUF dequant).  Compiled with hipcc for gfx950 only; importable on a
  CPU-only box (calls fail until a GPU is present).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(PKG_DIR, "csrc")

GFX_ARCH = os.environ.get("DEMODEL_GFX_ARCH", "gfx950")


def _py_includes() -> list[str]:
    import pybind11

    return [
        "-I" + sysconfig.get_paths()["include"],
        "-I" + pybind11.get_include(),
    ]


def _newer(target: str, sources: list[str]) -> bool:
    if not os.path.exists(target):
        return False
    t = os.path.getmtime(target)
    deps = list(sources) + [__file__]
    return all(os.path.getmtime(s) < t for s in deps if os.path.exists(s))


def _run(cmd: list[str]) -> None:
    print("[demodel-amd build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)


def build_native(force: bool = False) -> str:
    out = os.path.join(PKG_DIR, "_native.so")
    srcs = [os.path.join(CSRC, "certs.cpp"),
            os.path.join(CSRC, "zstd_host.cpp")]
    if not force and _newer(out, srcs):
        return out
    cmd = (
        ["c++", "-O2", "-std=c++17", "-shared", "-fPIC",
         "-fvisibility=hidden"]
        + _py_includes()
        + srcs
        + ["-lcrypto", "-o", out]
    )
    _run(cmd)
    return out


HIP_SRCS = [
    "hip_core.cpp",
    "sha256.hip",
    "scatter.hip",
    "gguf_dequant.hip",
    "inflate.hip",
    "zstd_kernel.hip",
    "snappy.hip",
    "lz4.hip",
]


def build_hip(force: bool = False) -> str:
    out = os.path.join(PKG_DIR, "_hip.so")
    srcs = [os.path.join(CSRC, s) for s in HIP_SRCS
            if os.path.exists(os.path.join(CSRC, s))]
    if not srcs:
        raise RuntimeError("no HIP sources present")
    if not force and _newer(out, srcs):
        return out
    hipcc = os.environ.get("HIPCC", "hipcc")
    objs = []
    os.makedirs(os.path.join(PKG_DIR, "_build"), exist_ok=True)
    for src in srcs:
        obj = os.path.join(
            PKG_DIR, "_build", os.path.basename(src).rsplit(".", 1)[0] + ".o"
        )
        if not _newer(obj, [src]):
            cmd = (
                [hipcc, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17",
                 "-fPIC", "-fvisibility=hidden", "-c", src, "-o", obj]
                + _py_includes()
            )
            _run(cmd)
        objs.append(obj)
    _run([hipcc, f"--offload-arch={GFX_ARCH}", "-shared", "-fPIC"]
         + objs + ["-lamdhip64", "-o", out])
    return out


def build_all(force: bool = False) -> None:
    build_native(force=force)
    if any(os.path.exists(os.path.join(CSRC, s)) for s in HIP_SRCS):
        build_hip(force=force)


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
