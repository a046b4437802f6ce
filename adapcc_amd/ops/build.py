"""In-tree build of the native engine (hipcc, gfx950 only).

The built ``_core.so`` lives inside the package so it travels with repo
snapshots (gpurun) and is found without installation. Rebuilds only when a
source file changed (content hash stamp).
"""

from __future__ import annotations

import hashlib
import os
import subprocess
import sys
import sysconfig

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))  # adapcc_amd/
CSRC = os.path.join(PKG_DIR, "ops", "csrc")
SOURCES = ["kernels.hip", "engine.hip", "plan.cpp", "lnorm.hip", "ce.hip",
           "attn.hip", "gelu.hip", "bindings.hip"]
HEADERS = ["common.h", "engine.h", "plan.h"]
OUT_SO = os.path.join(PKG_DIR, "_core.so")
STAMP = os.path.join(PKG_DIR, "ops", ".build_stamp")

HIPCC = os.environ.get("ADAPCC_HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _source_hash() -> str:
    h = hashlib.sha256()
    for name in SOURCES + HEADERS:
        with open(os.path.join(CSRC, name), "rb") as f:
            h.update(f.read())
    h.update(ARCH.encode())
    return h.hexdigest()


def needs_build() -> bool:
    if not os.path.exists(OUT_SO):
        return True
    if not os.path.exists(STAMP):
        return True
    with open(STAMP) as f:
        return f.read().strip() != _source_hash()


def build(verbose: bool = True, force: bool = False) -> str:
    if not force and not needs_build():
        return OUT_SO
    import pybind11

    py_inc = sysconfig.get_paths()["include"]
    cmd = [
        HIPCC,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{CSRC}",
        f"-I{pybind11.get_include()}",
        f"-I{py_inc}",
    ]
    cmd += [os.path.join(CSRC, s) for s in SOURCES]
    cmd += ["-o", OUT_SO]
    if verbose:
        print("[adapcc_amd.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    with open(STAMP, "w") as f:
        f.write(_source_hash())
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
