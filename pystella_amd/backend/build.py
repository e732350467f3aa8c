"""In-tree build of the native extension (hipcc, gfx950).

The extension is torch-free (raw pointers + streams cross the boundary),
so it builds with plain hipcc in seconds and the resulting
``pystella_amd/_C.so`` travels with the source tree to GPU machines.
"""

from __future__ import annotations

import os
import subprocess
import sysconfig

PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CSRC = os.path.join(PKG_DIR, "csrc")
SO_PATH = os.path.join(PKG_DIR, "_C.so")

SOURCES = ["module.cpp", "derivs.hip", "tt_mfma.hip"]
ARCH = os.environ.get("PYSTELLA_ROCM_ARCH",
                      os.environ.get("PYTORCH_ROCM_ARCH", "gfx950"))


def _hipcc():
    for cand in (os.environ.get("HIPCC"), "/opt/rocm/bin/hipcc", "hipcc"):
        if cand and (os.path.exists(cand) or cand == "hipcc"):
            return cand
    raise RuntimeError("hipcc not found")


def needs_rebuild():
    if not os.path.exists(SO_PATH):
        return True
    so_mtime = os.path.getmtime(SO_PATH)
    for src in SOURCES:
        if os.path.getmtime(os.path.join(CSRC, src)) > so_mtime:
            return True
    return False


def build_extension(force=False, verbose=True):
    """Compile csrc/ into pystella_amd/_C.so for gfx950."""
    if not force and not needs_rebuild():
        return SO_PATH
    import pybind11
    cmd = [
        _hipcc(), "-O3", "-std=c++17", "-fPIC", "-shared",
        f"--offload-arch={ARCH}",
        "-Wno-unused-result",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
    ]
    cmd += [os.path.join(CSRC, s) for s in SOURCES]
    cmd += ["-L/opt/rocm/lib", "-lhiprtc", "-lamdhip64", "-o", SO_PATH]
    if verbose:
        print("[pystella_amd] building native extension:",
              " ".join(cmd), flush=True)
    subprocess.check_call(cmd)
    return SO_PATH


if __name__ == "__main__":
    build_extension(force="--force" in os.sys.argv)
