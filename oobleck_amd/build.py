# In-tree build of the HIP extension (hipcc cross-compiles gfx950 without a
# GPU; the built .so travels to the GPU box with the repo snapshot).
from __future__ import annotations

import pathlib
import subprocess

PKG_DIR = pathlib.Path(__file__).resolve().parent
SO_PATH = PKG_DIR / "libob_stage.so"
SOURCES = [PKG_DIR / "csrc" / "ob_kernels.hip",
           PKG_DIR / "csrc" / "ob_kernels_bf16.hip",
           PKG_DIR / "csrc" / "ob_blaslt.hip",
           PKG_DIR / "csrc" / "ob_layer.hip"]
HEADERS = [PKG_DIR.parent / "include" / "oobleck_stage.h",
           PKG_DIR / "csrc" / "ob_internal.h"]


def needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    return any(p.stat().st_mtime > so_mtime for p in SOURCES + HEADERS)


def build(force: bool = False) -> pathlib.Path:
    if force or needs_rebuild():
        cmd = ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
               "-shared"] + [str(s) for s in SOURCES] + \
              ["-L/opt/rocm/lib", "-lhipblaslt", "-o", str(SO_PATH)]
        subprocess.run(cmd, check=True, capture_output=True, text=True)
    return SO_PATH


if __name__ == "__main__":
    print(build(force=True))
