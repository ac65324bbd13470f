# Planner package: the pybind rebuild of the reference's
# oobleck.csrc.planning.pipeline_template module (see csrc/
# pipeline_template.cpp header for why it is rebuilt rather than compiled
# verbatim).  Import surface matches pipeline_template.pyi, so reference
# consumers swap `from oobleck.csrc.planning import pipeline_template` for
# `from oobleck_amd.planning import pipeline_template`.
from __future__ import annotations

import pathlib
import subprocess
import sys
import sysconfig

_PKG = pathlib.Path(__file__).resolve().parent
_SRC = _PKG / "csrc" / "pipeline_template.cpp"


def _so_path() -> pathlib.Path:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return _PKG / f"pipeline_template{suffix}"


def build(force: bool = False) -> pathlib.Path:
    so = _so_path()
    if not force and so.exists() and so.stat().st_mtime > _SRC.stat().st_mtime:
        return so
    import pybind11
    inc_py = sysconfig.get_paths()["include"]
    cmd = ["g++", "-O2", "-std=c++20", "-shared", "-fPIC",
           f"-I{pybind11.get_include()}", f"-I{inc_py}", str(_SRC),
           "-o", str(so)]
    subprocess.run(cmd, check=True, capture_output=True, text=True)
    return so


def load():
    build()
    if str(_PKG) not in sys.path:
        sys.path.insert(0, str(_PKG))
    import pipeline_template  # noqa: F401  (extension module)
    return pipeline_template
