"""Build the C++ index-builder extension (CPU, pybind11)."""

from __future__ import annotations

import os
import subprocess
import sysconfig

PKG = os.path.dirname(os.path.abspath(__file__))
SO = os.path.join(PKG, "helpers_cpp" + sysconfig.get_config_var("EXT_SUFFIX"))


def build(force: bool = False) -> str:
    src = os.path.join(PKG, "helpers.cpp")
    if not force and os.path.exists(SO) and os.path.getmtime(SO) > os.path.getmtime(src):
        return SO
    import pybind11

    cmd = [
        "g++", "-O3", "-std=c++17", "-shared", "-fPIC",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        src, "-o", SO,
    ]
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        raise RuntimeError(f"helpers_cpp build failed:\n{r.stderr}")
    return SO


def load_helpers():
    if not os.path.exists(SO):
        build()
    import importlib.util

    spec = importlib.util.spec_from_file_location("helpers_cpp", SO)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod
