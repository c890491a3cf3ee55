"""Build + ctypes access for the C API shared library."""

from __future__ import annotations

import os
import subprocess
import sysconfig

_THIS = os.path.dirname(os.path.abspath(__file__))
_SO = os.path.join(_THIS, "libmultiverso_amd.so")


def build(verbose: bool = False) -> str:
    """Compile libmultiverso_amd.so (C API embedding the Python runtime)."""
    import pybind11
    src = os.path.join(_THIS, "c_api.cpp")
    if (os.path.exists(_SO)
            and os.path.getmtime(_SO) > os.path.getmtime(src)):
        return _SO
    py_inc = sysconfig.get_paths()["include"]
    ldlib = sysconfig.get_config_var("LDLIBRARY") or ""
    libdir = sysconfig.get_config_var("LIBDIR") or "/usr/lib"
    pyver = f"python{sysconfig.get_python_version()}"
    cmd = ["g++", "-O2", "-std=c++17", "-shared", "-fPIC", src,
           f"-I{py_inc}", f"-I{pybind11.get_include()}",
           f"-L{libdir}", f"-l{pyver}", "-o", _SO]
    if verbose:
        print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return _SO


def so_path() -> str:
    return _SO
