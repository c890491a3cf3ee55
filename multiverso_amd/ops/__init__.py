"""In-tree HIP extension loader/builder for gfx950.

The compute path is hand-written CDNA4 HIP (ops/csrc/kernels.hip) compiled
by hipcc with --offload-arch=gfx950 — no hipify, no CUDA shims. The torch
binding layer (ops/csrc/bindings.cpp) is compiled by torch.utils.cpp_extension
and links the hipcc-built object, so tensors flow straight into the kernels.

The built .so lives IN-TREE (multiverso_amd/ops/_build/_mv_hip.so) so it
travels with the repo snapshot to GPU boxes. On a GPU host a missing
extension is a hard error — table ops never silently fall back to eager.
"""

from __future__ import annotations

import os
import subprocess
import sys
from typing import Optional

_THIS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_THIS_DIR, "csrc")
_BUILD = os.path.join(_THIS_DIR, "_build")
_MOD_NAME = "_mv_hip"

_mod = None
_load_error: Optional[str] = None


def _so_path() -> str:
    return os.path.join(_BUILD, _MOD_NAME + ".so")


def build(verbose: bool = False) -> str:
    """Compile the HIP kernels + torch bindings for gfx950, in-tree."""
    import torch  # noqa: F401  (must be imported before cpp_extension)
    from torch.utils import cpp_extension

    os.makedirs(_BUILD, exist_ok=True)
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    rocm_home = os.environ.get("ROCM_HOME", "/opt/rocm")

    kernels_src = os.path.join(_CSRC, "kernels.hip")
    kernels_obj = os.path.join(_BUILD, "kernels.hip.o")
    # Rebuild the pure-HIP object only when the source is newer.
    if (not os.path.exists(kernels_obj)
            or os.path.getmtime(kernels_src) > os.path.getmtime(kernels_obj)):
        cmd = ["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
               "-c", kernels_src, "-o", kernels_obj]
        if verbose:
            print("+", " ".join(cmd), flush=True)
        subprocess.run(cmd, check=True)

    # cpp_extension's ninja rules only see bindings.cpp; kernels.hip.o is
    # an opaque ldflag, so a kernel-only change would leave a STALE .so
    # (observed: an optimized kernel measuring identical to the old one).
    # Removing the out-of-date .so forces the relink.
    so = _so_path()
    if (os.path.exists(so)
            and os.path.getmtime(kernels_obj) > os.path.getmtime(so)):
        os.remove(so)

    mod = cpp_extension.load(
        name=_MOD_NAME,
        sources=[os.path.join(_CSRC, "bindings.cpp")],
        extra_cflags=["-O3", "-D__HIP_PLATFORM_AMD__=1"],
        extra_include_paths=[os.path.join(rocm_home, "include")],
        extra_ldflags=[kernels_obj,
                       f"-L{os.path.join(rocm_home, 'lib')}", "-lamdhip64"],
        build_directory=_BUILD,
        verbose=verbose,
    )
    global _mod
    _mod = mod
    return _so_path()


def module(required: bool = False):
    """Return the loaded extension module (or None on CPU-only hosts)."""
    global _mod, _load_error
    if _mod is not None:
        return _mod
    import torch  # noqa: F401 — torch symbols must be resolvable first
    so = _so_path()
    if os.path.exists(so):
        try:
            import importlib.util
            spec = importlib.util.spec_from_file_location(_MOD_NAME, so)
            m = importlib.util.module_from_spec(spec)
            spec.loader.exec_module(m)
            _mod = m
            sys.modules.setdefault(_MOD_NAME, m)
            return _mod
        except Exception as e:  # pragma: no cover
            _load_error = f"{type(e).__name__}: {e}"
    if required:
        raise RuntimeError(
            f"multiverso_amd HIP extension not available at {so} "
            f"(load error: {_load_error}). Run __graft_entry__.build() "
            "or multiverso_amd.ops.build(); GPU table ops refuse to run "
            "without the native gfx950 kernels.")
    return None


def scatter_add_rows(dst, idx, src, alpha: float = 1.0):
    """dst[idx[i], :] += alpha * src[i, :] with duplicate-index
    accumulation. On GPU this dispatches to k_row_scatter_add (atomic,
    thread-per-element): torch's index_add_ falls into its
    indexFuncLargeIndex path for few-column rows, measured 4-5x slower
    on gfx950 (profiles/round1_pmc.md). CPU falls back to index_add_."""
    import torch
    if dst.is_cuda and dst.dtype == torch.float32 and dst.dim() == 2:
        m = module(required=True)
        m.row_scatter_add(dst, idx, src.contiguous(), alpha)
        return
    dst.index_add_(0, idx, src if alpha == 1.0 else alpha * src)
