"""HIP extension loader.

The CDNA4 kernels live in ``automodel_amd/ops/csrc/*.hip`` and are compiled
IN-TREE to ``automodel_amd/ops/libamd_ops.so`` (see ``automodel_amd/ops/build.py``
and ``__graft_entry__.build``). Ops register into the ``amd_ops`` torch
namespace via TORCH_LIBRARY.

Policy: on a GPU box the HIP path is mandatory — a missing extension raises
instead of silently falling back to eager (so GPU tests can never pass on a
fallback). On CPU the wrappers use plain torch reference implementations.
"""

from __future__ import annotations

import os

import torch

_SO_NAME = "libamd_ops.so"
_loaded: bool | None = None


def so_path() -> str:
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), _SO_NAME)


def ops_available() -> bool:
    global _loaded
    if _loaded is None:
        path = so_path()
        if os.path.exists(path):
            torch.ops.load_library(path)
            _loaded = True
        else:
            _loaded = False
    return _loaded


def require_ops() -> None:
    if not ops_available():
        raise RuntimeError(
            f"automodel_amd HIP extension not found at {so_path()} — build it with "
            "`python -m automodel_amd.ops.build` (or __graft_entry__.build()). "
            "Refusing to run a silent eager fallback on GPU."
        )


def hip_ops():
    """Return the torch.ops.amd_ops namespace, loading the extension first."""
    require_ops()
    return torch.ops.amd_ops
