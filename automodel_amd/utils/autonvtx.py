"""Automatic rocTX range annotation of every submodule forward.

Reference behavior: nemo_automodel/autonvtx/__init__.py:22-97 (recursive
NVTX forward/backward hooks with a recursion guard, enabled via ``nvtx: true``
in the recipe YAML). torch.cuda.nvtx maps onto roctracer/rocTX on ROCm, so
ranges show up in rocprofv3 --marker-trace captures.
"""

from __future__ import annotations

import torch
import torch.nn as nn

_PATCHED_ATTR = "_amd_nvtx_hooks"


def patch(module: nn.Module, name: str | None = None) -> nn.Module:
    """Install nvtx range push/pop around every submodule forward."""
    if not torch.cuda.is_available():
        return module
    root = name or type(module).__name__
    for mod_name, mod in module.named_modules():
        if getattr(mod, _PATCHED_ATTR, False):
            continue
        full = f"{root}.{mod_name}" if mod_name else root

        def pre_hook(m, args, _full=full):
            torch.cuda.nvtx.range_push(_full)

        def post_hook(m, args, output, _full=full):
            torch.cuda.nvtx.range_pop()

        h1 = mod.register_forward_pre_hook(pre_hook)
        h2 = mod.register_forward_hook(post_hook)
        setattr(mod, _PATCHED_ATTR, (h1, h2))
    return module


def unpatch(module: nn.Module) -> None:
    for mod in module.modules():
        hooks = getattr(mod, _PATCHED_ATTR, None)
        if hooks:
            for h in hooks:
                h.remove()
            delattr(mod, _PATCHED_ATTR)
