"""Loader for the in-tree gfx950 HIP extension (vilbert_hip.so).

The extension is built IN-TREE (so the .so travels with the repo snapshot to
the GPU box) by ``python -m vilbert_multi_task_amd.ops.build`` or by
``__graft_entry__.build()``. Kernels register under ``torch.ops.vilbert_amd``.
"""

from __future__ import annotations

import os

import torch

_LOADED = None

_SO_DIR = os.path.join(os.path.dirname(__file__), "_C")


def so_path() -> str:
    return os.path.join(_SO_DIR, "vilbert_hip.so")


def load():
    """Load the .so and return a namespace of op callables.

    Raises if the library is absent or fails to load — callers decide whether
    that is fatal (it IS fatal on a GPU box: ops/functional.py refuses eager
    fallback on CUDA tensors).
    """
    global _LOADED
    if _LOADED is not None:
        return _LOADED
    path = so_path()
    if not os.path.exists(path):
        raise FileNotFoundError(
            f"{path} not built; run `python -m vilbert_multi_task_amd.ops.build`"
        )
    torch.ops.load_library(path)

    class _Ext:
        """Proxies every registered vilbert_amd op (no manual listing)."""

        lib_path = path

        def __getattr__(self, name):
            return getattr(torch.ops.vilbert_amd, name)

    _LOADED = _Ext()
    return _LOADED
