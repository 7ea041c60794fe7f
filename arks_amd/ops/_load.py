"""Imports the in-tree built HIP extension (arks_amd/ops/_build/arks_amd_C.so).

Raises ImportError if the extension has not been built — callers on a GPU box
must fail loudly rather than fall back to eager PyTorch.
"""

from __future__ import annotations

import importlib.util

import torch  # noqa: F401  (the extension links against libtorch)

from .build import built_so_path

_so = built_so_path()
if _so is None:
    raise ImportError(
        "arks_amd_C.so not found in arks_amd/ops/_build — run "
        "`python -m arks_amd.ops.build`"
    )

_spec = importlib.util.spec_from_file_location("arks_amd_C", _so)
C = importlib.util.module_from_spec(_spec)
_spec.loader.exec_module(C)
