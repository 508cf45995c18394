"""Loader for the in-tree gfx950 HIP extension (`_cosamd_hip.so`).

The extension is built in-tree by `setup.py build_ext --inplace` (or
`__graft_entry__.build()`), producing `caffeonspark_amd/ops/_cosamd_hip*.so`
which travels to GPU boxes with the repo snapshot.  GPU registrations into
the dispatcher happen here so `ops/__init__` stays backend-agnostic.
"""

from __future__ import annotations

import glob
import importlib
import os

import torch

_EXT_NAME = "_cosamd_hip"


def _find_ext():
    here = os.path.dirname(__file__)
    hits = glob.glob(os.path.join(here, _EXT_NAME + "*.so"))
    return hits[0] if hits else None


def load():
    if not torch.cuda.is_available():
        # CPU-only host (CI container): extension not needed
        return None
    so = _find_ext()
    if so is None:
        raise RuntimeError(
            "gfx950 HIP extension not built: expected "
            f"caffeonspark_amd/ops/{_EXT_NAME}*.so — run "
            "`python setup.py build_ext --inplace` (or __graft_entry__.build())")
    mod = importlib.import_module(f"caffeonspark_amd.ops.{_EXT_NAME}")
    from .gpu import register_all
    register_all(mod)
    return mod
