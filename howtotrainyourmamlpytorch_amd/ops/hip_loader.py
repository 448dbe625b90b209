"""Load the in-tree ``_maml_hip.so`` extension (built by ``ops/build.py``)."""

from __future__ import annotations

import os

import torch

PKG_DIR = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(PKG_DIR, "_maml_hip.so")

_module = None


def load():
    global _module
    if _module is not None:
        return _module
    if not os.path.isfile(SO_PATH):
        raise ImportError(
            f"HIP extension not built: {SO_PATH} missing. Run "
            "`python -m howtotrainyourmamlpytorch_amd.ops.build`.")
    assert torch is not None  # torch import loads libtorch symbols first
    import importlib.util
    spec = importlib.util.spec_from_file_location("_maml_hip", SO_PATH)
    _module = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(_module)
    return _module
