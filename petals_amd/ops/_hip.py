"""Loader for the in-tree HIP extension (.so built by petals_amd.ops.build)."""

from __future__ import annotations

import importlib.util
import os

SO_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), "_hip_ops.so")


def load():
    import torch  # noqa: F401 — the extension links against loaded libtorch

    if not os.path.exists(SO_PATH):
        raise ImportError(f"HIP extension not built: {SO_PATH} missing (run python -m petals_amd.ops.build)")
    spec = importlib.util.spec_from_file_location("petals_amd_hip_ops", SO_PATH)
    module = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(module)
    return module
