"""Native-extension loader.

On a GPU box (``/dev/kfd`` present) a missing native extension is a hard
error — the HIP/native path must be the one that runs, never a silent
Python fallback. On CPU-only dev boxes the loader can build on demand.
"""

from __future__ import annotations

import os
from pathlib import Path
from types import ModuleType
from typing import Optional

_native_mod: Optional[ModuleType] = None


def gpu_box() -> bool:
    return os.path.exists("/dev/kfd")


def load_native(build_if_missing: bool = True) -> ModuleType:
    global _native_mod
    if _native_mod is not None:
        return _native_mod
    try:
        from . import _native  # type: ignore[attr-defined]
        _native_mod = _native
        return _native
    except ImportError as exc:
        if gpu_box():
            raise RuntimeError(
                "k8s_runpod_kubelet_amd native extension (_native) is missing on a "
                "GPU box — build it with `python -m k8s_runpod_kubelet_amd.ops.build`"
            ) from exc
        if not build_if_missing:
            raise
        from . import build as build_mod

        build_mod.build_native()
        from . import _native  # type: ignore[attr-defined]  # noqa: F811
        _native_mod = _native
        return _native


def podworker_binary() -> str:
    from . import build as build_mod

    path = build_mod.podworker_path()
    if not path.exists():
        if gpu_box():
            raise RuntimeError(
                "podworker binary missing on a GPU box — build it with "
                "`python -m k8s_runpod_kubelet_amd.ops.build`"
            )
        build_mod.build_podworker()
    return str(path)
