"""HIP extension loader.

The native extension (`factorvae_hip.*.so`, built in-tree by setup.py /
__graft_entry__.build from ops/hip/*) holds every hand-written CDNA4
kernel. Policy: on a machine WITH a GPU the extension must load — ops
fail loudly rather than falling back to eager silently; on CPU-only
machines (CI) everything runs through the eager oracle path.
"""

from __future__ import annotations

import importlib
import os
import sys

import torch

_ext = None
_load_error: Exception | None = None


def _try_load():
    global _ext, _load_error
    if _ext is not None:
        return _ext
    try:
        # in-tree build: factorvae_amd/ops/factorvae_hip*.so
        from . import factorvae_hip as ext  # type: ignore
        _ext = ext
    except Exception as e:  # pragma: no cover - exercised on GPU boxes
        _load_error = e
        _ext = None
    return _ext


def extension_available() -> bool:
    return _try_load() is not None


def get_extension():
    ext = _try_load()
    if ext is None:
        if torch.cuda.is_available():
            raise RuntimeError(
                "factorvae_hip extension missing on a GPU machine — the HIP "
                "path must not silently fall back to eager. Build it with "
                "`python setup.py build_ext --inplace` "
                f"(original import error: {_load_error!r})"
            )
        raise RuntimeError(
            f"factorvae_hip extension not built (CPU machine): {_load_error!r}"
        )
    return ext
