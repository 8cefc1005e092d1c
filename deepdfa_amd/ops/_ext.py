"""Loader for the in-tree HIP extension (deepdfa_amd._C).

The extension is built IN-TREE (setup.py build_ext --inplace) so the .so
travels with the repo snapshot to GPU boxes. Policy (required by the build
contract): on a machine WITH a GPU the HIP path is mandatory — ops raise
rather than fall back to eager PyTorch, so a silently-missing extension can
never masquerade as a passing GPU run. On CPU-only machines the pure-torch
reference path is used.
"""

from __future__ import annotations

import importlib

import torch

_ext = None
_tried = False


def load_ext(required: bool = False):
    global _ext, _tried
    if _ext is None and not _tried:
        _tried = True
        try:
            _ext = importlib.import_module("deepdfa_amd._C")
        except ImportError as e:
            _ext = None
            _err = e
    if _ext is None and required:
        raise RuntimeError(
            "deepdfa_amd._C HIP extension is not built but a GPU tensor was "
            "passed. Build it in-tree with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950). Refusing to fall back to eager PyTorch "
            "on GPU."
        )
    return _ext


def has_ext() -> bool:
    return load_ext(required=False) is not None


def ext_for(t: torch.Tensor):
    """Return the extension module if `t` is on GPU (required), else None."""
    if t.is_cuda:
        return load_ext(required=True)
    return None
