"""Loader for the in-tree HIP extension (csrc/ -> _dmnist_hip.so).

The extension is built IN-TREE by ``__graft_entry__.build()`` (hipcc,
--offload-arch=gfx950) so the .so travels with the repo snapshot to GPU boxes.

Policy (fail loudly, no silent eager fallback): on a machine with a visible
GPU, ops MUST run through the HIP extension — if it cannot be imported we
raise at first use.  On a CPU-only machine the fp32 torch reference
implementations (ops/cpu_ref.py) are used instead; they also serve as the
numerics oracle for the kernel unit tests.
"""

from __future__ import annotations

import importlib
import importlib.util
import os
import sys

import torch

_ext = None
_tried = False


def _find_so():
    here = os.path.dirname(os.path.abspath(__file__))
    root = os.path.dirname(here)
    for d in (os.path.join(root, "csrc"), root, here):
        if not os.path.isdir(d):
            continue
        for fn in os.listdir(d):
            if fn.startswith("_dmnist_hip") and fn.endswith(".so"):
                return os.path.join(d, fn)
    return None


def try_load():
    """Import the HIP extension if present; returns module or None."""
    global _ext, _tried
    if _ext is not None or _tried:
        return _ext
    _tried = True
    so = _find_so()
    if so is None:
        return None
    spec = importlib.util.spec_from_file_location("_dmnist_hip", so)
    mod = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(mod)
    except ImportError as e:
        # Missing hip runtime etc. on CPU-only boxes is fine; on GPU boxes
        # ext() will raise below.
        sys.stderr.write(f"[distributedmnist_amd] HIP ext present but failed to load: {e}\n")
        return None
    _ext = mod
    return _ext


def ext():
    """The HIP extension module. Raises on GPU machines when missing."""
    m = try_load()
    if m is None and torch.cuda.is_available():
        raise RuntimeError(
            "distributedmnist_amd: HIP extension _dmnist_hip.so not found but a GPU "
            "is visible. Build it with `python __graft_entry__.py build` (or "
            "`python -c 'import __graft_entry__ as g; g.build()'`). Refusing to "
            "fall back to eager PyTorch on GPU."
        )
    return m


def has_ext() -> bool:
    return try_load() is not None
