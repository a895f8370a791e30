"""HIP kernel extension loading + native-path policy.

The compiled extension ``_dlb_kernels`` (built in-tree from ops/csrc/ for
gfx950 only) is the GPU compute path for the framework's hot ops.  Policy:

- On a CUDA(ROCm) device, an op that HAS a native kernel uses it
  unconditionally; if the extension failed to import, the op raises — no
  silent eager fallback on GPU (set ``DLB_ALLOW_EAGER=1`` only for
  bring-up/debugging).
- On CPU (the reference's `-d true` debug mode) ops run their plain
  PyTorch composition — that path is part of the reference API surface
  (parser.py:42-43), not a compatibility layer.
"""

from __future__ import annotations

import os

_EXT = None
_EXT_ERR: Exception | None = None


def _load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import _dlb_kernels  # type: ignore  # built .so lives in this dir

        _EXT = _dlb_kernels
    except ImportError:
        try:
            import importlib
            _EXT = importlib.import_module("_dlb_kernels")
        except ImportError as e:  # record, raise lazily at first GPU use
            _EXT_ERR = e
    return _EXT


def available() -> bool:
    return _load() is not None


def ext():
    """The extension module; raises loudly if missing when required."""
    m = _load()
    if m is None:
        if os.environ.get("DLB_ALLOW_EAGER") == "1":
            return None
        raise RuntimeError(
            "dlb HIP kernel extension (_dlb_kernels) is not built — run "
            "`python -c 'import __graft_entry__; __graft_entry__.build()'` "
            f"(import error: {_EXT_ERR})"
        )
    return m
