"""Loader for the in-tree HIP extension.

The extension is built IN-TREE (murmura_amd/ops/_murmura_hip.<abi>.so) by
``python -m murmura_amd.ops.build`` / ``__graft_entry__.build()`` so the
built .so travels to the GPU box with the repo snapshot. We import it from
the package directory, never from a JIT cache.
"""

from __future__ import annotations

import importlib
import importlib.util
from pathlib import Path

_ext = None


def load():
    """Import the built extension; raises ImportError if absent."""
    global _ext
    if _ext is not None:
        return _ext
    pkg_dir = Path(__file__).parent
    candidates = sorted(pkg_dir.glob("_murmura_hip*.so"))
    if not candidates:
        raise ImportError(
            f"no _murmura_hip*.so found in {pkg_dir}; build with "
            "`python -m murmura_amd.ops.build`"
        )
    spec = importlib.util.spec_from_file_location("_murmura_hip", candidates[0])
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _ext = mod
    return _ext
