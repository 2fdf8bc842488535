"""HIP extension loader.

The compute path for GPU tensors is the in-tree C++/HIP extension
``npairloss_amd/_C*.so`` (built by ``python setup.py build_ext --inplace``
or ``__graft_entry__.build()`` with PYTORCH_ROCM_ARCH=gfx950).

Policy: on a GPU tensor the HIP path is mandatory — if the extension is
missing we raise instead of silently falling back to eager PyTorch, so a
"passing" GPU run always means the native kernels ran.  CPU tensors use the
pure-torch implementation (mirroring ops/oracle.py) for tests and
multi-process gloo runs.
"""

from __future__ import annotations

import importlib
from typing import Optional

_EXT = None
_EXT_ERR: Optional[Exception] = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        _EXT = importlib.import_module("npairloss_amd._C")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e


def has_extension() -> bool:
    _try_load()
    return _EXT is not None


def ext():
    """Return the loaded extension module, raising loudly if unavailable."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "npairloss_amd HIP extension (npairloss_amd/_C) is not built — "
            "GPU tensors require the native gfx950 kernels. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_EXT_ERR!r}"
        )
    return _EXT
