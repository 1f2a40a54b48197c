"""Extension loader for apex_amd's in-tree HIP extensions.

Every native component is an independently built extension living inside the
package (``apex_amd/_amp_C*.so`` etc., built by ``setup.py build_ext
--inplace`` with ``PYTORCH_ROCM_ARCH=gfx950``).

Policy (this is deliberate and load-bearing):

* On a GPU box (``torch.cuda.is_available()``), a missing extension is a hard
  error at first use — there must never be a silent eager fallback on the
  device path.
* On a CPU-only box (CI), callers fall back to reference PyTorch math so the
  full test suite runs without a GPU.
"""

import importlib
from typing import Optional

import torch

_CACHE: dict = {}


class _MissingExt:
    """Placeholder that raises loudly on any attribute access."""

    def __init__(self, name: str, err: Exception):
        self._name = name
        self._err = err

    def __getattr__(self, item):
        raise RuntimeError(
            f"apex_amd extension '{self._name}' is required on a GPU box but "
            f"could not be imported ({self._err!r}). Build it in-tree with:\n"
            f"  PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace\n"
            f"(apex_amd never falls back to eager PyTorch on the GPU path)."
        )

    def __bool__(self):
        return False


def get_ext(short_name: str) -> Optional[object]:
    """Import ``apex_amd._<short_name>``.

    Returns the module, or ``None`` on a CPU-only machine when the extension
    is absent, or a loud-failing placeholder on a GPU machine.
    """
    if short_name in _CACHE:
        return _CACHE[short_name]
    fqname = f"apex_amd._{short_name}"
    try:
        mod = importlib.import_module(fqname)
    except ImportError as e:
        if torch.cuda.is_available():
            mod = _MissingExt(fqname, e)
        else:
            mod = None
    _CACHE[short_name] = mod
    return mod


def has_ext(short_name: str) -> bool:
    mod = get_ext(short_name)
    return mod is not None and not isinstance(mod, _MissingExt)


def require_ext(short_name: str):
    """Return the extension module or raise (used on device code paths)."""
    mod = get_ext(short_name)
    if mod is None:
        raise RuntimeError(
            f"apex_amd extension 'apex_amd._{short_name}' is not built. "
            f"Build with: PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace"
        )
    return mod
