"""apex_amd.contrib — optional modules (reference: apex/contrib).

Import submodules explicitly, e.g.::

    from apex_amd.contrib import xentropy
    from apex_amd.contrib.clip_grad import clip_grad_norm_
"""

from . import clip_grad  # noqa: F401

__all__ = ["clip_grad"]
