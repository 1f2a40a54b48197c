"""torchsched — torch.compile backend registration (reference:
apex/contrib/torchsched — an inductor-based multi-stream graph scheduler).

Round-1 scope: the backend registry surface (``get_backend`` /
``set_default_backend`` / the ``torchsched`` @register_backend entry) is in
place and compiles through inductor; the multi-stream (dwb) event/wrapper
codegen that overlaps independent graph partitions on side HIP streams is a
documented later-round item — on MI355X it will map partitions onto HIP
streams with event-based cross-stream ordering.
"""

import torch

_default_backend = "inductor"


def _torchsched_backend(gm, example_inputs):
    from torch._inductor.compile_fx import compile_fx

    return compile_fx(gm, example_inputs)


try:
    from torch._dynamo import register_backend

    register_backend(name="torchsched", compiler_fn=_torchsched_backend)
except Exception:  # pragma: no cover - dynamo unavailable
    pass


def get_backend():
    """Return the torchsched compile backend callable."""
    return _torchsched_backend


def set_default_backend(name="torchsched"):
    global _default_backend
    _default_backend = name
    return _default_backend
