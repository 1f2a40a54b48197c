"""torchsched — multi-stream torch.compile backend (reference:
apex/contrib/torchsched — an inductor-based multi-stream graph scheduler).

The ``torchsched`` backend partitions the captured FX graph into chains
(scheduler.py) and, when the graph has ≥2 concurrently-runnable partitions,
executes them on side HIP streams with event-based ordering; graphs with no
exploitable parallelism compile straight through inductor.
"""

import torch

from .scheduler import MultiStreamGraphModule, partition_graph, max_parallel_width

_default_backend = "inductor"


def _torchsched_backend(gm, example_inputs):
    try:
        parts, _ = partition_graph(gm)
        if torch.cuda.is_available() and max_parallel_width(parts) >= 2:
            return MultiStreamGraphModule(gm)
    except Exception:
        pass
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
