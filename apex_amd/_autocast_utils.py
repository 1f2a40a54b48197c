"""Autocast bridge utilities (reference: apex/_autocast_utils.py:40 —
``_cast_if_autocast_enabled`` used by the functional wrappers)."""

from typing import Sequence

import torch


def _get_autocast_dtypes() -> Sequence[torch.dtype]:
    if torch.cuda.is_bf16_supported():
        return [torch.half, torch.bfloat16]
    return [torch.half]


def _get_current_dtype(dtype=None) -> torch.dtype:
    if not torch.is_autocast_enabled():
        return torch.float32 if dtype is None else dtype
    return torch.get_autocast_dtype("cuda")


def _cast_if_autocast_enabled(*args):
    if not torch.is_autocast_enabled():
        return args
    return torch.amp.autocast_mode._cast(args, "cuda", torch.get_autocast_dtype("cuda"))
