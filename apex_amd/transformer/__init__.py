from .softmax import (
    ScaledMaskedSoftmax,
    ScaledSoftmax,
    ScaledUpperTriangMaskedSoftmax,
    GenericScaledMaskedSoftmax,
    scaled_masked_softmax,
    scaled_softmax,
    scaled_upper_triang_masked_softmax,
    generic_scaled_masked_softmax,
)
from .rope import (
    fused_apply_rotary_pos_emb,
    fused_apply_rotary_pos_emb_cached,
    fused_apply_rotary_pos_emb_thd,
    fused_apply_rotary_pos_emb_2d,
)
from .wgrad import wgrad_gemm_accum_fp32, wgrad_gemm_accum_fp16
from .fmha import flash_attention, flash_attention_forward, flash_attention_supported

__all__ = [
    "ScaledMaskedSoftmax",
    "ScaledSoftmax",
    "ScaledUpperTriangMaskedSoftmax",
    "GenericScaledMaskedSoftmax",
    "scaled_masked_softmax",
    "scaled_softmax",
    "scaled_upper_triang_masked_softmax",
    "generic_scaled_masked_softmax",
    "fused_apply_rotary_pos_emb",
    "fused_apply_rotary_pos_emb_cached",
    "fused_apply_rotary_pos_emb_thd",
    "fused_apply_rotary_pos_emb_2d",
    "wgrad_gemm_accum_fp32",
    "wgrad_gemm_accum_fp16",
    "flash_attention_forward",
    "flash_attention",
    "flash_attention_supported",
]
