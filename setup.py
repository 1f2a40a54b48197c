"""Build apex_amd's in-tree HIP extensions for MI355X (gfx950).

    PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace

Every extension is built into the package directory (apex_amd/_*.so) so the
snapshot that travels to a GPU box carries the binaries. No JIT cache, no
site-packages install. Mirrors the reference's per-extension registry
(setup.py:165-845) but with plain HIP sources.
"""

import os

from setuptools import setup
from torch.utils.cpp_extension import BuildExtension, CppExtension, CUDAExtension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

COMMON_FLAGS = ["-O3", "-std=c++17"]
HIP_FLAGS = ["-O3", "-std=c++17"]


def hip_ext(name, sources, extra_hip_flags=None, libraries=None, extra_link=None):
    return CUDAExtension(
        name=name,
        sources=sources,
        extra_compile_args={
            "cxx": COMMON_FLAGS,
            "nvcc": HIP_FLAGS + (extra_hip_flags or []),
        },
        libraries=libraries or [],
        extra_link_args=extra_link or [],
    )


ext_modules = [
    CppExtension(
        name="apex_amd._apex_C",
        sources=["csrc/flatten_unflatten.cpp"],
        extra_compile_args={"cxx": COMMON_FLAGS},
    ),
    hip_ext(
        "apex_amd._amp_C",
        [
            "csrc/amp_C_frontend.cpp",
            "csrc/multi_tensor_elementwise.hip",
            "csrc/multi_tensor_l2norm.hip",
            "csrc/multi_tensor_sgd.hip",
            "csrc/multi_tensor_adam.hip",
            "csrc/multi_tensor_opt.hip",
            "csrc/multi_tensor_lamb.hip",
            "csrc/update_scale_hysteresis.hip",
        ],
    ),
    hip_ext("apex_amd._fused_norm", ["csrc/fused_norm.hip"]),
    hip_ext("apex_amd._syncbn", ["csrc/syncbn.hip"]),
    hip_ext("apex_amd._softmax", ["csrc/softmax.hip"]),
    hip_ext("apex_amd._rope", ["csrc/rope.hip"]),
    hip_ext("apex_amd._fused_dense", ["csrc/fused_dense.hip"], libraries=["hipblaslt"]),
    hip_ext("apex_amd._mlp", ["csrc/mlp.hip"], libraries=["hipblaslt"]),
    hip_ext("apex_amd._xentropy", ["csrc/xentropy.hip"]),
    hip_ext("apex_amd._permutation_search", ["csrc/permutation_search.hip"]),
    hip_ext("apex_amd._focal_loss", ["csrc/focal_loss.hip"]),
    hip_ext("apex_amd._index_mul_2d", ["csrc/index_mul_2d.hip"]),
    hip_ext("apex_amd._group_norm", ["csrc/group_norm.hip"]),
    hip_ext("apex_amd._transducer", ["csrc/transducer.hip"]),
    hip_ext("apex_amd._peer_memory", ["csrc/peer_memory.hip"]),
    hip_ext("apex_amd._rccl_p2p", ["csrc/rccl_p2p.hip"], libraries=["rccl"]),
    hip_ext("apex_amd._rccl_allocator", ["csrc/rccl_allocator.cpp"], libraries=["rccl"]),
    hip_ext("apex_amd._mfma", ["csrc/mfma_gemm.hip", "csrc/mfma_probe.hip", "csrc/fmha.hip"]),
    hip_ext("apex_amd._tune", ["csrc/stream_tune.hip"]),
]

setup(
    name="apex_amd",
    version="0.1.0",
    description="MI355X-native mixed-precision and fused-kernel training library",
    packages=[
        "apex_amd",
        "apex_amd.amp",
        "apex_amd.contrib",
        "apex_amd.contrib.clip_grad",
        "apex_amd.contrib.xentropy",
        "apex_amd.contrib.focal_loss",
        "apex_amd.contrib.index_mul_2d",
        "apex_amd.contrib.optimizers",
        "apex_amd.contrib.group_norm",
        "apex_amd.contrib.groupbn",
        "apex_amd.contrib.layer_norm",
        "apex_amd.contrib.sparsity",
        "apex_amd.contrib.transducer",
        "apex_amd.contrib.peer_memory",
        "apex_amd.contrib.nccl_allocator",
        "apex_amd.contrib.rccl_p2p",
        "apex_amd.contrib.conv_bias_relu",
        "apex_amd.contrib.bottleneck",
        "apex_amd.contrib.gbn",
        "apex_amd.contrib.openfold",
        "apex_amd.contrib.torchsched",
        "apex_amd.contrib.fast_multihead_attn",
        "apex_amd.fused_dense",
        "apex_amd.mlp",
        "apex_amd.models",
        "apex_amd.multi_tensor_apply",
        "apex_amd.normalization",
        "apex_amd.optimizers",
        "apex_amd.parallel",
        "apex_amd.transformer",
    ],
    ext_modules=ext_modules,
    cmdclass={"build_ext": BuildExtension.with_options(use_ninja=True)},
)
