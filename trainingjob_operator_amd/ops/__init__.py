"""MI355X-native op library: CDNA4 HIP kernels + autograd wrappers.

GPU path: hand-written gfx950 kernels in hip/ops.hip (bf16x8 vectorized,
wave64 reductions), loaded from the in-tree libhipops.so. CPU path: the
fp32 PyTorch reference (reference.py). The two are cross-checked by the
numerics tests in tests/test_ops_gpu.py.
"""
from .functional import (
    apply_rope,
    decode_attention,
    decode_linear,
    decode_swiglu,
    fused_cross_entropy,
    fused_rmsnorm,
    dispatch_rows,
    gather_rows,
    make_inv_freq,
    moe_combine,
    swiglu,
    swiglu_packed,
)
from .native import HipOpsUnavailable, available, build_ops, load

__all__ = [
    "apply_rope",
    "decode_attention",
    "decode_linear",
    "decode_swiglu",
    "dispatch_rows",
    "gather_rows",
    "moe_combine",
    "fused_cross_entropy",
    "fused_rmsnorm",
    "make_inv_freq",
    "swiglu",
    "available",
    "build_ops",
    "load",
    "HipOpsUnavailable",
]
