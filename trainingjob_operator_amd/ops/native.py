"""ctypes bindings to the in-tree gfx950 HIP library (libhipops.so).

The library is built by ``build_ops()`` (one direct hipcc invocation,
``--offload-arch=gfx950`` — no hipify, no compatibility layers) and loaded
from the package directory so the ``.so`` travels with the repo snapshot.

Policy: on a GPU box these bindings are REQUIRED — ops fail loudly if the
extension is missing rather than silently falling back to eager PyTorch.
"""
from __future__ import annotations

import ctypes
import os
import subprocess
from typing import Optional

_PKG_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_PKG_DIR, "libhipops.so")
SRC_PATH = os.path.join(_PKG_DIR, "hip", "ops.hip")
ATTN_SRC_PATH = os.path.join(_PKG_DIR, "hip", "attention.hip")

_lib: Optional[ctypes.CDLL] = None


class HipOpsUnavailable(RuntimeError):
    pass


def build_ops(verbose: bool = False) -> str:
    """Compile ops.hip -> libhipops.so in-tree for gfx950."""
    cmd = [
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
        "-shared", "-fPIC", SRC_PATH, ATTN_SRC_PATH, "-o", LIB_PATH,
    ]
    if verbose:
        print("+", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return LIB_PATH


def _sig(fn, argtypes, restype=None):
    fn.argtypes = argtypes
    fn.restype = restype


def load(require: bool = True) -> Optional[ctypes.CDLL]:
    """Load (and memoize) the HIP ops library."""
    global _lib
    if _lib is not None:
        return _lib
    if not os.path.exists(LIB_PATH):
        if require:
            raise HipOpsUnavailable(
                f"{LIB_PATH} not built — run trainingjob_operator_amd.ops."
                "native.build_ops() (or __graft_entry__.build())")
        return None
    lib = ctypes.CDLL(LIB_PATH)
    vp, l, i, f = ctypes.c_void_p, ctypes.c_long, ctypes.c_int, ctypes.c_float
    # every launcher returns int: 0 = launched, nonzero = unsupported shape
    # (callers raise via check_rc — no silent no-op launches)
    _sig(lib.hipops_arch_check, [], i)
    _sig(lib.rmsnorm_fwd, [vp, vp, vp, vp, vp, vp, vp, l, i, f], i)
    _sig(lib.rmsnorm_bwd, [vp, vp, vp, vp, vp, vp, vp, vp, l, i], i)
    _sig(lib.rmsnorm_bwd_partials, [l, i], l)
    _sig(lib.rmsnorm_dw_reduce, [vp, vp, l, vp, i], i)
    _sig(lib.rope, [vp, vp, vp, vp, l, i, i, i, f], i)
    _sig(lib.rope_at, [vp, vp, vp, vp, l, i, i, i, f, i], i)
    _sig(lib.rope_at_dev, [vp, vp, vp, vp, l, i, i, i, f, vp], i)
    _sig(lib.gemv_bf16, [vp, vp, vp, vp, i, i, i], i)
    _sig(lib.gemv_swiglu, [vp, vp, vp, vp, i, i, i], i)
    _sig(lib.gemv_moe_swiglu, [vp, vp, vp, vp, vp, vp, i, i, i, i], i)
    _sig(lib.gemv_moe, [vp, vp, vp, vp, vp, i, i, i, i], i)
    _sig(lib.attn_decode, [vp, vp, vp, vp, vp, vp, vp,
                           i, i, i, i, i, f], i)
    _sig(lib.rope_cache, [vp, vp, vp, vp, vp, vp, vp, i, i, i, i, i], i)
    _sig(lib.swiglu_fwd, [vp, vp, vp, vp, l], i)
    _sig(lib.swiglu_bwd, [vp, vp, vp, vp, vp, vp, l], i)
    _sig(lib.swiglu_packed_fwd, [vp, vp, vp, l, i], i)
    _sig(lib.swiglu_packed_bwd, [vp, vp, vp, vp, l, i], i)
    _sig(lib.ce_fwd, [vp, vp, vp, vp, vp, l, i, i], i)
    _sig(lib.ce_bwd, [vp, vp, vp, vp, vp, vp, l, i, i], i)
    _sig(lib.l2normsq, [vp, vp, l, vp, i, vp], i)
    _sig(lib.rows_gather, [vp, vp, vp, vp, l, i], i)
    _sig(lib.rows_scatter, [vp, vp, vp, vp, l, i], i)
    _sig(lib.rows_scatter_add_f32, [vp, vp, vp, vp, l, i], i)
    _sig(lib.moe_combine, [vp, vp, vp, vp, vp, l, i, i], i)
    _sig(lib.moe_combine_bwd, [vp, vp, vp, vp, vp, vp, vp, l, i, i], i)
    _sig(lib.adamw_step, [vp, vp, vp, vp, vp, vp, vp, l,
                          f, f, f, f, f, f, f, f, f, vp], i)
    _sig(lib.adamw_step_bf16mom, [vp, vp, vp, vp, vp, vp, vp, l,
                                  f, f, f, f, f, f, f, f, f, vp], i)
    _sig(lib.mfma_probe, [vp, vp, vp, vp])
    _sig(lib.attn_fwd, [vp, vp, vp, vp, vp, vp,
                        l, l, l, l, l, l, l, l, l, i, i, i, i, f], i)
    _sig(lib.attn_fwd_v5, [vp, vp, vp, vp, vp, vp,
                           l, l, l, l, l, l, l, l, l, i, i, i, f], i)
    _sig(lib.attn_fwd_nw8, [vp, vp, vp, vp, vp, vp,
                            l, l, l, l, l, l, l, l, l, i, i, i, i, f], i)
    _sig(lib.attn_fwd_v7, [vp, vp, vp, vp, vp, vp,
                           l, l, l, l, l, l, l, l, l, i, i, i, i, f], i)
    _sig(lib.attn_bwd, [vp, vp, vp, vp, vp, vp, vp, vp, vp, vp, vp,
                        l, l, l, l, l, l, l, l, l, i, i, i, i, f], i)
    _sig(lib.mfma_probe32, [vp, vp, vp, vp])
    assert lib.hipops_arch_check() == 950
    _lib = lib
    return lib


def check_rc(rc: int, op: str, detail: str = "") -> None:
    """Raise on a nonzero launcher status (unsupported shape)."""
    if rc != 0:
        raise RuntimeError(
            f"hipops {op}: unsupported shape ({detail}) — the kernel "
            f"launcher refused to launch (rc={rc})")


def available() -> bool:
    return os.path.exists(LIB_PATH)


def stream_ptr() -> ctypes.c_void_p:
    """Current torch HIP stream as a raw handle."""
    import torch
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def tr_probe_maps(device="cuda"):
    """Dump ds_read_b64_tr_b16 lane/elem -> LDS element mappings."""
    import torch
    lib = load()
    lib.tr_probe.argtypes = [ctypes.c_void_p, ctypes.c_void_p,
                             ctypes.c_int, ctypes.c_int]
    results = {}
    for mode in (0, 1, 2, 3):
        out = torch.empty(64, 4, dtype=torch.int16, device=device)
        lib.tr_probe(stream_ptr(), ctypes.c_void_p(out.data_ptr()), 0, mode)
        torch.cuda.synchronize()
        results[mode] = out.to(torch.int32).cpu() & 0xFFFF
    return results
