"""ctypes loader for libdolomite_hip.so (the C-ABI declared in
include/dolomite_hip.h). The library is built in-tree by
`python -m dolomite_engine_amd.csrc.build` and travels with the repo.

On a GPU box, every op in this package REQUIRES the library: a missing or
stale .so raises immediately (no silent eager fallback).
"""

import ctypes
from pathlib import Path

import torch

_SO_PATH = Path(__file__).resolve().parent.parent / "libdolomite_hip.so"
_ABI_VERSION = 2

_lib = None

F32 = 0
BF16 = 1

_c = ctypes
_i32 = _c.c_int32
_i64 = _c.c_int64
_f32 = _c.c_float
_p = _c.c_void_p


_SIGNATURES = {
    "dolomite_hip_abi_version": ([], _i32),
    "dolomite_rmsnorm_fwd": ([_p, _p, _p, _p, _p, _p, _p, _i64, _i64, _f32, _i32], _i32),
    "dolomite_rmsnorm_bwd_nblocks": ([_i64], _i32),
    "dolomite_rmsnorm_bwd": ([_p, _p, _p, _p, _p, _p, _p, _p, _i64, _i64, _i32], _i32),
    "dolomite_layernorm_fwd": ([_p, _p, _p, _p, _p, _p, _p, _p, _p, _i64, _i64, _f32, _i32], _i32),
    "dolomite_layernorm_bwd": ([_p, _p, _p, _p, _p, _p, _p, _p, _p, _i64, _i64, _i32], _i32),
    "dolomite_reduce_partials": ([_p, _p, _p, _i64, _i64], _i32),
    "dolomite_rope_qkv": ([_p, _p, _p, _p, _p, _i64, _i64, _i32, _i32, _i32, _i32, _i64, _i64, _i64, _i32, _i32, _i32], _i32),
    "dolomite_fa_varlen_fwd": (
        [_p, _p, _p, _p, _p, _p, _p, _i32, _i32, _i64, _i32, _i32, _i32, _i32,
         _i64, _i64, _i64, _i64, _i64, _i64, _f32, _i32],
        _i32,
    ),
    "dolomite_fa_bwd_preprocess": ([_p, _p, _p, _p, _i64, _i32, _i32, _i64, _i64, _i32], _i32),
    "dolomite_fa_varlen_bwd": (
        [_p, _p, _p, _p, _p, _p, _p, _p, _p, _p, _p, _i32, _i32, _i64, _i32, _i32, _i32, _i32,
         _i64, _i64, _i64, _i64, _i64, _i64, _i64, _f32, _i32],
        _i32,
    ),
    "dolomite_fa_grad_finalize": ([_p, _p, _p, _p, _i64, _i32, _i32, _i32, _i64, _i64, _i64, _i64, _i32], _i32),
    "dolomite_ce_fwd": ([_p, _p, _p, _p, _p, _i64, _i64, _i64, _i32, _i32], _i32),
    "dolomite_ce_bwd": ([_p, _p, _p, _p, _p, _p, _i64, _i64, _i64, _i32, _i32], _i32),
    "dolomite_adamw_step": ([_p, _p, _p, _p, _i32, _p, _p, _i64, _f32, _f32, _f32, _f32, _f32, _i32, _p], _i32),
    "dolomite_sqsum": ([_p, _p, _i64, _p, _p, _i32], _i32),
    "dolomite_moe_rows_combine": ([_p, _p, _p, _p, _i64, _i32, _i32, _i32], _i32),
    "dolomite_scale_inplace": ([_p, _p, _i64, _f32, _i32], _i32),
    "dolomite_moe_gemm_fwd": ([_p, _p, _p, _p, _p, _p, _i32, _i32, _i32, _i32, _i32], _i32),
    "dolomite_moe_gemm_dgrad": ([_p, _p, _p, _p, _p, _i32, _i32, _i32, _i32, _i32], _i32),
    "dolomite_moe_gemm_wgrad": ([_p, _p, _p, _p, _p, _i32, _i32, _i32, _i32], _i32),
    "dolomite_mfma_probe": ([_p, _p, _p, _p], _i32),
    "dolomite_tr16_probe": ([_p, _p], _i32),
    "dolomite_tr16_bfrag_probe": ([_p, _p, _p, _i32, _i32, _i32, _i32], _i32),
    "dolomite_build_sample_idx_i32": ([_p, _p, _i32, _i32, _i64, _p, _i64], _i32),
    "dolomite_build_sample_idx_i64": ([_p, _p, _i32, _i32, _i64, _p, _i64], _i32),
    "dolomite_build_blending_indices": ([_p, _p, _p, _i32, _i64], _i32),
}


def so_path() -> Path:
    return _SO_PATH


def is_available() -> bool:
    return _SO_PATH.exists()


def lib() -> ctypes.CDLL:
    global _lib
    if _lib is None:
        if not _SO_PATH.exists():
            raise RuntimeError(
                f"dolomite HIP extension not found at {_SO_PATH}. The GPU hot path "
                "requires it — build with `python -m dolomite_engine_amd.csrc.build` "
                "(there is no eager fallback on GPU)."
            )
        l = ctypes.CDLL(str(_SO_PATH))
        for name, (argtypes, restype) in _SIGNATURES.items():
            fn = getattr(l, name)
            fn.argtypes = argtypes
            fn.restype = restype
        ver = l.dolomite_hip_abi_version()
        if ver != _ABI_VERSION:
            raise RuntimeError(f"dolomite_hip ABI mismatch: built {ver}, expected {_ABI_VERSION}")
        _lib = l
    return _lib


def stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def check(ret: int, what: str) -> None:
    if ret != 0:
        raise RuntimeError(f"dolomite_hip {what} failed with code {ret}")


def ptr(t: torch.Tensor | None, offset_elems: int = 0) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    return ctypes.c_void_p(t.data_ptr() + offset_elems * t.element_size())


def dt(t: torch.Tensor) -> int:
    if t.dtype == torch.bfloat16:
        return BF16
    if t.dtype == torch.float32:
        return F32
    raise TypeError(f"dolomite_hip ops support fp32/bf16, got {t.dtype}")


# ---------------------------------------------------------------------------
# Lightweight per-op HIP-event profiling (bench.py roofline evidence).
# ---------------------------------------------------------------------------

_PROF: dict | None = None


def enable_profiling() -> None:
    global _PROF
    _PROF = {}


def disable_profiling() -> None:
    global _PROF
    _PROF = None


class _Region:
    __slots__ = ("name", "start")

    def __init__(self, name):
        self.name = name

    def __enter__(self):
        if _PROF is not None:
            self.start = torch.cuda.Event(enable_timing=True)
            self.start.record()
        return self

    def __exit__(self, *a):
        if _PROF is not None:
            stop = torch.cuda.Event(enable_timing=True)
            stop.record()
            _PROF.setdefault(self.name, []).append((self.start, stop))
        return False


def prof(name: str) -> _Region:
    return _Region(name)


def collect_profile() -> dict:
    """{name: {"count": N, "total_ms": t, "avg_ms": t/N}} — syncs the device."""
    if _PROF is None:
        return {}
    torch.cuda.synchronize()
    out = {}
    for name, pairs in _PROF.items():
        total = sum(s.elapsed_time(e) for s, e in pairs)
        out[name] = {"count": len(pairs), "total_ms": total, "avg_ms": total / len(pairs)}
    return out
