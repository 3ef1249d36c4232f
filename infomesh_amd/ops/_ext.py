"""ctypes loader for the CDNA4 HIP extension (fail-loud policy).

On a machine with a GPU the extension MUST load — ops raise
GpuExtensionMissing rather than silently falling back to eager PyTorch
(the round-end check records which .so files the GPU processes load).
On CPU-only machines the torch reference implementations in
ops/reference.py are the test oracle instead.
"""
from __future__ import annotations

import ctypes
from ctypes import c_float, c_int, c_long, c_void_p
from pathlib import Path

from ..errors import GpuExtensionMissing

_LIB: ctypes.CDLL | None = None
_LOAD_ERROR: str | None = None

_SIGNATURES: dict[str, list] = {
    "infomesh_gemm_bf16_nt": [c_void_p, c_void_p, c_void_p, c_void_p,
                              c_int, c_int, c_int, c_int,
                              c_long, c_long, c_long,
                              c_int, c_float, c_int, c_void_p],
    "infomesh_layernorm": [c_void_p, c_void_p, c_void_p, c_void_p,
                           c_void_p, c_void_p, c_long, c_int, c_float,
                           c_void_p],
    "infomesh_rmsnorm": [c_void_p, c_void_p, c_void_p, c_void_p,
                         c_void_p, c_long, c_int, c_float, c_void_p],
    "infomesh_softmax": [c_void_p, c_void_p, c_void_p, c_long, c_int,
                         c_int, c_int, c_float, c_void_p],
    "infomesh_bias_act": [c_void_p, c_void_p, c_long, c_int, c_int, c_void_p],
    "infomesh_silu_mul": [c_void_p, c_void_p, c_void_p, c_long, c_void_p],
    "infomesh_add": [c_void_p, c_void_p, c_void_p, c_long, c_void_p],
    "infomesh_rope": [c_void_p, c_void_p, c_void_p, c_void_p, c_long,
                      c_int, c_int, c_int, c_void_p],
    "infomesh_gather": [c_void_p, c_void_p, c_void_p, c_long, c_int,
                        c_float, c_void_p],
    "infomesh_pool": [c_void_p, c_void_p, c_void_p, c_int, c_int, c_int,
                      c_int, c_int, c_void_p],
    "infomesh_argmax": [c_void_p, c_void_p, c_long, c_int, c_void_p],
    "infomesh_topk": [c_void_p, c_void_p, c_void_p, c_void_p, c_int,
                      c_long, c_int, c_int, c_void_p, c_void_p],
    "infomesh_bm25_block": [c_void_p, c_void_p, c_void_p, c_void_p,
                            c_void_p, c_void_p, c_void_p, c_void_p,
                            c_void_p, c_void_p, c_int, c_int, c_long,
                            c_long, c_long, c_int, c_float, c_float,
                            c_float, c_void_p],
    "infomesh_score_combine": [c_void_p, c_void_p, c_void_p, c_float,
                               c_float, c_long, c_void_p],
    "infomesh_simhash_fingerprint": [c_void_p, c_void_p, c_void_p, c_long,
                                     c_void_p],
    "infomesh_hamming_scan": [c_void_p, c_void_p, c_void_p, c_void_p,
                              c_void_p, c_void_p, c_int, c_long, c_int,
                              c_int, c_void_p],
    "infomesh_attn_decode": [c_void_p, c_void_p, c_void_p, c_void_p,
                             c_void_p, c_int, c_int, c_int, c_int, c_int,
                             c_float, c_void_p],
    "infomesh_attn_decode_split": [c_void_p, c_void_p, c_void_p, c_void_p,
                                   c_void_p, c_void_p, c_int, c_int, c_int,
                                   c_int, c_int, c_int, c_float, c_void_p],
    "infomesh_attn_fused": [c_void_p, c_void_p, c_void_p, c_void_p,
                            c_void_p, c_int, c_int, c_int, c_int, c_int,
                            c_int, c_long, c_long, c_long, c_long, c_long,
                            c_long, c_long, c_long, c_long,
                            c_int, c_float, c_void_p],
    "infomesh_kv_append": [c_void_p, c_void_p, c_void_p, c_void_p,
                           c_void_p, c_int, c_int, c_int, c_int, c_void_p],
    "infomesh_dense_scores": [c_void_p, c_void_p, c_void_p,
                              c_int, c_long, c_int, c_float, c_void_p],
    "infomesh_dense_scores_fp8": [c_void_p, c_void_p, c_void_p,
                                  c_int, c_long, c_int, c_float,
                                  c_void_p],
    "infomesh_gemm8_bf16_nt": [c_void_p, c_void_p, c_void_p, c_void_p,
                               c_int, c_int, c_int, c_int,
                               c_long, c_long, c_long,
                               c_int, c_float, c_int, c_void_p],
    "infomesh_gemv_bf16_nt": [c_void_p, c_void_p, c_void_p, c_void_p,
                              c_int, c_int, c_int, c_int,
                              c_long, c_long, c_long,
                              c_int, c_float, c_int, c_void_p],
    "infomesh_qkv_split": [c_void_p, c_void_p, c_void_p, c_void_p,
                           c_void_p, c_void_p, c_void_p,
                           c_int, c_int, c_int, c_int, c_int, c_int,
                           c_void_p],
    "infomesh_merge_heads": [c_void_p, c_void_p, c_int, c_int, c_int,
                             c_int, c_void_p],
    "infomesh_silu_mul_fused": [c_void_p, c_void_p, c_long, c_int,
                                c_void_p],
}

_RESTYPES = {"infomesh_topk_workspace_u32": c_long}


def _try_load() -> ctypes.CDLL | None:
    global _LOAD_ERROR
    so = Path(__file__).resolve().parent / "libinfomesh_hip.so"
    if not so.exists():
        _LOAD_ERROR = f"{so} not built"
        return None
    try:
        lib = ctypes.CDLL(str(so))
    except OSError as e:
        _LOAD_ERROR = str(e)
        return None
    for name, argtypes in _SIGNATURES.items():
        fn = getattr(lib, name)
        fn.argtypes = argtypes
        # dense_scores reports dispatch eligibility (0 = launched)
        fn.restype = c_int if name in ("infomesh_dense_scores",
                                       "infomesh_dense_scores_fp8") \
            else None
    ws = lib.infomesh_topk_workspace_u32
    ws.argtypes = [c_int]
    ws.restype = c_long
    import os
    if os.environ.get("INFOMESH_SYNC_DEBUG"):
        return _SyncDebugLib(lib)
    return lib


class _SyncDebugLib:
    """Sanitizer-style launch checking (INFOMESH_SYNC_DEBUG=1): after
    every kernel launch, synchronize the device so an async fault
    (bad pointer, OOB write surfacing later) raises AT THE LAUNCH SITE
    with the kernel's name, instead of poisoning a later op. The GPU
    analogue of the reference's structural concurrency checks
     — debugging only, serializes every launch."""

    def __init__(self, real: ctypes.CDLL):
        self._real = real

    def __getattr__(self, name: str):
        fn = getattr(self._real, name)
        if not name.startswith("infomesh_"):
            return fn

        def checked(*args, _fn=fn, _name=name):
            r = _fn(*args)
            import torch
            if torch.cuda.is_available():
                try:
                    torch.cuda.synchronize()
                except RuntimeError as e:
                    raise RuntimeError(
                        f"async GPU fault surfaced at {_name}: {e}"
                    ) from e
            return r
        return checked


def lib():
    """The loaded extension; raises GPU001 when missing."""
    global _LIB
    if _LIB is None:
        _LIB = _try_load()
    if _LIB is None:
        raise GpuExtensionMissing(_LOAD_ERROR or "unknown load failure")
    return _LIB


def available() -> bool:
    global _LIB
    if _LIB is None:
        _LIB = _try_load()
    return _LIB is not None


def stream_ptr() -> int:
    import torch
    return torch.cuda.current_stream().cuda_stream
