"""ctypes binding of the C-ABI library (include/flashmoe_abi.h).

The product path: fails loudly if the HIP library is missing on a GPU
machine — there is no CPU fallback anywhere in this package.
"""
from __future__ import annotations

import ctypes
import os

_LIB_NAME = "_libflashmoe.so"
_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)), _LIB_NAME)

_lib = None
_load_error: Exception | None = None


class FMConfig(ctypes.Structure):
    # mirrors fm_config in include/flashmoe_abi.h
    _fields_ = [
        ("num_experts", ctypes.c_int32),
        ("expert_top_k", ctypes.c_int32),
        ("capacity_factor", ctypes.c_int32),
        ("drop_tokens", ctypes.c_int32),
        ("hidden_act", ctypes.c_int32),
        ("hidden_size", ctypes.c_int32),
        ("intermediate_size", ctypes.c_int32),
        ("sequence_len", ctypes.c_int32),
        ("mini_batch", ctypes.c_int32),
        ("dtype", ctypes.c_int32),
        ("is_training", ctypes.c_int32),
    ]


def _bind(lib: ctypes.CDLL) -> ctypes.CDLL:
    p = ctypes.c_void_p
    i64 = ctypes.c_int64
    lib.fm_initialize.argtypes = [ctypes.POINTER(FMConfig), ctypes.c_int, ctypes.c_int]
    lib.fm_initialize.restype = ctypes.c_int
    lib.fm_finalize.argtypes = []
    lib.fm_finalize.restype = ctypes.c_int
    lib.fm_get_compiled_config.argtypes = [ctypes.POINTER(i64)] * 6
    lib.fm_get_compiled_config.restype = ctypes.c_int
    lib.fm_get_num_local_experts.argtypes = []
    lib.fm_get_num_local_experts.restype = ctypes.c_int
    lib.fm_moe_forward.argtypes = [p, p, p, p, p, p, p, p, i64]
    lib.fm_moe_forward.restype = ctypes.c_int
    lib.fm_moe_forward_phased.argtypes = [p, p, p, p, p, p, p, p, i64,
                                          ctypes.POINTER(ctypes.c_float)]
    lib.fm_moe_forward_phased.restype = ctypes.c_int
    lib.fm_gate_forward.argtypes = [p, p, p, p, i64]
    lib.fm_gate_forward.restype = ctypes.c_int
    lib.fm_read_routing.argtypes = [p, p, p, p]
    lib.fm_read_routing.restype = ctypes.c_int
    lib.fm_expert_ffn.argtypes = [p, p, p, p, p, p, i64, ctypes.c_int32]
    lib.fm_expert_ffn.restype = ctypes.c_int
    lib.fm_heap_init.argtypes = []
    lib.fm_heap_init.restype = ctypes.c_int
    lib.fm_heap_handle.argtypes = [p]
    lib.fm_heap_handle.restype = ctypes.c_int
    lib.fm_heap_connect.argtypes = [p]
    lib.fm_heap_connect.restype = ctypes.c_int
    lib.fm_heap_ptrs.argtypes = [ctypes.POINTER(ctypes.c_void_p)] * 2
    lib.fm_heap_ptrs.restype = ctypes.c_int
    lib.fm_dispatch_p2p.argtypes = [p, p]
    lib.fm_dispatch_p2p.restype = ctypes.c_int
    lib.fm_return_p2p.argtypes = [p, p]
    lib.fm_return_p2p.restype = ctypes.c_int
    lib.fm_p2p_error_check.argtypes = [p]
    lib.fm_p2p_error_check.restype = ctypes.c_int
    lib.fm_pack_dispatch.argtypes = [p, p, p]
    lib.fm_pack_dispatch.restype = ctypes.c_int
    lib.fm_expert_ffn_segments.argtypes = [p, p, p, ctypes.c_int32, p, p]
    lib.fm_expert_ffn_segments.restype = ctypes.c_int
    lib.fm_combine_padded.argtypes = [p, p, p, p, i64]
    lib.fm_combine_padded.restype = ctypes.c_int
    lib.fm_expert_ffn_grouped.argtypes = [p, p, p, ctypes.c_int32, p, p, p, p]
    lib.fm_expert_ffn_grouped.restype = ctypes.c_int
    lib.fm_combine.argtypes = [p, p, p, p, i64, ctypes.c_int32]
    lib.fm_combine.restype = ctypes.c_int
    lib.fm_combine_finalize.argtypes = [p, p, i64]
    lib.fm_combine_finalize.restype = ctypes.c_int
    lib.fm_export_routing.argtypes = [p, p, p]
    lib.fm_export_routing.restype = ctypes.c_int
    lib.fm_read_aux_loss.argtypes = [p, p, p]
    lib.fm_read_aux_loss.restype = ctypes.c_int
    lib.fm_last_error.argtypes = []
    lib.fm_last_error.restype = ctypes.c_char_p
    lib.fm_built_for_gfx950.argtypes = []
    lib.fm_built_for_gfx950.restype = ctypes.c_int
    lib.fm_debug_mfma.argtypes = [p, p, p, p]
    lib.fm_debug_mx_mfma.argtypes = [p, p, p, p, p, p, ctypes.c_int]
    lib.fm_debug_mx_mfma.restype = ctypes.c_int
    lib.fm_debug_mfma.restype = ctypes.c_int
    return lib


def load():
    global _lib, _load_error
    if _lib is not None:
        return _lib
    try:
        _lib = _bind(ctypes.CDLL(_LIB_PATH))
    except OSError as e:  # pragma: no cover
        _load_error = e
        raise ImportError(
            f"FlashMoE HIP library not found/loadable at {_LIB_PATH}: {e}. "
            "Build it with: python -c 'import __graft_entry__; __graft_entry__.build()' "
            "(hipcc --offload-arch=gfx950)."
        ) from e
    return _lib


def is_available() -> bool:
    try:
        load()
        return True
    except ImportError:
        return False


def check(rc: int, what: str):
    if rc != 0:
        msg = load().fm_last_error().decode()
        raise RuntimeError(f"{what} failed (rc={rc}): {msg}")
