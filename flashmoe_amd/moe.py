"""Host-side mirror of the reference's pybind11 module `flashmoe._C`
(csrc/python_bindings.cu:194-217): initialize / moe_forward / finalize /
get_compiled_config / get_num_local_experts — same names, argument meaning
and error behaviour, implemented over the C-ABI (include/flashmoe_abi.h).
"""
from __future__ import annotations

import ctypes
import os

from . import _ext
from .config import (element_size_of, load_config, torch_dtype_of,
                     weight_dtype_of)

_state = {
    "initialized": False,
    "cfg": None,
    "rank": 0,
    "world": 1,
    "gate_out": None,  # persistent [S, PX] buffer (internal, like the
    # reference's flat-buffer gate output, python_bindings.cu:70-74)
}


def _check_cuda():
    import torch

    if not torch.cuda.is_available():
        raise RuntimeError(
            "FlashMoE-AMD requires a ROCm GPU (gfx950); torch.cuda.is_available() is False"
        )


def initialize(config_path: str | None = None, rank: int | None = None,
               world_size: int | None = None):
    """Mirror of _C.initialize (python_bindings.cu:157-159).

    rank/world default from torch.distributed env vars (RANK/WORLD_SIZE)
    when present, else single process.
    """
    import torch

    if _state["initialized"]:
        raise RuntimeError("initialize() already called")
    _check_cuda()
    lib = _ext.load()  # fails loudly if the HIP extension is missing
    cfg = load_config(config_path)
    if rank is None:
        rank = int(os.environ.get("RANK", "0"))
    if world_size is None:
        world_size = int(os.environ.get("WORLD_SIZE", "1"))
    torch.cuda.set_device(rank % torch.cuda.device_count())
    c = _ext.FMConfig(
        num_experts=cfg["num_experts"],
        expert_top_k=cfg["expert_top_k"],
        capacity_factor=cfg["capacity_factor"],
        drop_tokens=cfg["drop_tokens"],
        hidden_act=cfg["hidden_act"],
        hidden_size=cfg["hidden_size"],
        intermediate_size=cfg["intermediate_size"],
        sequence_len=cfg["sequence_len"],
        mini_batch=cfg["mini_batch"],
        dtype=cfg["torch_dtype"],
        is_training=cfg["is_training"],
    )
    _ext.check(lib.fm_initialize(ctypes.byref(c), rank, world_size), "fm_initialize")
    _state.update(initialized=True, cfg=cfg, rank=rank, world=world_size)
    # persistent gate_out buffer
    S = cfg["sequence_len"] * cfg["mini_batch"]
    PX = (cfg["num_experts"] + 63) // 64 * 64
    _state["gate_out"] = torch.empty(
        S, PX, dtype=torch_dtype_of(cfg["torch_dtype"]), device="cuda"
    )


def finalize():
    """Mirror of _C.finalize (python_bindings.cu:163-168)."""
    if not _state["initialized"]:
        raise RuntimeError("finalize() before initialize()")
    _ext.check(_ext.load().fm_finalize(), "fm_finalize")
    _state.update(initialized=False, cfg=None, gate_out=None)
    _state.pop("p2p", None)        # heap freed by fm_finalize
    _state.pop("ep_buffers", None)
    # a P2P setup failure is scoped to one initialize/finalize cycle
    _state.pop("p2p_failed", None)
    _state.pop("p2p_validated", None)


def get_compiled_config() -> dict:
    """Mirror of _C.get_compiled_config (python_bindings.cu:170-183 /
    ops.py:63-71): dict with S, H, E, P, PX, element_size_bytes."""
    if not _state["initialized"]:
        # the reference reads compile-time constants without initialize;
        # we read the default config file (same contract)
        cfg = load_config()
        S = cfg["sequence_len"] * cfg["mini_batch"]
        return {
            "S": S,
            "H": cfg["hidden_size"],
            "E": cfg["num_experts"],
            "P": cfg["intermediate_size"],
            "PX": (cfg["num_experts"] + 63) // 64 * 64,
            "element_size_bytes": element_size_of(cfg["torch_dtype"]),
        }
    lib = _ext.load()
    vals = [ctypes.c_int64() for _ in range(6)]
    _ext.check(lib.fm_get_compiled_config(*[ctypes.byref(v) for v in vals]),
               "fm_get_compiled_config")
    S, H, E, P, PX, esz = [int(v.value) for v in vals]
    return {"S": S, "H": H, "E": E, "P": P, "PX": PX, "element_size_bytes": esz}


def get_num_local_experts() -> int:
    """Mirror of _C.get_num_local_experts (python_bindings.cu:185-189)."""
    if not _state["initialized"]:
        raise RuntimeError("Must call initialize() first")
    return _ext.load().fm_get_num_local_experts()


def _validate_forward_args(input, gate_weights, expert_weights):
    """Shape/device/contiguity validation mirroring the reference's
    TORCH_CHECKs (python_bindings.cu:22-70) against the frozen config."""
    import torch

    if not _state["initialized"]:
        raise RuntimeError("Must call initialize() before moe_forward")
    cc = get_compiled_config()
    S, H, E, P = cc["S"], cc["H"], cc["E"], cc["P"]
    nLx = get_num_local_experts()
    for name, t in (("Input", input), ("Gate weights", gate_weights),
                    ("Expert weights", expert_weights)):
        if not t.is_cuda:
            raise ValueError(f"{name} must be CUDA tensor")
        if not t.is_contiguous():
            raise ValueError(f"{name} must be contiguous")
    if input.dim() != 3:
        raise ValueError("Input must be 3D [batch, seq, H]")
    if input.size(0) * input.size(1) != S:
        raise ValueError(
            f"Input batch*seq must equal compiled S={S}. Got batch={input.size(0)}, "
            f"seq={input.size(1)} (product={input.size(0) * input.size(1)})"
        )
    if input.size(2) != H:
        raise ValueError(f"Input hidden_size must equal compiled H={H}. Got {input.size(2)}")
    if tuple(gate_weights.shape) != (H, E):
        raise ValueError(
            f"Gate weights must be [H={H}, E={E}]. Got "
            f"[{gate_weights.size(0)}, {gate_weights.size(1)}]"
        )
    if expert_weights.size(0) != nLx:
        raise ValueError(
            f"Expert count mismatch. Expected {nLx} local experts, got {expert_weights.size(0)}"
        )
    if expert_weights.size(1) != 2:
        raise ValueError("Expert weights must have up and down projections [nLx, 2, P, H]")
    if expert_weights.size(2) != P or expert_weights.size(3) != H:
        raise ValueError(
            f"Expert weights must be [*, 2, P={P}, H={H}]. Got "
            f"[*, 2, {expert_weights.size(2)}, {expert_weights.size(3)}]"
        )
    expect_dtype = torch_dtype_of(_state["cfg"]["torch_dtype"])
    expect_wdtype = weight_dtype_of(_state["cfg"]["torch_dtype"])
    for name, t, want in (("Input", input, expect_dtype),
                          ("Gate weights", gate_weights, expect_dtype),
                          ("Expert weights", expert_weights, expect_wdtype)):
        if t.dtype != want:
            raise ValueError(f"{name} dtype {t.dtype} != compiled {want}")
    return S, H, E, P, nLx


def moe_forward(input, gate_weights, expert_weights):
    """Mirror of _C.moe_forward (python_bindings.cu:17-151): one DMoE
    forward on this rank's tokens. Single-rank path; world>1 uses the EP
    pipeline in ep.py (same result per rank)."""
    import torch

    S, H, E, P, nLx = _validate_forward_args(input, gate_weights, expert_weights)
    if _state["world"] != 1:
        from . import ep

        return ep.moe_forward_ep(input, gate_weights, expert_weights)
    lib = _ext.load()
    out = torch.empty_like(input)
    gate_out = _state["gate_out"]
    stream = torch.cuda.current_stream().cuda_stream
    _ext.check(
        lib.fm_moe_forward(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(input.data_ptr()),
            ctypes.c_void_p(gate_weights.data_ptr()),
            ctypes.c_void_p(expert_weights.data_ptr()),
            None,
            None,
            ctypes.c_void_p(gate_out.data_ptr()),
            ctypes.c_void_p(out.data_ptr()),
            S,
        ),
        "fm_moe_forward",
    )
    return out


def gate_output():
    """The [S, PX] softmax probabilities of the last forward (the
    reference exposes these as the first slab of its output buffer,
    moe.cuh:128-131)."""
    return _state["gate_out"]
