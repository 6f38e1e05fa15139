"""Expert-parallel (multi-GPU) pipeline — SURVEY.md par.8e semantics.

The reference runs dispatch/FFN/combine inside one kernel with one-sided
NVSHMEM/P2P puts (os/packet.cuh, os/processor/processor.cuh); round 1
here is a host-orchestrated pipeline around torch.distributed all-to-all
(RCCL over xGMI): local gate -> pack rows per owner rank -> all_to_all ->
local-expert FFN -> all_to_all back -> combine at the source. Per-rank
results are identical to the single-rank path on that rank's tokens
(per-(rank, expert) capacity, combine at source — DESIGN.md par.4), which
is what the gloo CPU tests check. Round 2 replaces the exchange with
in-kernel xGMI stores.

`exchange_rows` doubles as the gloo (CPU test) emulation of the
all_to_all, so tests/test_ep_padded_gloo.py exercises the same exchange
semantics the RCCL path uses.
"""
from __future__ import annotations

import ctypes
import os

import torch


def exchange_rows(x_rows: torch.Tensor, send_counts: torch.Tensor,
                  recv_counts: torch.Tensor, group=None) -> torch.Tensor:
    """all_to_all of variable row counts. RCCL uses all_to_all_single;
    gloo (CPU tests) has no all_to_all, so it is emulated with
    isend/irecv pairs — same wire semantics."""
    import torch.distributed as dist

    out = x_rows.new_empty(int(recv_counts.sum()), *x_rows.shape[1:])
    x_rows = x_rows.contiguous()
    backend = dist.get_backend(group)
    if backend == "gloo":
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        s_off = [0] + send_counts.cumsum(0).tolist()
        r_off = [0] + recv_counts.cumsum(0).tolist()
        reqs = []
        # local copy
        out[r_off[rank]:r_off[rank + 1]] = x_rows[s_off[rank]:s_off[rank + 1]]
        for peer in range(world):
            if peer == rank:
                continue
            if send_counts[peer]:
                reqs.append(dist.isend(x_rows[s_off[peer]:s_off[peer + 1]],
                                       peer, group=group))
            if recv_counts[peer]:
                reqs.append(dist.irecv(out[r_off[peer]:r_off[peer + 1]],
                                       peer, group=group))
        for r in reqs:
            r.wait()
    else:
        dist.all_to_all_single(
            out, x_rows,
            output_split_sizes=recv_counts.tolist(),
            input_split_sizes=send_counts.tolist(),
            group=group,
        )
    return out


def _padded_buffers(moe, world, E, EC, H, dtype, dev):
    """Cached static exchange buffers for the padded EP pipeline."""
    import torch

    key = (world, E, EC, H, str(dtype))
    cache = moe._state.setdefault("ep_buffers", {})
    if cache.get("key") != key:
        nLx = E // world
        cache.clear()
        cache["key"] = key
        cache["send"] = torch.empty(E * EC, H, dtype=dtype, device=dev)
        cache["recv"] = torch.empty(E * EC, H, dtype=dtype, device=dev)
        cache["ffn_out"] = torch.empty(E * EC, H, dtype=dtype, device=dev)
        cache["ret"] = torch.empty(E * EC, H, dtype=dtype, device=dev)
        # segment s = (source rank, local expert): expert = s % nLx
        cache["seg_expert"] = (torch.arange(world * nLx, device=dev,
                                            dtype=torch.int32) % nLx).contiguous()
    return cache


def _p2p_setup(moe, lib, world, group):
    """One-time hipIpc heap exchange for the one-sided transport."""
    import torch.distributed as dist

    from . import _ext

    _ext.check(lib.fm_heap_init(), "fm_heap_init")
    if world > 1:
        h = (ctypes.c_char * 64)()
        _ext.check(lib.fm_heap_handle(h), "fm_heap_handle")
        gathered = [None] * world
        dist.all_gather_object(gathered, bytes(h), group=group)
        blob = (ctypes.c_char * (64 * world)).from_buffer_copy(
            b"".join(gathered))
        _ext.check(lib.fm_heap_connect(blob), "fm_heap_connect")
    else:
        _ext.check(lib.fm_heap_connect(None), "fm_heap_connect")
    recv = ctypes.c_void_p()
    ret = ctypes.c_void_p()
    _ext.check(lib.fm_heap_ptrs(ctypes.byref(recv), ctypes.byref(ret)),
               "fm_heap_ptrs")
    moe._state["p2p"] = {"recv": recv.value, "ret": ret.value}
    return moe._state["p2p"]


def moe_forward_ep_p2p(input, gate_weights, expert_weights, group=None):
    """EP forward over the one-sided xGMI transport (FLASHMOE_P2P=1):
    in-kernel stores into peers' heap cells + system-scope signals
    replace both all_to_alls (os/packet.cuh:214-258 semantics). Heap cell
    layout is identical to the padded path, so the FFN/combine stages are
    shared."""
    import torch.distributed as dist

    from . import _ext, moe

    lib = _ext.load()
    world = dist.get_world_size(group) if dist.is_initialized() else 1
    cc = moe.get_compiled_config()
    S, H, E = cc["S"], cc["H"], cc["E"]
    cfg = moe._state["cfg"]
    base = -(-S // E) if cfg["drop_tokens"] else S
    EC = base * cfg["capacity_factor"] * cfg["expert_top_k"]
    nLx = E // world
    gate_out = moe._state["gate_out"]
    stream = torch.cuda.current_stream().cuda_stream
    dev = input.device
    x2d = input.view(S, H)
    p2p = moe._state.get("p2p") or _p2p_setup(moe, lib, world, group)
    buf = _padded_buffers(moe, world, E, EC, H, input.dtype, dev)

    _ext.check(lib.fm_gate_forward(
        ctypes.c_void_p(stream), ctypes.c_void_p(x2d.data_ptr()),
        ctypes.c_void_p(gate_weights.data_ptr()),
        ctypes.c_void_p(gate_out.data_ptr()), S), "fm_gate_forward")
    # one-sided dispatch + in-kernel wait for my experts' cells
    _ext.check(lib.fm_dispatch_p2p(
        ctypes.c_void_p(stream), ctypes.c_void_p(x2d.data_ptr())),
        "fm_dispatch_p2p")
    _ext.check(lib.fm_expert_ffn_segments(
        ctypes.c_void_p(stream), ctypes.c_void_p(p2p["recv"]),
        ctypes.c_void_p(buf["seg_expert"].data_ptr()), world * nLx,
        ctypes.c_void_p(expert_weights.data_ptr()),
        ctypes.c_void_p(buf["ffn_out"].data_ptr())), "fm_expert_ffn_segments")
    _ext.check(lib.fm_return_p2p(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(buf["ffn_out"].data_ptr())), "fm_return_p2p")
    out = torch.empty_like(input)
    _ext.check(lib.fm_combine_padded(
        ctypes.c_void_p(stream), ctypes.c_void_p(p2p["ret"]),
        ctypes.c_void_p(gate_out.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), S), "fm_combine_padded")
    return out


def moe_forward_ep(input, gate_weights, expert_weights, group=None,
                   _stub_exchange=False):
    """Capacity-padded EP forward (the product multi-GPU path): the
    exchange unit is a fixed [E, EC, H] expert-major buffer (the
    reference's symmetric-heap cell layout, types.cuh:1014-1032), so the
    all_to_all has STATIC equal splits and the step performs NO host
    synchronization. Rows past each expert's routed count are
    garbage-in/garbage-out and dropped by the source-side combine.

    _stub_exchange=True skips the exchanges (compute-only arm of the
    overlap-efficiency metric; output is not the DMoE result).
    """
    import torch
    import torch.distributed as dist

    from . import _ext, moe

    # One-sided xGMI transport is the DEFAULT data plane (the
    # reference's intra-node mode); FLASHMOE_P2P=0 forces the RCCL
    # all_to_all pipeline, FLASHMOE_P2P=1 forces P2P with no fallback.
    # In the default "auto" mode the first P2P step is validated with
    # fm_p2p_error_check (one extra sync); any setup or timeout failure
    # falls back to the all_to_all path for the rest of the run.
    p2p_mode = os.environ.get("FLASHMOE_P2P", "auto")
    if p2p_mode != "0" and not _stub_exchange and \
            not moe._state.get("p2p_failed"):
        if p2p_mode == "1":
            return moe_forward_ep_p2p(input, gate_weights, expert_weights,
                                      group)
        try:
            out = moe_forward_ep_p2p(input, gate_weights, expert_weights,
                                     group)
            if not moe._state.get("p2p_validated"):
                lib = _ext.load()
                stream = torch.cuda.current_stream().cuda_stream
                _ext.check(lib.fm_p2p_error_check(ctypes.c_void_p(stream)),
                           "fm_p2p_error_check")
                moe._state["p2p_validated"] = True
            return out
        except Exception as exc:  # pragma: no cover - multi-GPU only
            moe._state["p2p_failed"] = True
            print(f"flashmoe: P2P transport failed ({exc}); "
                  "falling back to RCCL all_to_all", flush=True)
    lib = _ext.load()
    world = dist.get_world_size(group)
    cc = moe.get_compiled_config()
    S, H, E = cc["S"], cc["H"], cc["E"]
    cfg = moe._state["cfg"]
    k = cfg["expert_top_k"]
    base = -(-S // E) if cfg["drop_tokens"] else S
    EC = base * cfg["capacity_factor"] * k
    gate_out = moe._state["gate_out"]
    stream = torch.cuda.current_stream().cuda_stream
    dev = input.device
    x2d = input.view(S, H)
    nLx = E // world
    buf = _padded_buffers(moe, world, E, EC, H, input.dtype, dev)

    # 1. local gate (fills tokenIds/eC) + pack the dispatch cells
    _ext.check(lib.fm_gate_forward(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(x2d.data_ptr()),
        ctypes.c_void_p(gate_weights.data_ptr()),
        ctypes.c_void_p(gate_out.data_ptr()), S), "fm_gate_forward")
    _ext.check(lib.fm_pack_dispatch(
        ctypes.c_void_p(stream), ctypes.c_void_p(x2d.data_ptr()),
        ctypes.c_void_p(buf["send"].data_ptr())), "fm_pack_dispatch")

    # 2. dispatch all_to_all: equal static chunks of nLx*EC rows
    if _stub_exchange or world == 1:
        recv = buf["send"]
    else:
        dist.all_to_all_single(buf["recv"], buf["send"], group=group)
        recv = buf["recv"]

    # 3. grouped FFN over [world, nLx, EC, H] segments (2 launches)
    _ext.check(lib.fm_expert_ffn_segments(
        ctypes.c_void_p(stream), ctypes.c_void_p(recv.data_ptr()),
        ctypes.c_void_p(buf["seg_expert"].data_ptr()), world * nLx,
        ctypes.c_void_p(expert_weights.data_ptr()),
        ctypes.c_void_p(buf["ffn_out"].data_ptr())), "fm_expert_ffn_segments")

    # 4. return all_to_all (same static splits), combine at the source
    if _stub_exchange or world == 1:
        returned = buf["ffn_out"]
    else:
        dist.all_to_all_single(buf["ret"], buf["ffn_out"], group=group)
        returned = buf["ret"]
    out = torch.empty_like(input)
    _ext.check(lib.fm_combine_padded(
        ctypes.c_void_p(stream), ctypes.c_void_p(returned.data_ptr()),
        ctypes.c_void_p(gate_out.data_ptr()),
        ctypes.c_void_p(out.data_ptr()), S), "fm_combine_padded")
    return out
