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

The pure-torch helpers (plan_dispatch / segment_recv) are device-agnostic
so the CPU (gloo) tests cover the exchange logic exactly as the GPU path
runs it.
"""
from __future__ import annotations

import ctypes
from dataclasses import dataclass

import torch


@dataclass
class DispatchPlan:
    """Packing of this rank's routed tokens for the all-to-all.

    order: indices into the (expert-major, slot-minor) routed list such
    that rows are grouped by destination rank (= expert // nLx), expert
    ascending within a rank, slot (arrival order) ascending within an
    expert — the canonical order both sides agree on.
    """

    send_counts: torch.Tensor  # [world] rows sent to each rank
    send_expert_counts: torch.Tensor  # [world, nLx] rows per (dst, local expert)
    token_idx: torch.Tensor  # [n] source-local token index, canonical order
    prob_sum: torch.Tensor  # [n] float32
    expert_of_row: torch.Tensor  # [n] global expert id


def plan_dispatch(routed_counts: torch.Tensor, token_idx_2d: torch.Tensor,
                  prob_sum_2d: torch.Tensor, world: int) -> DispatchPlan:
    """routed_counts: [E]; token_idx_2d/prob_sum_2d: [E, EC] (slots beyond
    routed_counts[e] are junk). Experts are owned contiguously:
    rank r owns experts [r*nLx, (r+1)*nLx) (uniform split,
    bootstrap.cuh:36-52)."""
    E = routed_counts.numel()
    nLx = E // world
    toks, probs, experts = [], [], []
    send_counts = torch.zeros(world, dtype=torch.long)
    send_expert_counts = torch.zeros(world, nLx, dtype=torch.long)
    for e in range(E):
        n = int(routed_counts[e])
        if n == 0:
            continue
        toks.append(token_idx_2d[e, :n])
        probs.append(prob_sum_2d[e, :n])
        experts.append(torch.full((n,), e, dtype=torch.long))
        send_counts[e // nLx] += n
        send_expert_counts[e // nLx, e % nLx] += n
    cat = (lambda lst, dt: torch.cat(lst) if lst else torch.empty(0, dtype=dt))
    return DispatchPlan(
        send_counts=send_counts,
        send_expert_counts=send_expert_counts,
        token_idx=cat(toks, torch.int64).long(),
        prob_sum=cat(probs, torch.float32).float(),
        expert_of_row=cat(experts, torch.int64),
    )


def segment_recv(recv_expert_counts: torch.Tensor):
    """recv_expert_counts: [world, nLx] rows per (source, local expert) in
    the canonical receive order (source-major, expert-minor). Returns a
    permutation grouping the received rows by local expert (expert-major,
    source-minor, slot order preserved) and the per-expert row counts —
    the order the grouped-FFN consumes, inverse applied before the
    return exchange."""
    world, nLx = recv_expert_counts.shape
    counts = recv_expert_counts
    start = torch.cat([torch.zeros(1, dtype=torch.long), counts.flatten().cumsum(0)[:-1]])
    offs = start.reshape(world, nLx)
    perm = []
    per_expert = torch.zeros(nLx, dtype=torch.long)
    for le in range(nLx):
        for r in range(world):
            n = int(counts[r, le])
            if n:
                perm.append(torch.arange(offs[r, le], offs[r, le] + n))
                per_expert[le] += n
    perm_t = torch.cat(perm) if perm else torch.empty(0, dtype=torch.long)
    return perm_t, per_expert


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


def moe_forward_ep(input, gate_weights, expert_weights, group=None):
    """The EP forward on GPU ranks (called from moe.moe_forward when
    world > 1). Requires torch.distributed initialized with the nccl
    (RCCL) backend."""
    import numpy as np
    import torch.distributed as dist

    from . import _ext, moe

    lib = _ext.load()
    world = dist.get_world_size(group)
    cc = moe.get_compiled_config()
    S, H, E = cc["S"], cc["H"], cc["E"]
    nLx = E // world
    cfg = moe._state["cfg"]
    k = cfg["expert_top_k"]
    base = -(-S // E) if cfg["drop_tokens"] else S
    EC = base * cfg["capacity_factor"] * k
    gate_out = moe._state["gate_out"]
    stream = torch.cuda.current_stream().cuda_stream
    x2d = input.view(S, H)

    # 1. local gate (fills the library's tokenIds/eC)
    _ext.check(lib.fm_gate_forward(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(x2d.data_ptr()),
        ctypes.c_void_p(gate_weights.data_ptr()),
        ctypes.c_void_p(gate_out.data_ptr()), S), "fm_gate_forward")

    # 2. routing to host
    counts = np.zeros(E, dtype=np.uint32)
    tok = np.zeros(E * EC, dtype=np.uint32)
    ps = np.zeros(E * EC, dtype=np.float32)
    _ext.check(lib.fm_read_routing(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(counts.ctypes.data),
        ctypes.c_void_p(tok.ctypes.data),
        ctypes.c_void_p(ps.ctypes.data)), "fm_read_routing")
    plan = plan_dispatch(torch.from_numpy(counts.astype(np.int64)),
                         torch.from_numpy(tok.reshape(E, EC).astype(np.int64)),
                         torch.from_numpy(ps.reshape(E, EC)), world)

    # 3. exchange counts, then rows
    all_expert_counts = exchange_rows(
        plan.send_expert_counts.reshape(world, nLx).to(input.device),
        torch.ones(world, dtype=torch.long),
        torch.ones(world, dtype=torch.long), group).cpu()  # [world, nLx]
    recv_counts = all_expert_counts.sum(1)
    send_rows = x2d.index_select(0, plan.token_idx.to(input.device))
    recv_rows = exchange_rows(send_rows, plan.send_counts, recv_counts, group)

    # 4. group by local expert, FFN, restore order
    perm, per_expert = segment_recv(all_expert_counts)
    perm_d = perm.to(input.device)
    grouped = recv_rows.index_select(0, perm_d) if perm.numel() else recv_rows
    out_grouped = torch.empty_like(grouped)
    off = 0
    for le in range(nLx):
        n = int(per_expert[le])
        if n == 0:
            continue
        _ext.check(lib.fm_expert_ffn(
            ctypes.c_void_p(stream),
            ctypes.c_void_p(grouped[off:off + n].data_ptr()),
            ctypes.c_void_p(expert_weights.data_ptr()),
            None, None,
            ctypes.c_void_p(out_grouped[off:off + n].data_ptr()),
            n, le), "fm_expert_ffn")
        off += n
    result_rows = torch.empty_like(recv_rows)
    if perm.numel():
        result_rows.index_copy_(0, perm_d, out_grouped)
    else:
        result_rows = out_grouped

    # 5. return exchange (reverse splits), combine at source
    returned = exchange_rows(result_rows, recv_counts, plan.send_counts, group)
    n = returned.shape[0]
    tok_d = plan.token_idx.to(torch.int32).to(input.device)
    if k > 1:
        probs = gate_out[plan.token_idx.to(input.device),
                         plan.expert_of_row.to(input.device)].float()
        scale = (probs / plan.prob_sum.to(input.device)).contiguous()
    else:
        scale = torch.ones(n, dtype=torch.float32, device=input.device)
    _ext.check(lib.fm_combine(
        ctypes.c_void_p(stream),
        ctypes.c_void_p(returned.data_ptr()),
        ctypes.c_void_p(tok_d.data_ptr()),
        ctypes.c_void_p(scale.data_ptr()), n, 1), "fm_combine")
    out = torch.empty_like(input)
    _ext.check(lib.fm_combine_finalize(
        ctypes.c_void_p(stream), ctypes.c_void_p(out.data_ptr()), S),
        "fm_combine_finalize")
    return out
