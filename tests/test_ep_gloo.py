"""CPU multi-process tests of the expert-parallel pipeline (gloo, world 2).

Runs the exact exchange/packing/segmentation code the GPU EP path uses
(flashmoe_amd/ep.py helpers + exchange_rows over gloo), with the oracle
supplying gate/FFN compute, and checks each rank's output equals the
single-rank oracle on that rank's tokens (the EP identity of
DESIGN.md par.4 / SURVEY.md par.8e).
"""
import os

import numpy as np
import pytest
import torch

from oracle.moe_oracle import OracleConfig, expert_ffn, gate_forward, moe_forward, route_tokens

WORLD = 2


def _ep_pipeline_rank(rank, world, cfg, S, H, P, E, seed, out_q):
    import torch.distributed as dist

    from flashmoe_amd.ep import exchange_rows, plan_dispatch, segment_recv

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        nLx = E // world
        k = cfg.expert_top_k
        g = np.random.default_rng(1234)  # same weights everywhere
        gate_w = g.standard_normal((E, H), dtype=np.float32)
        expert_w = g.standard_normal((E, 2, P, H), dtype=np.float32)
        gx = np.random.default_rng(47 + rank)  # per-rank tokens
        x = gx.standard_normal((S, H), dtype=np.float32)

        # 1. local gate (oracle compute)
        gate_out, topk, mCw, _ = gate_forward(x, gate_w, cfg)
        lists, eC, _ = route_tokens(topk, cfg, S)
        from oracle.moe_oracle import expert_capacity

        EC = expert_capacity(S, cfg)
        counts = torch.tensor([len(l) for l in lists], dtype=torch.long)
        tok2d = torch.zeros(E, EC, dtype=torch.long)
        ps2d = torch.zeros(E, EC, dtype=torch.float32)
        for e in range(E):
            n = len(lists[e])
            if n:
                tok2d[e, :n] = torch.from_numpy(lists[e])
                ps2d[e, :n] = torch.from_numpy(mCw[lists[e]])

        # 2-3. plan + exchanges (the product helpers under test)
        plan = plan_dispatch(counts, tok2d, ps2d, world)
        all_counts = exchange_rows(plan.send_expert_counts,
                                   torch.ones(world, dtype=torch.long),
                                   torch.ones(world, dtype=torch.long))
        recv_counts = all_counts.sum(1)
        send_rows = torch.from_numpy(x)[plan.token_idx]
        recv_rows = exchange_rows(send_rows, plan.send_counts, recv_counts)

        # 4. group by local expert, oracle FFN, restore order
        perm, per_expert = segment_recv(all_counts)
        grouped = recv_rows[perm] if perm.numel() else recv_rows
        out_grouped = torch.empty_like(grouped)
        off = 0
        for le in range(nLx):
            n = int(per_expert[le])
            if n:
                ge = rank * nLx + le
                z = expert_ffn(grouped[off:off + n].numpy(), expert_w[ge, 0],
                               expert_w[ge, 1].reshape(-1), None, None, cfg)
                out_grouped[off:off + n] = torch.from_numpy(z.astype(np.float32))
                off += n
        result_rows = torch.empty_like(recv_rows)
        if perm.numel():
            result_rows.index_copy_(0, perm, out_grouped)

        # 5. return + combine at source
        returned = exchange_rows(result_rows, recv_counts, plan.send_counts)
        out = np.zeros((S, H), dtype=np.float32)
        toks = plan.token_idx.numpy()
        if k > 1:
            probs = gate_out[toks, plan.expert_of_row.numpy()]
            scale = probs / plan.prob_sum.numpy()
        else:
            scale = np.ones(len(toks), dtype=np.float32)
        np.add.at(out, toks, returned.numpy() * scale[:, None])

        # reference: single-rank oracle on this rank's tokens, full E
        ref = moe_forward(x, gate_w.reshape(-1), expert_w, cfg)
        np.testing.assert_allclose(out, ref["moe_out"], rtol=1e-4, atol=1e-4)
        out_q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        out_q.put((rank, f"FAIL: {type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("topk,E", [(1, 4), (2, 8)])
def test_ep_pipeline_matches_single_rank_oracle(topk, E):
    import torch.multiprocessing as mp

    cfg = OracleConfig(num_experts=E, expert_top_k=topk, capacity_factor=1,
                       drop_tokens=1, element="fp32")
    S, H, P = 128, 32, 48
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = str(29531 + topk)
    procs = [ctx.Process(target=_ep_pipeline_rank,
                         args=(r, WORLD, cfg, S, H, P, E, 47, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_plan_dispatch_shapes():
    from flashmoe_amd.ep import plan_dispatch, segment_recv

    E, EC, world = 4, 8, 2
    counts = torch.tensor([3, 0, 8, 1])
    tok = torch.arange(E * EC).reshape(E, EC)
    ps = torch.ones(E, EC)
    plan = plan_dispatch(counts, tok, ps, world)
    assert plan.send_counts.tolist() == [3, 9]
    assert plan.send_expert_counts.tolist() == [[3, 0], [8, 1]]
    assert plan.token_idx.numel() == 12
    # canonical order: expert 0 slots, then expert 2 slots, then expert 3
    assert plan.expert_of_row.tolist() == [0] * 3 + [2] * 8 + [3] * 1

    # receiver-side segmentation: 2 sources x 2 local experts
    recv = torch.tensor([[2, 1], [3, 0]])
    perm, per_expert = segment_recv(recv)
    assert per_expert.tolist() == [5, 1]
    # canonical recv order: src0(e0 x2, e1 x1), src1(e0 x3)
    assert perm.tolist() == [0, 1, 3, 4, 5, 2]
