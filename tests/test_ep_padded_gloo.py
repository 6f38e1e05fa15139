"""CPU (gloo, world 2) test of the capacity-padded EP pipeline layout:
fixed [E, EC, H] expert-major exchange cells with equal static splits
(the product multi-GPU path, flashmoe_amd/ep.py / fm_pack_dispatch /
fm_expert_ffn_segments / fm_combine_padded semantics), oracle compute.

Padding rows are POISONED with a large finite value: the test proves
they never leak into any rank's output (they are dropped by the
source-side combine), and that segment->expert and return routing are
correct across ranks.
"""
import os

import numpy as np
import pytest
import torch

from oracle.moe_oracle import OracleConfig, expert_capacity, expert_ffn, gate_forward, \
    moe_forward, route_tokens

WORLD = 2
POISON = 7777.0


def _padded_rank(rank, world, cfg, S, H, P, E, out_q):
    import torch.distributed as dist

    from flashmoe_amd.ep import exchange_rows

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        nLx = E // world
        k = cfg.expert_top_k
        EC = expert_capacity(S, cfg)
        g = np.random.default_rng(99)  # shared weights
        gate_w = g.standard_normal((E, H), dtype=np.float32)
        expert_w = g.standard_normal((E, 2, P, H), dtype=np.float32)
        gx = np.random.default_rng(7 + rank)
        x = gx.standard_normal((S, H), dtype=np.float32)

        # local gate + pack [E, EC, H] (fm_pack_dispatch semantics)
        gate_out, topk, mCw, _ = gate_forward(x, gate_w, cfg)
        lists, eC, _ = route_tokens(topk, cfg, S)
        send = np.full((E, EC, H), POISON, dtype=np.float32)
        for e in range(E):
            toks = lists[e]
            send[e, : len(toks)] = x[toks]

        # dispatch all_to_all: equal chunks of nLx*EC rows
        eq = torch.full((world,), nLx * EC, dtype=torch.long)
        recv = exchange_rows(torch.from_numpy(send.reshape(E * EC, H)), eq, eq)
        recv = recv.numpy().reshape(world, nLx, EC, H)

        # grouped FFN: segment (r, le) -> local expert le, ALL EC rows
        # (padding rows produce garbage outputs by design)
        ffn_out = np.empty_like(recv)
        for r in range(world):
            for le in range(nLx):
                ge = rank * nLx + le
                ffn_out[r, le] = expert_ffn(
                    recv[r, le], expert_w[ge, 0],
                    expert_w[ge, 1].reshape(-1), None, None, cfg)

        # return all_to_all (same splits) -> [E, EC, H] at the source
        ret = exchange_rows(torch.from_numpy(
            ffn_out.reshape(world * nLx * EC, H)), eq, eq)
        ret = ret.numpy().reshape(E, EC, H)

        # combine at source from LOCAL metadata only (fm_combine_padded)
        out = np.zeros((S, H), dtype=np.float32)
        for e in range(E):
            toks = lists[e]
            for i, t in enumerate(toks):
                sc = gate_out[t, e] / mCw[t] if k > 1 else 1.0
                out[t] += sc * ret[e, i]

        ref = moe_forward(x, gate_w.reshape(-1), expert_w, cfg)
        np.testing.assert_allclose(out, ref["moe_out"], rtol=1e-4, atol=1e-4)
        assert np.abs(out).max() < POISON / 2, "poisoned padding leaked"
        out_q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        out_q.put((rank, f"FAIL: {type(e).__name__}: {e}"))
        raise
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("topk,E", [(1, 4), (2, 8)])
def test_padded_ep_pipeline(topk, E):
    import torch.multiprocessing as mp

    cfg = OracleConfig(num_experts=E, expert_top_k=topk, capacity_factor=1,
                       drop_tokens=1, element="fp32")
    S, H, P = 128, 32, 48
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = str(29541 + topk)
    procs = [ctx.Process(target=_padded_rank, args=(r, WORLD, cfg, S, H, P, E, q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(WORLD)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"
