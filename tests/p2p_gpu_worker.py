"""GPU P2P worker: two ranks SHARING one GPU validate the one-sided
hipIpc transport protocol (fm_heap_* / fm_dispatch_p2p / fm_return_p2p)
at world size 2 — a real cross-PROCESS hipIpc mapping and system-scope
signal exchange, which world-1 self-mapping cannot exercise
(os/packet.cuh:214-258 semantics; the xGMI link itself needs the
driver's multi-GPU box, but the protocol — handle exchange, peer heap
stores, seq-tagged flags, bounded waits — is identical on one device).

Bootstrap runs over gloo (RCCL cannot place two ranks on one device);
the data plane is entirely in-kernel one-sided stores + signals.

Launched by tests/test_gpu_p2p.py via torch.distributed.run.
"""
import json
import os
import sys
import tempfile

import numpy as np
import torch
import torch.distributed as dist

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "2"))
    torch.cuda.set_device(0)  # both ranks on the same device
    dist.init_process_group("gloo")

    cfg = {
        "capacity_factor": 2, "drop_tokens": 1, "expert_top_k": 2,
        "global_batch": 256, "is_training": 0, "hidden_act": 0,
        "hidden_size": 256, "intermediate_size": 512, "mini_batch": 1,
        "moe_frequency": 1, "num_experts": 8, "num_layers": 1,
        "sequence_len": 512, "torch_dtype": 2, "vocab_size": 32000,
    }
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump(cfg, f)
        cfg_path = f.name

    from flashmoe_amd import ep, moe

    moe.initialize(cfg_path, rank=rank, world_size=world)

    S, H, P, E = 512, 256, 512, 8
    nLx = E // world
    torch.manual_seed(1234)  # identical global weights on both ranks
    gw = torch.randn(H, E, dtype=torch.bfloat16, device="cuda")
    ew_full = torch.randn(E, 2, P, H, dtype=torch.bfloat16, device="cuda")
    ew_local = ew_full[rank * nLx:(rank + 1) * nLx].contiguous()
    torch.manual_seed(47 + rank)  # per-rank tokens
    x = torch.randn(1, S, H, dtype=torch.bfloat16, device="cuda")

    out_p2p = ep.moe_forward_ep_p2p(x, gw, ew_local)
    torch.cuda.synchronize()
    dist.barrier()
    # two more forwards: the seq-tagged flags must keep working without
    # re-zeroing across calls (types.cuh:1045-1063 seqBit semantics)
    for _ in range(2):
        out_p2p = ep.moe_forward_ep_p2p(x, gw, ew_local)
    torch.cuda.synchronize()
    dist.barrier()

    # reference: the single-rank path on this rank's tokens, all experts
    moe.finalize()
    moe._state["initialized"] = False
    moe.initialize(cfg_path, rank=rank, world_size=1)
    out_ref = moe.moe_forward(x, gw, ew_full.contiguous())
    torch.cuda.synchronize()

    a = out_p2p.float().cpu().numpy()
    b = out_ref.float().cpu().numpy()
    scale = max(1.0, float(np.abs(b).max()))
    ok = np.allclose(a, b, rtol=2e-2, atol=2e-3 * scale)
    err = float(np.abs(a - b).max())
    print(f"rank {rank}: P2P world-{world} vs single-rank max abs err "
          f"{err:.5f} (scale {scale:.1f}) -> {'OK' if ok else 'FAIL'}",
          flush=True)
    dist.destroy_process_group()
    if not ok:
        sys.exit(1)


if __name__ == "__main__":
    main()
