"""GPU EP worker: runs the expert-parallel pipeline end-to-end under
torch.distributed (nccl/RCCL) and checks each rank's output against the
single-rank HIP path on the same tokens (the EP identity).
Launched by tests/test_gpu_ep.py via torch.distributed.run; works at any
world size that divides num_experts (world 1 covers the code path on a
single-GPU box; the driver's 8-GPU scale run covers world > 1).
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
    world = int(os.environ.get("WORLD_SIZE", "1"))
    dtype_code = int(os.environ.get("FM_TEST_DTYPE", "2"))
    cfg = {
        "capacity_factor": 2, "drop_tokens": 1, "expert_top_k": 2,
        "global_batch": 256, "is_training": 0, "hidden_act": 0,
        "hidden_size": 256, "intermediate_size": 512, "mini_batch": 1,
        "moe_frequency": 1, "num_experts": 8, "num_layers": 1,
        "sequence_len": 512, "torch_dtype": dtype_code, "vocab_size": 32000,
    }
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump(cfg, f)
        cfg_path = f.name

    from flashmoe_amd import ep, moe

    moe.initialize(cfg_path, rank=rank, world_size=world)
    if world > 1 or True:
        dist.init_process_group("nccl")

    S, H, P, E = 512, 256, 512, 8
    nLx = E // world
    from flashmoe_amd.config import weight_dtype_of

    torch.manual_seed(1234)  # global weights identical on all ranks
    gw = torch.randn(H, E, dtype=torch.bfloat16, device="cuda")
    ew_full = torch.randn(E, 2, P, H, dtype=torch.bfloat16,
                          device="cuda").to(weight_dtype_of(dtype_code))
    ew_local = ew_full[rank * nLx:(rank + 1) * nLx].contiguous()
    torch.manual_seed(47 + rank)  # per-rank tokens
    x = torch.randn(1, S, H, dtype=torch.bfloat16, device="cuda")

    out_ep = ep.moe_forward_ep(x, gw, ew_local)
    torch.cuda.synchronize()

    # exercise the compute-only overlap-probe arm (results unchecked -
    # it is a timing stub; just must run)
    ep.moe_forward_ep(x, gw, ew_local, _stub_exchange=True)
    torch.cuda.synchronize()

    # reference: the single-rank HIP path on this rank's tokens with ALL
    # experts (requires a world-1 re-init)
    moe.finalize()
    moe._state["initialized"] = False  # defensive
    moe.initialize(cfg_path, rank=rank, world_size=1)
    out_ref = moe.moe_forward(x, gw, ew_full.contiguous())
    torch.cuda.synchronize()

    a = out_ep.float().cpu().numpy()
    b = out_ref.float().cpu().numpy()
    scale = max(1.0, float(np.abs(b).max()))
    # dtype 5 (MX): the EP segments and single-rank paths tile the GEMMs
    # differently, so the re-quantized intermediate can flip an e4m3
    # rounding step (~blockscale*ulp on a few elements) - wider bar
    rtol, atol_s = (5e-2, 5e-3) if dtype_code == 5 else (2e-2, 2e-3)
    ok = np.allclose(a, b, rtol=rtol, atol=atol_s * scale)
    err = float(np.abs(a - b).max())
    print(f"rank {rank}: EP vs single-rank max abs err {err:.5f} "
          f"(scale {scale:.1f}) -> {'OK' if ok else 'FAIL'}", flush=True)
    dist.destroy_process_group()
    if not ok:
        sys.exit(1)


if __name__ == "__main__":
    main()
