"""Worker script for distributed execution — mirror of the reference's
flashmoe/worker.py:11-71: load config, pick the GPU from rank env vars,
create random tensors of the compiled shapes, run moe_forward."""
import json
import os
import sys


def main():
    import torch

    if len(sys.argv) < 2:
        print("ERROR: Config path not provided", file=sys.stderr)
        sys.exit(1)
    config_path = sys.argv[1]
    with open(config_path) as f:
        config = json.load(f)

    rank = int(os.environ.get("RANK", os.environ.get("LOCAL_RANK", "0")))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))

    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from flashmoe_amd import moe
    from flashmoe_amd.config import torch_dtype_of, weight_dtype_of

    moe.initialize(config_path, rank=rank, world_size=world_size)
    if world_size > 1:
        import torch.distributed as dist

        if not dist.is_initialized():
            dist.init_process_group("nccl")

    device_id = rank % torch.cuda.device_count()
    print(f"Process {rank}/{world_size} using GPU {device_id}", flush=True)

    mini_batch = config["mini_batch"]
    seq_len = config["sequence_len"]
    H = config["hidden_size"]
    inter = config["intermediate_size"]
    E = config["num_experts"]
    dtype = torch_dtype_of(config["torch_dtype"])
    wdtype = weight_dtype_of(config["torch_dtype"])
    nLx = moe.get_num_local_experts()
    print(f"Process {rank}: Creating {nLx} local experts (total {E})", flush=True)

    # reference protocol: torch.randn, per-rank seed (worker.py:56-58,
    # flash_bench.cu:40-41)
    torch.manual_seed(47 + rank)
    input_tensor = torch.randn(mini_batch, seq_len, H, dtype=dtype, device="cuda")
    gate_weights = torch.randn(H, E, dtype=dtype, device="cuda")
    expert_weights = torch.randn(
        nLx, 2, inter, H, dtype=dtype, device="cuda").to(wdtype)

    print(f"Process {rank}: Calling moe_forward...", flush=True)
    output = moe.moe_forward(input_tensor, gate_weights, expert_weights)
    torch.cuda.synchronize()
    print(f"Process {rank}: Completed! Output: {tuple(output.shape)}", flush=True)
    return output


if __name__ == "__main__":
    main()
