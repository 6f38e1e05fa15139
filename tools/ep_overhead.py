import json, sys, tempfile, time, os
sys.path.insert(0, "/root/repo")
import torch, torch.distributed as dist
from flashmoe_amd import moe, ep

cfg = {"capacity_factor": 1, "drop_tokens": 1, "expert_top_k": 2,
       "global_batch": 256, "is_training": 0, "hidden_act": 0,
       "hidden_size": 1024, "intermediate_size": 4096, "mini_batch": 1,
       "moe_frequency": 1, "num_experts": 8, "num_layers": 1,
       "sequence_len": 4096, "torch_dtype": 2, "vocab_size": 32000}
f = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False); json.dump(cfg, f); f.close()
os.environ.setdefault("MASTER_ADDR", "127.0.0.1"); os.environ.setdefault("MASTER_PORT", "29529")
os.environ.setdefault("RANK", "0"); os.environ.setdefault("WORLD_SIZE", "1")
moe.initialize(f.name, rank=0, world_size=1)
dist.init_process_group("nccl", rank=0, world_size=1)
torch.manual_seed(47)
S, H, P, E = 4096, 1024, 4096, 8
x = torch.randn(1, S, H, dtype=torch.bfloat16, device="cuda")
gw = torch.randn(H, E, dtype=torch.bfloat16, device="cuda")
ew = torch.randn(E, 2, P, H, dtype=torch.bfloat16, device="cuda")

def timeit(fn, n=40):
    for _ in range(10): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1e6

t_single = timeit(lambda: moe.moe_forward(x, gw, ew))
t_padded = timeit(lambda: ep.moe_forward_ep(x, gw, ew))       # world1: exchange skipped
t_p2p    = timeit(lambda: ep.moe_forward_ep_p2p(x, gw, ew))   # self-heap store+signal+wait
print(json.dumps({"single_rank_us": round(t_single,1), "ep_padded_world1_us": round(t_padded,1),
                  "ep_p2p_world1_us": round(t_p2p,1)}))
