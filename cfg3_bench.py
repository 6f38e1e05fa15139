import json, sys, tempfile, time, ctypes
sys.path.insert(0, "/root/repo")
import torch
from flashmoe_amd import moe
import flashmoe_amd._ext as _ext

# BASELINE config 3 shape on ONE GPU (full E=64): S=4096 H=2048 P=8192 bf16 top-2
cfg = {"capacity_factor": 1, "drop_tokens": 1, "expert_top_k": 2,
       "global_batch": 256, "is_training": 0, "hidden_act": 0,
       "hidden_size": 2048, "intermediate_size": 8192, "mini_batch": 1,
       "moe_frequency": 1, "num_experts": 64, "num_layers": 1,
       "sequence_len": 4096, "torch_dtype": 2, "vocab_size": 32000}
f = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False); json.dump(cfg, f); f.close()
moe.initialize(f.name, rank=0, world_size=1)
torch.manual_seed(47)
S, H, P, E = 4096, 2048, 8192, 64
x = torch.randn(1, S, H, dtype=torch.bfloat16, device="cuda")
gw = torch.randn(H, E, dtype=torch.bfloat16, device="cuda")
ew = torch.randn(E, 2, P, H, dtype=torch.bfloat16, device="cuda")
for _ in range(10): moe.moe_forward(x, gw, ew)
torch.cuda.synchronize(); t0 = time.perf_counter()
K = 30
for _ in range(K): moe.moe_forward(x, gw, ew)
torch.cuda.synchronize(); dt = (time.perf_counter() - t0) / K
# phases
lib = _ext.load(); ms = (ctypes.c_float * 4)(); acc = [0.0]*4
out = torch.empty_like(x); go = moe.gate_output()
for _ in range(5):
    _ext.check(lib.fm_moe_forward_phased(
        ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
        ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(gw.data_ptr()),
        ctypes.c_void_p(ew.data_ptr()), None, None,
        ctypes.c_void_p(go.data_ptr()), ctypes.c_void_p(out.data_ptr()), S, ms), "p")
    for i in range(4): acc[i] += ms[i]/5
flops = 4.0 * S * 2 * H * P  # upper bound (pre-drop)
print(json.dumps({"workload": "BASELINE cfg3 shape on 1 GPU (E=64, S=4096, H=2048, P=8192, bf16 top-2)",
  "us_per_fwd": round(dt*1e6,1), "tokens_per_s": round(S/dt),
  "gemm_tflops_at_routed<=": round(flops/ (acc[1]+acc[2]) / 1e9, 1),
  "phase_ms": {"gate": round(acc[0],4), "up": round(acc[1],4), "down": round(acc[2],4), "other": round(acc[3],4)}}))
