"""Single-GPU shape benchmark for any BASELINE-like MoE shape.

Usage: python tools/shape_bench.py [S H P E topk [steps [dtype]]]
Defaults to the BASELINE config-3 shape. Prints one JSON line with
per-phase HIP-event timing (fm_moe_forward_phased).
"""
import ctypes
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

import flashmoe_amd._ext as _ext
from flashmoe_amd import moe

args = [int(a) for a in sys.argv[1:]]
S, H, P, E, topk = (args + [4096, 2048, 8192, 64, 2][len(args):])[:5]
steps = args[5] if len(args) > 5 else 30
dtype_code = args[6] if len(args) > 6 else 2

cfg = {"capacity_factor": 1, "drop_tokens": 1, "expert_top_k": topk,
       "global_batch": 256, "is_training": 0, "hidden_act": 0,
       "hidden_size": H, "intermediate_size": P, "mini_batch": 1,
       "moe_frequency": 1, "num_experts": E, "num_layers": 1,
       "sequence_len": S, "torch_dtype": dtype_code, "vocab_size": 32000}
f = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False)
json.dump(cfg, f)
f.close()
moe.initialize(f.name, rank=0, world_size=1)
from flashmoe_amd.config import torch_dtype_of, weight_dtype_of

adt, wdt = torch_dtype_of(dtype_code), weight_dtype_of(dtype_code)
torch.manual_seed(47)
x = torch.randn(1, S, H, dtype=adt, device="cuda")
gw = torch.randn(H, E, dtype=adt, device="cuda")
ew = torch.randn(E, 2, P, H, dtype=torch.float32, device="cuda").to(wdt)
for _ in range(max(3, steps // 3)):
    moe.moe_forward(x, gw, ew)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(steps):
    moe.moe_forward(x, gw, ew)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / steps

lib = _ext.load()
ms = (ctypes.c_float * 4)()
acc = [0.0] * 4
out = torch.empty_like(x)
go = moe.gate_output()
for _ in range(5):
    _ext.check(lib.fm_moe_forward_phased(
        ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
        ctypes.c_void_p(x.data_ptr()), ctypes.c_void_p(gw.data_ptr()),
        ctypes.c_void_p(ew.data_ptr()), None, None,
        ctypes.c_void_p(go.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        S, ms), "phased")
    for i in range(4):
        acc[i] += ms[i] / 5
flops = 4.0 * S * topk * H * P  # upper bound (pre-capacity-drop)
gemm_ms = acc[1] + acc[2]
print(json.dumps({
    "workload": (f"1xMI355X S={S} H={H} P={P} E={E} top-{topk} "
                 f"dtype{dtype_code} CF=1 drop"),
    "us_per_fwd": round(dt * 1e6, 1),
    "tokens_per_s": round(S / dt),
    "gemm_tflops_upper": round(flops / gemm_ms / 1e9, 1) if gemm_ms else None,
    "phase_ms": {"gate": round(acc[0], 4), "up": round(acc[1], 4),
                 "down": round(acc[2], 4), "other": round(acc[3], 4)},
}))
