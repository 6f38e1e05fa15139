"""Diagnose the E=96 no-drop multi-tile parity failure (sweep seed 2024
case 8): dump mismatching rows, their top-k vs the oracle's, counts and
tie margins."""
import json
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from oracle.moe_oracle import OracleConfig, moe_forward as oracle_forward
from flashmoe_amd import moe
from flashmoe_amd.config import torch_dtype_of

S, H, P, E, k = 256, 512, 256, 96, 2
cfg = {"capacity_factor": 1, "drop_tokens": 0, "expert_top_k": k,
       "global_batch": 256, "is_training": 0, "hidden_act": 0,
       "hidden_size": H, "intermediate_size": P, "mini_batch": 1,
       "moe_frequency": 1, "num_experts": E, "num_layers": 1,
       "sequence_len": S, "torch_dtype": 2, "vocab_size": 32000}
f = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False)
json.dump(cfg, f)
f.close()
moe.initialize(f.name, rank=0, world_size=1)
dt = torch_dtype_of(2)
gten = torch.Generator().manual_seed(1000 + 8)
x = torch.randn(1, S, H, generator=gten).to(dt).cuda()
gw = torch.randn(H, E, generator=gten).to(dt).cuda()
ew = torch.randn(E, 2, P, H, generator=gten).to(dt).cuda()
out = moe.moe_forward(x, gw, ew)
gate_out = moe.gate_output().clone()
torch.cuda.synchronize()

ocfg = OracleConfig(num_experts=E, expert_top_k=k, capacity_factor=1,
                    drop_tokens=0, hidden_act=0, element="bf16")
ref = oracle_forward(x.view(S, H).float().cpu().numpy(),
                     gw.float().cpu().numpy().reshape(-1),
                     ew.float().cpu().numpy(), ocfg)
got = out.view(S, H).float().cpu().numpy()
want = ref["moe_out"]
scale = max(1.0, float(np.abs(want).max()))
ok = np.isclose(got, want, rtol=2e-2, atol=2e-3 * scale)
bad_rows = np.where(~ok.all(axis=1))[0]
print(f"scale {scale:.1f}; bad rows: {len(bad_rows)} -> {bad_rows[:20].tolist()}")

# routing comparison
import ctypes
import flashmoe_amd._ext as _ext
lib = _ext.load()
EC = S * k  # no-drop: S*cf*k
counts = np.zeros(E, dtype=np.uint32)
tok = np.zeros(E * EC, dtype=np.uint32)
ps = np.zeros(E * EC, dtype=np.float32)
_ext.check(lib.fm_read_routing(None, ctypes.c_void_p(counts.ctypes.data),
                               ctypes.c_void_p(tok.ctypes.data),
                               ctypes.c_void_p(ps.ctypes.data)), "routing")
tok = tok.reshape(E, EC)
print("counts equal:", np.array_equal(counts.astype(np.int64), ref["eC"]),
      " sum", counts.sum(), ref["eC"].sum())
# per bad row: oracle topk vs kernel membership
topk = ref["topk_idx"]
tm = ref["tie_margin"]
for r in bad_rows[:8]:
    in_experts = [e for e in range(E) if r in tok[e, :counts[e]].tolist()]
    err = np.abs(got[r] - want[r]).max()
    print(f"row {r}: oracle topk {topk[r].tolist()} kernel-lists {in_experts} "
          f"tie_margin {tm[r]:.5f} maxerr {err:.3f} "
          f"gate_out[r,topk] {gate_out[r, topk[r]].float().cpu().numpy()} "
          f"oracle probs {ref['gate_out'][r, topk[r]]}")
# find rows whose gate_out mismatches oracle
g_got = gate_out.float().cpu().numpy()[:, :E]
g_want = ref["gate_out"][:, :E]
gbad = np.where(~np.isclose(g_got, g_want, rtol=2e-2, atol=2e-3).all(axis=1))[0]
print("gate_out bad rows:", len(gbad), gbad[:10].tolist())
moe.finalize()
