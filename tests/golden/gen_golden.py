"""Generate committed golden vectors for oracle regression pinning.

Run from the repo root: python tests/golden/gen_golden.py
Inputs follow the reference's synthetic protocol: normal(0,1), seed 47
(flashmoe/worker.py:56-58, csrc/benchmarks/flash_bench.cu:40-41).
"""
import json
import os

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
REPO = os.path.dirname(os.path.dirname(HERE))
import sys

sys.path.insert(0, REPO)
from oracle.moe_oracle import OracleConfig, moe_forward  # noqa: E402

CASES = {
    # BASELINE.json config 1 shape, shrunk seq for a small committed file
    "small_fp32_top1": dict(
        S=128, H=64, P=96, cfg=dict(num_experts=4, expert_top_k=1, element="fp32")
    ),
    "small_fp32_top2": dict(
        S=128, H=64, P=96, cfg=dict(num_experts=8, expert_top_k=2, element="fp32")
    ),
    "small_bf16_top2": dict(
        S=128, H=64, P=96, cfg=dict(num_experts=8, expert_top_k=2, element="bf16")
    ),
    # dtype 5 (MX fp8): pins the e4m3 RNE + per-64-block E8M0 activation
    # quantization model (H, P multiples of 64 so blocks are whole)
    "small_mx_top2": dict(
        S=128, H=64, P=128, cfg=dict(num_experts=8, expert_top_k=2,
                                     element="bf16", mx_fp8=True)
    ),
}


def main():
    for name, spec in CASES.items():
        g = np.random.default_rng(47)
        S, H, P = spec["S"], spec["H"], spec["P"]
        cfg = OracleConfig(**spec["cfg"])
        E = cfg.num_experts
        x = g.standard_normal((S, H), dtype=np.float32)
        gate_w = g.standard_normal((H * E,), dtype=np.float32)
        expert_w = g.standard_normal((E, 2, P, H), dtype=np.float32)
        r = moe_forward(x, gate_w, expert_w, cfg)
        np.savez_compressed(
            os.path.join(HERE, f"{name}.npz"),
            x=x,
            gate_w=gate_w,
            expert_w=expert_w,
            moe_out=r["moe_out"],
            gate_out=r["gate_out"],
            topk_idx=r["topk_idx"],
            eC=r["eC"],
            mCw=r["mCw"],
        )
        with open(os.path.join(HERE, f"{name}.json"), "w") as f:
            json.dump({"cfg": spec["cfg"], "S": S, "H": H, "P": P, "seed": 47}, f, indent=1)
        print(name, "written; eC =", r["eC"].tolist())


if __name__ == "__main__":
    main()
