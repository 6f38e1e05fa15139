"""flashmoe_config.json contract (reference: csrc/flashmoe_config.json +
csrc/flashmoe_config.schema.json; loaded/validated at initialize time, the
MI355X analog of the reference's JSON -> CMake -D macro pipeline,
setup.py:223-292 / CMakeLists.txt:112-237)."""
from __future__ import annotations

import json
import os

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
DEFAULT_CONFIG_PATH = os.path.join(REPO_ROOT, "csrc", "flashmoe_config.json")

REQUIRED_KEYS = [
    # csrc/flashmoe_config.schema.json "required"
    "capacity_factor", "drop_tokens", "expert_top_k", "is_training",
    "hidden_act", "hidden_size", "intermediate_size", "mini_batch",
    "moe_frequency", "num_experts", "num_layers", "sequence_len",
    "torch_dtype", "vocab_size",
]

# torch_dtype enum (schema): 0 float, 1 tf32 (== fp32 on CDNA4: no xf32),
# 2 bf16, 3 fp16, 4 fp8e4m3 expert weights with bf16 activations (the
# config-5 "fp8 weights / bf16 accumulate" regime), 5 MX fp8: fp8e4m3
# weights AND runtime-quantized fp8 activations (per-64-element E8M0
# block scales) on the CDNA4 scaled MFMA - the 2x-rate path (extension
# codes; the reference schema stops at 3). torch_dtype_of() is the
# ACTIVATION dtype at the API boundary; weight_dtype_of() the
# expert-weight storage dtype.
def torch_dtype_of(code: int):
    import torch

    return {0: torch.float32, 1: torch.float32, 2: torch.bfloat16,
            3: torch.float16, 4: torch.bfloat16, 5: torch.bfloat16}[code]


def weight_dtype_of(code: int):
    import torch

    if code in (4, 5):
        return torch.float8_e4m3fn
    return torch_dtype_of(code)


def element_size_of(code: int) -> int:
    # activation element size (workspace sizing); dtype-4 weights are 1B
    return {0: 4, 1: 4, 2: 2, 3: 2, 4: 2, 5: 2}[code]


def load_config(path: str | None = None) -> dict:
    path = path or DEFAULT_CONFIG_PATH
    with open(path) as f:
        cfg = json.load(f)
    missing = [k for k in REQUIRED_KEYS if k not in cfg]
    if missing:
        raise ValueError(f"config {path} missing required keys: {missing}")
    if cfg["sequence_len"] % 128:
        raise ValueError("sequence_len must be a multiple of 128 (schema)")
    if cfg["hidden_size"] % 64 or cfg["intermediate_size"] % 64:
        raise ValueError("hidden_size/intermediate_size must be multiples of 64 (schema)")
    return cfg
