"""CPU-side checks of the C-ABI library: it loads and exports every symbol
include/flashmoe_abi.h declares (no compute calls without a GPU)."""
import ctypes
import os
import re
import subprocess

import pytest

from tests.conftest import REPO_ROOT

LIB = os.path.join(REPO_ROOT, "flashmoe_amd", "_libflashmoe.so")
HDR = os.path.join(REPO_ROOT, "include", "flashmoe_abi.h")


def _built():
    if not os.path.exists(LIB):
        subprocess.run(
            ["python", "-c", "import __graft_entry__ as g; g.build()"],
            cwd=REPO_ROOT, check=True, capture_output=True,
        )
    return LIB


def header_symbols():
    with open(HDR) as f:
        text = f.read()
    syms = re.findall(r"\b(fm_[a-z0-9_]+)\s*\(", text)
    return sorted(set(syms))


def test_header_declares_the_boundary():
    syms = header_symbols()
    for required in ["fm_initialize", "fm_finalize", "fm_moe_forward",
                     "fm_get_compiled_config", "fm_get_num_local_experts",
                     "fm_gate_forward", "fm_expert_ffn", "fm_combine"]:
        assert required in syms


def test_library_loads_and_exports_all_header_symbols():
    lib = ctypes.CDLL(_built())
    for sym in header_symbols():
        assert getattr(lib, sym, None) is not None, f"missing export: {sym}"
    assert lib.fm_built_for_gfx950() == 1


def test_ext_binding_loads():
    _built()
    import flashmoe_amd._ext as _ext

    lib = _ext.load()
    assert lib.fm_get_num_local_experts() == -1  # not initialized yet


def test_product_path_never_imports_oracle():
    """The oracle is test infrastructure; the product package must not
    reference it."""
    pkg = os.path.join(REPO_ROOT, "flashmoe_amd")
    for root, _, files in os.walk(pkg):
        for fn in files:
            if fn.endswith((".py", ".hip", ".h", ".cpp")):
                with open(os.path.join(root, fn), errors="ignore") as f:
                    assert "oracle" not in f.read(), f"{fn} references oracle/"
    shim = os.path.join(REPO_ROOT, "flashmoe", "__init__.py")
    with open(shim) as f:
        assert "oracle" not in f.read()


def test_config_contract():
    from flashmoe_amd.config import DEFAULT_CONFIG_PATH, load_config

    cfg = load_config()
    assert os.path.basename(DEFAULT_CONFIG_PATH) == "flashmoe_config.json"
    # BASELINE.json config 2 is the default workload
    assert cfg["num_experts"] == 8 and cfg["expert_top_k"] == 2
    assert cfg["hidden_size"] == 1024 and cfg["intermediate_size"] == 4096
    assert cfg["torch_dtype"] == 2
    with pytest.raises(ValueError):
        load_config(os.path.join(REPO_ROOT, "BASELINE.json"))


def test_gpu_less_moe_forward_fails_loudly():
    """On a machine with no GPU the product path raises, never falls back
    to CPU."""
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU present")
    _built()
    from flashmoe_amd import moe

    with pytest.raises(RuntimeError, match="ROCm GPU"):
        moe.initialize()
