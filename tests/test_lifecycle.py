"""Lifecycle and contract checks (CPU where possible, GPU where not)."""
import json
import os
import subprocess
import sys
import tempfile

import numpy as np
import pytest

torch = pytest.importorskip("torch")

from tests.conftest import REPO_ROOT


def test_bench_contract_static():
    """bench.py carries the driver contract: flags, JSON keys, workload."""
    src = open(os.path.join(REPO_ROOT, "bench.py")).read()
    for flag in ["--gpus", "--steps", "--warmup"]:
        assert flag in src
    for key in ['"metric"', '"value"', '"unit"', '"n_gpus"', '"steps"',
                '"warmup"', '"ms_per_step"', '"higher_is_better"',
                '"scaling"', '"vs_baseline"', '"dtype"', '"data"',
                '"config"', '"roofline"', '"cpu_baseline"',
                '"overlap_efficiency"']:
        assert key in src, key
    sys.path.insert(0, REPO_ROOT)
    import bench

    cfg = bench.build_config(1)
    assert cfg["num_experts"] == 8 and cfg["sequence_len"] == 4096
    assert bench.build_config(8)["num_experts"] == 64  # weak scaling: E=8N


def test_graft_entry_surface():
    import __graft_entry__ as g

    assert callable(g.build) and callable(g.smoke)


def test_config_rejects_bad_shapes(tmp_path):
    from flashmoe_amd.config import load_config

    bad = {
        "capacity_factor": 1, "drop_tokens": 1, "expert_top_k": 2,
        "global_batch": 256, "is_training": 0, "hidden_act": 0,
        "hidden_size": 100,  # not multiple of 64
        "intermediate_size": 4096, "mini_batch": 1, "moe_frequency": 1,
        "num_experts": 8, "num_layers": 1, "sequence_len": 4096,
        "torch_dtype": 2, "vocab_size": 32000,
    }
    p = tmp_path / "bad.json"
    p.write_text(json.dumps(bad))
    with pytest.raises(ValueError, match="multiples of 64"):
        load_config(str(p))


@pytest.mark.gpu
def test_initialize_finalize_cycles():
    """init/forward/finalize three times in one process (state machine,
    python_bindings.cu init-ordering semantics)."""
    from flashmoe_amd import moe

    cfg = {
        "capacity_factor": 1, "drop_tokens": 1, "expert_top_k": 2,
        "global_batch": 256, "is_training": 0, "hidden_act": 0,
        "hidden_size": 128, "intermediate_size": 256, "mini_batch": 1,
        "moe_frequency": 1, "num_experts": 8, "num_layers": 1,
        "sequence_len": 128, "torch_dtype": 2, "vocab_size": 32000,
    }
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump(cfg, f)
        path = f.name
    outs = []
    for cycle in range(3):
        moe.initialize(path, rank=0, world_size=1)
        torch.manual_seed(11)
        x = torch.randn(1, 128, 128, dtype=torch.bfloat16, device="cuda")
        gw = torch.randn(128, 8, dtype=torch.bfloat16, device="cuda")
        ew = torch.randn(8, 2, 256, 128, dtype=torch.bfloat16, device="cuda")
        out = moe.moe_forward(x, gw, ew)
        torch.cuda.synchronize()
        outs.append(out.float().cpu().numpy().copy())
        with pytest.raises(RuntimeError):
            moe.initialize(path)  # double-init must fail
        moe.finalize()
        with pytest.raises(RuntimeError):
            moe.finalize()  # double-finalize must fail
    assert np.array_equal(outs[0], outs[1]) and np.array_equal(outs[1], outs[2])


@pytest.mark.gpu
def test_run_moe_end_to_end():
    """The reference's own user entry: flashmoe.run_moe() single process
    (ops.py:18-59 / worker.py flow) using the default config contract."""
    env = dict(os.environ, PYTHONPATH=REPO_ROOT)
    r = subprocess.run(
        [sys.executable, "-c",
         "import flashmoe; flashmoe.run_moe(n_processes=1, "
         "config_path='csrc/flashmoe_config.json')"],
        capture_output=True, text=True, timeout=600, cwd=REPO_ROOT, env=env)
    sys.stdout.write(r.stdout[-1500:])
    sys.stderr.write(r.stderr[-1000:])
    assert r.returncode == 0
    assert "Completed! Output: (1, 4096, 1024)" in r.stdout
