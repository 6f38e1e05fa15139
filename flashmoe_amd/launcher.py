"""Multi-process launcher — replaces the reference's nvshmrun subprocess
builder (flashmoe/launcher.py:11-71) with torch.distributed.run over RCCL
(backend "nccl" IS RCCL on ROCm), one rank per GPU, single node."""
from __future__ import annotations

import subprocess
import sys
from pathlib import Path
from typing import Optional


def distributed_launcher(
    config_path: str = "csrc/flashmoe_config.json",
    n_processes: int = 1,
    processes_per_node: int = 1,
    hostfile: Optional[str] = None,
):
    config_path = Path(config_path).resolve()
    if not config_path.exists():
        raise FileNotFoundError(f"Config file not found: {config_path}")
    worker_script = Path(__file__).parent / "worker.py"
    if not worker_script.exists():
        raise FileNotFoundError(f"Worker script not found: {worker_script}")
    if hostfile is not None:
        raise NotImplementedError("multi-node launch is out of scope (single-node xGMI)")

    if n_processes == 1:
        cmd = [sys.executable, str(worker_script), str(config_path)]
    else:
        cmd = [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", f"--nproc-per-node={n_processes}",
            "--master-addr", "127.0.0.1", "--master-port", "29517",
            str(worker_script), str(config_path),
        ]
    print(f"Launching FlashMoE with: {' '.join(cmd)}")
    result = subprocess.run(cmd, capture_output=True, text=True)
    print(result.stdout)
    if result.stderr:
        print("STDERR:", result.stderr, file=sys.stderr)
    if result.returncode != 0:
        raise RuntimeError(f"worker failed with exit code {result.returncode}")
    return result
