"""Core FlashMoE functionality — mirror of the reference's flashmoe/ops.py
(flashmoe/ops.py:1-71), with the nvshmrun launcher replaced by a
torch.distributed spawner (one process per GPU over RCCL)."""
from __future__ import annotations

from typing import Optional

from .launcher import distributed_launcher
from .moe import get_compiled_config  # noqa: F401  (re-exported, ops.py:63-71)


def run_moe(
    n_processes: int = 1,
    processes_per_node: Optional[int] = None,
    hostfile: Optional[str] = None,
    config_path: str = "csrc/flashmoe_config.json",
):
    """Run MoE forward pass with random tensors for benchmarking/testing
    (mirror of ops.run_moe, flashmoe/ops.py:18-59). Tensors are created
    from the compiled configuration; single node, one process per GPU."""
    if processes_per_node is None:
        processes_per_node = n_processes
    return distributed_launcher(
        config_path=config_path,
        n_processes=n_processes,
        processes_per_node=processes_per_node,
        hostfile=hostfile,
    )
