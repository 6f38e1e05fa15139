"""FlashMoE-AMD: the FlashDMoE hot path, MI355X-native (gfx950).

Drop-in surface mirroring the reference package `flashmoe`
(flashmoe/__init__.py): run_moe(), get_compiled_config(), plus the module
`flashmoe_amd.moe` mirroring the `_C` extension entry points.
"""
from .moe import (  # noqa: F401
    finalize,
    get_compiled_config,
    get_num_local_experts,
    initialize,
    moe_forward,
)
from .ops import run_moe  # noqa: F401

__version__ = "0.1.0"
__all__ = [
    "run_moe",
    "get_compiled_config",
    "initialize",
    "finalize",
    "moe_forward",
    "get_num_local_experts",
]
