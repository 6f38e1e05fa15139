"""Drop-in alias: `import flashmoe` resolves to the MI355X-native
implementation (flashmoe_amd). Mirrors the reference package surface
(flashmoe/__init__.py: run_moe, get_compiled_config)."""
from flashmoe_amd import (  # noqa: F401
    finalize,
    get_compiled_config,
    get_num_local_experts,
    initialize,
    moe_forward,
    run_moe,
)
from flashmoe_amd import moe as _C  # noqa: F401  (the _C-equivalent surface)

__version__ = "0.1.0"
__all__ = ["run_moe", "get_compiled_config"]
