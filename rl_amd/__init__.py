"""rl_amd — MI355X-native reinforcement-learning framework.

A from-scratch framework with the capabilities of pytorch/rl (TorchRL),
built MI355X-first: PyTorch-ROCm for the module layer, hand-written
HIP/CDNA4 kernels for the hot ops (GAE/TD(λ)/V-trace scans, prioritized
sum-trees, fused actor epilogues, fused recurrent scans), and RCCL over
xGMI for every collective.
"""
__version__ = "0.1.0"

from ._utils import logger, timeit, seed_everything, set_profiling_enabled, warn
from .services import service_backend

__all__ = ["logger", "timeit", "seed_everything", "set_profiling_enabled", "warn", "service_backend"]

# reference-parity re-exports (torchrl subpackage-level __all__)
from ._utils import (  # noqa: F401
    implement_for,
)
__all__ = sorted(set(list(globals().get('__all__', [])) + ['implement_for']))
from ._utils import (  # noqa: F401
    auto_unwrap_transformed_env,
    compile_with_warmup,
    cuda_memory_profile,
    cuda_memory_stats,
    get_ray_default_runtime_env,
    merge_ray_runtime_env,
    reset_cuda_peak_stats,
    set_auto_unwrap_transformed_env,
    torchrl_logger,
    transport_backend,
)
__all__ = sorted(set(__all__) | {
    "auto_unwrap_transformed_env", "compile_with_warmup", "cuda_memory_profile",
    "cuda_memory_stats", "get_ray_default_runtime_env", "merge_ray_runtime_env",
    "reset_cuda_peak_stats", "set_auto_unwrap_transformed_env",
    "torchrl_logger", "transport_backend",
})
