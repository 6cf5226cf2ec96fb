"""autodist_amd — an MI355X-native distributed training engine with
petuum/autodist's capabilities.

Built from scratch for one-process-per-GPU over RCCL/xGMI on AMD Instinct
MI355X (gfx950): PyTorch-ROCm front end, hand-written CDNA4 HIP kernels for
the hot synchronization/optimizer ops, strategy-proto-compatible per-variable
synchronization plans (PS / AllReduce / partitioned / hybrid).

Reference API parity: `AutoDist(resource_spec_file, strategy_builder)`,
`.scope()`, `.create_distributed_session()`, `.function(fn)`,
`.build_strategy()` (reference autodist/autodist.py:297-322).
"""
__version__ = "0.1.0"

# torch version gate (reference gates TF 1.15-2.2, __init__.py:35-42)
import torch as _torch

_major, _minor = (int(x) for x in _torch.__version__.split(".")[:2])
if (_major, _minor) < (2, 1):
    raise RuntimeError(
        f"autodist_amd requires torch>=2.1 (post-accumulate-grad hooks); "
        f"found {_torch.__version__}")

from autodist_amd.autodist import AutoDist, get_default_autodist
from autodist_amd.const import ENV

__all__ = ["AutoDist", "ENV", "get_default_autodist", "__version__"]
